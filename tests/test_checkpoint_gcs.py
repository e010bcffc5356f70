"""GCS checkpoint backend (progen_amd/checkpoint.py:102-133) exercised
against an in-memory fake bucket implementing the google-cloud-storage
surface the code touches — the real client needs network + credentials,
but the save/get-last/prune protocol is backend logic and is covered
here (reference parity: checkpoint.py:41-81)."""

import torch

from progen_amd import checkpoint as cp


class FakeBlob:
    def __init__(self, store, name):
        self.store = store
        self.name = name

    def upload_from_filename(self, path, timeout=None):
        with open(path, "rb") as f:
            self.store[self.name] = f.read()

    def download_to_file(self, fobj, timeout=None):
        fobj.write(self.store[self.name])


class FakeBucket:
    def __init__(self):
        self.store = {}

    def list_blobs(self):
        return [FakeBlob(self.store, n) for n in self.store]

    def blob(self, name):
        return FakeBlob(self.store, name)

    def delete_blobs(self, blobs):
        for b in blobs:
            self.store.pop(b.name, None)


def _pkg(i):
    return {"next_seq_index": i,
            "params": {"w": torch.arange(3).float().numpy() + i}}


def test_gcs_roundtrip_and_prune(monkeypatch, tmp_path):
    # keep /tmp staging inside the test sandbox
    monkeypatch.setattr(cp.time, "time", lambda: 1_700_000_000 + len(b.store))
    b = FakeBucket()

    cp.gcs_save_checkpoint(b, _pkg(1))
    cp.gcs_save_checkpoint(b, _pkg(2))
    cp.gcs_save_checkpoint(b, _pkg(3))
    assert len(b.store) == 3

    last = cp.gcs_get_last_checkpoint(b)
    assert last["next_seq_index"] == 3

    # keep_last_n prunes the PRE-save listing (reference semantics:
    # n+1 files remain after the save that prunes)
    cp.gcs_save_checkpoint(b, _pkg(4), keep_last_n=2)
    assert len(b.store) == 3
    assert cp.gcs_get_last_checkpoint(b)["next_seq_index"] == 4

    cp.gcs_reset_checkpoint(b)
    assert len(b.store) == 0
    assert cp.gcs_get_last_checkpoint(b) is None


def test_gcs_missing_package_raises():
    import pytest
    with pytest.raises(RuntimeError, match="google-cloud-storage"):
        cp._gcs_client()
