"""TFRecord wire format, collate, tokenizer, resume-skip tests."""

import gzip

import numpy as np
import pytest

from progen_amd import data as D


def test_crc32c_known_vectors():
    # standard CRC-32C test vectors
    assert D.crc32c(b"") == 0x00000000
    assert D.crc32c(b"123456789") == 0xE3069283
    assert D.crc32c(b"\x00" * 32) == 0x8A9136AA


def test_example_roundtrip():
    payload = bytes(range(200))
    enc = D.encode_example(payload)
    assert D.decode_example(enc) == payload


def test_tfrecord_roundtrip(tmp_path):
    path = str(tmp_path / "0.3.train.tfrecord.gz")
    seqs = [b"MKV", b"ACDEFGHIKLMNPQRSTVWY", b"GG"]
    with D.with_tfrecord_writer(path) as write:
        for s in seqs:
            write(s)
    got = list(D.iter_tfrecord_file(path))
    assert got == seqs


def test_tfrecord_crc_detects_corruption(tmp_path):
    path = str(tmp_path / "0.1.train.tfrecord.gz")
    with D.with_tfrecord_writer(path) as write:
        write(b"HELLO")
    raw = gzip.open(path, "rb").read()
    bad = raw[:14] + bytes([raw[14] ^ 0xFF]) + raw[15:]
    import io
    with pytest.raises(IOError):
        list(D.read_records(io.BytesIO(bad)))


def test_collate_semantics():
    """uint8 -> uint16, +1 offset, truncate, zero-pad, BOS column
    (reference: data.py:30-35,67-69)."""
    batch = [bytes([10, 20, 30]), bytes(range(50))]
    out = D.collate_fn(batch, pad_length=8, offset=1)
    assert out.dtype == np.uint16
    assert out.shape == (2, 8)
    assert out[0].tolist() == [11, 21, 31, 0, 0, 0, 0, 0]
    assert out[1].tolist() == [1, 2, 3, 4, 5, 6, 7, 8]  # truncated to 8, +1


def test_iterator_and_filename_count_contract(tmp_path):
    # shard name {idx}.{count}.{type}.tfrecord.gz; count parsed from field -4
    # (reference: data.py:46, generate_data.py:142)
    for idx, n in [(0, 2), (1, 3)]:
        path = str(tmp_path / f"{idx}.{n}.train.tfrecord.gz")
        with D.with_tfrecord_writer(path) as write:
            for i in range(n):
                write(bytes([65 + idx * 10 + i] * (3 + i)))
    num_seqs, iter_fn = D.iterator_from_tfrecords_folder(str(tmp_path), "train")
    assert num_seqs == 5

    batches = list(iter_fn(seq_len=6, batch_size=2))
    assert sum(b.shape[0] for b in batches) == 5
    for b in batches:
        assert b.shape[1] == 7  # seq_len + BOS
        assert (b[:, 0] == 0).all()  # BOS column

    # skip semantics: skip=2 drops the first two SEQUENCES (resume contract,
    # reference: data.py:56, train.py:163)
    all_rows = np.concatenate([b for b in iter_fn(seq_len=6, batch_size=1)])
    skipped = np.concatenate([b for b in iter_fn(seq_len=6, batch_size=1, skip=2)])
    np.testing.assert_array_equal(skipped, all_rows[2:])


def test_iterator_loop(tmp_path):
    path = str(tmp_path / "0.1.valid.tfrecord.gz")
    with D.with_tfrecord_writer(path) as write:
        write(b"AB")
    _, iter_fn = D.iterator_from_tfrecords_folder(str(tmp_path), "valid")
    it = iter_fn(seq_len=4, batch_size=1, loop=True)
    rows = [next(it) for _ in range(3)]
    assert len(rows) == 3
    np.testing.assert_array_equal(rows[0], rows[2])


def test_tokenizer_roundtrip():
    s = "[tax=Bacteria] # MKVL"
    toks = D.encode_tokens(s)
    assert toks[0] == ord("[") + 1
    assert D.decode_tokens(np.array(toks)) == s
    # token 0 decodes to '' (PAD/BOS/EOS; reference data.py:79-81)
    assert D.decode_tokens(np.array([0])) == ""


def test_prefetch_iter_order_and_errors():
    from progen_amd.data import prefetch_iter
    assert list(prefetch_iter(iter(range(100)), depth=3)) == list(range(100))
    assert list(prefetch_iter(iter([]), depth=2)) == []

    def boom():
        yield 1
        raise ValueError("producer failed")

    it = prefetch_iter(boom(), depth=2)
    assert next(it) == 1
    import pytest
    with pytest.raises(ValueError, match="producer failed"):
        next(it)
