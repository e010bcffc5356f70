"""Checkpoint subsystem (reference: progen_transformer/checkpoint.py).

Layout preserved exactly:
  - ``{path}/ckpt_{unix_time}.pkl`` pickled dict (reference:
    checkpoint.py:25-37, package schema train.py:196-202);
  - lexically-last checkpoint wins on resume (checkpoint.py:16-23);
  - prune to ``keep_last_n`` (checkpoint.py:33-37);
  - factory dispatches on a ``gs://`` prefix (checkpoint.py:85-109) — the
    GCS backend needs google-cloud-storage and raises cleanly when the
    dependency is absent (offline image).

Torch tensors are stored as numpy arrays inside the pickle for
cross-framework readability (a reference-style consumer can read the
params without torch).
"""

from __future__ import annotations

import pickle
import time
from functools import partial
from pathlib import Path
from typing import Any, Dict, Optional, Tuple

import numpy as np

from .utils import clear_directory_, silentremove


def tensors_to_numpy(obj: Any) -> Any:
    import torch

    if isinstance(obj, torch.Tensor):
        t = obj.detach().cpu()
        if t.dtype == torch.bfloat16:
            t = t.float()  # numpy has no bf16; store fp32
        return t.numpy()
    if isinstance(obj, dict):
        return {k: tensors_to_numpy(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return type(obj)(tensors_to_numpy(v) for v in obj)
    return obj


def numpy_to_tensors(obj: Any) -> Any:
    import torch

    if isinstance(obj, np.ndarray):
        return torch.from_numpy(obj.copy())
    if isinstance(obj, dict):
        return {k: numpy_to_tensors(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return type(obj)(numpy_to_tensors(v) for v in obj)
    return obj


# -- local filesystem backend (reference: checkpoint.py:12-37) ---------------

def file_reset_checkpoint(path: Path) -> None:
    clear_directory_(path)


def file_get_last_checkpoint(path: Path) -> Optional[Dict]:
    checkpoints = sorted(path.glob("**/ckpt_*"))
    if len(checkpoints) == 0:
        return None
    with open(str(checkpoints[-1]), "rb") as f:
        return pickle.load(f)


def file_save_checkpoint(path: Path, package: Dict,
                         keep_last_n: Optional[int] = None) -> None:
    unix_time = int(time.time())
    checkpoints = sorted(path.glob("**/ckpt_*"))
    num_checkpoints = len(checkpoints)

    with open(str(path / f"ckpt_{unix_time}.pkl"), "wb") as f:
        pickle.dump(package, f)

    if keep_last_n is None:
        return
    for p in checkpoints[: max(0, num_checkpoints - keep_last_n)]:
        silentremove(p)


# -- GCS backend (reference: checkpoint.py:41-81) ----------------------------

GCS_READ_TIMEOUT = 60 * 30
GCS_WRITE_TIMEOUT = 60 * 30


def _gcs_client():
    try:
        from google.cloud import storage  # type: ignore
    except ImportError as e:
        raise RuntimeError(
            "gs:// checkpoint paths require google-cloud-storage, which is "
            "not installed in this offline image") from e
    return storage.Client()


def gcs_reset_checkpoint(bucket) -> None:
    bucket.delete_blobs(list(bucket.list_blobs()))


def gcs_get_last_checkpoint(bucket) -> Optional[Dict]:
    blobs = sorted(bucket.list_blobs(), key=lambda b: b.name)
    if len(blobs) == 0:
        return None
    last = blobs[-1]
    filename = f"/tmp/{last.name}"
    with open(filename, "wb") as f:
        last.download_to_file(f, timeout=GCS_READ_TIMEOUT)
    with open(filename, "rb") as f:
        return pickle.load(f)


def gcs_save_checkpoint(bucket, package: Dict,
                        keep_last_n: Optional[int] = None) -> None:
    unix_time = int(time.time())
    blobs = sorted(bucket.list_blobs(), key=lambda b: b.name)
    num_checkpoints = len(blobs)

    filename = f"ckpt_{unix_time}.pkl"
    tmp_path = f"/tmp/{filename}"
    with open(tmp_path, "wb") as f:
        pickle.dump(package, f)
    blob = bucket.blob(filename)
    blob.upload_from_filename(tmp_path, timeout=GCS_WRITE_TIMEOUT)

    if keep_last_n is None:
        return
    bucket.delete_blobs(blobs[: max(0, num_checkpoints - keep_last_n)])


# -- factory (reference: checkpoint.py:85-109) -------------------------------

def get_checkpoint_fns(path: str) -> Tuple:
    """Returns (reset, get_last, save) partials bound to the backend."""
    use_gcs = path.startswith("gs://")
    if not use_gcs:
        obj: Any = Path(path)
        obj.mkdir(exist_ok=True, parents=True)
        fns = (file_reset_checkpoint, file_get_last_checkpoint,
               file_save_checkpoint)
    else:
        client = _gcs_client()
        obj = client.get_bucket(path[5:])
        fns = (gcs_reset_checkpoint, gcs_get_last_checkpoint,
               gcs_save_checkpoint)
    return tuple(partial(fn, obj) for fn in fns)
