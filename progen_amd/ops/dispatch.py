"""HIP extension loading and dispatch policy.

Policy (MI355X-native, no multi-backend dispatch):
  - on a GPU (ROCm) device, ops MUST run the hand-written CDNA4 HIP
    kernels: if the in-tree extension is missing we raise rather than
    silently fall back to eager PyTorch;
  - on CPU, ops run the pure-PyTorch reference implementations
    (progen_amd/ops/reference.py) — that is the test oracle, not a
    compatibility layer.

The extension is built IN-TREE as ``progen_amd/_C*.so`` by
``python setup.py build_ext --inplace`` (driven by __graft_entry__.build),
with PYTORCH_ROCM_ARCH=gfx950, so the .so travels with the repo snapshot.
"""

from __future__ import annotations

import importlib
import os
from typing import Any, Optional

_EXT: Optional[Any] = None
_EXT_ERR: Optional[str] = None
_TRIED = False


def _try_load() -> None:
    global _EXT, _EXT_ERR, _TRIED
    if _TRIED:
        return
    _TRIED = True
    try:
        _EXT = importlib.import_module("progen_amd._C")
    except Exception as e:  # noqa: BLE001 - record and re-raise on GPU use
        _EXT = None
        _EXT_ERR = f"{type(e).__name__}: {e}"


def ext() -> Any:
    """Return the loaded HIP extension, raising loudly if unavailable.

    Called only on the GPU path; a GPU box without the built extension is
    a deployment error, never a silent eager fallback."""
    _try_load()
    if _EXT is None:
        raise RuntimeError(
            "progen_amd HIP extension (progen_amd._C) is not available "
            f"(import error: {_EXT_ERR}). Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950)."
        )
    return _EXT


def has_ext() -> bool:
    _try_load()
    return _EXT is not None


def use_hip(t, op: str = None) -> bool:
    """True when tensor t lives on a ROCm GPU (→ HIP kernels are mandatory).

    Debug-only escape hatches (never the default on GPU):
    - PROGEN_FORCE_EAGER=1 routes EVERY op to its torch reference;
    - PROGEN_EAGER_OPS="attn,sgu" routes only the named ops eager —
      the per-kernel bisect lever for the graphed-replay investigation
      (profiles/r02_graphed_nan_investigation.md). Tags: ln, attn, glu,
      sgu, ce, adamw.
    """
    if not t.is_cuda:
        return False
    if os.environ.get("PROGEN_FORCE_EAGER") == "1":
        return False
    if op is not None:
        eager = os.environ.get("PROGEN_EAGER_OPS")
        if eager and op in {x.strip() for x in eager.split(",")}:
            return False
    return True
