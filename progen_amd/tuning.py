"""hipBLASLt/rocBLAS GEMM algorithm selection (PyTorch TunableOp).

The library-GEMM path (projections, SGU spatial matmul) goes through
hipBLASLt; the default heuristic algorithm choice leaves 10-30% on the
table for some ProGen shapes. Tuned selections for gfx950 are shipped at
``progen_amd/tuned/tunableop_gfx950.csv`` and loaded at startup (tuning
itself stays OFF in production runs).

To retune (e.g. new shapes):  PYTORCH_TUNABLEOP_TUNING=1 python bench.py ...
then merge the emitted tunableop*.csv into the shipped file.
"""

from __future__ import annotations

import os
from pathlib import Path

TUNED_CSV = Path(__file__).parent / "tuned" / "tunableop_gfx950.csv"


def enable_tuned_gemms() -> bool:
    """Enable TunableOp with the shipped tuned results. Returns True when
    the tuned file was loaded. No-op on CPU."""
    import torch

    if not torch.cuda.is_available():
        return False
    if os.environ.get("PROGEN_NO_TUNABLEOP") == "1":
        return False
    import torch.cuda.tunable as tunable

    tunable.enable(True)
    if os.environ.get("PYTORCH_TUNABLEOP_TUNING") == "1":
        return True  # caller is retuning; keep torch defaults for output
    tunable.tuning_enable(False)
    if TUNED_CSV.exists():
        try:
            tunable.read_file(str(TUNED_CSV))
            return True
        except Exception:  # noqa: BLE001 — fall back to heuristics
            return False
    return False
