from . import cp, tp, tp_model, zero1  # noqa: F401 — parallelism modules
from .ddp import DistributedTrainer, init_distributed, is_distributed

__all__ = ["DistributedTrainer", "init_distributed", "is_distributed",
           "cp", "tp", "tp_model", "zero1"]
