from .ddp import DistributedTrainer, init_distributed, is_distributed

__all__ = ["DistributedTrainer", "init_distributed", "is_distributed"]
