from .trainer import Trainer, move_batch

__all__ = ["Trainer", "move_batch"]
