from .callbacks import EarlyStopping, ModelCheckpoint
from .trainer import Trainer, move_batch

__all__ = ["EarlyStopping", "ModelCheckpoint", "Trainer", "move_batch"]
