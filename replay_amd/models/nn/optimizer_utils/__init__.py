from replay_amd.utils import TORCH_AVAILABLE
from .optimizer_factory import FatLRSchedulerFactory, FatOptimizerFactory, LRSchedulerFactory, OptimizerFactory

__all__ = [
    "TORCH_AVAILABLE","FatLRSchedulerFactory", "FatOptimizerFactory", "LRSchedulerFactory", "OptimizerFactory"]
