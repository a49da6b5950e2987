from .optimizer_factory import FatLRSchedulerFactory, FatOptimizerFactory, LRSchedulerFactory, OptimizerFactory

__all__ = ["FatLRSchedulerFactory", "FatOptimizerFactory", "LRSchedulerFactory", "OptimizerFactory"]
