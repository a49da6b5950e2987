from .seen_items import BasePostProcessor, SampleItemsFilter, SeenItemsFilter

PostprocessorBase = BasePostProcessor  # reference name

__all__ = ["BasePostProcessor", "PostprocessorBase", "SampleItemsFilter", "SeenItemsFilter"]
