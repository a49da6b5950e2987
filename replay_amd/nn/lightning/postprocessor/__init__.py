from .seen_items import BasePostProcessor, SampleItemsFilter, SeenItemsFilter

__all__ = ["BasePostProcessor", "SampleItemsFilter", "SeenItemsFilter"]
