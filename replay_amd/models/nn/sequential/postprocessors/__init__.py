from replay_amd.utils import TORCH_AVAILABLE
"""Legacy postprocessors namespace (reference replay/models/nn/sequential/
postprocessors/postprocessors.py: RemoveSeenItems, SampleItems)."""

from replay_amd.nn.lightning.postprocessor.seen_items import (
    BasePostProcessor,
    SampleItemsFilter as SampleItems,
    SeenItemsFilter as RemoveSeenItems,
)

__all__ = [
    "TORCH_AVAILABLE","BasePostProcessor", "SampleItems", "RemoveSeenItems"]
