from .converter import CSRConverter
from .discretizer import Discretizer, GreedyDiscretizingRule, QuantileDiscretizingRule
from .filters import (
    ConsecutiveDuplicatesFilter,
    EntityDaysFilter,
    GlobalDaysFilter,
    InteractionEntriesFilter,
    LowRatingFilter,
    MinCountFilter,
    NumInteractionsFilter,
    QuantileItemsFilter,
    TimePeriodFilter,
)
from .label_encoder import (
    LabelEncoder,
    LabelEncoderTransformWarning,
    LabelEncodingRule,
    SequenceEncodingRule,
)
from .sessionizer import Sessionizer

__all__ = [
    "CSRConverter",
    "Discretizer",
    "GreedyDiscretizingRule",
    "QuantileDiscretizingRule",
    "ConsecutiveDuplicatesFilter",
    "EntityDaysFilter",
    "GlobalDaysFilter",
    "InteractionEntriesFilter",
    "LowRatingFilter",
    "MinCountFilter",
    "NumInteractionsFilter",
    "QuantileItemsFilter",
    "TimePeriodFilter",
    "LabelEncoder",
    "LabelEncoderTransformWarning",
    "LabelEncodingRule",
    "SequenceEncodingRule",
    "Sessionizer",
]
