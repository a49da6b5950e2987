from .converter import CSRConverter
from .discretizer import (
    Discretizer,
    GreedyDiscretizingRule,
    HandleInvalidStrategies,
    QuantileDiscretizingRule,
)
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
from .history_based_fp import (
    ConditionalPopularityProcessor,
    EmptyFeatureProcessor,
    HistoryBasedFeaturesProcessor,
    LogStatFeaturesProcessor,
)
from .label_encoder import (
    LabelEncoder,
    LabelEncoderPartialFitWarning,
    LabelEncoderTransformWarning,
    LabelEncodingRule,
    SequenceEncodingRule,
)
from .sessionizer import Sessionizer

__all__ = [
    "CSRConverter",
    "Discretizer",
    "GreedyDiscretizingRule",
    "HandleInvalidStrategies",
    "QuantileDiscretizingRule",
    "ConditionalPopularityProcessor",
    "EmptyFeatureProcessor",
    "HistoryBasedFeaturesProcessor",
    "LogStatFeaturesProcessor",
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
    "LabelEncoderPartialFitWarning",
    "LabelEncoderTransformWarning",
    "LabelEncodingRule",
    "SequenceEncodingRule",
    "Sessionizer",
]
