from .base_splitter import Splitter, SplitterReturnType
from .splitters import (
    ColdUserRandomSplitter,
    KFolds,
    LastNSplitter,
    NewUsersSplitter,
    RandomNextNSplitter,
    RandomSplitter,
    RatioSplitter,
    TimeSplitter,
    TwoStageSplitter,
)

__all__ = [
    "Splitter",
    "SplitterReturnType",
    "ColdUserRandomSplitter",
    "KFolds",
    "LastNSplitter",
    "NewUsersSplitter",
    "RandomNextNSplitter",
    "RandomSplitter",
    "RatioSplitter",
    "TimeSplitter",
    "TwoStageSplitter",
]
