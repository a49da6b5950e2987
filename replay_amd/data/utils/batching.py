"""Uniform batch-window arithmetic (reference replay/data/utils/batching.py):
a length is carved into ceil(length / batch_size) half-open [first, last)
windows; the parquet reader uses these to slice row groups."""

from __future__ import annotations

from typing import Iterator, Tuple


def validate_length(length: int) -> int:
    if length < 1:
        raise ValueError(f"Length is invalid. Got {length}.")
    return length


def validate_batch_size(batch_size: int) -> int:
    if batch_size < 1:
        raise ValueError(f"Batch Size is invalid. Got {batch_size}.")
    return batch_size


def uniform_batch_count(length: int, batch_size: int) -> int:
    validate_length(length)
    validate_batch_size(batch_size)
    return (length + batch_size - 1) // batch_size


class UniformBatching:
    """Index -> [first, last) window over a fixed-length collection."""

    def __init__(self, length: int, batch_size: int) -> None:
        self.length = validate_length(length)
        self.batch_size = validate_batch_size(batch_size)

    @property
    def batch_count(self) -> int:
        return uniform_batch_count(self.length, self.batch_size)

    def __len__(self) -> int:
        return self.batch_count

    def get_limits(self, index: int) -> Tuple[int, int]:
        if index < 0 or index >= self.batch_count:
            raise IndexError(f"Batching Index is invalid. Got {index}.")
        first = index * self.batch_size
        return first, min(self.length, first + self.batch_size)

    def __getitem__(self, index: int) -> Tuple[int, int]:
        return self.get_limits(index)

    def __iter__(self) -> Iterator[Tuple[int, int]]:
        for index in range(self.batch_count):
            yield self.get_limits(index)
