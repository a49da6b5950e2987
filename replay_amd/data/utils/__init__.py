from .batching import UniformBatching, uniform_batch_count, validate_batch_size, validate_length

__all__ = ["UniformBatching", "uniform_batch_count", "validate_batch_size", "validate_length"]
