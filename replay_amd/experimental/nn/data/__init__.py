from .schema_builder import TensorSchemaBuilder

__all__ = ["TensorSchemaBuilder"]
