from .base_compiled_model import BaseCompiledModel, Bert4RecCompiled, SasRecCompiled

__all__ = ["BaseCompiledModel", "Bert4RecCompiled", "SasRecCompiled"]
