from replay_amd.utils import OPENVINO_AVAILABLE
from .base_compiled_model import BaseCompiledModel, Bert4RecCompiled, SasRecCompiled

__all__ = [
    "OPENVINO_AVAILABLE","BaseCompiledModel", "Bert4RecCompiled", "SasRecCompiled"]
