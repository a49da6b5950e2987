from .padder import Padder
from .sequence_generator import SequenceGenerator

__all__ = ["Padder", "SequenceGenerator"]
