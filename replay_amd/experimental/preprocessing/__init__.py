from .data_preparator import DataPreparator, Indexer
from .padder import Padder
from .sequence_generator import SequenceGenerator

__all__ = ["DataPreparator", "Indexer", "Padder", "SequenceGenerator"]
