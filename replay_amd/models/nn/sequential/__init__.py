from .bert4rec import Bert4Rec
from .sasrec import SasRec
from .tisasrec import TiSasRec

__all__ = ["Bert4Rec", "SasRec", "TiSasRec"]
