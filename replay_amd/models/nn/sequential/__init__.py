from replay_amd.utils import TORCH_AVAILABLE

from .bert4rec import Bert4Rec
from .sasrec import SasRec
from .tisasrec import TiSasRec

__all__ = [
    "TORCH_AVAILABLE","Bert4Rec", "SasRec", "TiSasRec"]
