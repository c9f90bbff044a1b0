from .lbfgsnew import LBFGSNew
from .fusedadam import FusedAdam

__all__ = ["LBFGSNew", "FusedAdam"]
