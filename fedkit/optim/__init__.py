from .lbfgsnew import LBFGSNew

__all__ = ["LBFGSNew"]
