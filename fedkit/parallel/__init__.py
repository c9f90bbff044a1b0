from .comm import Communicator, LocalComm, DistComm, make_comm
from .runtime import FederatedJob, FedConfig

__all__ = ["Communicator", "LocalComm", "DistComm", "make_comm",
           "FederatedJob", "FedConfig"]
