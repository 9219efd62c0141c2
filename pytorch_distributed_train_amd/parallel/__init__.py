from .flat import FlatParams
from .comm import make_communicator, NullComm, TorchDistComm
from .ddp import GradReducer, DistributedDataParallel
from .optim import FusedSGD

__all__ = ["FlatParams", "make_communicator", "NullComm", "TorchDistComm",
           "GradReducer", "DistributedDataParallel", "FusedSGD"]
