from .comm import init_distributed  # noqa: F401
from .ddp import DistributedDataParallel, average_gradients  # noqa: F401
from .rccl_comm import RcclComm  # noqa: F401
