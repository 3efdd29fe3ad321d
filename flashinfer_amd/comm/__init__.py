"""Communication & parallelism over RCCL/xGMI (reference parity:
flashinfer/comm/). Control plane = torch.distributed; data plane = RCCL
collectives shaped for the 8xMI355X xGMI topology (7 direct links per GPU:
prefer all-to-all / one-shot patterns over rings)."""

from .allreduce import AllReduceFusionPattern, allreduce, allreduce_fusion
from .comm_backend import TorchDistBackend, init_distributed
from .mapping import Mapping
from .moe_alltoall import MoeAlltoAll
from .ulysses import UlyssesCommunicator
from .dcp import dcp_gather_o, dcp_scatter_q
from .all_gather_matmul import all_gather_matmul
from .quantized_allreduce import quantized_all_reduce
from .hip_ipc import create_shared_buffer, free_shared_buffer
from .custom_ar import CustomAllReduce

__all__ = [
    "AllReduceFusionPattern",
    "allreduce",
    "allreduce_fusion",
    "TorchDistBackend",
    "init_distributed",
    "Mapping",
    "MoeAlltoAll",
    "UlyssesCommunicator",
    "dcp_scatter_q",
    "dcp_gather_o",
]
