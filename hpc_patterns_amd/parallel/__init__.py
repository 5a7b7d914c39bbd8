"""parallel — distributed patterns: one process per GPU over
torch.distributed (RCCL on ROCm, gloo for CPU tests), plus HIP-IPC and
hipMemcpyPeerAsync direct-xGMI transports from the native core.

This is the MI355X-native replacement of the reference's GPU-aware-MPI layer
(SURVEY.md §2.7): MPI pt2pt -> dist.send/recv (ncclSend/Recv over xGMI);
MPI_Allreduce -> dist.all_reduce (RCCL); MPI_Win/MPI_Put -> HIP-IPC one-sided
put; Level-Zero topology -> rocm_smi/HIP link matrix.
"""

from .init import init_distributed, rank_world  # noqa: F401
from .ring import ring_allreduce, ring_allreduce_pipelined  # noqa: F401
from .p2p import pairwise_exchange, pingpong  # noqa: F401
from .placement import map_rank_to_gpu  # noqa: F401
from . import topology  # noqa: F401
