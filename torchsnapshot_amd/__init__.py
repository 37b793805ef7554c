"""torchsnapshot_amd: MI355X-native distributed checkpointing.

A from-scratch checkpointing framework with the capabilities of
pytorch/torchsnapshot, built for AMD Instinct MI355X nodes: device tensors
stage through a HIP/CDNA4 gather-pack kernel + SDMA D2H copies into pinned
host memory, overlapped with parallel storage writes under a host-memory
budget; metadata coordination runs over RCCL (torch.distributed "nccl"
backend on ROCm) or gloo.
"""

from .rng_state import RNGState
from .snapshot import PendingSnapshot, Snapshot
from .state_dict import StateDict
from .stateful import AppState, Stateful
from .version import __version__

__all__ = [
    "Snapshot",
    "PendingSnapshot",
    "Stateful",
    "AppState",
    "StateDict",
    "RNGState",
    "__version__",
]
