"""Adapters for training-framework state layouts (DDP, FSDP, DeepSpeed)."""

from .ddp import DDPWrappedAdapter, StripDDPPrefixAdapter
from .fsdp import FSDPOptimizerAdapter

__all__ = [
    "StripDDPPrefixAdapter",
    "DDPWrappedAdapter",
    "FSDPOptimizerAdapter",
]
