"""DTensor/ShardedTensor runtime predicates.

Parity with reference torchsnapshot/dtensor_utils.py:17-69.
"""

from __future__ import annotations

from typing import Any


def is_sharded(obj: Any) -> bool:
    """True for ShardedTensor, or a DTensor with at least one Shard
    placement."""
    try:
        from torch.distributed._shard.sharded_tensor import ShardedTensor

        if isinstance(obj, ShardedTensor):
            return True
    except ImportError:
        pass
    return is_sharded_dtensor(obj)


def is_sharded_dtensor(obj: Any) -> bool:
    try:
        from torch.distributed.tensor import DTensor
        from torch.distributed.tensor.placement_types import Shard
    except ImportError:
        return False
    if not isinstance(obj, DTensor):
        return False
    return any(isinstance(p, Shard) for p in obj.placements)


def is_replicated_dtensor(obj: Any) -> bool:
    """True for a DTensor with at least one Replicate placement."""
    try:
        from torch.distributed.tensor import DTensor
        from torch.distributed.tensor.placement_types import Replicate
    except ImportError:
        return False
    if not isinstance(obj, DTensor):
        return False
    return any(isinstance(p, Replicate) for p in obj.placements)
