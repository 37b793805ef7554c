"""PGWrapper: uniform collectives whether torch.distributed is
uninitialized (world of 1, all no-ops), using the default process group, or
an explicit group.

All snapshot collectives are small object collectives (pickled metadata);
on ROCm the "nccl" backend is RCCL over xGMI, and gloo serves CPU-only
runs. Parity with reference torchsnapshot/pg_wrapper.py:17-91, including
the NCCL/RCCL scatter_object_list fallback (RCCL has no scatter for
objects, so rank 0 broadcasts the full list and each rank picks its slot).
"""

from __future__ import annotations

from typing import Any, List, Optional

import torch.distributed as dist


class PGWrapper:
    def __init__(self, pg: Optional[dist.ProcessGroup] = None) -> None:
        if pg is None and dist.is_available() and dist.is_initialized():
            pg = dist.group.WORLD
        self.pg = pg

    def get_rank(self) -> int:
        if self.pg is None:
            return 0
        return dist.get_rank(group=self.pg)

    def get_world_size(self) -> int:
        if self.pg is None:
            return 1
        return dist.get_world_size(group=self.pg)

    def barrier(self) -> None:
        if self.pg is None:
            return
        dist.barrier(group=self.pg)

    def all_gather_object(self, obj_list: List[Any], obj: Any) -> None:
        if self.pg is None:
            obj_list[0] = obj
            return
        dist.all_gather_object(obj_list, obj, group=self.pg)

    def broadcast_object_list(self, obj_list: List[Any], src: int = 0) -> None:
        if self.pg is None:
            return
        dist.broadcast_object_list(obj_list, src=src, group=self.pg)

    def scatter_object_list(
        self,
        output_list: List[Any],
        input_list: Optional[List[Any]],
        src: int = 0,
    ) -> None:
        if self.pg is None:
            output_list[0] = (input_list or [None])[0]
            return
        backend = dist.get_backend(self.pg)
        if backend == dist.Backend.NCCL:
            # RCCL: no object scatter — broadcast the whole list instead.
            world_size = self.get_world_size()
            if self.get_rank() == src:
                if input_list is None or len(input_list) != world_size:
                    raise ValueError(
                        "scatter_object_list requires input_list of length "
                        f"world_size on src (got {input_list!r})"
                    )
                payload = [input_list]
            else:
                payload = [None]
            self.broadcast_object_list(payload, src=src)
            output_list[0] = payload[0][self.get_rank()]
        else:
            dist.scatter_object_list(output_list, input_list, src=src, group=self.pg)
