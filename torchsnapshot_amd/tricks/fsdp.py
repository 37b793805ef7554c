"""FSDP optimizer-state adapter.

FSDP optimizer state must be saved/loaded through FSDP's own
consolidation APIs so it reshards with the parameters (parity with
reference torchsnapshot/tricks/fsdp.py:39-51).
"""

from __future__ import annotations

from typing import Any, Dict

import torch.nn as nn
from torch.optim import Optimizer


class FSDPOptimizerAdapter:
    def __init__(self, module: nn.Module, optimizer: Optimizer) -> None:
        from torch.distributed.fsdp import FullyShardedDataParallel as FSDP

        if not isinstance(module, FSDP):
            raise TypeError(
                "FSDPOptimizerAdapter expects an FSDP-wrapped module"
            )
        self.module = module
        self.optimizer = optimizer

    def state_dict(self) -> Dict[str, Any]:
        from torch.distributed.fsdp import FullyShardedDataParallel as FSDP

        return FSDP.optim_state_dict(self.module, self.optimizer)

    def load_state_dict(self, state_dict: Dict[str, Any]) -> None:
        from torch.distributed.fsdp import FullyShardedDataParallel as FSDP

        load_sd = FSDP.optim_state_dict_to_load(
            self.module, self.optimizer, state_dict
        )
        self.optimizer.load_state_dict(load_sd)
