"""DDP prefix adapters.

A module saved while wrapped in DistributedDataParallel has every key
prefixed with ``module.``; these adapters bridge the two layouts in either
direction (parity with reference torchsnapshot/tricks/ddp.py:17-47).
"""

from __future__ import annotations

from typing import Any, Dict

import torch.nn as nn

_PREFIX = "module."


class StripDDPPrefixAdapter:
    """Restore a DDP-saved snapshot into a PLAIN module: the adapter
    presents DDP-style (prefixed) keys to the snapshot so in-place targets
    line up, and strips the prefix when loading."""

    def __init__(self, module: nn.Module) -> None:
        self.module = module

    def state_dict(self) -> Dict[str, Any]:
        return {_PREFIX + k: v for k, v in self.module.state_dict().items()}

    def load_state_dict(self, state_dict: Dict[str, Any]) -> None:
        stripped = {
            (k[len(_PREFIX):] if k.startswith(_PREFIX) else k): v
            for k, v in state_dict.items()
        }
        self.module.load_state_dict(stripped)


class DDPWrappedAdapter:
    """Restore a plain-module snapshot into a DDP-WRAPPED module: presents
    unprefixed keys, adds the prefix when loading into the wrapper."""

    def __init__(self, ddp_module: nn.Module) -> None:
        self.ddp_module = ddp_module

    def state_dict(self) -> Dict[str, Any]:
        return {
            (k[len(_PREFIX):] if k.startswith(_PREFIX) else k): v
            for k, v in self.ddp_module.state_dict().items()
        }

    def load_state_dict(self, state_dict: Dict[str, Any]) -> None:
        self.ddp_module.load_state_dict(
            {_PREFIX + k: v for k, v in state_dict.items()}
        )
