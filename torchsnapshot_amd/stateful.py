"""The Stateful protocol: anything with state_dict()/load_state_dict().

Parity with reference torchsnapshot/stateful.py:15-23.
"""

from __future__ import annotations

from typing import Any, Dict, Protocol, runtime_checkable


@runtime_checkable
class Stateful(Protocol):
    def state_dict(self) -> Dict[str, Any]:
        ...

    def load_state_dict(self, state_dict: Dict[str, Any]) -> None:
        ...


# The application state captured by a snapshot: a str-keyed mapping of
# stateful objects (modules, optimizers, dataloaders, custom objects...).
AppState = Dict[str, Stateful]
