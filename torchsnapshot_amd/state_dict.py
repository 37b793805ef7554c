"""StateDict: a dict that is itself Stateful, so plain values can be
checkpointed alongside modules.

Parity with reference torchsnapshot/state_dict.py:15-29.

Example::

    progress = StateDict(epoch=0, step=0)
    Snapshot.take(path, {"progress": progress})
"""

from __future__ import annotations

from collections import UserDict
from typing import Any, Dict


class StateDict(UserDict):
    def state_dict(self) -> Dict[str, Any]:
        return dict(self.data)

    def load_state_dict(self, state_dict: Dict[str, Any]) -> None:
        self.data = dict(state_dict)
