"""RNGState: a Stateful that captures/restores the torch CPU RNG state.

When an RNGState instance is present in the app state, Snapshot guarantees
RNG invariance: taking a snapshot captures the RNG state as of entry into
take(), and restoring sets the RNG state last so later draws reproduce
exactly (parity with reference torchsnapshot/rng_state.py:43-47 and the
special-casing in snapshot.py:539-574,371-381).
"""

from __future__ import annotations

from typing import Any, Dict

import torch


class RNGState:
    def state_dict(self) -> Dict[str, Any]:
        return {"rng_state": torch.get_rng_state()}

    def load_state_dict(self, state_dict: Dict[str, Any]) -> None:
        torch.set_rng_state(state_dict["rng_state"])
