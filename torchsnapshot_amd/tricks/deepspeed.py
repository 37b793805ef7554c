"""DeepSpeed ZeRO integration.

``patch_deepspeed_engine`` rewires a DeepSpeedEngine's zero-checkpoint
save/load to torchsnapshot_amd (async save; per-rank ZeRO partitions are
exactly the per-rank entries this library persists natively). Gated on
deepspeed being importable (parity with reference
torchsnapshot/tricks/deepspeed.py:30-103).
"""

from __future__ import annotations

import logging
from typing import Any, Dict, Optional

from ..snapshot import PendingSnapshot, Snapshot

logger = logging.getLogger(__name__)


class Zero3StateAdapter:
    """Stateful exposing a DeepSpeed ZeRO-3 engine's partitioned state
    (fp16 flat partitions + optimizer state) as per-rank entries."""

    def __init__(self, engine: Any) -> None:
        self.engine = engine

    def state_dict(self) -> Dict[str, Any]:
        sd: Dict[str, Any] = {
            "module": self.engine.module.state_dict(),
        }
        if getattr(self.engine, "optimizer", None) is not None:
            sd["optimizer"] = self.engine.optimizer.state_dict()
        return sd

    def load_state_dict(self, state_dict: Dict[str, Any]) -> None:
        self.engine.module.load_state_dict(state_dict["module"])
        if "optimizer" in state_dict and self.engine.optimizer is not None:
            self.engine.optimizer.load_state_dict(state_dict["optimizer"])


_pending: Optional[PendingSnapshot] = None


def patch_deepspeed_engine() -> None:
    """Replace DeepSpeedEngine._save_zero_checkpoint /
    _load_zero_checkpoint with torchsnapshot_amd async snapshots."""
    try:
        from deepspeed.runtime.engine import DeepSpeedEngine
    except ImportError as e:
        raise RuntimeError(
            "deepspeed is not installed; patch_deepspeed_engine is a no-op "
            "without it"
        ) from e

    def _save_zero_checkpoint(self: Any, save_dir: str, tag: str) -> None:
        global _pending
        if _pending is not None:
            _pending.wait()
        path = f"{save_dir}/{tag}/tsamd_zero"
        _pending = Snapshot.async_take(path, {"zero": Zero3StateAdapter(self)})

    def _load_zero_checkpoint(
        self: Any, load_dir: str, tag: str, load_optimizer_states: bool = True
    ) -> bool:
        path = f"{load_dir}/{tag}/tsamd_zero"
        try:
            Snapshot(path).restore({"zero": Zero3StateAdapter(self)})
        except Exception:
            logger.exception("failed to restore zero checkpoint from %s", path)
            return False
        return True

    DeepSpeedEngine._save_zero_checkpoint = _save_zero_checkpoint
    DeepSpeedEngine._load_zero_checkpoint = _load_zero_checkpoint


def wait_for_pending() -> None:
    global _pending
    if _pending is not None:
        _pending.wait()
        _pending = None
