"""Pluggable telemetry events.

An Event is emitted around every public Snapshot API call (take /
async_take / restore / read_object) with a unique id, rank, and success
flag. Handlers register via the ``tsamd_event_handlers`` entry-point group
(parity with reference torchsnapshot/event.py + event_handlers.py).
"""

from __future__ import annotations

import logging
from dataclasses import dataclass, field
from importlib.metadata import entry_points
from typing import Any, Dict, List, Optional, Protocol

logger = logging.getLogger(__name__)

_ENTRY_POINT_GROUP = "tsamd_event_handlers"


@dataclass
class Event:
    name: str
    metadata: Dict[str, Any] = field(default_factory=dict)


class EventHandler(Protocol):
    def handle_event(self, event: Event) -> None:
        ...


_handlers: Optional[List[EventHandler]] = None
# process-local handlers added programmatically (tests, embedding apps)
_local_handlers: List[EventHandler] = []


def register_event_handler(handler: EventHandler) -> None:
    _local_handlers.append(handler)


def unregister_event_handler(handler: EventHandler) -> None:
    _local_handlers.remove(handler)


def _discovered_handlers() -> List[EventHandler]:
    global _handlers
    if _handlers is None:
        _handlers = []
        try:
            eps = entry_points(group=_ENTRY_POINT_GROUP)
        except TypeError:
            eps = entry_points().get(_ENTRY_POINT_GROUP, [])  # type: ignore[attr-defined]
        for ep in eps:
            try:
                obj = ep.load()
                _handlers.append(obj() if isinstance(obj, type) else obj)
            except Exception:
                logger.exception("failed to load event handler %s", ep.name)
    return _handlers


def log_event(event: Event) -> None:
    for handler in _discovered_handlers() + _local_handlers:
        try:
            handler.handle_event(event)
        except Exception:
            logger.exception("event handler failed for event %s", event.name)
