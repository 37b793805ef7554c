"""TCPStore bootstrap + a store-based barrier usable off the main thread.

The async-snapshot commit path runs on a background thread where
RCCL/process-group collectives are forbidden, so cross-rank synchronization
there goes through a key-value store instead (parity with reference
torchsnapshot/dist_store.py:24-196).

``LinearBarrier`` is a two-phase (arrive/depart) barrier: every rank
arrives, rank 0 observes all arrivals and performs its privileged action
(writing the snapshot metadata), then departs everyone. Errors reported by
any rank propagate to all peers via an error counter + pickled payload.
"""

from __future__ import annotations

import pickle
import socket
import time
from datetime import timedelta
from typing import Any, List

import torch.distributed as dist

from .pg_wrapper import PGWrapper

_STORE_BOOTSTRAP_KEY_PREFIX = "tsamd_store"


def _find_free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("", 0))
        return s.getsockname()[1]


def get_or_create_store(pg_wrapper: PGWrapper) -> dist.Store:
    """Reuse the process group's default store when one exists; otherwise
    bootstrap a TCPStore (rank 0 hosts, address broadcast to peers)."""
    if dist.is_available() and dist.is_initialized():
        try:
            store = dist.distributed_c10d._get_default_store()
            if store is not None:
                return store
        except (RuntimeError, AttributeError):
            pass
    world_size = pg_wrapper.get_world_size()
    rank = pg_wrapper.get_rank()
    if world_size == 1:
        return dist.HashStore()
    # Bootstrap: rank 0 picks a free port and broadcasts (addr, port).
    if rank == 0:
        addr = socket.gethostname()
        port = _find_free_port()
        payload: List[Any] = [(addr, port)]
    else:
        payload = [None]
    pg_wrapper.broadcast_object_list(payload, src=0)
    addr, port = payload[0]
    return dist.TCPStore(
        host_name=addr,
        port=port,
        world_size=world_size,
        is_master=(rank == 0),
        timeout=timedelta(seconds=600),
    )


class LinearBarrier:
    """Two-phase store barrier with error propagation.

    Usage::

        barrier = LinearBarrier(prefix, store, rank, world_size)
        try:
            barrier.arrive(timeout)
            if rank == 0:
                ...privileged action...
            barrier.depart(timeout)
        except Exception as e:
            barrier.report_error(e)
            raise
    """

    _POLL_INTERVAL_S = 0.02

    def __init__(
        self,
        prefix: str,
        store: dist.Store,
        rank: int,
        world_size: int,
    ) -> None:
        self.store = dist.PrefixStore(prefix, store)
        self.rank = rank
        self.world_size = world_size

    # -- error channel -------------------------------------------------------

    def report_error(self, exc: BaseException) -> None:
        try:
            payload = pickle.dumps(exc)
        except Exception:
            payload = pickle.dumps(RuntimeError(repr(exc)))
        self.store.set("error_payload", payload)
        self.store.add("error_flag", 1)

    def _check_error(self) -> None:
        if self.store.add("error_flag", 0) > 0:
            exc = pickle.loads(self.store.get("error_payload"))
            raise RuntimeError(
                f"[LinearBarrier] peer rank reported an error: {exc!r}"
            ) from exc

    # -- phases --------------------------------------------------------------

    def _wait_counter(self, key: str, target: int, timeout_s: float) -> None:
        deadline = time.monotonic() + timeout_s
        while True:
            self._check_error()
            if self.store.add(key, 0) >= target:
                return
            if time.monotonic() > deadline:
                raise TimeoutError(
                    f"[LinearBarrier] timed out waiting for {key} to reach "
                    f"{target} within {timeout_s}s"
                )
            time.sleep(self._POLL_INTERVAL_S)

    def arrive(self, timeout_s: float = 1800.0) -> None:
        self._check_error()
        if self.rank == 0:
            # wait for all non-zero ranks to arrive
            self._wait_counter("arrived", self.world_size - 1, timeout_s)
        else:
            self.store.add("arrived", 1)

    def depart(self, timeout_s: float = 1800.0) -> None:
        if self.rank == 0:
            self.store.add("departed", 1)
            self._check_error()
        else:
            self._wait_counter("departed", 1, timeout_s)
