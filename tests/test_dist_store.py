"""LinearBarrier + store bootstrap tests."""

import tempfile
import threading

import pytest
import torch.distributed as dist

from torchsnapshot_amd.dist_store import LinearBarrier, get_or_create_store
from torchsnapshot_amd.pg_wrapper import PGWrapper
from torchsnapshot_amd.test_utils import run_multiprocess

pytestmark = pytest.mark.timeout(120)


def test_linear_barrier_single_rank():
    store = dist.HashStore()
    barrier = LinearBarrier("p", store, rank=0, world_size=1)
    barrier.arrive(timeout_s=5)
    barrier.depart(timeout_s=5)


def test_linear_barrier_threads():
    """Simulate 3 ranks with threads over one HashStore."""
    store = dist.HashStore()
    order = []
    lock = threading.Lock()

    def rank_fn(rank):
        barrier = LinearBarrier("p", store, rank=rank, world_size=3)
        barrier.arrive(timeout_s=20)
        if rank == 0:
            with lock:
                order.append("rank0-privileged")
        barrier.depart(timeout_s=20)
        with lock:
            order.append(f"rank{rank}-done")

    threads = [threading.Thread(target=rank_fn, args=(r,)) for r in range(3)]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert order[0] == "rank0-privileged"
    assert len(order) == 4


def test_linear_barrier_error_propagation():
    store = dist.HashStore()
    results = {}

    def rank0():
        barrier = LinearBarrier("p", store, rank=0, world_size=2)
        try:
            barrier.arrive(timeout_s=20)
            barrier.depart(timeout_s=20)
        except Exception as e:
            results[0] = e

    def rank1():
        barrier = LinearBarrier("p", store, rank=1, world_size=2)
        barrier.report_error(ValueError("injected"))

    t0 = threading.Thread(target=rank0)
    t1 = threading.Thread(target=rank1)
    t1.start()
    t1.join()
    t0.start()
    t0.join()
    assert isinstance(results.get(0), RuntimeError)
    assert "injected" in str(results[0].__cause__ or results[0])


def test_linear_barrier_timeout():
    store = dist.HashStore()
    barrier = LinearBarrier("p", store, rank=0, world_size=2)
    with pytest.raises(TimeoutError):
        barrier.arrive(timeout_s=0.2)


def _store_bootstrap(tmpdir: str) -> None:
    pgw = PGWrapper(None)
    store = get_or_create_store(pgw)
    rank = pgw.get_rank()
    store.set(f"k{rank}", f"v{rank}")
    # both ranks see each other's writes
    assert store.get(f"k{1 - rank}") == f"v{1 - rank}".encode()


def test_get_or_create_store_dist():
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(2, _store_bootstrap, d)
