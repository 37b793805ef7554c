"""Test utilities shipped with the package (like the reference's
torchsnapshot/test_utils.py): state-dict equality oracles, random tensors
of every dtype, and multi-process launchers for distributed CPU tests.
"""

from __future__ import annotations

import contextlib
import functools
import os
import socket
import tempfile
from typing import Any, Callable, Dict, Iterator

import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def tensor_eq(a: torch.Tensor, b: torch.Tensor) -> bool:
    """Equality for tensors of any flavor (dense, quantized, ShardedTensor,
    DTensor)."""
    try:
        from torch.distributed._shard.sharded_tensor import ShardedTensor

        if isinstance(a, ShardedTensor) or isinstance(b, ShardedTensor):
            if not (isinstance(a, ShardedTensor) and isinstance(b, ShardedTensor)):
                return False
            a_shards = {tuple(s.metadata.shard_offsets): s.tensor for s in a.local_shards()}
            b_shards = {tuple(s.metadata.shard_offsets): s.tensor for s in b.local_shards()}
            if a_shards.keys() != b_shards.keys():
                return False
            return all(tensor_eq(a_shards[k], b_shards[k]) for k in a_shards)
    except ImportError:
        pass
    try:
        from torch.distributed.tensor import DTensor

        if isinstance(a, DTensor) or isinstance(b, DTensor):
            if not (isinstance(a, DTensor) and isinstance(b, DTensor)):
                return False
            return tensor_eq(a.full_tensor(), b.full_tensor())
    except ImportError:
        pass
    if a.is_quantized != b.is_quantized:
        return False
    if a.is_quantized:
        if a.qscheme() != b.qscheme():
            return False
        return torch.equal(a.dequantize(), b.dequantize())
    if a.dtype != b.dtype or a.shape != b.shape:
        return False
    return torch.equal(a.cpu(), b.cpu())


def _value_eq(a: Any, b: Any) -> bool:
    if isinstance(a, torch.Tensor) or isinstance(b, torch.Tensor):
        if not (torch.is_tensor(a) or _is_dist_tensor(a)) or not (
            torch.is_tensor(b) or _is_dist_tensor(b)
        ):
            return False
        return tensor_eq(a, b)
    if isinstance(a, dict) and isinstance(b, dict):
        if a.keys() != b.keys():
            return False
        return all(_value_eq(a[k], b[k]) for k in a)
    if isinstance(a, (list, tuple)) and isinstance(b, (list, tuple)):
        if len(a) != len(b):
            return False
        return all(_value_eq(x, y) for x, y in zip(a, b))
    return bool(a == b)


def _is_dist_tensor(x: Any) -> bool:
    try:
        from torch.distributed._shard.sharded_tensor import ShardedTensor

        if isinstance(x, ShardedTensor):
            return True
    except ImportError:
        pass
    try:
        from torch.distributed.tensor import DTensor

        return isinstance(x, DTensor)
    except ImportError:
        return False


def check_state_dict_eq(a: Dict[str, Any], b: Dict[str, Any]) -> bool:
    return _value_eq(a, b)


def assert_state_dict_eq(tc, a: Dict[str, Any], b: Dict[str, Any]) -> None:
    tc.assertTrue(
        check_state_dict_eq(a, b),
        f"state dicts differ:\n{a}\n---\n{b}",
    )


def rand_tensor(shape, dtype: torch.dtype) -> torch.Tensor:
    """Random tensor of any dtype, including int/bool/complex/quantized."""
    if dtype in (torch.qint8, torch.quint8, torch.qint32):
        f = torch.rand(shape)
        return torch.quantize_per_tensor(f, scale=0.1, zero_point=1, dtype=dtype)
    if dtype == torch.bool:
        return torch.rand(shape) > 0.5
    if not dtype.is_floating_point and not dtype.is_complex:
        return torch.randint(0, 64, shape, dtype=dtype)
    if dtype in (
        getattr(torch, "float8_e4m3fn", None),
        getattr(torch, "float8_e5m2", None),
        getattr(torch, "float8_e4m3fnuz", None),
        getattr(torch, "float8_e5m2fnuz", None),
    ):
        return torch.rand(shape, dtype=torch.float32).to(dtype)
    return torch.rand(shape, dtype=torch.float32).to(dtype)


# ---------------------------------------------------------------------------
# multi-process launching (CPU/gloo; the reference uses torchelastic for
# the same purpose, test_utils.py:210-270)
# ---------------------------------------------------------------------------


def _find_free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _mp_entry(
    rank: int,
    world_size: int,
    port: int,
    module_name: str,
    qualname: str,
    args: tuple,
) -> None:
    import importlib

    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group(backend="gloo", rank=rank, world_size=world_size)
    try:
        module = importlib.import_module(module_name)
        obj: Any = module
        for part in qualname.split("."):
            obj = getattr(obj, part)
        fn = getattr(obj, "__wrapped__", obj)
        fn(*args)
    finally:
        dist.destroy_process_group()


def run_multiprocess(nproc: int, fn: Callable[..., None], *args: Any) -> None:
    """Run ``fn(*args)`` in ``nproc`` spawned processes with a gloo process
    group initialized (rendezvous on 127.0.0.1). ``fn`` must be resolvable
    as a module-level attribute (the children re-import its module)."""
    port = _find_free_port()
    mp.start_processes(
        _mp_entry,
        args=(nproc, port, fn.__module__, fn.__qualname__, args),
        nprocs=nproc,
        start_method="spawn",
        join=True,
    )


def _mp_entry_gpu(
    rank: int,
    world_size: int,
    port: int,
    backend: str,
    share_device: bool,
    module_name: str,
    qualname: str,
    args: tuple,
) -> None:
    import importlib

    device_index = 0 if share_device else rank
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["LOCAL_RANK"] = str(device_index)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.cuda.set_device(device_index)
    kwargs: Dict[str, Any] = {}
    if backend == "nccl":
        kwargs["device_id"] = torch.device("cuda", device_index)
    dist.init_process_group(
        backend=backend, rank=rank, world_size=world_size, **kwargs
    )
    try:
        module = importlib.import_module(module_name)
        obj: Any = module
        for part in qualname.split("."):
            obj = getattr(obj, part)
        fn = getattr(obj, "__wrapped__", obj)
        fn(*args)
    finally:
        dist.destroy_process_group()


def run_multiprocess_gpu(
    nproc: int,
    fn: Callable[..., None],
    *args: Any,
    backend: str = "nccl",
    share_device: bool = False,
) -> None:
    """GPU flavor of :func:`run_multiprocess`: rank r uses cuda:r (RCCL
    over xGMI on a multi-GPU node). With ``share_device=True`` every rank
    uses cuda:0 and the backend must be gloo (RCCL refuses multiple ranks
    on one device) — this exercises the multi-rank save/restore logic
    with device tensors even on a 1-GPU box."""
    if share_device and backend == "nccl":
        raise ValueError("share_device requires the gloo backend")
    port = _find_free_port()
    mp.start_processes(
        _mp_entry_gpu,
        args=(
            nproc,
            port,
            backend,
            share_device,
            fn.__module__,
            fn.__qualname__,
            args,
        ),
        nprocs=nproc,
        start_method="spawn",
        join=True,
    )


def run_with_dist(nproc: int):
    """Decorator form: the wrapped test launches itself in N gloo-connected
    processes. The test function must be module-level (re-importable)."""

    def decorator(fn: Callable[..., None]) -> Callable[..., None]:
        @functools.wraps(fn)
        def wrapper(*args: Any, **kwargs: Any) -> None:
            assert not kwargs, "kwargs unsupported in run_with_dist tests"
            run_multiprocess(nproc, wrapper, *args)

        wrapper.__wrapped__ = fn  # type: ignore[attr-defined]
        return wrapper

    return decorator


@contextlib.contextmanager
def tmp_snapshot_path() -> Iterator[str]:
    with tempfile.TemporaryDirectory() as d:
        yield os.path.join(d, "snapshot")
