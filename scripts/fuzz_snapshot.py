#!/usr/bin/env python3
"""Randomized snapshot fuzzer: random nested state dicts (all dtypes,
views, aliases, empties, primitives, odd shapes), random knob settings,
take/async_take -> restore -> bitwise compare, plus read_object spot
checks. Exits nonzero on the first mismatch/crash with the seed printed.

Usage: python scripts/fuzz_snapshot.py [--iters N] [--seed S]
"""

import argparse
import os
import random
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

DTYPES = [
    torch.float32, torch.float64, torch.float16, torch.bfloat16,
    torch.int8, torch.uint8, torch.int16, torch.int32, torch.int64,
    torch.bool, torch.complex64,
    torch.float8_e4m3fn, torch.float8_e5m2,
]


USE_GPU = False  # set by --gpu: tensors randomly live on cuda:0


def rand_tensor(rng: random.Random):
    dtype = rng.choice(DTYPES)
    ndim = rng.randint(0, 3)
    shape = tuple(rng.randint(0, 9) for _ in range(ndim))
    base = torch.randn(*(s + 2 for s in shape)) if ndim else torch.randn(())
    t = base.to(dtype) if dtype != torch.bool else base > 0
    if USE_GPU and rng.random() < 0.6:
        t = t.cuda()
    kind = rng.random()
    if ndim >= 1 and kind < 0.2:
        t = t[tuple(slice(0, s) for s in shape)]  # view into larger
    elif ndim >= 2 and kind < 0.35:
        t = t.transpose(0, 1)
    elif ndim >= 1 and kind < 0.45:
        t = t[tuple(slice(None, None, 2) for _ in shape)]
    return t


def rand_leaf(rng: random.Random, pool):
    r = rng.random()
    if r < 0.55:
        t = rand_tensor(rng)
        if pool and rng.random() < 0.15:
            return rng.choice(pool)  # alias an earlier tensor
        pool.append(t)
        return t
    if r < 0.65:
        return rng.randint(-(10**12), 10**12)
    if r < 0.72:
        return rng.uniform(-1e30, 1e30)
    if r < 0.8:
        return "".join(rng.choice("ab/%.0 é") for _ in range(rng.randint(0, 12)))
    if r < 0.85:
        return bool(rng.getrandbits(1))
    if r < 0.9:
        return bytes(rng.getrandbits(8) for _ in range(rng.randint(0, 20)))
    if r < 0.95:
        return None
    return {rng.randint(0, 9) for _ in range(3)}  # non-tensor object leaf


def rand_state(rng: random.Random, depth=0):
    pool = []

    def build(d):
        n = rng.randint(1, 5)
        if d < 2 and rng.random() < 0.4:
            out = {}
            for i in range(n):
                key = rng.choice(
                    [f"k{i}", f"odd/{i}", f"p%{i}", str(rng.randint(0, 99))]
                )
                out[key] = build(d + 1) if rng.random() < 0.3 else rand_leaf(rng, pool)
            return out
        if d < 2 and rng.random() < 0.25:
            return [rand_leaf(rng, pool) for _ in range(n)]
        return rand_leaf(rng, pool)

    sd = {}
    for i in range(rng.randint(1, 6)):
        sd[f"top{i}"] = build(0)
    return sd


def eq(a, b, path=""):
    if isinstance(a, torch.Tensor):
        assert isinstance(b, torch.Tensor), f"{path}: {type(b)}"
        ac = a.detach().contiguous().cpu()
        bc = b.detach().contiguous().cpu()
        assert ac.dtype == bc.dtype, f"{path}: dtype {ac.dtype} vs {bc.dtype}"
        assert ac.shape == bc.shape, f"{path}: shape"
        if ac.numel():
            assert torch.equal(
                ac.view(torch.uint8) if ac.dtype.is_floating_point and ac.dtype not in (torch.float32, torch.float64, torch.float16, torch.bfloat16) else ac,
                bc.view(torch.uint8) if bc.dtype.is_floating_point and bc.dtype not in (torch.float32, torch.float64, torch.float16, torch.bfloat16) else bc,
            ), f"{path}: values"
        return
    if isinstance(a, dict):
        assert set(a) == set(b), f"{path}: keys {set(a)} vs {set(b)}"
        for k in a:
            eq(a[k], b[k], f"{path}/{k}")
        return
    if isinstance(a, list):
        assert isinstance(b, list) and len(a) == len(b), f"{path}: list"
        for i, (x, y) in enumerate(zip(a, b)):
            eq(x, y, f"{path}[{i}]")
        return
    if isinstance(a, float):
        assert a == b or (a != a and b != b), f"{path}: {a} vs {b}"
        return
    assert a == b, f"{path}: {a!r} vs {b!r}"


class Holder:
    def __init__(self, sd):
        self.sd = sd

    def state_dict(self):
        return self.sd

    def load_state_dict(self, sd):
        self.sd = sd


def one_case(seed: int) -> None:
    from torchsnapshot_amd import Snapshot

    rng = random.Random(seed)
    torch.manual_seed(seed)
    env = {}
    if rng.random() < 0.3:
        env["TSAMD_DISABLE_BATCHING"] = "1"
    if rng.random() < 0.3:
        env["TSAMD_MAX_CHUNK_SIZE_BYTES"] = str(rng.choice([128, 4096, 65536]))
    if rng.random() < 0.3:
        env["TSAMD_SLAB_SIZE_THRESHOLD_BYTES"] = str(rng.choice([64, 1024, 1 << 20]))
    if rng.random() < 0.5:
        env["TSAMD_CHECKSUM"] = "1"
        env["TSAMD_VERIFY_CHECKSUM"] = "1"
    if rng.random() < 0.5:
        env["TSAMD_ASYNC_SHADOW"] = rng.choice(["0", "1", "auto"])
    old = {k: os.environ.get(k) for k in env}
    os.environ.update(env)
    try:
        sd = rand_state(rng)
        holder = Holder(sd)
        with tempfile.TemporaryDirectory() as d:
            path = os.path.join(d, "snap")
            if rng.random() < 0.5:
                snap = Snapshot.take(path, {"app": holder})
            else:
                snap = Snapshot.async_take(path, {"app": holder}).wait()
            out = Holder({})
            snap.restore({"app": out})
            eq(sd, out.sd, "app")
            # read_object spot check on one random manifest path
            man = snap.get_manifest()
            payloads = [
                k for k, v in man.items()
                if isinstance(v, dict) and v.get("kind") in ("tensor", "object", "primitive")
            ]
            if payloads:
                snap.read_object(rng.choice(payloads))
    finally:
        for k, v in old.items():
            if v is None:
                os.environ.pop(k, None)
            else:
                os.environ[k] = v


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--iters", type=int, default=50)
    ap.add_argument("--seed", type=int, default=None)
    ap.add_argument("--dist", action="store_true", help="world-2 gloo fuzz")
    ap.add_argument("--gpu", action="store_true", help="mix cuda tensors in")
    ap.add_argument("--world", type=int, default=2, help="dist world size")
    ap.add_argument(
        "--cross-world",
        type=str,
        default=None,
        metavar="A:B",
        help="save at world A, restore at world B",
    )
    args = ap.parse_args()
    if args.gpu:
        global USE_GPU
        USE_GPU = True
    if args.cross_world:
        a, b = (int(x) for x in args.cross_world.split(":"))
        base = (
            args.seed
            if args.seed is not None
            else random.SystemRandom().randint(0, 1 << 30)
        )
        run_cross_world(args.iters, base, a, b)
        return
    if args.dist:
        base = (
            args.seed
            if args.seed is not None
            else random.SystemRandom().randint(0, 1 << 30)
        )
        run_dist(args.iters, base, args.world)
        return
    if args.seed is not None:
        one_case(args.seed)
        print(f"seed {args.seed}: OK")
        return
    base = random.SystemRandom().randint(0, 1 << 30)
    for i in range(args.iters):
        seed = base + i
        try:
            one_case(seed)
        except Exception:
            print(f"FUZZ FAILURE at seed {seed}", file=sys.stderr)
            raise
        if (i + 1) % 10 == 0:
            print(f"{i + 1}/{args.iters} ok (base seed {base})")
    print(f"all {args.iters} cases ok (base seed {base})")




# ---------------------------------------------------------------------------
# distributed fuzzing (world 2, gloo): replicated globs + per-rank leaves
# stress the partitioner / dedup / manifest-merge machinery
# ---------------------------------------------------------------------------


def _dist_case(seed: int) -> None:
    import torch.distributed as dist

    from torchsnapshot_amd import Snapshot

    rank = dist.get_rank()
    rng = random.Random(seed)  # same stream on every rank
    torch.manual_seed(seed)
    env = {}
    if rng.random() < 0.3:
        env["TSAMD_DISABLE_BATCHING"] = "1"
    if rng.random() < 0.4:
        env["TSAMD_MAX_CHUNK_SIZE_BYTES"] = str(rng.choice([256, 8192]))
    if rng.random() < 0.4:
        env["TSAMD_SLAB_SIZE_THRESHOLD_BYTES"] = str(rng.choice([256, 1 << 16]))
    if rng.random() < 0.4:
        env["TSAMD_CHECKSUM"] = "1"
        env["TSAMD_VERIFY_CHECKSUM"] = "1"
    if rng.random() < 0.5:
        env["TSAMD_ASYNC_SHADOW"] = rng.choice(["0", "1"])
    old = {k: os.environ.get(k) for k in env}
    os.environ.update(env)
    try:
        # replicated leaves: identical on all ranks (seeded); per-rank
        # leaves keyed off the rank. Tied aliases included.
        pool = []
        sd = {}
        repl_paths = []
        for i in range(rng.randint(1, 6)):
            t = rand_tensor(rng)
            if pool and rng.random() < 0.2:
                t = rng.choice(pool)
            pool.append(t)
            sd[f"rep{i}"] = t
            repl_paths.append(f"app/rep{i}")
        n_local = rng.randint(0, 3)
        for i in range(n_local):
            # rank-dependent content AND size
            torch.manual_seed(seed * 1000 + rank * 7 + i)
            sd[f"mine{i}"] = torch.rand(rng.randint(1, 50) + rank * 3)
        expected = {
            k: (v.clone() if isinstance(v, torch.Tensor) else v)
            for k, v in sd.items()
        }
        holder = Holder(sd)
        # every rank must agree on the replicated list
        globs = rng.choice(
            [repl_paths, ["app/rep*"], ["**/rep*"], repl_paths[:1]]
        )
        with tempfile.TemporaryDirectory() as local_d:
            box = [local_d if rank == 0 else None]
            dist.broadcast_object_list(box, src=0)
            path = os.path.join(box[0], "snap")
            if rng.random() < 0.5:
                snap = Snapshot.take(path, {"app": holder}, replicated=globs)
            else:
                snap = Snapshot.async_take(
                    path, {"app": holder}, replicated=globs
                ).wait()
            out = Holder(
                {
                    k: (torch.zeros_like(v) if isinstance(v, torch.Tensor) else None)
                    for k, v in expected.items()
                }
            )
            snap.restore({"app": out})
            for k, v in expected.items():
                eq(v, out.sd[k], f"rank{rank}/{k}")
            dist.barrier()
    finally:
        for k, v in old.items():
            if v is None:
                os.environ.pop(k, None)
            else:
                os.environ[k] = v


def _dist_worker(seeds) -> None:
    for s in seeds:
        try:
            _dist_case(s)
        except Exception:
            print(f"DIST FUZZ FAILURE at seed {s}", file=sys.stderr)
            raise


def run_dist(iters: int, base: int, world: int = 2) -> None:
    from torchsnapshot_amd.test_utils import run_multiprocess

    seeds = [base + i for i in range(iters)]
    run_multiprocess(world, _dist_worker, seeds)
    print(f"dist world {world}: all {iters} cases ok (base seed {base})")




# ---------------------------------------------------------------------------
# cross-world elasticity fuzzing: save replicated state at world A,
# restore at world B (borrowing for B > A, subset reads for B < A)
# ---------------------------------------------------------------------------


def _xw_state(seed: int):
    rng = random.Random(seed)
    torch.manual_seed(seed)
    pool = []
    sd = {}
    for i in range(rng.randint(1, 6)):
        t = rand_tensor(rng)
        if pool and rng.random() < 0.2:
            t = rng.choice(pool)
        pool.append(t)
        sd[f"rep{i}"] = t
    return sd, rng


def _xw_save(seeds, shared_dir) -> None:
    from torchsnapshot_amd import Snapshot

    for s in seeds:
        sd, rng = _xw_state(s)
        globs = rng.choice([["**"], [f"app/rep{i}" for i in range(len(sd))]])
        path = os.path.join(shared_dir, f"snap_{s}")
        if rng.random() < 0.5:
            Snapshot.take(path, {"app": Holder(sd)}, replicated=globs)
        else:
            Snapshot.async_take(
                path, {"app": Holder(sd)}, replicated=globs
            ).wait()


def _xw_restore(seeds, shared_dir) -> None:
    from torchsnapshot_amd import Snapshot

    for s in seeds:
        expected, _ = _xw_state(s)
        out = Holder(
            {
                k: (torch.zeros_like(v) if isinstance(v, torch.Tensor) else None)
                for k, v in expected.items()
            }
        )
        Snapshot(os.path.join(shared_dir, f"snap_{s}")).restore({"app": out})
        for k, v in expected.items():
            eq(v, out.sd[k], f"xw/{k}")


def run_cross_world(iters: int, base: int, w_save: int, w_restore: int) -> None:
    from torchsnapshot_amd.test_utils import run_multiprocess

    seeds = [base + i for i in range(iters)]
    with tempfile.TemporaryDirectory() as d:
        run_multiprocess(w_save, _xw_save, seeds, d)
        run_multiprocess(w_restore, _xw_restore, seeds, d)
    print(
        f"cross-world {w_save}->{w_restore}: all {iters} cases ok "
        f"(base seed {base})"
    )


if __name__ == "__main__":
    main()
