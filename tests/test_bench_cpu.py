"""Smoke the bench pipeline on CPU: tiny model, world 1 and world 2
(gloo), exactly the code path the driver exercises on the GPU node."""

import json
import os
import subprocess
import sys
import tempfile

import pytest

pytestmark = pytest.mark.timeout(300)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(cmd, env_extra=None):
    env = dict(os.environ)
    env.update(env_extra or {})
    return subprocess.run(
        cmd, cwd=REPO, env=env, capture_output=True, text=True, timeout=240
    )


def test_bench_tiny_single():
    with tempfile.TemporaryDirectory() as d:
        r = _run(
            [
                sys.executable,
                "bench.py",
                "--device",
                "cpu",
                "--model",
                "tiny",
                "--steps",
                "2",
                "--warmup",
                "1",
                "--dir",
                d,
            ]
        )
        assert r.returncode == 0, r.stderr[-2000:]
        line = r.stdout.strip().splitlines()[-1]
        out = json.loads(line)
        assert out["metric"] == "checkpoint_save_GBps"
        assert out["n_gpus"] == 1
        assert out["ms_per_step"] > 0
        assert out["stall_sec"] >= 0


def test_bench_tiny_world2_gloo():
    with tempfile.TemporaryDirectory() as d:
        r = _run(
            [
                sys.executable,
                "-m",
                "torch.distributed.run",
                "--nnodes=1",
                "--nproc-per-node",
                "2",
                "--master-addr",
                "127.0.0.1",
                "--master-port",
                "29511",
                "bench.py",
                "--device",
                "cpu",
                "--model",
                "tiny",
                "--steps",
                "2",
                "--warmup",
                "1",
                "--dir",
                d,
            ]
        )
        assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
        lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
        out = json.loads(lines[-1])
        assert out["n_gpus"] == 2
        assert out["config"]["parallelism"] == "fsdp2"
        assert out["ms_per_step"] > 0


def test_bench_tiny_world3_uneven_gloo():
    """World 3 makes every tiny shape split unevenly (64 -> 22/22/20,
    16 -> 6/6/4): the bench (and the DTensor save path under it) must not
    require divisible shards."""
    with tempfile.TemporaryDirectory() as d:
        r = _run(
            [
                sys.executable,
                "-m",
                "torch.distributed.run",
                "--nnodes=1",
                "--nproc-per-node",
                "3",
                "--master-addr",
                "127.0.0.1",
                "--master-port",
                "29513",
                "bench.py",
                "--device",
                "cpu",
                "--model",
                "tiny",
                "--steps",
                "1",
                "--warmup",
                "1",
                "--dir",
                d,
            ]
        )
        assert r.returncode == 0, (r.stdout[-1000:], r.stderr[-2000:])
        lines = [l for l in r.stdout.strip().splitlines() if l.startswith("{")]
        out = json.loads(lines[-1])
        assert out["n_gpus"] == 3
        assert out["ms_per_step"] > 0
