"""The shipped examples must run (CPU)."""

import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.timeout(300)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(script):
    return subprocess.run(
        [sys.executable, os.path.join(REPO, "examples", script)],
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=240,
    )


def test_example_simple():
    r = _run("simple.py")
    assert r.returncode == 0, r.stderr[-2000:]
    assert "resumed at step" in r.stdout


def test_example_train_loop():
    r = _run("train_loop.py")
    assert r.returncode == 0, r.stderr[-2000:]
    assert "restored step: 30" in r.stdout


def test_example_ddp_world2():
    r = subprocess.run(
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
            "29571",
            os.path.join(REPO, "examples", "ddp.py"),
        ],
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=240,
    )
    assert r.returncode == 0, (r.stdout[-800:], r.stderr[-1500:])
    assert "restored; step = 1" in r.stdout


def test_example_sharded_embedding_world2():
    r = subprocess.run(
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
            "29572",
            os.path.join(REPO, "examples", "sharded_embedding.py"),
        ],
        cwd=REPO,
        capture_output=True,
        text=True,
        timeout=240,
    )
    assert r.returncode == 0, (r.stdout[-800:], r.stderr[-1500:])
    assert "restored; all table rows verified" in r.stdout
