"""Crash consistency: a process SIGKILLed mid-save leaves an UNCOMMITTED
snapshot (no metadata), and the same path can be re-used afterwards."""

import os
import signal
import subprocess
import sys
import tempfile
import time

import pytest
import torch

from torchsnapshot_amd import Snapshot, StateDict

pytestmark = pytest.mark.timeout(180)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_sigkill_mid_save_leaves_no_commit():
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "snap")
        env = dict(os.environ, PYTHONPATH=REPO)
        proc = subprocess.Popen(
            [sys.executable, os.path.join(REPO, "tests", "_crash_child.py"), path],
            cwd=REPO,
            env=env,
            stdout=subprocess.PIPE,
            stderr=subprocess.PIPE,
            text=True,
        )
        # wait until the child is inside take(), then kill it hard
        line = proc.stdout.readline()
        assert "taking" in line
        time.sleep(0.3)
        proc.send_signal(signal.SIGKILL)
        proc.wait(timeout=30)

        # no commit happened
        assert not os.path.exists(os.path.join(path, ".snapshot_metadata"))
        with pytest.raises((RuntimeError, ValueError)):
            _ = Snapshot(path).metadata

        # the same path is reusable for a healthy snapshot
        sd = StateDict(a=torch.rand(16))
        snap = Snapshot.take(path, {"sd": sd})
        out = StateDict()
        snap.restore({"sd": out})
        assert torch.equal(out["a"], sd["a"])
