"""Execute the DeepSpeed ZeRO trick against a vendored engine stub.

deepspeed is not installable in this environment, so a minimal fake
``deepspeed.runtime.engine.DeepSpeedEngine`` reproducing the
`_save_zero_checkpoint`/`_load_zero_checkpoint` call contract (mirroring
reference torchsnapshot/tricks/deepspeed.py:30-103) stands in: the patch
must replace both methods, the save must produce a restorable snapshot,
and the load must return True and restore module + optimizer state.
"""

import os
import sys
import tempfile
import types

import pytest
import torch


@pytest.fixture()
def fake_deepspeed(monkeypatch):
    deepspeed = types.ModuleType("deepspeed")
    runtime = types.ModuleType("deepspeed.runtime")
    engine_mod = types.ModuleType("deepspeed.runtime.engine")

    class DeepSpeedEngine:
        """Just enough surface for the trick: .module, .optimizer, and the
        two zero-checkpoint methods the patch replaces."""

        def __init__(self, module, optimizer=None):
            self.module = module
            self.optimizer = optimizer

        def _save_zero_checkpoint(self, save_dir, tag):
            raise AssertionError("unpatched _save_zero_checkpoint called")

        def _load_zero_checkpoint(
            self, load_dir, tag, load_optimizer_states=True
        ):
            raise AssertionError("unpatched _load_zero_checkpoint called")

    engine_mod.DeepSpeedEngine = DeepSpeedEngine
    runtime.engine = engine_mod
    deepspeed.runtime = runtime
    monkeypatch.setitem(sys.modules, "deepspeed", deepspeed)
    monkeypatch.setitem(sys.modules, "deepspeed.runtime", runtime)
    monkeypatch.setitem(sys.modules, "deepspeed.runtime.engine", engine_mod)
    yield DeepSpeedEngine


def test_patch_and_roundtrip(fake_deepspeed):
    from torchsnapshot_amd.tricks import deepspeed as trick

    trick.patch_deepspeed_engine()
    DeepSpeedEngine = fake_deepspeed
    assert DeepSpeedEngine._save_zero_checkpoint is not None

    torch.manual_seed(0)
    model = torch.nn.Linear(32, 32)
    optim = torch.optim.Adam(model.parameters(), lr=1e-3)
    model(torch.rand(4, 32)).sum().backward()
    optim.step()
    engine = DeepSpeedEngine(model, optim)

    saved_w = model.weight.detach().clone()
    saved_exp_avg = {
        id(p): optim.state[p]["exp_avg"].clone()
        for p in model.parameters()
        if p in optim.state
    }
    assert saved_exp_avg, "optimizer must have state after a step"

    with tempfile.TemporaryDirectory() as d:
        engine._save_zero_checkpoint(d, "step10")
        trick.wait_for_pending()
        assert os.path.exists(
            os.path.join(d, "step10", "tsamd_zero", ".snapshot_metadata")
        )

        with torch.no_grad():
            model.weight.add_(1.0)
            for p in model.parameters():
                if p in optim.state:
                    optim.state[p]["exp_avg"].zero_()

        ok = engine._load_zero_checkpoint(d, "step10")
        assert ok is True
        assert torch.equal(model.weight, saved_w)
        for p in model.parameters():
            if p in optim.state:
                assert torch.equal(
                    optim.state[p]["exp_avg"], saved_exp_avg[id(p)]
                )


def test_load_missing_checkpoint_returns_false(fake_deepspeed):
    from torchsnapshot_amd.tricks import deepspeed as trick

    trick.patch_deepspeed_engine()
    engine = fake_deepspeed(torch.nn.Linear(4, 4))
    with tempfile.TemporaryDirectory() as d:
        assert engine._load_zero_checkpoint(d, "nope") is False


def test_pending_chained_saves(fake_deepspeed):
    """A second save waits for the first pending snapshot (the trick keeps
    at most one outstanding async snapshot)."""
    from torchsnapshot_amd.tricks import deepspeed as trick

    trick.patch_deepspeed_engine()
    model = torch.nn.Linear(16, 16)
    engine = fake_deepspeed(model)
    with tempfile.TemporaryDirectory() as d:
        engine._save_zero_checkpoint(d, "t1")
        engine._save_zero_checkpoint(d, "t2")
        trick.wait_for_pending()
        for tag in ("t1", "t2"):
            assert os.path.exists(
                os.path.join(d, tag, "tsamd_zero", ".snapshot_metadata")
            )
