"""GPU tests: HIP staging engine numerics vs plain torch fp32 reference
paths, and end-to-end snapshots with device tensors."""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():
    pytest.skip("needs a ROCm GPU", allow_module_level=True)

from torchsnapshot_amd import Snapshot, StateDict  # noqa: E402
from torchsnapshot_amd.ops import staging  # noqa: E402
from torchsnapshot_amd.test_utils import tmp_snapshot_path  # noqa: E402


def _ref_bytes(t: torch.Tensor) -> bytes:
    """Reference serialization: plain torch contiguous + cpu."""
    return bytes(
        t.detach().contiguous().cpu().reshape(-1).view(torch.uint8).numpy()
    )


def test_ext_available():
    assert staging.HIP_EXT_AVAILABLE, (
        "_csnap must be importable on a GPU box; a torch fallback would "
        "invalidate every number measured above it"
    )


@pytest.mark.parametrize(
    "shape,dtype,make",
    [
        ((1024, 1024), torch.float32, "contig"),
        ((4096, 4096), torch.bfloat16, "contig"),
        ((333, 777), torch.float16, "contig"),
        ((512, 512), torch.float32, "transpose"),
        ((1000, 64), torch.bfloat16, "narrow"),
        ((100, 100), torch.float32, "strided"),
        ((128,), torch.float64, "contig"),
        ((0, 4), torch.float32, "contig"),
        ((77, 3), torch.uint8, "contig"),
        ((4096, 512), torch.float8_e4m3fn, "contig"),
        ((255, 17), torch.float8_e5m2, "contig"),
    ],
)
def test_stage_single(shape, dtype, make):
    t = torch.randn(shape, dtype=torch.float32).to(dtype).cuda()
    if make == "transpose":
        t = t.t()
    elif make == "narrow":
        t = t[100:900]
    elif make == "strided":
        t = t[::2, ::3]
    engine = staging.get_staging_engine(t.device)
    batch = engine.stage([t])
    batch.wait()
    got = bytes(batch.memoryview_of(0))
    assert got == _ref_bytes(t)
    batch.release()


def test_stage_batch_mixed():
    torch.manual_seed(0)
    tensors = [
        torch.randn(257, 129, device="cuda"),
        torch.randn(64, 64, device="cuda").t(),
        torch.randn(1000, device="cuda", dtype=torch.float32).to(torch.bfloat16),
        torch.randn(33, 5, 7, device="cuda").permute(2, 0, 1),
        torch.randn(1, device="cuda"),
        torch.empty(0, device="cuda"),
    ]
    engine = staging.get_staging_engine(tensors[0].device)
    batch = engine.stage(tensors)
    batch.wait()
    for i, t in enumerate(tensors):
        assert bytes(batch.memoryview_of(i)) == _ref_bytes(t), f"tensor {i}"
    batch.release()


def test_stage_direct_mode(monkeypatch):
    monkeypatch.setenv("TSAMD_STAGE_MODE", "direct")
    tensors = [
        torch.randn(256, 256, device="cuda"),
        torch.randn(128, 128, device="cuda").t(),
    ]
    engine = staging.get_staging_engine(tensors[0].device)
    batch = engine.stage(tensors)
    batch.wait()
    for i, t in enumerate(tensors):
        assert bytes(batch.memoryview_of(i)) == _ref_bytes(t), f"tensor {i}"
    batch.release()


def test_scatter_h2d_round_trip():
    """pack_d2h then scatter_h2d must reproduce the tensors exactly."""
    from torchsnapshot_amd import _csnap
    from torchsnapshot_amd.ops.staging import _items_to_flat, build_pack_items

    torch.manual_seed(1)
    srcs = [
        torch.randn(300, 200, device="cuda"),
        torch.randn(64, 32, device="cuda").t().contiguous().t(),  # contig again
        torch.randn(50, 60, device="cuda")[:, ::2],
    ]
    dsts = [torch.zeros_like(s) for s in srcs]
    items, offsets, total = build_pack_items(srcs)
    pinned = torch.empty(total, dtype=torch.uint8, pin_memory=True)
    slab = torch.empty(total, dtype=torch.uint8, device="cuda")
    h = _csnap.pack_d2h(
        _items_to_flat(items), len(items), slab.data_ptr(),
        pinned.data_ptr(), total, torch.cuda.current_stream().cuda_stream, 0,
    )
    _csnap.wait(h)
    # scatter back into the zeroed destinations
    ditems, _, _ = build_pack_items(dsts)
    slab2 = torch.empty(total, dtype=torch.uint8, device="cuda")
    h = _csnap.scatter_h2d(
        _items_to_flat(ditems), len(ditems), slab2.data_ptr(),
        pinned.data_ptr(), total, torch.cuda.current_stream().cuda_stream, 0,
    )
    _csnap.wait(h)
    torch.cuda.synchronize()
    for s, d in zip(srcs, dsts):
        assert torch.equal(s, d)


def test_snapshot_gpu_round_trip():
    sd = StateDict(
        fp8=torch.randn(512, 256).to(torch.float8_e4m3fn).cuda(),
        big=torch.randn(2048, 2048, dtype=torch.bfloat16, device="cuda"),
        small1=torch.randn(100, device="cuda"),
        small2=torch.randn(64, 64, device="cuda").t(),
        scalar=torch.tensor(3.25, device="cuda"),
        n=5,
    )
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        out = StateDict(
            fp8=torch.zeros(512, 256, dtype=torch.float8_e4m3fn, device="cuda"),
            big=torch.zeros(2048, 2048, dtype=torch.bfloat16, device="cuda"),
            small1=torch.zeros(100, device="cuda"),
            small2=torch.zeros(64, 64, device="cuda"),
            scalar=torch.tensor(0.0, device="cuda"),
            n=0,
        )
        snap.restore({"sd": out})
        assert torch.equal(
            out["fp8"].view(torch.uint8), sd["fp8"].view(torch.uint8)
        )
        assert torch.equal(out["big"], sd["big"])
        assert torch.equal(out["small1"], sd["small1"])
        assert torch.equal(out["small2"], sd["small2"].contiguous())
        assert out["scalar"].item() == 3.25
        assert out["n"] == 5


def test_snapshot_gpu_chunked(monkeypatch):
    monkeypatch.setenv("TSAMD_MAX_CHUNK_SIZE_BYTES", str(1024 * 1024))
    big = torch.randn(1024, 1024, device="cuda")  # 4 MB -> 4 chunks
    sd = StateDict(big=big)
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        out = StateDict(big=torch.zeros(1024, 1024, device="cuda"))
        snap.restore({"sd": out})
        assert torch.equal(out["big"], big)


def test_async_take_gpu():
    sd = StateDict(w=torch.randn(512, 512, device="cuda"))
    with tmp_snapshot_path() as path:
        pending = Snapshot.async_take(path, {"sd": sd})
        # mutation after async_take returns must not corrupt the snapshot
        saved = sd["w"].clone()
        sd["w"].fill_(0.0)
        snap = pending.wait()
        out = StateDict(w=torch.zeros(512, 512, device="cuda"))
        snap.restore({"sd": out})
        assert torch.equal(out["w"], saved)


def test_uvm_managed_tensor():
    """hipMallocManaged tensor: CPU-addressable, staged zero-copy."""
    from torchsnapshot_amd.uvm_tensor import (
        is_uvm_tensor,
        new_managed_tensor,
        prefetch_to_device,
        uvm_to_cpu,
    )

    t = new_managed_tensor((128, 64), dtype=torch.float32)
    t.uniform_(-1, 1)
    # the CPU view addresses managed pages
    assert t.device.type == "cpu"
    prefetch_to_device(t, 0)
    prefetch_to_device(t, -1)
    # snapshot it and restore
    sd = StateDict(emb=t)
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        out = StateDict(emb=torch.zeros(128, 64))
        snap.restore({"sd": out})
        assert torch.equal(out["emb"], t)


def test_dtensor_on_gpu_single_rank():
    """DTensor save/restore on a CUDA 1-rank mesh (RCCL world of 1) —
    exercises the DTensor preparer through the HIP staging engine."""
    import torch.distributed as dist

    created = False
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29531")
        dist.init_process_group(
            "nccl", rank=0, world_size=1, device_id=torch.device("cuda", 0)
        )
        created = True
    try:
        from torch.distributed.device_mesh import init_device_mesh
        from torch.distributed.tensor import distribute_tensor
        from torch.distributed.tensor.placement_types import Shard

        mesh = init_device_mesh("cuda", (1,))
        full = torch.rand(512, 64, device="cuda")
        dt = distribute_tensor(full, mesh, [Shard(0)])

        class Holder:
            def __init__(self, dt):
                self.dt = dt

            def state_dict(self):
                return {"dt": self.dt}

            def load_state_dict(self, sd):
                self.dt = sd["dt"]

        with tmp_snapshot_path() as path:
            snap = Snapshot.take(path, {"obj": Holder(dt)})
            dt2 = distribute_tensor(
                torch.zeros(512, 64, device="cuda"), mesh, [Shard(0)]
            )
            holder = Holder(dt2)
            snap.restore({"obj": holder})
            assert torch.equal(holder.dt.to_local(), full)
    finally:
        if created:
            dist.destroy_process_group()


def test_stage_large_noncontig_materialized():
    """Non-contiguous tensors >=2 GiB exceed the pack kernel's u32
    within-tensor byte indexing and must be materialized (contiguous copy)
    before staging — the engine's only remaining materialization branch
    (torch itself forbids negative strides, so no flip case exists).
    Exercises ops/staging.py's size guard on device."""
    rows = 32768
    cols = 32772  # rows*cols*2 B = 2 GiB + 256 KiB > 2**31
    t = torch.empty(cols, rows, dtype=torch.bfloat16, device="cuda")
    torch.manual_seed(3)
    t.view(torch.int16).random_()
    v = t.t()  # non-contiguous view over > 2**31 bytes
    assert v.numel() * v.element_size() >= 2**31 and not v.is_contiguous()
    engine = staging.get_staging_engine(t.device)
    batch = engine.stage([v])
    batch.wait()
    got = torch.frombuffer(
        bytearray(batch.memoryview_of(0)), dtype=torch.bfloat16
    ).view(rows, cols)
    batch.release()
    ref = v.contiguous().cpu().view(torch.bfloat16)
    assert torch.equal(got.view(torch.int16), ref.view(torch.int16))
    del t, v, ref
    torch.cuda.empty_cache()


def test_gpu_saved_cpu_restored():
    """Cross-device: snapshot taken from device tensors restores into CPU
    tensors (and vice versa)."""
    sd = StateDict(
        big=torch.randn(1024, 1024, dtype=torch.bfloat16, device="cuda"),
        small1=torch.randn(100, device="cuda"),
        small2=torch.randn(64, 64, device="cuda"),
    )
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        out = StateDict(
            big=torch.zeros(1024, 1024, dtype=torch.bfloat16),
            small1=torch.zeros(100),
            small2=torch.zeros(64, 64),
        )
        snap.restore({"sd": out})
        assert torch.equal(out["big"], sd["big"].cpu())
        assert torch.equal(out["small1"], sd["small1"].cpu())
        assert torch.equal(out["small2"], sd["small2"].cpu())

    # CPU-saved -> GPU-restored
    cpu_sd = StateDict(w=torch.rand(256, 128))
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": cpu_sd})
        out = StateDict(w=torch.zeros(256, 128, device="cuda"))
        snap.restore({"sd": out})
        assert torch.equal(out["w"].cpu(), cpu_sd["w"])


def test_tied_weights_on_gpu():
    class Tied(torch.nn.Module):
        def __init__(self):
            super().__init__()
            self.embed = torch.nn.Embedding(128, 64)
            self.head = torch.nn.Linear(64, 128, bias=False)
            self.head.weight = self.embed.weight

    m = Tied().cuda()
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"m": m})
        man = snap.get_manifest()
        assert (
            man["0/m/embed.weight"]["location"]
            == man["0/m/head.weight"]["location"]
        )
        m2 = Tied().cuda()
        snap.restore({"m": m2})
        assert torch.equal(m2.embed.weight, m.embed.weight)


def test_async_stall_much_smaller_than_sync():
    """The headline property: async_take returns (staging done) in a small
    fraction of the full sync save time."""
    import time

    sd = StateDict(
        **{
            f"w{i}": torch.randn(16, 1024, 1024, dtype=torch.bfloat16, device="cuda")
            for i in range(32)
        }
    )  # 1 GB
    with tmp_snapshot_path() as path:
        t0 = time.monotonic()
        Snapshot.take(path, {"sd": sd})
        sync_s = time.monotonic() - t0
    with tmp_snapshot_path() as path:
        t0 = time.monotonic()
        pending = Snapshot.async_take(path, {"sd": sd})
        stall_s = time.monotonic() - t0
        pending.wait()
    # staging 1 GB at ~50 GB/s is ~0.02 s; storage is the long pole.
    # generous bound to stay robust on shared boxes:
    assert stall_s < max(0.5 * sync_s, 0.5), (stall_s, sync_s)


def test_device_psum64_matches_cpu(monkeypatch):
    """The gather kernel's psum64 must match the CPU verifier exactly, for
    every layout class and both stage modes."""
    from torchsnapshot_amd import integrity

    torch.manual_seed(3)
    tensors = [
        torch.randn(333, 257, device="cuda"),                      # contig
        torch.randn(256, 256, device="cuda").t(),                  # transpose
        torch.randn(100, 64, device="cuda")[:, ::2],               # strided
        torch.randn(4096, 64, dtype=torch.bfloat16, device="cuda"),# bf16
        torch.randn(17, device="cuda"),                            # tail < 8B mult
        (torch.rand(77, 3, device="cuda") * 255).to(torch.uint8),  # 1-byte
    ]
    for mode in ("direct", "slab"):
        monkeypatch.setenv("TSAMD_STAGE_MODE", mode)
        engine = staging.get_staging_engine(tensors[0].device)
        batch = engine.stage(tensors, compute_checksums=True)
        batch.wait()
        assert batch.checksums is not None
        total = 0
        for i, t in enumerate(tensors):
            mv = batch.memoryview_of(i)
            # per-tensor checksum uses FILE offsets; recompute via shifting:
            # the CPU check below covers the whole-file identity instead
            total = (total + batch.checksums[i]) % (1 << 64)
        want = integrity.psum64_hexdigest(batch.slab_memoryview())
        got = "psum64:" + format(total, "016x")
        assert got == want, (mode, got, want)
        batch.release()


def test_checksummed_gpu_snapshot_verifies(monkeypatch):
    monkeypatch.setenv("TSAMD_CHECKSUM", "1")
    monkeypatch.setenv("TSAMD_VERIFY_CHECKSUM", "1")
    sd = StateDict(
        big=torch.randn(2048, 2048, dtype=torch.bfloat16, device="cuda"),
        small=torch.randn(100, device="cuda"),
        tr=torch.randn(128, 128, device="cuda").t(),
    )
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        out = StateDict(
            big=torch.zeros(2048, 2048, dtype=torch.bfloat16, device="cuda"),
            small=torch.zeros(100, device="cuda"),
            tr=torch.zeros(128, 128, device="cuda"),
        )
        snap.restore({"sd": out})
        assert torch.equal(out["big"], sd["big"])
        assert torch.equal(out["tr"], sd["tr"].contiguous())


def test_device_psum64_kernel_matches_cpu():
    """The device psum64 reduction agrees with the numpy verifier for
    arbitrary sizes incl. non-multiple-of-8 tails and nonzero word_base."""
    from torchsnapshot_amd.integrity import psum64_hexdigest
    from torchsnapshot_amd.ops.staging import device_psum64

    torch.manual_seed(0)
    for nbytes in (0, 1, 7, 8, 64, 4097, 1 << 20):
        for word_base in (0, 8, 123 * 8 // 8):
            host = torch.randint(
                0, 256, (max(nbytes, 1),), dtype=torch.uint8
            )[:nbytes]
            dev = host.cuda()
            got = device_psum64(dev, word_base * 8)
            want = psum64_hexdigest(
                host.numpy().tobytes(), word_base=word_base
            )
            assert f"psum64:{got:016x}" == want, (nbytes, word_base)


def test_device_verified_restore_roundtrip(monkeypatch):
    """Checksummed save of device tensors + verify-enabled restore into
    device tensors: verification runs ON the GPU (consumer device path)
    and a corrupted payload fails the restore."""
    monkeypatch.setenv("TSAMD_CHECKSUM", "1")
    sd = StateDict(
        big=torch.randn(2048, 1024, dtype=torch.bfloat16, device="cuda"),
        w2=torch.randn(1024, 512, device="cuda"),
    )
    with tmp_snapshot_path() as path:
        snap = Snapshot.take(path, {"sd": sd})
        monkeypatch.setenv("TSAMD_VERIFY_CHECKSUM", "1")
        out = StateDict(
            big=torch.zeros(2048, 1024, dtype=torch.bfloat16, device="cuda"),
            w2=torch.zeros(1024, 512, device="cuda"),
        )
        snap.restore({"sd": out})
        assert torch.equal(out["big"], sd["big"])
        assert torch.equal(out["w2"], sd["w2"])

        # corrupt one byte of one payload (standalone or slab)
        import glob

        payloads = [
            p
            for p in glob.glob(os.path.join(path, "**"), recursive=True)
            if os.path.isfile(p) and not p.endswith((".snapshot_metadata", ".checksums"))
        ]
        target = max(payloads, key=os.path.getsize)
        with open(target, "r+b") as f:
            f.seek(1024)
            b = f.read(1)
            f.seek(1024)
            f.write(bytes([b[0] ^ 0x40]))
        with pytest.raises(RuntimeError, match="checksum mismatch"):
            snap.restore(
                {
                    "sd": StateDict(
                        big=torch.zeros(
                            2048, 1024, dtype=torch.bfloat16, device="cuda"
                        ),
                        w2=torch.zeros(1024, 512, device="cuda"),
                    )
                }
            )


def test_kitchen_sink_gpu_everything_on(monkeypatch):
    """Maximal integration: chunking + batching + checksums + async take +
    device-verified restore + tiled read_object, all in one snapshot."""
    monkeypatch.setenv("TSAMD_CHECKSUM", "1")
    monkeypatch.setenv("TSAMD_MAX_CHUNK_SIZE_BYTES", str(2 * 1024 * 1024))
    torch.manual_seed(7)
    sd = StateDict(
        big=torch.randn(2048, 1024, device="cuda"),  # 8 MB -> 4 chunks
        **{f"s{i}": torch.randn(199, 33, device="cuda") for i in range(9)},
        host=torch.randn(321, 5),
        t_view=torch.randn(128, 128, device="cuda").t(),
        n=41,
        f=2.5,
    )
    saved = {
        k: (v.clone() if isinstance(v, torch.Tensor) else v)
        for k, v in sd.items()
    }
    with tmp_snapshot_path() as path:
        pending = Snapshot.async_take(path, {"sd": sd})
        # mutate everything while I/O is in flight
        with torch.no_grad():
            for v in sd.values():
                if isinstance(v, torch.Tensor):
                    v.zero_()
        snap = pending.wait()

        monkeypatch.setenv("TSAMD_VERIFY_CHECKSUM", "1")
        out = StateDict(
            big=torch.zeros(2048, 1024, device="cuda"),
            **{f"s{i}": torch.zeros(199, 33, device="cuda") for i in range(9)},
            host=torch.zeros(321, 5),
            t_view=torch.zeros(128, 128, device="cuda"),
            n=0,
            f=0.0,
        )
        snap.restore({"sd": out})
        for k, v in saved.items():
            if isinstance(v, torch.Tensor):
                got = out[k]
                want = v.contiguous().to(got.device)
                assert torch.equal(got, want), k
            else:
                assert out[k] == v, k

        # tiled random access with verification active
        big = snap.read_object("0/sd/big", memory_budget_bytes=1024 * 1024)
        assert torch.equal(big.cpu(), saved["big"].cpu())
