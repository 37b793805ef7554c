"""Property-based tests (hypothesis): flatten/inflate round-trips over
arbitrary nestings, and pack-layout logical-order over random views."""

import hypothesis.strategies as st
import torch
from hypothesis import given, settings

from torchsnapshot_amd.flatten import flatten, inflate
from torchsnapshot_amd.ops.staging import build_pack_items, collapse_layout

# -- flatten/inflate ---------------------------------------------------------

_keys = st.one_of(
    st.text(
        alphabet=st.characters(
            blacklist_categories=("Cs",), blacklist_characters="\x00"
        ),
        max_size=8,
    ),
    st.integers(min_value=-3, max_value=99),
)
_leaves = st.one_of(
    st.integers(min_value=-(2**40), max_value=2**40),
    st.floats(allow_nan=False),
    st.booleans(),
    st.text(max_size=6),
    st.binary(max_size=6),
)


def _nested(depth: int):
    if depth == 0:
        return _leaves
    sub = _nested(depth - 1)
    return st.one_of(
        _leaves,
        st.lists(sub, max_size=4),
        st.dictionaries(_keys, sub, max_size=4),
    )


@settings(max_examples=80, deadline=None)
@given(obj=st.dictionaries(st.text(max_size=6), _nested(3), max_size=4))
def test_flatten_inflate_round_trip(obj):
    manifest, flattened = flatten(obj, prefix="root")
    rebuilt = inflate(manifest, flattened, prefix="root")
    assert rebuilt == obj


# -- pack layout -------------------------------------------------------------


@settings(max_examples=60, deadline=None)
@given(
    dims=st.lists(st.integers(min_value=1, max_value=7), min_size=1, max_size=4),
    data=st.data(),
)
def test_collapse_layout_any_view(dims, data):
    t = torch.arange(int(torch.tensor(dims).prod()), dtype=torch.float32).reshape(
        dims
    )
    # random slicing per dim
    view = t
    for d in range(t.dim()):
        size = view.shape[d]
        start = data.draw(st.integers(0, max(size - 1, 0)))
        step = data.draw(st.integers(1, 3))
        view = view.narrow(d, start, size - start)
        idx = [slice(None)] * view.dim()
        idx[d] = slice(None, None, step)
        view = view[tuple(idx)]
    maybe_permute = data.draw(st.booleans())
    if maybe_permute and view.dim() > 1:
        perm = data.draw(st.permutations(list(range(view.dim()))))
        view = view.permute(perm)

    sizes, strides = collapse_layout(view)
    v2 = torch.as_strided(view, sizes, strides, view.storage_offset())
    assert torch.equal(
        v2.contiguous().reshape(-1), view.contiguous().reshape(-1)
    )
    # descriptor invariants
    items, offsets, total = build_pack_items([view])
    it = items[0]
    if it.nbytes:
        assert it.row_bytes % it.vec == 0
        assert it.nbytes % it.row_bytes == 0
        rows = it.nbytes // it.row_bytes
        outer = 1
        for s in it.outer_sizes:
            outer *= s
        assert outer == rows or (not it.outer_sizes and rows == 1)


@settings(max_examples=60, deadline=None)
@given(data=st.data())
def test_manifest_entry_json_round_trip(data):
    import json

    from torchsnapshot_amd.manifest import (
        ChunkedTensorEntry,
        PrimitiveEntry,
        Shard,
        TensorEntry,
        entry_from_dict,
    )

    kind = data.draw(st.sampled_from(["tensor", "chunked", "primitive"]))
    if kind == "tensor":
        entry = TensorEntry(
            location=data.draw(st.text(max_size=12)),
            serializer=data.draw(st.sampled_from(["buffer", "torch_save", "qtensor"])),
            dtype="float32",
            shape=data.draw(st.lists(st.integers(0, 64), max_size=4)),
            replicated=data.draw(st.booleans()),
            byte_range=data.draw(
                st.one_of(st.none(), st.tuples(st.integers(0, 100), st.integers(100, 200)).map(list))
            ),
        )
    elif kind == "chunked":
        n = data.draw(st.integers(1, 3))
        entry = ChunkedTensorEntry(
            dtype="bfloat16",
            shape=[n * 4, 2],
            chunks=[
                Shard(
                    offsets=[i * 4, 0],
                    sizes=[4, 2],
                    tensor=TensorEntry(
                        location=f"l{i}", serializer="buffer",
                        dtype="bfloat16", shape=[4, 2],
                    ),
                )
                for i in range(n)
            ],
            replicated=data.draw(st.booleans()),
        )
    else:
        value = data.draw(
            st.one_of(
                st.integers(-(2**50), 2**50),
                st.floats(allow_nan=False),
                st.booleans(),
                st.text(max_size=12),
                st.binary(max_size=12),
            )
        )
        entry = PrimitiveEntry.from_object(value)
        rebuilt = entry_from_dict(json.loads(json.dumps(entry.to_dict())))
        out = rebuilt.get_value()
        assert type(out) is type(value) and out == value
        return

    rebuilt = entry_from_dict(json.loads(json.dumps(entry.to_dict())))
    assert rebuilt.to_dict() == entry.to_dict()


@given(
    nbytes=st.integers(min_value=0, max_value=4096),
    seed=st.integers(min_value=0, max_value=2**31),
)
@settings(max_examples=60, deadline=None)
def test_psum64_subrange_additivity(nbytes, seed):
    """psum64 with file-global word indexing is additive over disjoint
    8-aligned subranges — the invariant the batched-slab ranged checksum
    verification relies on (integrity.verify_ranged_buffer)."""
    import random

    from torchsnapshot_amd.integrity import psum64_hexdigest

    rng = random.Random(seed)
    buf = bytes(rng.getrandbits(8) for _ in range(nbytes))
    whole = psum64_hexdigest(buf)

    # random 8-aligned cut points
    n_cuts = rng.randint(0, 5)
    cuts = sorted({rng.randrange(0, nbytes + 1) & ~7 for _ in range(n_cuts)})
    bounds = [0] + cuts + [nbytes]
    total = 0
    for lo, hi in zip(bounds, bounds[1:]):
        if hi <= lo:
            continue
        part = psum64_hexdigest(buf[lo:hi], word_base=lo // 8)
        total = (total + int(part[len("psum64:"):], 16)) % (1 << 64)
    assert "psum64:" + format(total, "016x") == whole


@given(
    pad=st.integers(min_value=0, max_value=64),
    seed=st.integers(min_value=0, max_value=2**31),
)
@settings(max_examples=30, deadline=None)
def test_psum64_zero_padding_invariant(pad, seed):
    """Zero bytes contribute nothing: the checksum of a buffer equals the
    checksum of the buffer + trailing zeros (slab padding invariant)."""
    import random

    from torchsnapshot_amd.integrity import psum64_hexdigest

    rng = random.Random(seed)
    buf = bytes(rng.getrandbits(8) for _ in range(256))
    assert psum64_hexdigest(buf) == psum64_hexdigest(buf + b"\0" * pad)


import pytest


@pytest.mark.parametrize("seed", [11, 222, 3333, 44444, 555555])
def test_fuzz_snapshot_seeded(seed):
    """Deterministic slices of the randomized snapshot fuzzer
    (scripts/fuzz_snapshot.py): random nested state dicts across dtypes,
    views, aliases, knob combinations — take/restore must round-trip
    bit-exactly."""
    import importlib.util
    import os as _os

    spec = importlib.util.spec_from_file_location(
        "fuzz_snapshot",
        _os.path.join(
            _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))),
            "scripts",
            "fuzz_snapshot.py",
        ),
    )
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    mod.one_case(seed)


def _dist_fuzz_seeds(seeds):
    import importlib.util
    import os as _os

    spec = importlib.util.spec_from_file_location(
        "fuzz_snapshot",
        _os.path.join(
            _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))),
            "scripts",
            "fuzz_snapshot.py",
        ),
    )
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    mod._dist_worker(seeds)


def test_fuzz_distributed_seeded():
    """World-2 slices of the randomized fuzzer: replicated globs,
    tied aliases, per-rank uneven leaves through partitioner + dedup +
    manifest merge."""
    from torchsnapshot_amd.test_utils import run_multiprocess

    run_multiprocess(2, _dist_fuzz_seeds, [77, 7878, 787878])


def _xw_phase(phase, seeds, d):
    import importlib.util
    import os as _os

    spec = importlib.util.spec_from_file_location(
        "fuzz_snapshot",
        _os.path.join(
            _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))),
            "scripts",
            "fuzz_snapshot.py",
        ),
    )
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    (mod._xw_save if phase == "save" else mod._xw_restore)(seeds, d)


def test_fuzz_cross_world_seeded():
    """Elasticity fuzz slice: replicated state saved at world 2 restores
    bit-exactly at world 3 (borrowing) and vice versa."""
    import tempfile as _tf

    from torchsnapshot_amd.test_utils import run_multiprocess

    for w_save, w_restore, base in ((2, 3, 4242), (3, 2, 2424)):
        seeds = [base, base + 1]
        with _tf.TemporaryDirectory() as d:
            run_multiprocess(w_save, _xw_phase, "save", seeds, d)
            run_multiprocess(w_restore, _xw_phase, "restore", seeds, d)
