"""Unit tests for load-path manifest transformations."""

from torchsnapshot_amd.flatten import inflate
from torchsnapshot_amd.manifest import (
    ChunkedTensorEntry,
    DictEntry,
    PrimitiveEntry,
    Shard,
    ShardedTensorEntry,
    SnapshotMetadata,
    TensorEntry,
)
from torchsnapshot_amd.manifest_ops import (
    get_manifest_for_rank,
    handle_sharded_tensor_elasticity,
    remove_entry_from_manifest,
)


def _tensor_entry(loc, shape=(4,)):
    return TensorEntry(
        location=loc, serializer="buffer", dtype="float32", shape=list(shape)
    )


def _metadata(manifest, world_size=2):
    return SnapshotMetadata(
        version="0.1.0", world_size=world_size, manifest=manifest
    )


def test_replicated_borrowed_by_other_ranks():
    # rank 1 wrote the replicated payload (partitioner assigned it there);
    # every rank's view must include it
    md = _metadata(
        {
            "0/app": DictEntry(keys=["x"]),
            "1/app": DictEntry(keys=["x"]),
            "1/app/x": TensorEntry(
                location="replicated/app/x",
                serializer="buffer",
                dtype="float32",
                shape=[4],
                replicated=True,
            ),
        }
    )
    for rank in (0, 1):
        manifest, payloads = get_manifest_for_rank(md, rank)
        assert "app/x" in payloads
        assert payloads["app/x"].location == "replicated/app/x"


def test_upscaled_rank_gets_containers_and_replicated():
    md = _metadata(
        {
            "0/app": DictEntry(keys=["x", "s"]),
            "0/app/x": TensorEntry(
                location="replicated/app/x",
                serializer="buffer",
                dtype="float32",
                shape=[4],
                replicated=True,
            ),
            "0/app/s": ShardedTensorEntry(
                shards=[Shard([0], [4], _tensor_entry("sharded/app/s.0"))],
                dtype="float32",
                shape=[8],
            ),
            "1/app": DictEntry(keys=["x", "s"]),
            "1/app/s": ShardedTensorEntry(
                shards=[Shard([4], [4], _tensor_entry("sharded/app/s.4"))],
                dtype="float32",
                shape=[8],
            ),
        }
    )
    manifest, payloads = get_manifest_for_rank(md, rank=5)
    assert "app" in manifest  # container structure borrowed from rank 0
    assert "app/x" in payloads
    # merged shard set visible to the new rank
    assert len(payloads["app/s"].shards) == 2


def test_sharded_entries_merged_across_ranks():
    md = _metadata(
        {
            "0/app/s": ShardedTensorEntry(
                shards=[Shard([0], [4], _tensor_entry("sharded/app/s.0"))],
                dtype="float32",
                shape=[8],
            ),
            "1/app/s": ShardedTensorEntry(
                shards=[Shard([4], [4], _tensor_entry("sharded/app/s.4"))],
                dtype="float32",
                shape=[8],
            ),
        }
    )
    _, payloads = get_manifest_for_rank(md, rank=0)
    offsets = sorted(tuple(s.offsets) for s in payloads["app/s"].shards)
    assert offsets == [(0,), (4,)]


def test_partial_chunked_replicated_merge():
    # the partitioner split a replicated chunked tensor: rank 0 wrote chunk
    # 0, rank 1 wrote chunk 8; readers see the union
    c0 = Shard([0], [8], _tensor_entry("replicated/app/c_c0", (8,)))
    c8 = Shard([8], [8], _tensor_entry("replicated/app/c_c8", (8,)))
    md = _metadata(
        {
            "0/app/c": ChunkedTensorEntry(
                dtype="float32", shape=[16], chunks=[c0], replicated=True
            ),
            "1/app/c": ChunkedTensorEntry(
                dtype="float32", shape=[16], chunks=[c8], replicated=True
            ),
        }
    )
    for rank in (0, 1, 3):
        _, payloads = get_manifest_for_rank(md, rank)
        chunks = payloads["app/c"].chunks
        assert [c.offsets[0] for c in chunks] == [0, 8]


def test_remove_entry_prunes_parent_keys():
    manifest = {
        "app": DictEntry(keys=["a", "b"]),
        "app/a": PrimitiveEntry.from_object(1),
        "app/b": PrimitiveEntry.from_object(2),
    }
    remove_entry_from_manifest(manifest, "app/a")
    assert "app/a" not in manifest
    assert manifest["app"].keys == ["b"]
    # manifest stays inflatable
    out = inflate(manifest, {"app/b": 2}, prefix="app")
    assert out == {"b": 2}


def test_elasticity_drops_unrequested_sharded_entry():
    sharded = ShardedTensorEntry(
        shards=[Shard([0], [4], _tensor_entry("sharded/app/s.0"))],
        dtype="float32",
        shape=[4],
    )
    manifest = {
        "app": DictEntry(keys=["s", "k"]),
        "app/s": sharded,
        "app/k": PrimitiveEntry.from_object(3),
    }
    payloads = {"app/s": sharded, "app/k": manifest["app/k"]}
    target = {"app/k": 0}  # the target never asks for app/s
    handle_sharded_tensor_elasticity(manifest, payloads, target)
    assert "app/s" not in payloads
    assert manifest["app"].keys == ["k"]
    out = inflate(manifest, {"app/k": 3}, prefix="app")
    assert out == {"k": 3}


def test_elasticity_drops_unavailable_target():
    from unittest import mock

    manifest = {"app": DictEntry(keys=["k"])}
    payloads = {}

    class FakeSharded:
        pass

    st = FakeSharded()
    target = {"app/missing": st, "app/k": 1}
    with mock.patch(
        "torchsnapshot_amd.dtensor_utils.is_sharded",
        side_effect=lambda o: isinstance(o, FakeSharded),
    ):
        handle_sharded_tensor_elasticity(manifest, payloads, target)
    assert "app/missing" not in target
    assert "app/k" in target


def test_get_replicated_ranks():
    from torchsnapshot_amd.manifest import DTensorEntry
    from torchsnapshot_amd.manifest_utils import (
        get_replicated_ranks,
        is_partially_replicated_entry,
    )

    # 2x2 mesh, sharded on mesh dim 1 only -> replica sets across dim 0
    e = DTensorEntry(
        shards=[], mesh=[[0, 1], [2, 3]], dim_map=[[1], []],
        dtype="float32", shape=[8, 8],
    )
    assert is_partially_replicated_entry(e)
    sets = get_replicated_ranks(e)
    assert sorted(sorted(s) for s in sets) == [[0, 2], [1, 3]]

    # fully sharded: singleton sets
    e2 = DTensorEntry(
        shards=[], mesh=[[0, 1], [2, 3]], dim_map=[[0], [1]],
        dtype="float32", shape=[8, 8],
    )
    sets = get_replicated_ranks(e2)
    assert sorted(sorted(s) for s in sets) == [[0], [1], [2], [3]]
