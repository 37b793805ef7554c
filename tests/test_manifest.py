import json

import pytest
import yaml

from torchsnapshot_amd.manifest import (
    ChunkedTensorEntry,
    DictEntry,
    DTensorEntry,
    ListEntry,
    ObjectEntry,
    OrderedDictEntry,
    PrimitiveEntry,
    Shard,
    ShardedTensorEntry,
    SnapshotMetadata,
    TensorEntry,
    entry_from_dict,
)


def _sample_manifest():
    t = TensorEntry(
        location="0/model/weight",
        serializer="buffer",
        dtype="float32",
        shape=[4, 4],
    )
    return {
        "0/model": DictEntry(keys=["weight", "chunked", "sharded", "dt", "obj", "p"]),
        "0/model/weight": t,
        "0/model/chunked": ChunkedTensorEntry(
            dtype="bfloat16",
            shape=[100, 8],
            chunks=[
                Shard(
                    offsets=[0, 0],
                    sizes=[50, 8],
                    tensor=TensorEntry(
                        location="0/model/chunked_c0",
                        serializer="buffer",
                        dtype="bfloat16",
                        shape=[50, 8],
                    ),
                ),
                Shard(
                    offsets=[50, 0],
                    sizes=[50, 8],
                    tensor=TensorEntry(
                        location="0/model/chunked_c50",
                        serializer="buffer",
                        dtype="bfloat16",
                        shape=[50, 8],
                        byte_range=[0, 800],
                    ),
                ),
            ],
        ),
        "0/model/sharded": ShardedTensorEntry(
            shards=[
                Shard(
                    offsets=[0, 0],
                    sizes=[2, 4],
                    tensor=TensorEntry(
                        location="sharded/model/sharded.0",
                        serializer="buffer",
                        dtype="float32",
                        shape=[2, 4],
                    ),
                )
            ],
            dtype="float32",
            shape=[4, 4],
        ),
        "0/model/dt": DTensorEntry(
            shards=[
                Shard(
                    offsets=[0],
                    sizes=[2],
                    tensor=TensorEntry(
                        location="replicated_sharded/model/dt.0",
                        serializer="buffer",
                        dtype="float32",
                        shape=[2],
                    ),
                )
            ],
            mesh=[[0, 1], [2, 3]],
            dim_map=[[0]],
            dtype="float32",
            shape=[4],
        ),
        "0/model/obj": ObjectEntry(location="0/model/obj", obj_type="dict"),
        "0/model/p": PrimitiveEntry.from_object(3.14159),
        "0/l": ListEntry(),
        "0/od": OrderedDictEntry(keys=["x", 3]),
    }


def test_metadata_json_round_trip():
    md = SnapshotMetadata(version="0.1.0", world_size=2, manifest=_sample_manifest())
    s = md.to_json_str()
    md2 = SnapshotMetadata.from_str(s)
    assert md2.version == "0.1.0"
    assert md2.world_size == 2
    assert md2.manifest.keys() == md.manifest.keys()
    for k in md.manifest:
        assert md2.manifest[k].to_dict() == md.manifest[k].to_dict()


def test_metadata_is_valid_yaml():
    md = SnapshotMetadata(version="0.1.0", world_size=1, manifest=_sample_manifest())
    s = md.to_json_str()
    d = yaml.safe_load(s)
    assert d["world_size"] == 1
    assert json.loads(s) == d


def test_primitive_round_trip():
    for value in [1, -7, True, False, "hello", b"\x00\xffbytes", 3.14, -0.0, 1e-308]:
        e = PrimitiveEntry.from_object(value)
        e2 = entry_from_dict(json.loads(json.dumps(e.to_dict())))
        restored = e2.get_value()
        assert type(restored) is type(value)
        assert restored == value or (restored != restored and value != value)


def test_float_exactness():
    import math

    v = math.pi / 3
    e = PrimitiveEntry.from_object(v)
    assert e.get_value() == v  # bit-exact via float.hex


def test_unknown_kind_raises():
    with pytest.raises(ValueError):
        entry_from_dict({"kind": "nope"})


def test_bool_vs_int_primitive():
    e = PrimitiveEntry.from_object(True)
    assert e.ptype == "bool"
    assert e.get_value() is True
    e = PrimitiveEntry.from_object(1)
    assert e.ptype == "int"
    assert e.get_value() == 1
