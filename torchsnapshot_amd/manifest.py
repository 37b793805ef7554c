"""Snapshot manifest schema.

The manifest maps logical paths to typed entries. It is persisted (by rank 0
only) as the ``.snapshot_metadata`` file at the snapshot root, encoded as
JSON (which is also valid YAML, so either loader works — parity with
reference torchsnapshot/manifest.py:442-475). Global manifest keys are
``"<rank>/<logical_path>"``.

Entry kinds:

- containers: ``dict`` / ``ordered_dict`` (ordered keys preserved),
  ``list`` (children at ``<path>/<index>``)
- payloads: ``tensor``, ``chunked_tensor`` (big tensor split for pipelined
  I/O), ``sharded_tensor`` (one shard set spanning all ranks),
  ``dtensor`` (shards + device-mesh/dim-map for replica analysis),
  ``object`` (torch.save fallback)
- ``primitive``: int/float/str/bool/bytes inlined into the metadata itself
  (floats kept exact via ``float.hex()``, bytes via base64)
"""

from __future__ import annotations

import base64
import json
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Union

METADATA_FILENAME = ".snapshot_metadata"


# ---------------------------------------------------------------------------
# entries
# ---------------------------------------------------------------------------


@dataclass
class TensorEntry:
    """A single tensor payload written to ``location``.

    ``serializer`` is "buffer" (raw contiguous bytes, zero-copy) or
    "torch_save" (pickled, used for exotic dtypes / quantized tensors).
    ``byte_range`` is set when the payload lives inside a batched slab file:
    [start, end) byte offsets within ``location``.
    """

    location: str
    serializer: str
    dtype: str
    shape: List[int]
    replicated: bool = False
    byte_range: Optional[List[int]] = None

    KIND = "tensor"

    def nbytes_estimate(self) -> int:
        from .serialization import dtype_size_bytes

        n = 1
        for s in self.shape:
            n *= s
        return n * dtype_size_bytes(self.dtype)

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {
            "kind": self.KIND,
            "location": self.location,
            "serializer": self.serializer,
            "dtype": self.dtype,
            "shape": self.shape,
            "replicated": self.replicated,
        }
        if self.byte_range is not None:
            d["byte_range"] = self.byte_range
        return d

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "TensorEntry":
        return cls(
            location=d["location"],
            serializer=d["serializer"],
            dtype=d["dtype"],
            shape=list(d["shape"]),
            replicated=bool(d.get("replicated", False)),
            byte_range=list(d["byte_range"]) if d.get("byte_range") else None,
        )


@dataclass
class Shard:
    """One shard of a sharded/chunked tensor: its global offsets and sizes
    along each dim, and the TensorEntry holding its payload."""

    offsets: List[int]
    sizes: List[int]
    tensor: TensorEntry

    def to_dict(self) -> Dict[str, Any]:
        return {
            "offsets": self.offsets,
            "sizes": self.sizes,
            "tensor": self.tensor.to_dict(),
        }

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "Shard":
        return cls(
            offsets=list(d["offsets"]),
            sizes=list(d["sizes"]),
            tensor=TensorEntry.from_dict(d["tensor"]),
        )


@dataclass
class ChunkedTensorEntry:
    """A large tensor split along dim 0 into chunks for pipelined I/O."""

    dtype: str
    shape: List[int]
    chunks: List[Shard]
    replicated: bool = False

    KIND = "chunked_tensor"

    def to_dict(self) -> Dict[str, Any]:
        return {
            "kind": self.KIND,
            "dtype": self.dtype,
            "shape": self.shape,
            "chunks": [c.to_dict() for c in self.chunks],
            "replicated": self.replicated,
        }

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ChunkedTensorEntry":
        return cls(
            dtype=d["dtype"],
            shape=list(d["shape"]),
            chunks=[Shard.from_dict(c) for c in d["chunks"]],
            replicated=bool(d.get("replicated", False)),
        )


@dataclass
class ShardedTensorEntry:
    """A ShardedTensor: shards contributed by (possibly many) ranks.

    In the global manifest each rank's entry lists only its local shards;
    on load the per-rank views are merged so every rank sees all shards
    (see manifest_ops.get_manifest_for_rank)."""

    shards: List[Shard]
    dtype: str = ""
    shape: List[int] = field(default_factory=list)

    KIND = "sharded_tensor"

    def to_dict(self) -> Dict[str, Any]:
        return {
            "kind": self.KIND,
            "shards": [s.to_dict() for s in self.shards],
            "dtype": self.dtype,
            "shape": self.shape,
        }

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ShardedTensorEntry":
        return cls(
            shards=[Shard.from_dict(s) for s in d["shards"]],
            dtype=d.get("dtype", ""),
            shape=list(d.get("shape", [])),
        )


@dataclass
class DTensorEntry:
    """A DTensor: local shards plus mesh/dim_map metadata.

    ``mesh`` is the device mesh as a nested int list. ``dim_map`` has one
    inner list per tensor dim, naming the mesh dims that shard it (empty =
    not sharded along that dim). A DTensor whose dim_map is all-empty is
    fully replicated across the mesh."""

    shards: List[Shard]
    mesh: List[Any]
    dim_map: List[List[int]]
    dtype: str = ""
    shape: List[int] = field(default_factory=list)

    KIND = "dtensor"

    def to_dict(self) -> Dict[str, Any]:
        return {
            "kind": self.KIND,
            "shards": [s.to_dict() for s in self.shards],
            "mesh": self.mesh,
            "dim_map": self.dim_map,
            "dtype": self.dtype,
            "shape": self.shape,
        }

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "DTensorEntry":
        return cls(
            shards=[Shard.from_dict(s) for s in d["shards"]],
            mesh=d["mesh"],
            dim_map=[list(x) for x in d["dim_map"]],
            dtype=d.get("dtype", ""),
            shape=list(d.get("shape", [])),
        )


@dataclass
class ObjectEntry:
    """Arbitrary picklable object saved via torch.save."""

    location: str
    serializer: str = "torch_save"
    obj_type: str = ""
    replicated: bool = False

    KIND = "object"

    def to_dict(self) -> Dict[str, Any]:
        return {
            "kind": self.KIND,
            "location": self.location,
            "serializer": self.serializer,
            "obj_type": self.obj_type,
            "replicated": self.replicated,
        }

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ObjectEntry":
        return cls(
            location=d["location"],
            serializer=d.get("serializer", "torch_save"),
            obj_type=d.get("obj_type", ""),
            replicated=bool(d.get("replicated", False)),
        )


@dataclass
class PrimitiveEntry:
    """int/float/str/bool/bytes inlined into the metadata (no payload file).

    Floats are stored as ``float.hex()`` so round-trips are bit-exact;
    bytes as base64."""

    ptype: str  # "int" | "float" | "str" | "bool" | "bytes"
    serialized_value: Union[int, str, bool]
    replicated: bool = False

    KIND = "primitive"

    @classmethod
    def supported(cls, obj: Any) -> bool:
        return type(obj) in (int, float, str, bool, bytes)

    @classmethod
    def from_object(cls, obj: Any, replicated: bool = False) -> "PrimitiveEntry":
        t = type(obj)
        if t is bool:
            return cls("bool", obj, replicated)
        if t is int:
            return cls("int", obj, replicated)
        if t is float:
            return cls("float", obj.hex(), replicated)
        if t is str:
            return cls("str", obj, replicated)
        if t is bytes:
            return cls("bytes", base64.b64encode(obj).decode("ascii"), replicated)
        raise TypeError(f"unsupported primitive type: {t}")

    def get_value(self) -> Any:
        if self.ptype == "bool":
            return bool(self.serialized_value)
        if self.ptype == "int":
            return int(self.serialized_value)
        if self.ptype == "float":
            return float.fromhex(self.serialized_value)
        if self.ptype == "str":
            return str(self.serialized_value)
        if self.ptype == "bytes":
            return base64.b64decode(self.serialized_value)
        raise TypeError(f"unsupported primitive ptype: {self.ptype}")

    def to_dict(self) -> Dict[str, Any]:
        return {
            "kind": self.KIND,
            "ptype": self.ptype,
            "value": self.serialized_value,
            "replicated": self.replicated,
        }

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "PrimitiveEntry":
        return cls(
            ptype=d["ptype"],
            serialized_value=d["value"],
            replicated=bool(d.get("replicated", False)),
        )


@dataclass
class DictEntry:
    keys: List[Union[str, int]] = field(default_factory=list)

    KIND = "dict"

    def to_dict(self) -> Dict[str, Any]:
        return {"kind": self.KIND, "keys": self.keys}

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "DictEntry":
        return cls(keys=list(d["keys"]))


@dataclass
class OrderedDictEntry(DictEntry):
    KIND = "ordered_dict"

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "OrderedDictEntry":
        return cls(keys=list(d["keys"]))


@dataclass
class ListEntry:
    KIND = "list"

    def to_dict(self) -> Dict[str, Any]:
        return {"kind": self.KIND}

    @classmethod
    def from_dict(cls, d: Dict[str, Any]) -> "ListEntry":
        return cls()


Entry = Union[
    TensorEntry,
    ChunkedTensorEntry,
    ShardedTensorEntry,
    DTensorEntry,
    ObjectEntry,
    PrimitiveEntry,
    DictEntry,
    OrderedDictEntry,
    ListEntry,
]

Manifest = Dict[str, Entry]

_KIND_REGISTRY = {
    TensorEntry.KIND: TensorEntry,
    ChunkedTensorEntry.KIND: ChunkedTensorEntry,
    ShardedTensorEntry.KIND: ShardedTensorEntry,
    DTensorEntry.KIND: DTensorEntry,
    ObjectEntry.KIND: ObjectEntry,
    PrimitiveEntry.KIND: PrimitiveEntry,
    DictEntry.KIND: DictEntry,
    OrderedDictEntry.KIND: OrderedDictEntry,
    ListEntry.KIND: ListEntry,
}


def entry_from_dict(d: Dict[str, Any]) -> Entry:
    kind = d.get("kind")
    cls = _KIND_REGISTRY.get(kind)
    if cls is None:
        raise ValueError(f"unknown manifest entry kind: {kind!r}")
    return cls.from_dict(d)


def is_container_entry(entry: Entry) -> bool:
    return isinstance(entry, (DictEntry, OrderedDictEntry, ListEntry))


# ---------------------------------------------------------------------------
# snapshot metadata
# ---------------------------------------------------------------------------


@dataclass
class SnapshotMetadata:
    version: str
    world_size: int
    manifest: Manifest

    def to_json_str(self) -> str:
        return json.dumps(
            {
                "version": self.version,
                "world_size": self.world_size,
                "manifest": {k: v.to_dict() for k, v in self.manifest.items()},
            }
        )

    @classmethod
    def from_str(cls, s: str) -> "SnapshotMetadata":
        # JSON is a strict subset of YAML; try the fast parser first and
        # fall back to YAML for hand-edited metadata.
        try:
            d = json.loads(s)
        except json.JSONDecodeError:
            import yaml

            try:
                loader = yaml.CSafeLoader
            except AttributeError:
                loader = yaml.SafeLoader
            try:
                d = yaml.load(s, Loader=loader)
            except yaml.YAMLError as e:
                raise ValueError(
                    "snapshot metadata is corrupted (neither valid JSON "
                    f"nor YAML): {e}"
                ) from e
        try:
            return cls(
                version=d["version"],
                world_size=int(d["world_size"]),
                manifest={
                    k: entry_from_dict(v) for k, v in d["manifest"].items()
                },
            )
        except (KeyError, TypeError, AttributeError) as e:
            raise ValueError(
                f"snapshot metadata is corrupted (missing/invalid field: {e})"
            ) from e
