"""Snapshot inspection CLI.

    python -m torchsnapshot_amd inspect <path>       # manifest tree + sizes
    python -m torchsnapshot_amd cat <path> <entry>   # print one object
"""

from __future__ import annotations

import argparse
import sys
from collections import defaultdict

from .manifest import (
    ChunkedTensorEntry,
    DTensorEntry,
    ObjectEntry,
    PrimitiveEntry,
    ShardedTensorEntry,
    TensorEntry,
    is_container_entry,
)
from .snapshot import Snapshot


def _entry_bytes(entry) -> int:
    if isinstance(entry, TensorEntry):
        return entry.nbytes_estimate()
    if isinstance(entry, ChunkedTensorEntry):
        return sum(c.tensor.nbytes_estimate() for c in entry.chunks)
    if isinstance(entry, (ShardedTensorEntry, DTensorEntry)):
        return sum(s.tensor.nbytes_estimate() for s in entry.shards)
    return 0


def _describe(entry) -> str:
    if isinstance(entry, TensorEntry):
        return f"tensor {entry.dtype}{entry.shape} -> {entry.location}"
    if isinstance(entry, ChunkedTensorEntry):
        return (
            f"chunked_tensor {entry.dtype}{entry.shape} "
            f"({len(entry.chunks)} chunks)"
        )
    if isinstance(entry, ShardedTensorEntry):
        return (
            f"sharded_tensor {entry.dtype}{entry.shape} "
            f"({len(entry.shards)} shards)"
        )
    if isinstance(entry, DTensorEntry):
        return (
            f"dtensor {entry.dtype}{entry.shape} mesh={entry.mesh} "
            f"dim_map={entry.dim_map}"
        )
    if isinstance(entry, ObjectEntry):
        return f"object ({entry.obj_type}) -> {entry.location}"
    if isinstance(entry, PrimitiveEntry):
        return f"primitive {entry.ptype} = {entry.get_value()!r}"
    return type(entry).__name__


def cmd_inspect(args: argparse.Namespace) -> int:
    snapshot = Snapshot(args.path)
    md = snapshot.metadata
    print(f"snapshot: {args.path}")
    print(f"version: {md.version}  world_size: {md.world_size}")
    per_rank_bytes: dict = defaultdict(int)
    n_payloads = 0
    for key, entry in sorted(md.manifest.items()):
        if is_container_entry(entry):
            continue
        rank = key.partition("/")[0]
        nbytes = _entry_bytes(entry)
        per_rank_bytes[rank] += nbytes
        n_payloads += 1
        if not args.summary:
            print(f"  {key}: {_describe(entry)} [{nbytes / 1e6:.2f} MB]")
    total = sum(per_rank_bytes.values())
    print(f"{n_payloads} payload entries, {total / 1e9:.3f} GB logical")
    for rank in sorted(per_rank_bytes, key=int):
        print(f"  rank {rank}: {per_rank_bytes[rank] / 1e9:.3f} GB")
    return 0


def cmd_cat(args: argparse.Namespace) -> int:
    snapshot = Snapshot(args.path)
    obj = snapshot.read_object(args.entry)
    print(obj)
    return 0


def main(argv=None) -> int:
    parser = argparse.ArgumentParser(prog="python -m torchsnapshot_amd")
    sub = parser.add_subparsers(dest="cmd", required=True)
    p_inspect = sub.add_parser("inspect", help="print the manifest tree")
    p_inspect.add_argument("path")
    p_inspect.add_argument("--summary", action="store_true")
    p_inspect.set_defaults(fn=cmd_inspect)
    p_cat = sub.add_parser("cat", help="print one object by manifest path")
    p_cat.add_argument("path")
    p_cat.add_argument("entry")
    p_cat.set_defaults(fn=cmd_cat)
    args = parser.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
