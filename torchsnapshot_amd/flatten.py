"""Flatten nested state dicts into {logical_path: leaf} + container manifest.

A state dict is an arbitrary nesting of dict / OrderedDict / list containers
with leaves of any type. ``flatten`` walks the nesting and produces:

- a manifest mapping each *container's* logical path to a container entry
  (DictEntry / OrderedDictEntry / ListEntry), and
- a mapping of each *leaf's* logical path to the leaf object.

``inflate`` is the exact inverse. Logical paths join keys with "/", with
"/" and "%" occurrences inside keys percent-escaped so the mapping is
injective (parity with reference torchsnapshot/flatten.py:20,79,215-226).

Containers are only flattened when doing so is unambiguous:

- every key must be str or int (else the container becomes a leaf),
- no two keys may escape to the same path segment (e.g. int 1 vs str "1";
  the container becomes a leaf in that case).
"""

from __future__ import annotations

from collections import OrderedDict
from typing import Any, Dict, List, Tuple, Union

from .manifest import (
    DictEntry,
    ListEntry,
    Manifest,
    OrderedDictEntry,
)

Flattened = Dict[str, Any]


def escape_key(key: Union[str, int]) -> str:
    """Escape a container key for use as one logical-path segment."""
    s = str(key)
    return s.replace("%", "%25").replace("/", "%2F")


def unescape_key(segment: str) -> str:
    return segment.replace("%2F", "/").replace("%25", "%")


def _join(prefix: str, segment: str) -> str:
    return f"{prefix}/{segment}" if prefix else segment


def _flattenable_keys(d: Dict[Any, Any]) -> bool:
    seen = set()
    for k in d.keys():
        if not isinstance(k, (str, int)) or isinstance(k, bool):
            return False
        esc = escape_key(k)
        if esc in seen:
            return False
        seen.add(esc)
    return True


def flatten(obj: Any, prefix: str = "") -> Tuple[Manifest, Flattened]:
    """Flatten ``obj`` rooted at logical path ``prefix``.

    Returns (manifest of container entries, {leaf_path: leaf}).
    """
    manifest: Manifest = {}
    flattened: Flattened = {}
    _flatten_inner(obj, prefix, manifest, flattened)
    return manifest, flattened


def _flatten_inner(
    obj: Any, path: str, manifest: Manifest, flattened: Flattened
) -> None:
    if isinstance(obj, OrderedDict) and _flattenable_keys(obj):
        manifest[path] = OrderedDictEntry(keys=list(obj.keys()))
        for k, v in obj.items():
            _flatten_inner(v, _join(path, escape_key(k)), manifest, flattened)
    elif isinstance(obj, dict) and _flattenable_keys(obj):
        manifest[path] = DictEntry(keys=list(obj.keys()))
        for k, v in obj.items():
            _flatten_inner(v, _join(path, escape_key(k)), manifest, flattened)
    elif isinstance(obj, list):
        manifest[path] = ListEntry()
        for i, v in enumerate(obj):
            _flatten_inner(v, _join(path, str(i)), manifest, flattened)
    else:
        flattened[path] = obj


def inflate(manifest: Manifest, flattened: Flattened, prefix: str = "") -> Any:
    """Rebuild the nested object rooted at ``prefix`` from container entries
    and flattened leaves. The inverse of :func:`flatten`."""
    # Index children by parent path for O(1) traversal. A payload path can
    # appear in both the manifest (its entry) and flattened (its value);
    # de-duplicate.
    children: Dict[str, List[str]] = {}
    for path in dict.fromkeys(list(manifest.keys()) + list(flattened.keys())):
        if path == prefix:
            continue
        parent = path.rsplit("/", 1)[0] if "/" in path else ""
        children.setdefault(parent, []).append(path)

    def build(path: str) -> Any:
        if path in flattened:
            return flattened[path]
        entry = manifest.get(path)
        if entry is None:
            raise KeyError(f"no entry or value for logical path '{path}'")
        if isinstance(entry, ListEntry):
            idx_paths = children.get(path, [])
            pairs = sorted(
                (int(p.rsplit("/", 1)[-1]), p) for p in idx_paths
            )
            return [build(p) for _, p in pairs]
        if isinstance(entry, (DictEntry, OrderedDictEntry)):
            cls = OrderedDict if isinstance(entry, OrderedDictEntry) else dict
            out = cls()
            for key in entry.keys:
                child = _join(path, escape_key(key))
                out[key] = build(child)
            return out
        raise TypeError(
            f"logical path '{path}' maps to non-container entry "
            f"{type(entry).__name__} but no flattened value was provided"
        )

    return build(prefix)
