"""Storage plugin registry + URL dispatch.

``url_to_storage_plugin("protocol://path")`` instantiates the backend for a
snapshot location. Built-ins: fs (default), s3, gs. Third-party backends
register via the ``tsamd_storage_plugins`` entry-point group (parity with
reference torchsnapshot/storage_plugin.py:20-67).
"""

from __future__ import annotations

from importlib.metadata import entry_points
from typing import Optional

from ..io_types import StoragePlugin

__all__ = ["url_to_storage_plugin", "StoragePlugin"]

_ENTRY_POINT_GROUP = "tsamd_storage_plugins"


def _split_url(url: str) -> tuple[str, str]:
    if "://" in url:
        protocol, _, path = url.partition("://")
        if protocol == "":
            protocol = "fs"
        return protocol, path
    return "fs", url


def url_to_storage_plugin(
    url: str, storage_options: Optional[dict] = None
) -> StoragePlugin:
    protocol, path = _split_url(url)
    if protocol in ("fs", "file"):
        from .fs import FSStoragePlugin

        return FSStoragePlugin(path, storage_options)
    if protocol == "s3":
        from .s3 import S3StoragePlugin

        return S3StoragePlugin(path, storage_options)
    if protocol in ("gs", "gcs"):
        from .gcs import GCSStoragePlugin

        return GCSStoragePlugin(path, storage_options)

    # third-party plugins
    try:
        eps = entry_points(group=_ENTRY_POINT_GROUP)
    except TypeError:  # older importlib API
        eps = entry_points().get(_ENTRY_POINT_GROUP, [])  # type: ignore[attr-defined]
    for ep in eps:
        if ep.name == protocol:
            cls = ep.load()
            return cls(path, storage_options)
    raise ValueError(
        f"no storage plugin registered for protocol {protocol!r} (url {url!r})"
    )
