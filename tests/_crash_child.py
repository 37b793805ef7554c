"""Child process for the crash-consistency test: starts a slow snapshot
and never finishes (the parent SIGKILLs it mid-write)."""

import sys
from unittest import mock

import torch

from torchsnapshot_amd import Snapshot
from torchsnapshot_amd.state_dict import StateDict
from torchsnapshot_amd.storage.fs import FSStoragePlugin


class CrawlFS(FSStoragePlugin):
    async def write(self, write_io):
        import asyncio

        await asyncio.sleep(0.5)  # long enough for the parent to kill us
        await super().write(write_io)


def fake(url, storage_options=None):
    return CrawlFS(url.split("://")[-1], storage_options)


if __name__ == "__main__":
    path = sys.argv[1]
    sd = StateDict(**{f"t{i}": torch.rand(256, 256) for i in range(8)})
    with mock.patch(
        "torchsnapshot_amd.snapshot.url_to_storage_plugin", side_effect=fake
    ):
        print("taking", flush=True)
        Snapshot.take(path, {"sd": sd})
        print("UNEXPECTED: take completed", flush=True)
