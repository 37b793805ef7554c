"""Build the _csnap HIP extension for gfx950, in-tree.

Direct hipcc invocation (no hipify, no torch linkage): the extension is
pure HIP + pybind11 and talks to torch only through raw pointers/stream
handles, so it is immune to torch ABI churn and builds in seconds.

Usage:  python -m torchsnapshot_amd.ops.build
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

PKG_ROOT = Path(__file__).resolve().parent.parent
SRC = PKG_ROOT / "ops" / "hip" / "staging.hip"
OUT = PKG_ROOT / f"_csnap{sysconfig.get_config_var('EXT_SUFFIX') or '.so'}"

GFX_ARCH = os.environ.get("TSAMD_GFX_ARCH", "gfx950")


def _pybind11_includes() -> list[str]:
    import pybind11

    return [f"-I{pybind11.get_include()}"]


def needs_rebuild() -> bool:
    if not OUT.exists():
        return True
    return SRC.stat().st_mtime > OUT.stat().st_mtime


def build(force: bool = False, verbose: bool = True) -> Path:
    if not force and not needs_rebuild():
        if verbose:
            print(f"[tsamd] {OUT.name} up to date")
        return OUT
    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    cmd = [
        hipcc,
        f"--offload-arch={GFX_ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-fvisibility=hidden",
        f"-I{sysconfig.get_paths()['include']}",
        *_pybind11_includes(),
        str(SRC),
        "-o",
        str(OUT),
    ]
    if verbose:
        print("[tsamd] building:", " ".join(cmd))
    subprocess.run(cmd, check=True)
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
