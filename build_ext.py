#!/usr/bin/env python3
"""Build the _swarmq HIP extension in-tree for gfx950.

Invokes hipcc directly (no torch cpp_extension, no hipify — the sources
are native HIP). The built .so lands in swarmdb_amd/ so it travels with
the repo snapshot to GPU boxes.

Usage: python build_ext.py [--force]
"""

from __future__ import annotations

import subprocess
import sys
import sysconfig
from pathlib import Path

REPO = Path(__file__).resolve().parent
SRC = [REPO / "csrc" / "swarmq_module.hip"]
HDRS = [REPO / "csrc" / "swarmq_common.h"]
EXT_SUFFIX = sysconfig.get_config_var("EXT_SUFFIX") or ".so"
OUT = REPO / "swarmdb_amd" / f"_swarmq{EXT_SUFFIX}"

ARCH = "gfx950"


def pybind11_includes() -> list[str]:
    import pybind11

    return [f"-I{pybind11.get_include()}"]


def python_includes() -> list[str]:
    return [f"-I{sysconfig.get_path('include')}"]


def needs_build() -> bool:
    if not OUT.exists():
        return True
    mtime = OUT.stat().st_mtime
    return any(p.stat().st_mtime > mtime for p in SRC + HDRS)


def build(force: bool = False) -> Path:
    if not force and not needs_build():
        print(f"[build_ext] up to date: {OUT}")
        return OUT
    cmd = [
        "hipcc",
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-fvisibility=hidden",
        "-DNDEBUG",
        *pybind11_includes(),
        *python_includes(),
        *[str(s) for s in SRC],
        "-o",
        str(OUT),
    ]
    print("[build_ext]", " ".join(cmd))
    subprocess.run(cmd, check=True)
    print(f"[build_ext] built {OUT}")
    return OUT


if __name__ == "__main__":
    build(force="--force" in sys.argv)
