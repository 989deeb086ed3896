"""In-tree build driver for the HIP extension.

Compiles ops/csrc/engine.cpp (which #includes the kernels — single TU, no
RDC link) with hipcc for gfx950 into ggrmcp_amd/ops/_jsonproto.so.  Built
in-tree so the .so travels to GPU boxes with the repo snapshot.  hipcc
cross-compiles fine on GPU-less machines; only import-time device calls need
a GPU.
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
CSRC = OPS_DIR / "csrc"
SO_PATH = OPS_DIR / "_jsonproto.so"

GFX_ARCH = os.environ.get("GGRMCP_GFX_ARCH", "gfx950")


def source_files():
    return sorted(CSRC.glob("*.cpp")) + sorted(CSRC.glob("*.hip")) + sorted(
        CSRC.glob("*.h")
    )


def needs_build() -> bool:
    if not SO_PATH.exists():
        return True
    so_mtime = SO_PATH.stat().st_mtime
    return any(f.stat().st_mtime > so_mtime for f in source_files())


def build(verbose: bool = True, force: bool = False) -> Path:
    if not force and not needs_build():
        return SO_PATH
    import pybind11

    hipcc = os.environ.get("HIPCC", "hipcc")
    py_include = sysconfig.get_paths()["include"]
    cmd = [
        hipcc,
        "-x", "hip",
        str(CSRC / "engine.cpp"),
        f"--offload-arch={GFX_ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        "-shared",
        "-fvisibility=hidden",
        f"-I{pybind11.get_include()}",
        f"-I{py_include}",
        f"-I{CSRC}",
        "-o", str(SO_PATH),
    ]
    if verbose:
        print("[ggrmcp-amd build]", " ".join(cmd), file=sys.stderr, flush=True)
    subprocess.run(cmd, check=True)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(SO_PATH)
