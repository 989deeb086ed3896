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
H2_SO_PATH = OPS_DIR / "_h2grpc.so"

GFX_ARCH = os.environ.get("GGRMCP_GFX_ARCH", "gfx950")
NGHTTP2_INCLUDE = os.environ.get("GGRMCP_NGHTTP2_INCLUDE", "/opt/conda/include")


def _needs(so: Path, sources) -> bool:
    if not so.exists():
        return True
    so_mtime = so.stat().st_mtime
    return any(f.stat().st_mtime > so_mtime for f in sources)


def _common_flags():
    import pybind11

    py_include = sysconfig.get_paths()["include"]
    return [
        "-O3", "-std=c++17", "-fPIC", "-shared", "-fvisibility=hidden",
        f"-I{pybind11.get_include()}", f"-I{py_include}", f"-I{CSRC}",
    ]


def build_jsonproto(verbose: bool = True, force: bool = False) -> Path:
    sources = [CSRC / "engine.cpp", CSRC / "json2pb.hip", CSRC / "pb2json.hip",
               CSRC / "common.h", CSRC / "h2grpc_impl.h", CSRC / "span_api.h"]
    if not force and not _needs(SO_PATH, sources):
        return SO_PATH
    hipcc = os.environ.get("HIPCC", "hipcc")
    cmd = [hipcc, "-x", "hip", str(CSRC / "engine.cpp"),
           f"--offload-arch={GFX_ARCH}"] + _common_flags() + [
           f"-I{NGHTTP2_INCLUDE}", "-l:libnghttp2.so.14", "-pthread",
           "-o", str(SO_PATH)]
    if verbose:
        print("[ggrmcp-amd build]", " ".join(cmd), file=sys.stderr, flush=True)
    subprocess.run(cmd, check=True)
    return SO_PATH


def build_h2grpc(verbose: bool = True, force: bool = False) -> Path:
    sources = [CSRC / "h2grpc.cpp", CSRC / "h2grpc_impl.h"]
    if not force and not _needs(H2_SO_PATH, sources):
        return H2_SO_PATH
    cxx = os.environ.get("CXX", "g++")
    cmd = [cxx, str(CSRC / "h2grpc.cpp")] + _common_flags() + [
        f"-I{NGHTTP2_INCLUDE}",
        "-l:libnghttp2.so.14",
        "-pthread",
        "-o", str(H2_SO_PATH),
    ]
    if verbose:
        print("[ggrmcp-amd build]", " ".join(cmd), file=sys.stderr, flush=True)
    subprocess.run(cmd, check=True)
    return H2_SO_PATH


HOSTSIM_SO_PATH = OPS_DIR / "_hostsim.so"


def build_hostsim(verbose: bool = True, force: bool = False) -> Path:
    """Single-lane CPU build of the kernels (csrc/host_shim.h)."""
    sources = [CSRC / "host_sim.cpp", CSRC / "json2pb.hip", CSRC / "pb2json.hip",
               CSRC / "common.h", CSRC / "host_shim.h"]
    if not force and not _needs(HOSTSIM_SO_PATH, sources):
        return HOSTSIM_SO_PATH
    cxx = os.environ.get("CXX", "g++")
    extra = os.environ.get("GGRMCP_HOSTSIM_FLAGS", "").split()
    cmd = [cxx, str(CSRC / "host_sim.cpp"), "-DGGRMCP_HOST_SIM", "-DWAVE=1",
           "-x", "c++"] + _common_flags() + extra + ["-o", str(HOSTSIM_SO_PATH)]
    if verbose:
        print("[ggrmcp-amd build]", " ".join(cmd), file=sys.stderr, flush=True)
    subprocess.run(cmd, check=True)
    return HOSTSIM_SO_PATH


FRONTEND_SO_PATH = OPS_DIR / "_frontend.so"


def build_frontend(verbose: bool = True, force: bool = False) -> Path:
    sources = [CSRC / "frontend.cpp", CSRC / "session_table.h",
               CSRC / "span_api.h"]
    if not force and not _needs(FRONTEND_SO_PATH, sources):
        return FRONTEND_SO_PATH
    cxx = os.environ.get("CXX", "g++")
    cmd = [cxx, str(CSRC / "frontend.cpp")] + _common_flags() + [
        "-pthread", "-o", str(FRONTEND_SO_PATH),
    ]
    if verbose:
        print("[ggrmcp-amd build]", " ".join(cmd), file=sys.stderr, flush=True)
    subprocess.run(cmd, check=True)
    return FRONTEND_SO_PATH


def build(verbose: bool = True, force: bool = False) -> Path:
    build_h2grpc(verbose, force)
    build_hostsim(verbose, force)
    build_frontend(verbose, force)
    return build_jsonproto(verbose, force)


def needs_build() -> bool:
    return _needs(SO_PATH, list(CSRC.glob("*")))


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(SO_PATH, H2_SO_PATH)
