"""Build the CDNA4 HIP extension in-tree (gfx950 only, no hipify).

The built ``.so`` lives next to this file so it travels with the repo
snapshot to GPU boxes.  Invoked by ``__graft_entry__.build()``, ``setup.py``
and on-demand by ``ops.__init__`` when the library is missing but hipcc is
available (CPU boxes cross-compile in seconds).
"""
from __future__ import annotations

import os
import subprocess
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
CSRC = OPS_DIR / "csrc"
LIB_NAME = "libfedops_gfx950.so"
LIB_PATH = OPS_DIR / LIB_NAME

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTENSOR_FEDERATED_AMD_ARCH", "gfx950")


def sources():
    return sorted(CSRC.glob("*.hip"))


def needs_rebuild() -> bool:
    if not LIB_PATH.exists():
        return True
    lib_mtime = LIB_PATH.stat().st_mtime
    return any(src.stat().st_mtime > lib_mtime for src in sources())


WORKER_SRC = CSRC / "fed_worker.cpp"
WORKER_BIN = OPS_DIR / "fed_worker"


def build(force: bool = False, verbose: bool = True) -> Path:
    """Compile the .hip sources into one shared library (gfx950) and the
    native worker daemon binary."""
    if force or needs_rebuild():
        cmd = [
            HIPCC,
            f"--offload-arch={ARCH}",
            "-O3",
            "-std=c++17",
            "-shared",
            "-fPIC",
            "-o",
            str(LIB_PATH),
        ] + [str(s) for s in sources()]
        if verbose:
            print("[pytensor_federated_amd.ops.build]", " ".join(cmd))
        subprocess.run(cmd, check=True)
    if WORKER_SRC.exists() and (
        force
        or not WORKER_BIN.exists()
        or WORKER_SRC.stat().st_mtime > WORKER_BIN.stat().st_mtime
    ):
        cmd = [
            HIPCC,
            f"--offload-arch={ARCH}",
            "-O3",
            "-std=c++17",
            str(WORKER_SRC),
            "-ldl",
            "-o",
            str(WORKER_BIN),
        ]
        # gRPC edge: HTTP/2 framing + HPACK via libnghttp2 when the image
        # provides it (header from conda, ABI-stable runtime lib from the
        # system); without it the worker builds with the fast transport only.
        nghttp2_inc = Path("/opt/conda/include/nghttp2/nghttp2.h")
        nghttp2_lib = Path("/usr/lib/x86_64-linux-gnu/libnghttp2.so.14")
        if nghttp2_inc.exists() and nghttp2_lib.exists():
            cmd[cmd.index(str(WORKER_SRC)) : cmd.index(str(WORKER_SRC)) + 1] = [
                "-I/opt/conda/include",
                str(WORKER_SRC),
                f"-L{nghttp2_lib.parent}",
                f"-l:{nghttp2_lib.name}",
            ]
        if verbose:
            print("[pytensor_federated_amd.ops.build]", " ".join(cmd))
        subprocess.run(cmd, check=True)
    return LIB_PATH


if __name__ == "__main__":
    build(force=True)
