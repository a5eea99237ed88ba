"""In-tree build of the gfx950 HIP extension.

Compiles ``kernels/siglip_kernels.hip`` with hipcc into
``ops/_siglip_hip.so`` (plain C ABI, loaded via ctypes — no torch C++ ABI
coupling, so a single .so works across torch builds and travels with the
repo snapshot to GPU boxes).

Usage:  python -m distributed_sigmoid_loss_amd.ops.build
"""

from __future__ import annotations

import os
import subprocess
import sys

_OPS_DIR = os.path.dirname(os.path.abspath(__file__))
SO_PATH = os.path.join(_OPS_DIR, "_siglip_hip.so")
_SRCS = [os.path.join(_OPS_DIR, "kernels", "siglip_kernels.hip")]

HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")


def _stale() -> bool:
    if not os.path.exists(SO_PATH):
        return True
    so_mtime = os.path.getmtime(SO_PATH)
    return any(os.path.getmtime(s) > so_mtime for s in _SRCS)


def build(force: bool = False, verbose: bool = False) -> str:
    """Compile the extension if missing or out of date; returns the .so path."""
    if not force and not _stale():
        return SO_PATH
    cmd = [
        HIPCC, f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-shared", "-fPIC", *_SRCS, "-o", SO_PATH,
    ]
    if verbose:
        print("+", " ".join(cmd), file=sys.stderr)
    res = subprocess.run(cmd, capture_output=True, text=True)
    if res.returncode != 0:
        raise RuntimeError(
            f"hipcc build failed (rc={res.returncode}):\n{res.stdout}\n{res.stderr}")
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv, verbose=True)
    print(SO_PATH)
