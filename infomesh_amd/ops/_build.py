"""Build driver for the CDNA4 HIP extension.

Compiles every csrc/*.hip with hipcc --offload-arch=gfx950 into an
in-tree shared library (libinfomesh_hip.so) bound via ctypes — no
hipify, no CUDA-compat layer, no torch C++ ABI dependency (kernels take
raw device pointers + a HIP stream).

The .so is git-ignored but ships to the GPU box with the gpurun snapshot.
"""
from __future__ import annotations

import subprocess
import sys
from pathlib import Path

OPS_DIR = Path(__file__).resolve().parent
CSRC = OPS_DIR / "csrc"
SO_PATH = OPS_DIR / "libinfomesh_hip.so"
ARCH = "gfx950"

SOURCES = sorted(CSRC.glob("*.hip"))


def _hipcc() -> str:
    for cand in ("/opt/rocm/bin/hipcc", "hipcc"):
        try:
            subprocess.run([cand, "--version"], capture_output=True, check=True)
            return cand
        except (OSError, subprocess.CalledProcessError):
            continue
    raise RuntimeError("hipcc not found — ROCm required to build the extension")


def needs_rebuild() -> bool:
    if not SO_PATH.exists():
        return True
    so_mtime = SO_PATH.stat().st_mtime
    deps = list(SOURCES) + list(CSRC.glob("*.h")) + [Path(__file__)]
    return any(p.stat().st_mtime > so_mtime for p in deps)


def build(force: bool = False, verbose: bool = True) -> Path:
    if not force and not needs_rebuild():
        return SO_PATH
    cmd = [
        _hipcc(), f"--offload-arch={ARCH}", "-O3", "-std=c++17",
        "-fPIC", "-shared", "-Wall",
        *[str(s) for s in SOURCES],
        "-o", str(SO_PATH),
    ]
    if verbose:
        print(f"[infomesh-amd] building HIP extension ({len(SOURCES)} sources, "
              f"{ARCH}) …", file=sys.stderr)
    res = subprocess.run(cmd, capture_output=True, text=True)
    if res.returncode != 0:
        raise RuntimeError(
            f"hipcc build failed (exit {res.returncode}):\n{res.stderr[-4000:]}")
    if verbose and res.stderr.strip():
        print(res.stderr[-2000:], file=sys.stderr)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(SO_PATH)
