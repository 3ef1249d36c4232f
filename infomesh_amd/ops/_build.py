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
    """Build the extension (idempotent, multi-process safe).

    At N-rank bench start every rank calls build(); an flock serializes
    them and late arrivals see a fresh .so after re-checking."""
    if not force and not needs_rebuild():
        return SO_PATH
    import fcntl
    lock_path = OPS_DIR / ".build.lock"
    with open(lock_path, "w") as lock:
        fcntl.flock(lock, fcntl.LOCK_EX)
        try:
            if not force and not needs_rebuild():
                return SO_PATH  # another rank built it while we waited
            tmp = OPS_DIR / f".libinfomesh_hip.tmp{subprocess.os.getpid()}.so"
            cmd = [
                _hipcc(), f"--offload-arch={ARCH}", "-O3", "-std=c++17",
                "-fPIC", "-shared", "-Wall",
                *[str(s) for s in SOURCES],
                "-o", str(tmp),
            ]
            if verbose:
                print(f"[infomesh-amd] building HIP extension "
                      f"({len(SOURCES)} sources, {ARCH}) …", file=sys.stderr)
            res = subprocess.run(cmd, capture_output=True, text=True)
            if res.returncode != 0:
                tmp.unlink(missing_ok=True)
                raise RuntimeError(
                    f"hipcc build failed (exit {res.returncode}):\n"
                    f"{res.stderr[-4000:]}")
            tmp.replace(SO_PATH)
            if verbose and res.stderr.strip():
                print(res.stderr[-2000:], file=sys.stderr)
            return SO_PATH
        finally:
            fcntl.flock(lock, fcntl.LOCK_UN)


if __name__ == "__main__":
    build(force="--force" in sys.argv)
    print(SO_PATH)
