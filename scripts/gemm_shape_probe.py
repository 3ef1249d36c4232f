#!/usr/bin/env python3
"""Per-shape GEMM timing: our kernels vs torch(rocBLAS/hipBLASLt) as a
ceiling reference (torch is NOT used in the product path — this only
tells us how far the hand-written tiles are from library peak on the
encoder shapes)."""
from __future__ import annotations

import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import os
SHAPES = [
    (4096, 1152, 384),   # encoder QKV
    (4096, 384, 384),    # attn out
    (4096, 1536, 384),   # FFN up
    (4096, 384, 1536),   # FFN down
]
if os.environ.get("PROBE_RERANK"):
    # bge-reranker-base shapes at 12800 pairs x 160 tokens (M=2.05M);
    # probe at M/4 to keep memory/time sane (same regime)
    M = 512_000
    SHAPES = [(M, 2304, 768), (M, 768, 768), (M, 3072, 768),
              (M, 768, 3072)]


def bench(fn, iters=200):
    for _ in range(20):
        fn()
    torch.cuda.synchronize()
    e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
    e0.record()
    for _ in range(iters):
        fn()
    e1.record()
    torch.cuda.synchronize()
    return e0.elapsed_time(e1) / iters * 1e3  # us


def main() -> None:
    from infomesh_amd.ops import _build
    _build.build()
    from infomesh_amd.ops import kernels as K
    for (M, N, Kd) in SHAPES:
        a = torch.randn(M, Kd, device="cuda").bfloat16()
        b = torch.randn(N, Kd, device="cuda").bfloat16()
        out = torch.empty(1, M, N, device="cuda", dtype=torch.bfloat16)
        t_ours = bench(lambda: K.gemm_nt(a, b, out=out))
        bt = b.T.contiguous().T  # [N,K] -> matmul needs [K,N]; keep NT
        t_torch = bench(lambda: torch.matmul(a, b.T))
        fl = 2 * M * N * Kd / 1e9
        print(f"[{M}x{N}x{Kd}] ours {t_ours:7.1f} us ({fl/t_ours*1e3:6.0f} GF/s)"
              f"   torch {t_torch:7.1f} us ({fl/t_torch*1e3:6.0f} GF/s)")


if __name__ == "__main__":
    main()
