#!/usr/bin/env python3
"""Dense-plane probe: time the scoring GEMM + top-k at flagship shapes.
Round-1 baseline: 529 us at 128x1.25Mx384 (~3.1 TB/s effective) vs a
~5.4 TB/s measured mixed ceiling (BACKLOG / VERDICT #6)."""
from __future__ import annotations

import argparse
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--docs", type=int, default=1_250_000)
    ap.add_argument("--batch", type=int, default=128)
    ap.add_argument("--dim", type=int, default=384)
    ap.add_argument("--k", type=int, default=100)
    ap.add_argument("--iters", type=int, default=50)
    args = ap.parse_args()
    from infomesh_amd.ops import _build
    _build.build()
    from infomesh_amd.ops import kernels as K

    dev = "cuda"
    M, N, D = args.batch, args.docs, args.dim
    a = torch.randn(M, D, device=dev).bfloat16()
    b = torch.randn(N, D, device=dev).bfloat16()
    out = torch.empty(1, M, N, device=dev, dtype=torch.float32)
    tk = K.TopK(dev)

    # parity spot-check first
    got = K.gemm_nt(a, b, out_f32=True, out=out).reshape(M, N)
    ref = (a[:4].float() @ b[:4096].float().T)
    err = (got[:4, :4096] - ref).abs().max().item()
    print(f"parity max err (4x4096 slice): {err:.4f}")

    for name, fn in [
        ("gemm", lambda: K.gemm_nt(a, b, out_f32=True, out=out)),
        ("topk", lambda: tk(out.reshape(M, N), args.k)),
        ("gemm+topk", lambda: tk(K.gemm_nt(a, b, out_f32=True,
                                           out=out).reshape(M, N),
                                 args.k)),
    ]:
        for _ in range(5):
            fn()
        torch.cuda.synchronize()
        e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
        e0.record()
        for _ in range(args.iters):
            fn()
        e1.record()
        torch.cuda.synchronize()
        ms = e0.elapsed_time(e1) / args.iters
        if name == "gemm":
            gb = (N * D * 2 + M * D * 2 + M * N * 4) / 1e9
        elif name == "topk":
            gb = 3 * M * N * 4 / 1e9
        else:
            gb = (N * D * 2 + 4 * M * N * 4) / 1e9
        print(f"{name}: {ms * 1e3:.0f} us  ({gb / (ms / 1e3):.2f} GB/s)")


if __name__ == "__main__":
    main()
