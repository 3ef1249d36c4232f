#!/usr/bin/env python3
"""A/B the GEMM tile heuristic on the encoder/dense-plane hot shapes.

Run on a GPU box:
  for t in auto 64 128; do INFOMESH_GEMM_TILE=$t python scripts/gemm_tile_probe.py; done
(The override is read once per process, hence one process per setting.)
"""
from __future__ import annotations

import os
import pathlib
import sys
import time

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))
import torch  # noqa: E402


def main():
    from infomesh_amd.ops import _build, _ext
    _build.build(verbose=False)
    lib = _ext.lib()
    tag = os.environ.get("INFOMESH_GEMM_TILE", "auto")
    # (M, N, K, out_f32): encoder qkv/attn-out/ffn1/ffn2 at B=128 S=32,
    # then the dense cosine plane (128 queries x 1.25M docs).
    shapes = [(4096, 1152, 384, 0), (4096, 384, 384, 0),
              (4096, 1536, 384, 0), (4096, 384, 1536, 0),
              (128, 1_250_000, 384, 1)]
    for (M, N, K, f32) in shapes:
        a = torch.randn(M, K, device="cuda").bfloat16()
        b = torch.randn(N, K, device="cuda").bfloat16()
        out = torch.empty(M, N, device="cuda",
                          dtype=torch.float32 if f32 else torch.bfloat16)

        def run():
            lib.infomesh_gemm_bf16_nt(a.data_ptr(), b.data_ptr(),
                                      out.data_ptr(), None, M, N, K, 1,
                                      M * K, N * K, M * N, 0, 1.0, f32,
                                      _ext.stream_ptr())
        for _ in range(5):
            run()
        torch.cuda.synchronize()
        n = 50
        t0 = time.perf_counter()
        for _ in range(n):
            run()
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / n * 1e6
        gf = 2.0 * M * N * K / (us * 1e-6) / 1e9
        print(f"tile={tag:>4} {M}x{N}x{K}{'f32' if f32 else ''}: "
              f"{us:8.1f} us  {gf:7.0f} GF/s")
        if f32 and M <= 128:
            def run_ds():
                rc = lib.infomesh_dense_scores(
                    a.data_ptr(), b.data_ptr(), out.data_ptr(),
                    M, N, K, 1.0, _ext.stream_ptr())
                assert rc == 0
            for _ in range(5):
                run_ds()
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(n):
                run_ds()
            torch.cuda.synchronize()
            us = (time.perf_counter() - t0) / n * 1e6
            bw = (M + N) * K * 2 / (us * 1e-6) / 1e12
            print(f"  densescore      {M}x{N}x{K}f32: {us:8.1f} us "
                  f"  B-stream {bw:5.2f} TB/s")


if __name__ == "__main__":
    main()
