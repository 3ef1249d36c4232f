#!/usr/bin/env python3
"""Standalone GPU probes for model-path tuning (not the driver bench):
times encoder, reranker chunk, phi-3 prefill + per-token decode.

Usage: python scripts/gpu_probe.py [encoder|rerank|decode|all]
"""
from __future__ import annotations

import pathlib
import sys
import time

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parents[1]))

import torch


def _t(fn, n=5, warmup=2):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def probe_gemm():
    from infomesh_amd.ops import _ext, kernels as K
    lib = _ext.lib()
    for (M, N, Kd) in [(40960, 2304, 768), (40960, 3072, 768),
                       (8192, 8192, 8192), (4096, 4096, 4096)]:
        a = torch.randn(M, Kd, device="cuda").bfloat16()
        b = torch.randn(N, Kd, device="cuda").bfloat16()
        out = torch.empty(M, N, device="cuda", dtype=torch.bfloat16)
        flops = 2.0 * M * N * Kd

        def run128():
            lib.infomesh_gemm_bf16_nt(a.data_ptr(), b.data_ptr(),
                                      out.data_ptr(), None, M, N, Kd, 1,
                                      M * Kd, N * Kd, M * N, 0, 1.0, 0,
                                      _ext.stream_ptr())

        def run256():
            lib.infomesh_gemm8_bf16_nt(a.data_ptr(), b.data_ptr(),
                                       out.data_ptr(), None, M, N, Kd, 1,
                                       M * Kd, N * Kd, M * N, 0, 1.0, 0,
                                       _ext.stream_ptr())
        t1 = _t(run128, n=10, warmup=3)
        t2 = _t(run256, n=10, warmup=3)
        # correctness spot check vs each other (bf16 rounding identical path)
        run128(); torch.cuda.synchronize(); o1 = out.clone()
        run256(); torch.cuda.synchronize(); o2 = out.clone()
        diff = (o1.float() - o2.float()).abs().max().item()
        print(f"gemm {M}x{N}x{Kd}: 128tile {flops/t1/1e9:.0f} GF "
              f"| 256tile {flops/t2/1e9:.0f} GF | maxdiff {diff:.4f}")


def probe_encoder():
    from infomesh_amd.models.encoder import EmbeddingEncoder
    enc = EmbeddingEncoder(device="cuda")
    ids = torch.randint(4, 30522, (64, 32), dtype=torch.int32, device="cuda")
    lens = torch.full((64,), 32, dtype=torch.int32, device="cuda")
    ms = _t(lambda: enc.encode_ids(ids, lens), n=20, warmup=5)
    print(f"encoder B=64 S=32: {ms:.3f} ms ({64 / ms * 1000:.0f} enc/s)")


def probe_rerank():
    from infomesh_amd.models.reranker import CrossEncoderReranker
    rr = CrossEncoderReranker(device="cuda")
    for B in (1600, 6400):
        ids = torch.randint(4, 250002, (B, 160), dtype=torch.int32,
                            device="cuda")
        ids[:, 0] = 1
        lens = torch.full((B,), 160, dtype=torch.int32, device="cuda")
        ms = _t(lambda: rr.score_ids(ids, lens), n=3, warmup=1)
        toks = B * 160
        print(f"reranker B={B} S=160: {ms:.1f} ms "
              f"({toks / ms * 1000 / 1e6:.2f}M tok/s)")


def probe_decode():
    from infomesh_amd.models.phi3 import PHI3_MINI, Phi3Decoder
    dec = Phi3Decoder(PHI3_MINI, device="cuda", max_batch=1, max_seq=2304)
    prompt = torch.randint(0, 32064, (1, 1024), dtype=torch.int32,
                           device="cuda")
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    dec.reset()
    logits = dec.prefill(prompt)
    torch.cuda.synchronize()
    prefill_ms = (time.perf_counter() - t0) * 1e3
    print(f"phi3 prefill S=1024: {prefill_ms:.1f} ms "
          f"({1024 / prefill_ms * 1000:.0f} tok/s)")
    from infomesh_amd.ops import kernels as K
    tok = K.argmax(logits)
    # first decode includes graph capture
    t0 = time.perf_counter()
    logits = dec.decode_step(tok)
    torch.cuda.synchronize()
    print(f"phi3 first decode (graph capture): "
          f"{(time.perf_counter() - t0) * 1e3:.1f} ms")
    n = 32
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        logits = dec.decode_step(tok)
    torch.cuda.synchronize()
    per_tok = (time.perf_counter() - t0) / n * 1e3
    print(f"phi3 decode: {per_tok:.2f} ms/token "
          f"({1000 / per_tok:.0f} tok/s)")


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    assert torch.cuda.is_available()
    from infomesh_amd.ops import _build
    _build.build()
    if which in ("gemm",):
        probe_gemm()
    if which in ("encoder", "all"):
        probe_encoder()
    if which in ("rerank", "all"):
        probe_rerank()
    if which in ("decode", "all"):
        probe_decode()
