#!/usr/bin/env python3
"""Encoder forward probe: bge-small-shaped batch encode latency.
Round-1 baseline ~1.2 ms/batch at B=128,S=32 (VERDICT #6 target <0.8)."""
from __future__ import annotations

import argparse
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=128)
    ap.add_argument("--seq", type=int, default=32)
    ap.add_argument("--iters", type=int, default=100)
    args = ap.parse_args()
    from infomesh_amd.ops import _build
    _build.build()
    from infomesh_amd.models.encoder import EmbeddingEncoder

    enc = EmbeddingEncoder(device="cuda")
    g = torch.Generator().manual_seed(3)
    ids = torch.randint(4, 30522, (args.batch, args.seq), generator=g,
                        dtype=torch.int32).cuda()
    ids[:, 0] = 1
    lens = torch.full((args.batch,), args.seq, dtype=torch.int32,
                      device="cuda")
    # parity vs CPU fp32 reference on a small slice
    out = enc.encode_ids(ids, lens)
    ref = enc.encode_ids_reference(ids[:4].cpu(), lens[:4].cpu())
    err = (out[:4].cpu().float() - ref.float()).abs().max().item()
    print(f"parity vs fp32 reference (4 rows): max err {err:.4f}")
    for _ in range(10):
        enc.encode_ids(ids, lens)
    torch.cuda.synchronize()
    e0, e1 = torch.cuda.Event(True), torch.cuda.Event(True)
    e0.record()
    for _ in range(args.iters):
        enc.encode_ids(ids, lens)
    e1.record()
    torch.cuda.synchronize()
    ms = e0.elapsed_time(e1) / args.iters
    print(f"encode B={args.batch} S={args.seq}: {ms * 1e3:.0f} us/batch")


if __name__ == "__main__":
    main()
