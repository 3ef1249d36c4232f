#!/usr/bin/env python3
"""GPU engine example (requires an MI355X): build a hybrid shard,
search it, append incrementally, save/load the warm-start manifest.

    python examples/gpu_engine.py          # on a GPU box
"""
from __future__ import annotations

import sys
import tempfile
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import numpy as np
import torch


def main() -> None:
    assert torch.cuda.is_available(), "this example needs an MI355X GPU"
    from infomesh_amd.ops import _build
    _build.build()
    from infomesh_amd.index.gpu_index import GpuShard, bm25_term_ids
    from infomesh_amd.index.manifest import load_shard, save_shard

    shard = GpuShard("cuda")
    corpus = [f"document {i} about mfma kernels wave {i % 7}"
              for i in range(10_000)]
    for i, text in enumerate(corpus):
        shard.add_document(i, bm25_term_ids(text), None)
    shard.build()
    print(f"built: {shard.n_docs} docs, {shard.hbm_bytes()/1e6:.1f} MB HBM")

    vals, idx = shard.search_bm25([bm25_term_ids("mfma kernels wave 3")],
                                  k=5)
    torch.cuda.synchronize()
    print("top-5 ids:", idx[0].tolist(), "scores:",
          [round(v, 3) for v in vals[0].tolist()])

    # O(new) incremental append — a new posting segment, not a rebuild
    shard.add_document(10_000, bm25_term_ids("fresh appended doc"), None)
    shard.build()
    print("after append:", shard.n_docs, "docs,",
          len(shard.segments), "segments")

    with tempfile.TemporaryDirectory() as d:
        meta = save_shard(shard, Path(d) / "shard.pt")
        back = load_shard(Path(d) / "shard.pt")
        print("warm start:", back.n_docs, "docs restored,",
              meta["bytes"], "bytes on disk")


if __name__ == "__main__":
    main()
