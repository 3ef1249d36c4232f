"""GPU shard manifest: warm-start persistence of built shards.

Reference-parity intent: GPU shards are rebuildable caches
over the SQLite/snapshot ground truth; a sidecar binary manifest makes
restarts warm. Tensors are saved with torch.save to a per-rank file;
load_shard re-uploads them to HBM without re-tokenizing/re-embedding.

v2: segmented postings (offsets/doc_ids/tfdl per segment) + host doc
lengths, matching gpu_index.PostingSegment — doc_lens persistence means
incremental appends after a warm start work (round-1 ADVICE fix).
"""
from __future__ import annotations

import json
import time
from pathlib import Path

import numpy as np
import torch

from ..hashing import content_hash
from .gpu_index import CpuShard, GpuShard, PostingSegment

MANIFEST_VERSION = 2
# bump when bm25_term_ids tokenization changes — old term ids
# become incompatible and saved shards must be rebuilt
TOKENIZER_VERSION = 2  # v2: unicode \w + NFKD folding + CJK bigrams


def save_shard(shard: GpuShard, path: str | Path, rank: int = 0,
               world: int = 1) -> dict:
    path = Path(path)
    path.parent.mkdir(parents=True, exist_ok=True)
    segs = [{
        "offsets": s.offsets.cpu(), "doc_ids": s.doc_ids.cpu(),
        "tfdl": s.tfdl.cpu(), "doc_base": s.doc_base, "n_docs": s.n_docs,
    } for s in shard.segments]
    payload: dict = {"segments": segs}
    if shard.global_ids is not None:
        payload["global_ids"] = shard.global_ids.cpu()
    if shard.embeddings is not None:
        payload["embeddings"] = shard.embeddings.cpu()
    meta = {
        "version": MANIFEST_VERSION,
        "tokenizer_version": TOKENIZER_VERSION,
        "rank": rank, "world": world,
        "n_docs": shard.n_docs, "vocab": shard.vocab,
        "avgdl": shard.avgdl,
        "emb_dtype": shard.emb_dtype,
        "n_segments": len(shard.segments),
        "created_at": time.time(),
    }
    torch.save({"meta": meta, "df": shard.df,
                "doc_lens": shard._doc_lens, **payload}, path)
    meta["bytes"] = path.stat().st_size
    meta["checksum"] = content_hash(str(path.stat().st_size) +
                                    str(shard.n_docs))
    with open(path.with_suffix(".json"), "w") as f:
        json.dump(meta, f, indent=2)
    return meta


def load_shard(path: str | Path, device: str = "cuda") -> GpuShard:
    path = Path(path)
    blob = torch.load(path, map_location="cpu", weights_only=False)
    meta = blob["meta"]
    if meta.get("version") != MANIFEST_VERSION:
        raise ValueError(f"manifest version {meta.get('version')}")
    tv = meta.get("tokenizer_version", 1)
    if tv != TOKENIZER_VERSION:
        raise ValueError(
            f"shard was built with tokenizer v{tv}, current is "
            f"v{TOKENIZER_VERSION}: term ids are incompatible — rebuild "
            "the shard from the LocalStore/snapshot ground truth")
    shard = GpuShard(device) if device.startswith("cuda") else CpuShard()
    shard.n_docs = meta["n_docs"]
    shard.vocab = meta["vocab"]
    shard.avgdl = meta["avgdl"]
    shard.emb_dtype = meta.get("emb_dtype", "bf16")
    shard.df = np.asarray(blob["df"], dtype=np.int64)
    shard._doc_lens = np.asarray(blob["doc_lens"], dtype=np.int64)
    dev = shard.device
    for s in blob["segments"]:
        offs = s["offsets"]
        shard.segments.append(PostingSegment(
            offsets=offs.to(dev), doc_ids=s["doc_ids"].to(dev),
            tfdl=s["tfdl"].to(dev), doc_base=s["doc_base"],
            n_docs=s["n_docs"], h_offs=offs.numpy()))
    gids = blob.get("global_ids")
    if gids is not None:
        shard._gid_buf = gids.to(dev)
    emb = blob.get("embeddings")
    if emb is not None:
        shard._emb_buf = emb.to(dev).contiguous()
    return shard


def manifest_info(path: str | Path) -> dict | None:
    side = Path(path).with_suffix(".json")
    if not side.exists():
        return None
    return json.loads(side.read_text())
