"""Cross-encoder reranker: bge-reranker-base shape (XLM-R base: L=12,
H=768, 12 heads, FFN 3072, vocab 250002), random-init. Scores
(query, passage) pairs; top-100 -> top-10 batched rerank (BASELINE
config 3, bf16 MFMA).

Replaces: the reference's prompt-an-LLM reranker
(infomesh/search/reranker.py:20-163) with an in-process cross-encoder
forward on MFMA kernels.
"""
from __future__ import annotations

import torch

from ..ops import kernels as K
from .bert import BertConfig, BertEncoder, init_bert_weights
from .tokenizer import HashTokenizer, PAD

BGE_RERANKER_BASE = BertConfig(vocab_size=250002, hidden=768, layers=12,
                               heads=12, ffn=3072, max_pos=512)


class CrossEncoderReranker:
    def __init__(self, device: str = "cuda", seed: int = 4321,
                 cfg: BertConfig = BGE_RERANKER_BASE, max_len: int = 160,
                 batch: int = 256):
        self.cfg = cfg
        self.device = device
        self.max_len = max_len
        self.batch = batch
        self.tokenizer = HashTokenizer(cfg.vocab_size)
        self.bert = BertEncoder(cfg, init_bert_weights(cfg, seed, device))
        g = torch.Generator().manual_seed(seed + 1)
        self.cls_w = (torch.randn(1, cfg.hidden, generator=g) * 0.02)\
            .to(device).bfloat16()
        self.cls_b = torch.zeros(1, device=device)

    def score_ids(self, ids: torch.Tensor, lens: torch.Tensor) -> torch.Tensor:
        """[B,S] i32 pair encodings -> [B] f32 relevance logits."""
        hidden = self.bert.forward(ids, lens)          # [B,S,H]
        cls = K.pool(hidden, lens, mode="cls", l2=False)  # [B,H] f32
        logits = K.gemm_nt(cls.bfloat16(), self.cls_w, bias=self.cls_b,
                           out_f32=True)               # [B,1]
        return logits.view(-1)

    def score_pairs(self, query: str, passages: list[str]) -> torch.Tensor:
        out = []
        for i in range(0, len(passages), self.batch):
            chunk = passages[i:i + self.batch]
            enc = [self.tokenizer.encode_pair(query, p, self.max_len)
                   for p in chunk]
            lens = [len(e) for e in enc]
            width = max(lens)
            ids = torch.tensor([e + [PAD] * (width - len(e)) for e in enc],
                               dtype=torch.int32, device=self.device)
            lens_t = torch.tensor(lens, dtype=torch.int32, device=self.device)
            out.append(self.score_ids(ids, lens_t))
        return torch.cat(out) if out else torch.empty(0)

    def rerank(self, query: str, passages: list[str],
               keep: int = 10) -> list[tuple[int, float]]:
        """Returns [(passage_idx, logit)] best-first, truncated to keep."""
        if not passages:
            return []
        scores = self.score_pairs(query, passages).cpu()
        order = torch.argsort(scores, descending=True)[:keep]
        return [(int(i), float(scores[i])) for i in order]
