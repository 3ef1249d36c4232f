"""Text-embedding encoder: bge-small-en-v1.5 shape (L=12, H=384, 12 heads,
FFN 1536, 384-d CLS embedding, L2-normalized), random-init.

Replaces: sentence-transformers all-MiniLM in the reference
(infomesh/index/vector_store.py:26,104-125). Dense-index doc & query
encoding both go through this on MFMA kernels.
"""
from __future__ import annotations

import torch

from ..ops import kernels as K
from ..ops import reference as R
from .bert import BertConfig, BertEncoder, init_bert_weights
from .tokenizer import HashTokenizer

BGE_SMALL = BertConfig(vocab_size=30522, hidden=384, layers=12, heads=12,
                       ffn=1536, max_pos=512)

EMBED_DIM = BGE_SMALL.hidden


class EmbeddingEncoder:
    """CLS-pooled, L2-normalized text embeddings on MI355X kernels."""

    def __init__(self, device: str = "cuda", seed: int = 1234,
                 cfg: BertConfig = BGE_SMALL, max_len: int = 128,
                 use_graph: bool = True):
        self.cfg = cfg
        self.device = device
        self.max_len = max_len
        self.tokenizer = HashTokenizer(cfg.vocab_size)
        self.bert = BertEncoder(cfg, init_bert_weights(cfg, seed, device))
        from ..ops.graphs import GraphedCallable
        self._graphed = GraphedCallable(self._encode_impl, enabled=use_graph)

    def _encode_impl(self, ids: torch.Tensor,
                     lens: torch.Tensor) -> torch.Tensor:
        hidden = self.bert.forward(ids, lens)
        return K.pool(hidden, lens, mode="cls", l2=True)

    def encode_ids(self, ids: torch.Tensor, lens: torch.Tensor) -> torch.Tensor:
        """[B,S] i32 -> [B, H] f32 L2-normalized embeddings.

        Fixed-shape batches replay a captured hipGraph (launch-bound:
        ~150 kernel launches per forward otherwise)."""
        return self._graphed(ids, lens)

    def encode_texts(self, texts: list[str]) -> torch.Tensor:
        ids_l, lens = self.tokenizer.encode_batch(texts, self.max_len)
        ids = torch.tensor(ids_l, dtype=torch.int32, device=self.device)
        lens_t = torch.tensor(lens, dtype=torch.int32, device=self.device)
        return self.encode_ids(ids, lens_t)

    # CPU fp32 oracle for parity tests.
    def encode_ids_reference(self, ids: torch.Tensor,
                             lens: torch.Tensor) -> torch.Tensor:
        hidden = self.bert.forward_reference(ids, lens)
        return R.pool(hidden.bfloat16(), lens.cpu(), mode="cls", l2=True)
