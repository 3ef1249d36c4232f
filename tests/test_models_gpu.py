"""Model-level GPU parity tests vs the same-weights fp32 CPU reference.

Small configs keep runtime low; the shapes exercise every kernel on the
real model paths (SURVEY.md §4 test strategy).
"""
from __future__ import annotations

import pytest
import torch

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from infomesh_amd.ops import _build
    _build.build()

from infomesh_amd.models.bert import BertConfig
from infomesh_amd.models.encoder import EmbeddingEncoder
from infomesh_amd.models.phi3 import PHI3_TINY, Phi3Decoder
from infomesh_amd.models.reranker import CrossEncoderReranker


@pytest.fixture(autouse=True)
def _gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(0)


TINY_BERT = BertConfig(vocab_size=2048, hidden=128, layers=2, heads=4,
                       ffn=256, max_pos=64)


def test_encoder_parity_tiny():
    enc = EmbeddingEncoder(device="cuda", cfg=TINY_BERT, max_len=32)
    ids = torch.randint(4, 2048, (3, 17), dtype=torch.int32, device="cuda")
    lens = torch.tensor([17, 9, 3], dtype=torch.int32, device="cuda")
    ids[1, 9:] = 0
    ids[2, 3:] = 0
    out = enc.encode_ids(ids, lens).cpu()
    ref = enc.encode_ids_reference(ids.cpu(), lens.cpu())
    # bf16 through 2 layers: generous but meaningful bound; embeddings
    # are unit-norm so absolute error is the right measure.
    assert (out - ref).abs().max() < 0.05
    assert torch.allclose(out.norm(dim=-1), torch.ones(3), atol=1e-3)


def test_encoder_padding_invariance():
    """Extra padding must not change embeddings (mask correctness)."""
    enc = EmbeddingEncoder(device="cuda", cfg=TINY_BERT, max_len=32)
    ids = torch.randint(4, 2048, (1, 10), dtype=torch.int32, device="cuda")
    lens = torch.tensor([10], dtype=torch.int32, device="cuda")
    out1 = enc.encode_ids(ids, lens).cpu()
    padded = torch.zeros(1, 24, dtype=torch.int32, device="cuda")
    padded[:, :10] = ids
    out2 = enc.encode_ids(padded, lens).cpu()
    assert (out1 - out2).abs().max() < 2e-2


def test_encoder_text_api():
    enc = EmbeddingEncoder(device="cuda", cfg=TINY_BERT, max_len=32)
    e = enc.encode_texts(["gpu kernels", "gpu kernels", "pasta recipe"])
    assert e.shape == (3, 128)
    sim_same = torch.dot(e[0], e[1]).item()
    assert sim_same > 0.999  # identical text -> identical embedding


def test_reranker_scores():
    cfg = BertConfig(vocab_size=4096, hidden=128, layers=2, heads=4,
                     ffn=256, max_pos=128)
    rr = CrossEncoderReranker(device="cuda", cfg=cfg, max_len=64)
    ranked = rr.rerank("gpu kernels", ["doc one text", "doc two text",
                                      "third document"], keep=2)
    assert len(ranked) == 2
    scores = rr.score_pairs("gpu kernels", ["same passage"] * 3)
    assert torch.allclose(scores, scores[0].expand(3), atol=1e-3)


def test_phi3_tiny_prefill_parity():
    dec = Phi3Decoder(PHI3_TINY, device="cuda", max_batch=2, max_seq=64)
    ids = torch.randint(0, 1024, (2, 12), dtype=torch.int32, device="cuda")
    logits = dec.prefill(ids).cpu()
    ref = dec.forward_reference(ids.cpu())[:, -1, :]
    scale = ref.abs().max().item()
    assert (logits - ref).abs().max() < 0.05 * scale + 0.05


def test_phi3_decode_matches_prefill():
    """Decode-step logits must match a prefill over the same prefix."""
    dec = Phi3Decoder(PHI3_TINY, device="cuda", max_batch=2, max_seq=64)
    ids = torch.randint(0, 1024, (2, 9), dtype=torch.int32, device="cuda")
    dec.reset()
    dec.prefill(ids[:, :8])
    step_logits = dec.decode_step(ids[:, 8].contiguous()).cpu()

    dec2 = Phi3Decoder(PHI3_TINY, device="cuda", max_batch=2, max_seq=64)
    full_logits = dec2.prefill(ids).cpu()
    scale = full_logits.abs().max().item()
    assert (step_logits - full_logits).abs().max() < 0.05 * scale + 0.05


def test_phi3_generate_deterministic():
    dec = Phi3Decoder(PHI3_TINY, device="cuda", max_batch=1, max_seq=64)
    ids = torch.randint(0, 1024, (1, 6), dtype=torch.int32, device="cuda")
    g1 = dec.generate_greedy(ids, max_new_tokens=5).cpu()
    g2 = dec.generate_greedy(ids, max_new_tokens=5).cpu()
    assert torch.equal(g1, g2)
    assert g1.shape[1] == 5
