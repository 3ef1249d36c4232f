"""CPU-side model tests: tokenizer, weight init determinism, reference
forwards (shape/sanity — GPU parity lives in test_models_gpu.py)."""
from __future__ import annotations

import torch

from infomesh_amd.models.bert import BertConfig, init_bert_weights
from infomesh_amd.models.phi3 import PHI3_TINY, init_phi3_weights, rope_tables
from infomesh_amd.models.tokenizer import CLS, PAD, SEP, HashTokenizer, tokenize


def test_tokenize_basic():
    assert tokenize("Hello, World!") == ["hello", ",", "world", "!"]
    assert tokenize("日本語 text") == ["日", "本", "語", "text"]


def test_hash_tokenizer_deterministic():
    t = HashTokenizer(30522)
    a = t.encode("gpu kernels are fast")
    b = t.encode("gpu kernels are fast")
    assert a == b
    assert a[0] == CLS and a[-1] == SEP
    assert all(0 <= i < 30522 for i in a)


def test_encode_batch_padding():
    t = HashTokenizer(1000)
    ids, lens = t.encode_batch(["one two three", "one"], max_len=16)
    assert len(ids[0]) == len(ids[1])
    assert lens[0] > lens[1]
    assert ids[1][lens[1]:] == [PAD] * (len(ids[1]) - lens[1])


def test_encode_pair_budget():
    t = HashTokenizer(1000)
    pair = t.encode_pair("short query", "word " * 500, max_len=64)
    assert len(pair) <= 64
    assert pair.count(SEP) == 2


def test_bert_weights_deterministic():
    cfg = BertConfig(vocab_size=100, hidden=32, layers=1, heads=4, ffn=64)
    w1 = init_bert_weights(cfg, seed=7)
    w2 = init_bert_weights(cfg, seed=7)
    assert torch.equal(w1["layer.0.qkv.w"], w2["layer.0.qkv.w"])
    w3 = init_bert_weights(cfg, seed=8)
    assert not torch.equal(w1["layer.0.qkv.w"], w3["layer.0.qkv.w"])
    assert w1["layer.0.qkv.b"].dtype == torch.float32  # biases stay f32
    assert w1["layer.0.qkv.w"].dtype == torch.bfloat16


def test_phi3_weights_shapes():
    w = init_phi3_weights(PHI3_TINY, seed=1)
    assert w["embed"].shape == (1024, 256)
    assert w["layer.0.qkv.w"].shape == (3 * 256, 256)
    assert w["layer.1.gate_up.w"].shape == (1024, 256)
    cos, sin = rope_tables(PHI3_TINY, "cpu")
    assert cos.shape == (256, 16)
    assert torch.allclose(cos[0], torch.ones(16))


def test_phi3_kv_cache_bound_guard():
    """decode_step refuses when the KV cache is full instead of letting
    kv_append write out of bounds on device."""
    import pytest
    import torch
    from infomesh_amd.models.phi3 import PHI3_TINY, Phi3Decoder
    dec = Phi3Decoder(PHI3_TINY, device="cpu", max_batch=1, max_seq=8,
                      use_graph=False)
    dec._len_host = 8  # cache full
    with pytest.raises(RuntimeError, match="KV cache full"):
        dec.decode_step(torch.zeros(1, dtype=torch.int32))
