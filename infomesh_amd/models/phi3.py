"""Phi-3-mini-shaped decoder (L=32, H=3072, 32 heads, d=96, FFN 8192,
vocab 32064, RoPE, RMSNorm, SwiGLU, fused qkv/gate_up), random-init.
In-process RAG summarizer decode on MI355X kernels: prefill = batched
MFMA GEMMs + causal softmax; decode = KV-cache attention kernel.

Replaces: the reference's external Ollama/llama.cpp/vLLM HTTP backends
(infomesh/summarizer/engine.py:111-318) — SURVEY.md §2.7.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch

from ..ops import kernels as K
from ..ops import reference as R


@dataclass(frozen=True)
class Phi3Config:
    vocab_size: int = 32064
    hidden: int = 3072
    layers: int = 32
    heads: int = 32
    kv_heads: int = 32
    ffn: int = 8192
    max_pos: int = 4096
    rope_theta: float = 10000.0
    eps: float = 1e-5

    @property
    def head_dim(self) -> int:
        return self.hidden // self.heads


PHI3_MINI = Phi3Config()
PHI3_TINY = Phi3Config(vocab_size=1024, hidden=256, layers=2, heads=8,
                       kv_heads=8, ffn=512, max_pos=256)  # test shape


def init_phi3_weights(cfg: Phi3Config, seed: int = 777,
                      device: str = "cpu") -> dict[str, torch.Tensor]:
    g = torch.Generator().manual_seed(seed)
    H, F = cfg.hidden, cfg.ffn
    qkv_out = (cfg.heads + 2 * cfg.kv_heads) * cfg.head_dim

    def w(*shape):
        return (torch.randn(*shape, generator=g) * 0.02)

    ws: dict[str, torch.Tensor] = {"embed": w(cfg.vocab_size, H)}
    for i in range(cfg.layers):
        p = f"layer.{i}."
        ws[p + "ln1.g"] = torch.ones(H)
        ws[p + "qkv.w"] = w(qkv_out, H)
        ws[p + "o.w"] = w(H, cfg.heads * cfg.head_dim)
        ws[p + "ln2.g"] = torch.ones(H)
        ws[p + "gate_up.w"] = w(2 * F, H)
        ws[p + "down.w"] = w(H, F)
    ws["final_ln.g"] = torch.ones(H)
    ws["lm_head"] = w(cfg.vocab_size, H)
    return {k: v.to(device).bfloat16() for k, v in ws.items()}


def rope_tables(cfg: Phi3Config, device: str) -> tuple[torch.Tensor, torch.Tensor]:
    half = cfg.head_dim // 2
    inv = 1.0 / (cfg.rope_theta **
                 (torch.arange(half, dtype=torch.float32) * 2 / cfg.head_dim))
    t = torch.arange(cfg.max_pos, dtype=torch.float32)
    ang = torch.outer(t, inv)
    return ang.cos().to(device), ang.sin().to(device)


class Phi3Decoder:
    def __init__(self, cfg: Phi3Config = PHI3_MINI, device: str = "cuda",
                 seed: int = 777, max_batch: int = 8,
                 max_seq: int = 2304, use_graph: bool = True):
        self.cfg = cfg
        self.device = device
        self.max_seq = min(max_seq, cfg.max_pos)
        self.max_batch = max_batch
        self.w = init_phi3_weights(cfg, seed, device)
        self.cos, self.sin = rope_tables(cfg, device)
        d = cfg.head_dim
        self.k_cache = [torch.zeros(max_batch, cfg.kv_heads, self.max_seq, d,
                                    device=device, dtype=torch.bfloat16)
                        for _ in range(cfg.layers)]
        self.v_cache = [torch.zeros_like(self.k_cache[0])
                        for _ in range(cfg.layers)]
        self.lens = torch.zeros(max_batch, dtype=torch.int32, device=device)
        self._len_host = 0
        self.use_graph = use_graph and device.startswith("cuda")
        self._graphs: dict[int, tuple] = {}

    def reset(self) -> None:
        self.lens.zero_()
        self._len_host = 0  # host shadow of max(lens): bounds decode

    # ------------------------------------------------------------ layers
    def _split_qkv(self, qkv: torch.Tensor, B: int, S: int):
        cfg = self.cfg
        d, nh, nkv = cfg.head_dim, cfg.heads, cfg.kv_heads
        qkv = qkv.view(B, S, nh + 2 * nkv, d)
        q = qkv[:, :, :nh]
        k = qkv[:, :, nh:nh + nkv]
        v = qkv[:, :, nh + nkv:]
        return q, k, v

    def prefill(self, ids: torch.Tensor) -> torch.Tensor:
        """ids [B,S] i32 (no padding: same length rows) -> logits [B,V] f32
        for the last position. Fills the KV cache positions [0, S)."""
        cfg = self.cfg
        B, S_true = ids.shape
        if S_true % 32 != 0:
            # Pad to the MFMA K granularity; causal masking keeps padded
            # (future) keys invisible to real queries, padded cache rows
            # are overwritten by the first decode_step.
            ids = torch.nn.functional.pad(ids, (0, 32 - S_true % 32))
        B, S = ids.shape
        assert B <= self.max_batch and S <= self.max_seq
        H, nh, nkv, d = cfg.hidden, cfg.heads, cfg.kv_heads, cfg.head_dim
        pos = torch.arange(S, device=self.device, dtype=torch.int32).repeat(B)
        x = K.gather(self.w["embed"], ids.reshape(-1))       # [B*S, H]
        scale = d ** -0.5
        fused = d in (32, 64, 96, 128)
        for i in range(cfg.layers):
            p = f"layer.{i}."
            h = K.rmsnorm(x, self.w[p + "ln1.g"], eps=cfg.eps)
            qkv = K.gemm_nt(h, self.w[p + "qkv.w"])
            qh, kh, vt = K.qkv_split(qkv, B, S, nh, nkv, d,
                                     cos_t=self.cos, sin_t=self.sin,
                                     pos=pos, want_vt=not fused)
            v_view = qkv.view(B, S, nh + 2 * nkv, d)[:, :, nh + nkv:]\
                .permute(0, 2, 1, 3)                    # [B, nkv, S, d]
            self.k_cache[i][:B, :, :S] = kh.view(B, nkv, S, d)
            self.v_cache[i][:B, :, :S] = v_view
            if fused:
                ctx = K.attn_fused(qh.view(B, nh, S, d),
                                   kh.view(B, nkv, S, d), v_view,
                                   causal=True, scale=scale)
            else:
                if nh != nkv:
                    rep = nh // nkv
                    kh = kh.view(B, nkv, 1, S, d)\
                        .expand(B, nkv, rep, S, d)\
                        .reshape(B * nh, S, d).contiguous()
                    vt = vt.view(B, nkv, 1, d, S)\
                        .expand(B, nkv, rep, d, S)\
                        .reshape(B * nh, d, S).contiguous()
                scores = K.gemm_nt(qh, kh, out_f32=True, alpha=scale)
                probs = K.softmax(scores, causal=True)
                ctx = K.gemm_nt(probs, vt)
            merged = K.merge_heads(ctx, B, S, nh, d)
            attn = K.gemm_nt(merged, self.w[p + "o.w"])
            x = K.add(x, attn)
            h2 = K.rmsnorm(x, self.w[p + "ln2.g"], eps=cfg.eps)
            gu = K.gemm_nt(h2, self.w[p + "gate_up.w"])
            mlp = K.gemm_nt(K.silu_mul_fused(gu, cfg.ffn),
                            self.w[p + "down.w"])
            x = K.add(x, mlp)
        self.lens[:B] = S_true
        self._len_host = max(self._len_host, S_true)
        x_last = x.view(B, S, H)[:, S_true - 1, :].contiguous()
        h = K.rmsnorm(x_last, self.w["final_ln.g"], eps=cfg.eps)
        return K.gemm_nt(h, self.w["lm_head"], out_f32=True)

    def decode_step(self, ids: torch.Tensor) -> torch.Tensor:
        """ids [B] i32 (one token per row) -> logits [B,V] f32.
        Appends to the KV cache at self.lens and increments it.

        Decode is launch-bound (~400 launches per token at L=32), so
        fixed-batch steps replay a captured hipGraph: all mutable state
        (lens, caches) lives in device tensors the graph reads/writes."""
        B = ids.shape[0]
        # host-side bound: kv_append at lens == max_seq would write out
        # of the cache (device-side, unchecked by design in the kernel)
        if self._len_host + 1 > self.max_seq:
            raise RuntimeError(
                f"KV cache full ({self._len_host}/{self.max_seq}); "
                "raise max_seq or reset()")
        self._len_host += 1
        if not self.use_graph:
            return self._decode_impl(ids)
        entry = self._graphs.get(B)
        if entry is None:
            entry = self._capture_decode(ids)
            self._graphs[B] = entry
        graph, static_ids, static_out = entry
        static_ids.copy_(ids, non_blocking=True)
        graph.replay()
        return static_out

    def _capture_decode(self, ids: torch.Tensor):
        assert int(self.lens[: ids.shape[0]].max().item()) + 4 <= self.max_seq, \
            "no headroom to capture the decode graph"
        lens_save = self.lens.clone()
        static_ids = ids.clone()
        stream = torch.cuda.Stream()
        stream.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(stream):
            for _ in range(2):
                self._decode_impl(static_ids)   # warmup (state restored below)
        torch.cuda.current_stream().wait_stream(stream)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            static_out = self._decode_impl(static_ids)
        self.lens.copy_(lens_save)  # undo warmup/capture side effects
        # (_len_host is managed by decode_step; capture ran 3 impls on
        # restored device lens, so the host shadow needs no change)
        return graph, static_ids, static_out

    def _decode_impl(self, ids: torch.Tensor) -> torch.Tensor:
        cfg = self.cfg
        B = ids.shape[0]
        H, nh, nkv, d = cfg.hidden, cfg.heads, cfg.kv_heads, cfg.head_dim
        pos = self.lens[:B].contiguous()
        x = K.gather(self.w["embed"], ids)
        scale = d ** -0.5
        new_lens = (pos + 1).contiguous()
        # residual carried through fused rmsnorm(+add) — no standalone adds
        delta, res = x, None
        for i in range(cfg.layers):
            p = f"layer.{i}."
            h, res = K.rmsnorm(delta, self.w[p + "ln1.g"], residual=res,
                               eps=cfg.eps, return_residual=True)
            qkv = K.gemm_nt(h, self.w[p + "qkv.w"])
            q, k, vt = K.qkv_split(qkv, B, 1, nh, nkv, d,
                                   cos_t=self.cos, sin_t=self.sin, pos=pos)
            K.kv_append(k.view(B, nkv, d), vt.view(B, nkv, d),
                        self.k_cache[i], self.v_cache[i], pos)
            ctx = K.attn_decode(q.view(B, nh, d), self.k_cache[i][:B],
                                self.v_cache[i][:B], new_lens, scale)
            attn = K.gemm_nt(ctx.view(B, nh * d), self.w[p + "o.w"])
            h2, res = K.rmsnorm(attn, self.w[p + "ln2.g"], residual=res,
                                eps=cfg.eps, return_residual=True)
            gu = K.gemm_nt(h2, self.w[p + "gate_up.w"])
            delta = K.gemm_nt(K.silu_mul_fused(gu, cfg.ffn),
                              self.w[p + "down.w"])
        self.lens[:B] += 1
        h = K.rmsnorm(delta, self.w["final_ln.g"], residual=res, eps=cfg.eps)
        return K.gemm_nt(h, self.w["lm_head"], out_f32=True)

    def generate_greedy(self, prompt_ids: torch.Tensor,
                        max_new_tokens: int = 32,
                        eos_id: int | None = None) -> torch.Tensor:
        """prompt_ids [B,S] -> generated [B, max_new_tokens] i32."""
        self.reset()
        logits = self.prefill(prompt_ids)
        out = []
        tok = K.argmax(logits)
        for _ in range(max_new_tokens):
            out.append(tok)
            logits = self.decode_step(tok)
            nxt = K.argmax(logits)
            if eos_id is not None and bool((nxt == eos_id).all()):
                out.append(nxt)
                break
            tok = nxt
        return torch.stack(out, dim=1)

    # -------------------------------------------------- CPU fp32 oracle
    def forward_reference(self, ids: torch.Tensor) -> torch.Tensor:
        """Full-sequence fp32 reference logits [B,S,V] (small cfgs only)."""
        cfg = self.cfg
        w = {k: v.float().cpu() for k, v in self.w.items()}
        B, S = ids.shape
        H, nh, nkv, d = cfg.hidden, cfg.heads, cfg.kv_heads, cfg.head_dim
        cos, sin = self.cos.cpu(), self.sin.cpu()
        pos = torch.arange(S, dtype=torch.int32).repeat(B)
        x = w["embed"][ids.cpu().long()].view(B * S, H)
        for i in range(cfg.layers):
            p = f"layer.{i}."
            h = R.rmsnorm(x, w[p + "ln1.g"], eps=cfg.eps)
            qkv = R.gemm_nt(h, w[p + "qkv.w"]).view(B, S, nh + 2 * nkv, d)
            q = qkv[:, :, :nh].reshape(B * S, nh, d)
            k = qkv[:, :, nh:nh + nkv].reshape(B * S, nkv, d)
            v = qkv[:, :, nh + nkv:].reshape(B * S, nkv, d)
            q = R.rope(q.bfloat16(), cos, sin, pos)
            k = R.rope(k.bfloat16(), cos, sin, pos)
            qh = q.view(B, S, nh, d).permute(0, 2, 1, 3)
            kh = k.view(B, S, nkv, d).permute(0, 2, 1, 3)
            vh = v.view(B, S, nkv, d).permute(0, 2, 1, 3)
            if nh != nkv:
                rep = nh // nkv
                kh = kh.repeat_interleave(rep, dim=1)
                vh = vh.repeat_interleave(rep, dim=1)
            scores = (qh @ kh.transpose(-1, -2) * d ** -0.5)\
                .reshape(B * nh, S, S)
            probs = R.softmax(scores, causal=True)
            ctx = (probs.view(B, nh, S, S) @ vh).permute(0, 2, 1, 3)\
                .reshape(B * S, nh * d)
            x = x + R.gemm_nt(ctx, w[p + "o.w"])
            h2 = R.rmsnorm(x, w[p + "ln2.g"], eps=cfg.eps)
            gu = R.gemm_nt(h2, w[p + "gate_up.w"])
            mlp = R.gemm_nt(
                R.silu_mul(gu[:, :cfg.ffn], gu[:, cfg.ffn:]).bfloat16(),
                w[p + "down.w"])
            x = x + mlp
        h = R.rmsnorm(x, w["final_ln.g"], eps=cfg.eps)
        return R.gemm_nt(h, w["lm_head"]).view(B, S, cfg.vocab_size)
