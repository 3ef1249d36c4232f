"""BERT-family encoder on hand-written CDNA4 kernels (inference, bf16).

Used by the bge-small-shaped embedding encoder and the
bge-reranker-base-shaped cross-encoder. Forward path is entirely
infomesh HIP kernels: gather/add/layernorm, fused-QKV MFMA GEMM,
batched QK^T GEMM -> masked softmax -> PV GEMM, fused bias+GELU FFN.
A same-weights fp32 reference forward (ops/reference.py) is the parity
oracle for tests.
"""
from __future__ import annotations

from dataclasses import dataclass

import torch

from ..ops import kernels as K
from ..ops import reference as R


@dataclass(frozen=True)
class BertConfig:
    vocab_size: int
    hidden: int
    layers: int
    heads: int
    ffn: int
    max_pos: int = 512
    eps: float = 1e-12

    @property
    def head_dim(self) -> int:
        return self.hidden // self.heads


def init_bert_weights(cfg: BertConfig, seed: int = 1234,
                      device: str = "cpu",
                      dtype: torch.dtype = torch.bfloat16) -> dict[str, torch.Tensor]:
    """Deterministic random init (BASELINE: random-init weights)."""
    g = torch.Generator().manual_seed(seed)
    H, F = cfg.hidden, cfg.ffn
    std = 0.02

    def w(*shape, s=std):
        return torch.randn(*shape, generator=g) * s

    ws: dict[str, torch.Tensor] = {
        "embed.word": w(cfg.vocab_size, H),
        "embed.pos": w(cfg.max_pos, H),
        "embed.ln.g": torch.ones(H),
        "embed.ln.b": torch.zeros(H),
    }
    for i in range(cfg.layers):
        p = f"layer.{i}."
        ws[p + "qkv.w"] = w(3 * H, H)
        ws[p + "qkv.b"] = torch.zeros(3 * H)
        ws[p + "attn_out.w"] = w(H, H)
        ws[p + "attn_out.b"] = torch.zeros(H)
        ws[p + "ln1.g"] = torch.ones(H)
        ws[p + "ln1.b"] = torch.zeros(H)
        ws[p + "ffn_in.w"] = w(F, H)
        ws[p + "ffn_in.b"] = torch.zeros(F)
        ws[p + "ffn_out.w"] = w(H, F)
        ws[p + "ffn_out.b"] = torch.zeros(H)
        ws[p + "ln2.g"] = torch.ones(H)
        ws[p + "ln2.b"] = torch.zeros(H)
    out = {}
    for k, v in ws.items():
        t = v.to(device)
        # biases stay f32 (kernel epilogue reads f32 bias); rest bf16.
        out[k] = t if k.endswith(".b") and "ln" not in k else t.to(dtype)
        if "ln" in k:
            out[k] = t.to(dtype)
    return out


class BertEncoder:
    def __init__(self, cfg: BertConfig, weights: dict[str, torch.Tensor],
                 use_fused_attn: bool = True):
        self.cfg = cfg
        self.w = weights
        # fused flash kernel supports d in {32,64,96,128}
        self.use_fused_attn = use_fused_attn and cfg.head_dim in (32, 64,
                                                                  96, 128)

    # ------------------------------------------------------------- GPU
    def forward(self, ids: torch.Tensor, lens: torch.Tensor) -> torch.Tensor:
        """ids [B,S] int32 (padded), lens [B] int32 -> [B,S,H] bf16."""
        cfg, w = self.cfg, self.w
        B, S = ids.shape
        if S % 32 != 0:
            # PV GEMM contracts over S — pad to the MFMA K granularity;
            # padded keys are zeroed by the valid_len softmax mask.
            pad = 32 - S % 32
            ids = torch.nn.functional.pad(ids, (0, pad))
            S += pad
        H, nh, d = cfg.hidden, cfg.heads, cfg.head_dim
        pos_ids = torch.arange(S, device=ids.device, dtype=torch.int32)\
            .repeat(B)
        x = K.gather(w["embed.word"], ids.reshape(-1))
        x = K.add(x, K.gather(w["embed.pos"], pos_ids))
        x = K.layernorm(x, w["embed.ln.g"], w["embed.ln.b"], eps=cfg.eps)

        # per-(batch*head) valid key lengths for the softmax mask
        vl = lens.repeat_interleave(nh).contiguous()
        scale = d ** -0.5
        vl_b = lens.contiguous()
        for i in range(cfg.layers):
            p = f"layer.{i}."
            qkv = K.gemm_nt(x, w[p + "qkv.w"], bias=w[p + "qkv.b"])
            if self.use_fused_attn:
                # zero-copy strided head views straight out of qkv
                qkv5 = qkv.view(B, S, 3, nh, d)
                qv = qkv5[:, :, 0].permute(0, 2, 1, 3)
                kv = qkv5[:, :, 1].permute(0, 2, 1, 3)
                vv = qkv5[:, :, 2].permute(0, 2, 1, 3)
                ctx = K.attn_fused(qv, kv, vv, valid_len=vl_b, scale=scale)
            else:
                q, k, vt = K.qkv_split(qkv, B, S, nh, nh, d)
                scores = K.gemm_nt(q, k, out_f32=True, alpha=scale)
                probs = K.softmax(scores, valid_len=vl)
                ctx = K.gemm_nt(probs, vt)               # [G, S, d]
            merged = K.merge_heads(ctx, B, S, nh, d)
            attn = K.gemm_nt(merged, w[p + "attn_out.w"],
                             bias=w[p + "attn_out.b"])
            x = K.layernorm(attn, w[p + "ln1.g"], w[p + "ln1.b"],
                            residual=x, eps=cfg.eps)
            h = K.gemm_nt(x, w[p + "ffn_in.w"], bias=w[p + "ffn_in.b"],
                          act="gelu")
            o = K.gemm_nt(h, w[p + "ffn_out.w"], bias=w[p + "ffn_out.b"])
            x = K.layernorm(o, w[p + "ln2.g"], w[p + "ln2.b"],
                            residual=x, eps=cfg.eps)
        return x.view(B, S, H)

    # -------------------------------------------------- CPU fp32 oracle
    def forward_reference(self, ids: torch.Tensor,
                          lens: torch.Tensor) -> torch.Tensor:
        cfg = self.cfg
        w = {k: v.float().cpu() for k, v in self.w.items()}
        B, S = ids.shape
        H, nh, d = cfg.hidden, cfg.heads, cfg.head_dim
        ids_l = ids.cpu().long()
        x = w["embed.word"][ids_l] + w["embed.pos"][:S].unsqueeze(0)
        x = R.layernorm(x, w["embed.ln.g"], w["embed.ln.b"], eps=cfg.eps)
        vl = lens.cpu().repeat_interleave(nh)
        for i in range(cfg.layers):
            p = f"layer.{i}."
            qkv = R.gemm_nt(x.view(B * S, H), w[p + "qkv.w"], w[p + "qkv.b"])
            qkv = qkv.view(B, S, 3, nh, d)
            q = qkv[:, :, 0].permute(0, 2, 1, 3).reshape(B * nh, S, d)
            k = qkv[:, :, 1].permute(0, 2, 1, 3).reshape(B * nh, S, d)
            v = qkv[:, :, 2].permute(0, 2, 1, 3).reshape(B * nh, S, d)
            scores = torch.matmul(q, k.transpose(1, 2)) * d ** -0.5
            probs = R.softmax(scores, valid_len=vl)
            ctx = torch.matmul(probs, v)
            merged = ctx.view(B, nh, S, d).permute(0, 2, 1, 3).reshape(B * S, H)
            attn = R.gemm_nt(merged, w[p + "attn_out.w"], w[p + "attn_out.b"])
            x = R.layernorm(attn, w[p + "ln1.g"], w[p + "ln1.b"],
                            residual=x.view(B * S, H), eps=cfg.eps)
            h = R.gemm_nt(x, w[p + "ffn_in.w"], w[p + "ffn_in.b"], act="gelu")
            o = R.gemm_nt(h, w[p + "ffn_out.w"], w[p + "ffn_out.b"])
            x = R.layernorm(o, w[p + "ln2.g"], w[p + "ln2.b"],
                            residual=x, eps=cfg.eps)
        return x.view(B, S, H)
