"""Plain-PyTorch fp32 reference implementations of every HIP kernel.

These are the numerics oracle for the GPU parity tests (SURVEY.md §4:
kernel-vs-CPU-reference strategy) and the executable spec of each op.
They are NOT used on the GPU path.
"""
from __future__ import annotations

import torch


def gemm_nt(a: torch.Tensor, b: torch.Tensor,
            bias: torch.Tensor | None = None, act: str = "none",
            alpha: float = 1.0) -> torch.Tensor:
    out = alpha * torch.matmul(a.float(), b.float().transpose(-1, -2))
    if bias is not None:
        out = out + bias.float()
    return apply_act(out, act)


def apply_act(x: torch.Tensor, act: str) -> torch.Tensor:
    if act == "gelu":
        return torch.nn.functional.gelu(x, approximate="tanh")
    if act == "silu":
        return torch.nn.functional.silu(x)
    if act == "relu":
        return torch.relu(x)
    if act == "tanh":
        return torch.tanh(x)
    return x


def layernorm(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
              residual: torch.Tensor | None = None,
              eps: float = 1e-12) -> torch.Tensor:
    xf = x.float()
    if residual is not None:
        xf = xf + residual.float()
    return torch.nn.functional.layer_norm(
        xf, (x.shape[-1],), gamma.float(), beta.float(), eps)


def rmsnorm(x: torch.Tensor, gamma: torch.Tensor,
            residual: torch.Tensor | None = None,
            eps: float = 1e-5) -> torch.Tensor:
    xf = x.float()
    if residual is not None:
        xf = xf + residual.float()
    var = xf.pow(2).mean(-1, keepdim=True)
    return xf * torch.rsqrt(var + eps) * gamma.float()


def softmax(scores: torch.Tensor, scale: float = 1.0, causal: bool = False,
            valid_len: torch.Tensor | None = None) -> torch.Tensor:
    G, Sq, Sk = scores.shape
    s = scores.float() * scale
    mask = torch.zeros_like(s, dtype=torch.bool)
    if causal:
        i = torch.arange(Sq).unsqueeze(1)
        j = torch.arange(Sk).unsqueeze(0)
        mask |= (j > i + (Sk - Sq)).unsqueeze(0)
    if valid_len is not None:
        j = torch.arange(Sk).view(1, 1, Sk)
        mask |= j >= valid_len.view(G, 1, 1)
    s = s.masked_fill(mask, float("-inf"))
    out = torch.softmax(s, dim=-1)
    return torch.nan_to_num(out, nan=0.0)  # fully-masked rows -> 0


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    return torch.nn.functional.silu(gate.float()) * up.float()


def rope(x: torch.Tensor, cos_t: torch.Tensor, sin_t: torch.Tensor,
         pos: torch.Tensor, rot_dim: int | None = None) -> torch.Tensor:
    """NeoX half-rotation: pairs (d, d+rot/2)."""
    rows, H, D = x.shape
    rot = rot_dim or D
    half = rot // 2
    xf = x.float().clone()
    c = cos_t[pos.long()].view(rows, 1, half)
    s = sin_t[pos.long()].view(rows, 1, half)
    x0 = xf[..., :half].clone()
    x1 = xf[..., half:rot].clone()
    xf[..., :half] = x0 * c - x1 * s
    xf[..., half:rot] = x0 * s + x1 * c
    return xf


def pool(x: torch.Tensor, lens: torch.Tensor | None = None,
         mode: str = "cls", l2: bool = True) -> torch.Tensor:
    B, S, H = x.shape
    xf = x.float()
    if mode == "cls":
        out = xf[:, 0, :]
    else:
        if lens is None:
            out = xf.mean(1)
        else:
            mask = (torch.arange(S, device=x.device).view(1, S, 1)
                    < lens.view(B, 1, 1)).float()
            out = (xf * mask).sum(1) / mask.sum(1).clamp(min=1)
    if l2:
        out = torch.nn.functional.normalize(out, dim=-1)
    return out


def bm25_scores(postings: dict[int, list[tuple[int, int]]],
                doc_lens: torch.Tensor, queries: list[list[int]],
                n_docs: int, k1: float = 1.2, b: float = 0.75
                ) -> torch.Tensor:
    """Okapi BM25 over a {term: [(doc, tf)]} dict — the scalar spec of
    the CSR kernel. idf = ln(1 + (N - df + .5)/(df + .5))."""
    import math
    avgdl = float(doc_lens.float().mean()) if n_docs else 1.0
    norm = k1 * (1 - b + b * doc_lens.float() / avgdl)
    out = torch.zeros((len(queries), n_docs))
    for qi, terms in enumerate(queries):
        for t in terms:
            plist = postings.get(t, [])
            if not plist:
                continue
            df = len(plist)
            idf = math.log(1.0 + (n_docs - df + 0.5) / (df + 0.5))
            for doc, tf in plist:
                out[qi, doc] += idf * tf * (k1 + 1) / (tf + float(norm[doc]))
    return out


def simhash_fingerprint(hashes_per_doc: list[list[int]]) -> list[int]:
    """Charikar bit-vote over 64-bit shingle hashes."""
    fps = []
    for hashes in hashes_per_doc:
        fp = 0
        for bit in range(64):
            vote = sum(1 if (h >> bit) & 1 else -1 for h in hashes)
            if vote > 0:
                fp |= 1 << bit
        fps.append(fp)
    return fps


def hamming_matches(queries: list[int], table: list[int],
                    radius: int = 3) -> set[tuple[int, int]]:
    out = set()
    for m, q in enumerate(queries):
        for n, t in enumerate(table):
            if bin(q ^ t).count("1") <= radius:
                out.add((m, n))
    return out


def attn_decode(q: torch.Tensor, k_cache: torch.Tensor,
                v_cache: torch.Tensor, lens: torch.Tensor,
                scale: float) -> torch.Tensor:
    B, H, D = q.shape
    _, Hkv, Smax, _ = k_cache.shape
    group = H // Hkv
    out = torch.zeros_like(q, dtype=torch.float32)
    for b in range(B):
        L = int(lens[b])
        for h in range(H):
            kv = h // group
            k = k_cache[b, kv, :L].float()
            v = v_cache[b, kv, :L].float()
            p = torch.softmax(q[b, h].float() @ k.T * scale, dim=-1)
            out[b, h] = p @ v
    return out
