"""Typed torch-tensor wrappers over the HIP extension.

Every function validates layout/dtype, then launches the hand-written
CDNA4 kernel on the current HIP stream. No eager-PyTorch fallback on
GPU — a missing extension raises GPU001 (errors.GpuExtensionMissing).
"""
from __future__ import annotations

import torch

from . import _ext

ACT = {"none": 0, "gelu": 1, "silu": 2, "relu": 3, "tanh": 4}


def _ptr(t: torch.Tensor | None) -> int | None:
    return None if t is None else t.data_ptr()


def _check(t: torch.Tensor, dtype: torch.dtype, name: str) -> None:
    assert t.is_cuda, f"{name} must be on GPU"
    assert t.dtype == dtype, f"{name} must be {dtype}, got {t.dtype}"
    assert t.is_contiguous(), f"{name} must be contiguous"


def gemm_nt(a: torch.Tensor, b: torch.Tensor,
            bias: torch.Tensor | None = None,
            act: str = "none", alpha: float = 1.0,
            out_f32: bool = False,
            out: torch.Tensor | None = None) -> torch.Tensor:
    """C[...,M,N] = act(alpha * A[...,M,K] @ B[...,N,K]^T + bias[N]).

    A: [M,K] or [G,M,K] bf16; B: [N,K] or [G,N,K] bf16 (shared across the
    batch when 2-D). bias: [N] f32 or None."""
    sq_a = a.dim() == 2
    if sq_a:
        a = a.unsqueeze(0)
    if b.dim() == 2:
        b = b.unsqueeze(0).expand(a.shape[0], *b.shape)
    G, M, K = a.shape
    Gb, N, Kb = b.shape
    assert K == Kb and Gb == G, f"shape mismatch {a.shape} x {b.shape}"
    assert K % 32 == 0, f"K={K} must be a multiple of 32"
    _check(a, torch.bfloat16, "A")
    assert b.dtype == torch.bfloat16
    strideB = 0 if (G > 1 and b.stride(0) == 0) else N * K
    if strideB != 0:
        assert b.is_contiguous()
    else:
        assert b[0].is_contiguous()
    if bias is not None:
        _check(bias, torch.float32, "bias")
        assert bias.numel() == N
    dtype = torch.float32 if out_f32 else torch.bfloat16
    if out is None:
        out = torch.empty((G, M, N), device=a.device, dtype=dtype)
    else:
        assert out.shape == (G, M, N) and out.dtype == dtype and out.is_contiguous()
    if M <= 16:
        # skinny-M decode path: wave-per-column GEMV (csrc/gemv.hip)
        _ext.lib().infomesh_gemv_bf16_nt(
            a.data_ptr(), b.data_ptr(), out.data_ptr(), _ptr(bias),
            M, N, K, G, M * K, strideB, M * N,
            ACT[act], alpha, int(out_f32), _ext.stream_ptr())
    elif (M >= 4096 and N >= 512 and K % 64 == 0
          and ((M + 255) // 256) * ((N + 255) // 256) * max(G, 1) >= 224):
        # big projection shapes: deep-pipelined 256x256 tile (gemm8.hip)
        # (tile count must fill the chip at 1 block/CU)
        _ext.lib().infomesh_gemm8_bf16_nt(
            a.data_ptr(), b.data_ptr(), out.data_ptr(), _ptr(bias),
            M, N, K, G, M * K, strideB, M * N,
            ACT[act], alpha, int(out_f32), _ext.stream_ptr())
    else:
        _ext.lib().infomesh_gemm_bf16_nt(
            a.data_ptr(), b.data_ptr(), out.data_ptr(), _ptr(bias),
            M, N, K, G, M * K, strideB, M * N,
            ACT[act], alpha, int(out_f32), _ext.stream_ptr())
    return out.squeeze(0) if sq_a else out


def dense_scores(a: torch.Tensor, b: torch.Tensor, alpha: float = 1.0,
                 out: torch.Tensor | None = None) -> torch.Tensor | None:
    """Streaming dense-score GEMM (densescore.hip): C[M,N] f32 =
    alpha * A[M,K]bf16 @ B[N,K]^T for M<=128, K%128==0, huge N — the
    whole query block in LDS, docs streamed HBM->registers.

    Measured on MI355X at 128 x 1.25M x 384: 529 us vs 511 us for the
    generic 128^2 tile (both ~3.1 TB/s effective against a ~5.4 TB/s
    practical mixed-stream ceiling), so gemm_nt does NOT auto-dispatch
    here; this stays an explicit opt-in and a tuning baseline.
    Returns None when the shape is ineligible."""
    M, K = a.shape
    N = b.shape[0]
    _check(a, torch.bfloat16, "A")
    _check(b, torch.bfloat16, "B")
    if out is None:
        out = torch.empty(M, N, device=a.device, dtype=torch.float32)
    rc = _ext.lib().infomesh_dense_scores(
        a.data_ptr(), b.data_ptr(), out.data_ptr(), M, N, K, alpha,
        _ext.stream_ptr())
    return out if rc == 0 else None


def dense_scores_fp8(a: torch.Tensor, b: torch.Tensor,
                     alpha: float = 1.0,
                     out: torch.Tensor | None = None
                     ) -> torch.Tensor | None:
    """FP8 (OCP e4m3) streaming dense-score GEMM (densescore8.hip):
    C[M,N] f32 = A[M,K] @ B[N,K]^T with fp8 operands — half the read
    traffic and half the HBM footprint of the bf16 plane (the plane is
    bandwidth-bound; non-scaled fp8 MFMA matches the bf16 rate). Both
    operands must be torch.float8_e4m3fn. Returns None when the shape
    is ineligible."""
    M, K = a.shape
    N = b.shape[0]
    assert a.dtype == torch.float8_e4m3fn and a.is_cuda and a.is_contiguous()
    assert b.dtype == torch.float8_e4m3fn and b.is_contiguous()
    if out is None:
        out = torch.empty(M, N, device=a.device, dtype=torch.float32)
    rc = _ext.lib().infomesh_dense_scores_fp8(
        a.data_ptr(), b.data_ptr(), out.data_ptr(), M, N, K, alpha,
        _ext.stream_ptr())
    return out if rc == 0 else None


def layernorm(x: torch.Tensor, gamma: torch.Tensor, beta: torch.Tensor,
              residual: torch.Tensor | None = None, eps: float = 1e-12,
              return_residual: bool = False):
    """LayerNorm over the last dim; optionally fused residual add
    (y = LN(x + residual)); returns (y, x+residual) if requested."""
    H = x.shape[-1]
    assert H % 8 == 0
    rows = x.numel() // H
    _check(x, torch.bfloat16, "x")
    out = torch.empty_like(x)
    res_out = None
    if residual is not None:
        assert residual.shape == x.shape
        if return_residual:
            res_out = torch.empty_like(x)
    _ext.lib().infomesh_layernorm(
        x.data_ptr(), _ptr(residual), out.data_ptr(), _ptr(res_out),
        gamma.data_ptr(), beta.data_ptr(), rows, H, eps, _ext.stream_ptr())
    return (out, res_out) if return_residual else out


def rmsnorm(x: torch.Tensor, gamma: torch.Tensor,
            residual: torch.Tensor | None = None, eps: float = 1e-5,
            return_residual: bool = False):
    H = x.shape[-1]
    assert H % 8 == 0
    rows = x.numel() // H
    _check(x, torch.bfloat16, "x")
    out = torch.empty_like(x)
    res_out = None
    if residual is not None and return_residual:
        res_out = torch.empty_like(x)
    _ext.lib().infomesh_rmsnorm(
        x.data_ptr(), _ptr(residual), out.data_ptr(), _ptr(res_out),
        gamma.data_ptr(), rows, H, eps, _ext.stream_ptr())
    if return_residual:
        # with no residual input, the running residual IS the input
        return out, (res_out if res_out is not None else x)
    return out


def softmax(scores: torch.Tensor, scale: float = 1.0, causal: bool = False,
            valid_len: torch.Tensor | None = None,
            sq_dim: int | None = None) -> torch.Tensor:
    """Masked row softmax: f32 [G,Sq,Sk] -> bf16 [G,Sq,Sk]."""
    assert scores.dim() == 3
    _check(scores, torch.float32, "scores")
    G, Sq, Sk = scores.shape
    out = torch.empty_like(scores, dtype=torch.bfloat16)
    if valid_len is not None:
        _check(valid_len, torch.int32, "valid_len")
    _ext.lib().infomesh_softmax(
        scores.data_ptr(), out.data_ptr(), _ptr(valid_len),
        G, Sq, Sk, int(causal), scale, _ext.stream_ptr())
    return out


def bias_act(x: torch.Tensor, bias: torch.Tensor | None,
             act: str = "none") -> torch.Tensor:
    """In-place x = act(x + bias)."""
    N = x.shape[-1]
    assert N % 8 == 0
    _check(x, torch.bfloat16, "x")
    if bias is not None:
        _check(bias, torch.float32, "bias")
    _ext.lib().infomesh_bias_act(
        x.data_ptr(), _ptr(bias), x.numel() // N, N, ACT[act],
        _ext.stream_ptr())
    return x


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    assert gate.shape == up.shape and gate.numel() % 8 == 0
    out = torch.empty_like(gate)
    _ext.lib().infomesh_silu_mul(gate.data_ptr(), up.data_ptr(),
                                 out.data_ptr(), gate.numel(),
                                 _ext.stream_ptr())
    return out


def add(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    assert a.shape == b.shape and a.numel() % 8 == 0
    out = torch.empty_like(a)
    _ext.lib().infomesh_add(a.data_ptr(), b.data_ptr(), out.data_ptr(),
                            a.numel(), _ext.stream_ptr())
    return out


def rope(x: torch.Tensor, cos_t: torch.Tensor, sin_t: torch.Tensor,
         pos: torch.Tensor, rot_dim: int | None = None) -> torch.Tensor:
    """In-place NeoX-style RoPE on x [rows, H, D] with pos [rows] i32 and
    host-precomputed cos/sin [max_pos, rot/2] f32 tables."""
    rows, H, D = x.shape
    rot = rot_dim or D
    _check(x, torch.bfloat16, "x")
    _check(pos, torch.int32, "pos")
    _ext.lib().infomesh_rope(
        x.data_ptr(), cos_t.data_ptr(), sin_t.data_ptr(), pos.data_ptr(),
        rows, H, D, rot, _ext.stream_ptr())
    return x


def gather(table: torch.Tensor, ids: torch.Tensor,
           scale: float = 1.0) -> torch.Tensor:
    """out[r] = table[ids[r]] * scale (embedding lookup)."""
    _check(table, torch.bfloat16, "table")
    _check(ids, torch.int32, "ids")
    H = table.shape[1]
    out = torch.empty((ids.numel(), H), device=table.device,
                      dtype=torch.bfloat16)
    _ext.lib().infomesh_gather(table.data_ptr(), ids.data_ptr(),
                               out.data_ptr(), ids.numel(), H, scale,
                               _ext.stream_ptr())
    return out


def pool(x: torch.Tensor, lens: torch.Tensor | None = None,
         mode: str = "cls", l2: bool = True) -> torch.Tensor:
    """[B,S,H] bf16 -> [B,H] f32 (CLS or masked-mean pooling, L2 option)."""
    B, S, H = x.shape
    _check(x, torch.bfloat16, "x")
    out = torch.empty((B, H), device=x.device, dtype=torch.float32)
    _ext.lib().infomesh_pool(
        x.data_ptr(), _ptr(lens), out.data_ptr(), B, S, H,
        {"cls": 0, "mean": 1}[mode], int(l2), _ext.stream_ptr())
    return out


def argmax(logits: torch.Tensor) -> torch.Tensor:
    rows, V = logits.shape
    _check(logits, torch.float32, "logits")
    out = torch.empty((rows,), device=logits.device, dtype=torch.int32)
    _ext.lib().infomesh_argmax(logits.data_ptr(), out.data_ptr(), rows, V,
                               _ext.stream_ptr())
    return out


# Workspace layout offsets (u32 units) — keep in sync with topk.hip:
# hist1[B*256] | hist2[B*256] | bin1 | chi1 | thresh16 | cnt | cnt_eq
# | overflow. Exposed as functions because the dense plane's
# graph-replay path re-arms the deferred overflow check by hand (a
# hardcoded copy of these offsets once went stale and read cnt_eq as
# the overflow flag — round-2 fix).
def topk_cnt_off(B: int) -> int:
    return B * 512 + 3 * B


def topk_flag_off(B: int) -> int:
    return B * 512 + 5 * B


class TopK:
    """Reusable top-k selector (keeps its workspace allocated).

    With defer_check=True the overflow flag is NOT read per call (a
    host sync that would break stream overlap); the caller invokes
    check_pending() after its next natural sync point."""

    def __init__(self, device: torch.device | str = "cuda"):
        self.device = torch.device(device)
        self._ws: torch.Tensor | None = None
        self._ws_b = 0
        self.defer_check = False
        self._pending: list[tuple[torch.Tensor, int]] = []

    def __call__(self, scores: torch.Tensor, k: int,
                 sampled: bool | None = None,
                 ext_hist1: torch.Tensor | None = None
                 ) -> tuple[torch.Tensor, torch.Tensor]:
        assert scores.dim() == 2
        _check(scores, torch.float32, "scores")
        B, N = scores.shape
        assert 1 <= k <= 1024
        if sampled is None:
            # Default EXACT. The sampled threshold additionally assumes
            # (a) a near-continuous score distribution (massively tied
            # scores — e.g. BM25 tails — overflow the candidate cap) and
            # (b) N <~ 2.7M: beyond that the sample stride makes the
            # Gamma(Kp) candidate-count tails collide with the cap.
            # Failures are loud (overflow/undershoot RuntimeError), and
            # in-engine benchmarks showed no shape where it is a net
            # win, so nothing auto-enables it.
            sampled = False
        lib = _ext.lib()
        nu32 = lib.infomesh_topk_workspace_u32(B)
        if self._ws is None or self._ws_b < nu32:
            self._ws = torch.empty(nu32, device=scores.device,
                                   dtype=torch.int32)
            self._ws_b = nu32
        self._ws[: B * 512 + 5 * B + 2].zero_()  # hists + counters region
        vals = torch.empty((B, k), device=scores.device, dtype=torch.float32)
        idx = torch.empty((B, k), device=scores.device, dtype=torch.int32)
        if ext_hist1 is not None:
            # producer-fused pass 1 (bm25_block hist1=): exact counts,
            # mutually exclusive with the sampled threshold
            assert not sampled
            assert ext_hist1.dtype == torch.int32 \
                and ext_hist1.numel() >= B * 256
        lib.infomesh_topk(scores.data_ptr(), self._ws.data_ptr(),
                          vals.data_ptr(), idx.data_ptr(), B, N, k,
                          int(sampled),
                          0 if ext_hist1 is None else ext_hist1.data_ptr(),
                          _ext.stream_ptr())
        # cnt[B], cnt_eq[B], then the overflow flag after the histograms.
        cnt_off = topk_cnt_off(B)
        flag_off = topk_flag_off(B)
        need = min(k, N) if sampled else 0
        if self.defer_check:
            self._pending.append((self._ws, cnt_off, flag_off, B, need))
        else:
            self._verify(self._ws, cnt_off, flag_off, B, need)
        return vals, idx

    @staticmethod
    def _verify(ws, cnt_off, flag_off, B, need) -> None:
        if int(ws[flag_off].item()) != 0:
            raise RuntimeError(
                "topk strictly-above-threshold candidates exceeded the "
                "reserve (an unlucky sampled threshold); rerun with "
                "sampled=False")
        if need:
            # available candidates = strictly-above + threshold-equal
            avail = ws[cnt_off:cnt_off + B] + ws[cnt_off + B:cnt_off + 2 * B]
            if int(avail.min().item()) < need:
                raise RuntimeError(
                    "topk sampled threshold undershot (cnt < k); "
                    "rerun with sampled=False")

    def check_pending(self) -> None:
        pending, self._pending = self._pending, []
        for ws, cnt_off, flag_off, B, need in pending:
            self._verify(ws, cnt_off, flag_off, B, need)


def topk(scores: torch.Tensor, k: int) -> tuple[torch.Tensor, torch.Tensor]:
    return TopK(scores.device)(scores, k)


def bm25_block(doc_ids: torch.Tensor, tfdl: torch.Tensor,
               qt_off: torch.Tensor, qt_ut: torch.Tensor,
               qt_idf: torch.Tensor, u_begin: torch.Tensor,
               u_end: torch.Tensor, bounds: torch.Tensor,
               scores: torch.Tensor, doc_base: int, nseg: int,
               bd: int, avgdl: float, k1: float = 1.2,
               b: float = 0.75,
               hist1: torch.Tensor | None = None) -> torch.Tensor:
    """Doc-block LDS-accumulated BM25 for ONE posting segment.

    Two launches: a bounds pre-pass binary-searching each (unique term,
    doc-block) posting sub-range once, then the block kernel (one
    workgroup per (query, doc-block), query-major so adjacent
    workgroups reuse posting reads through L2). Writes every element of
    scores[:, doc_base:doc_base+nseg] exactly once (zero where no
    posting hits) — no pre-zeroing needed when segments partition the
    doc axis. Norm is computed in-kernel from the packed per-posting
    doc length and the current global avgdl.

    bounds: caller-provided i32 workspace of at least
    U * ceil(nseg/bd) * 2 elements."""
    B, N = scores.shape
    U = u_begin.numel()
    nblocks = (nseg + bd - 1) // bd
    _check(scores, torch.float32, "scores")
    assert nblocks <= 65535 and qt_off.numel() == B + 1
    assert doc_ids.dtype == torch.int32 and tfdl.dtype == torch.int32
    assert bounds.numel() >= U * nblocks * 2
    assert bd * 4 + (8 * 264 * 4 if hist1 is not None else 0) <= 160 * 1024
    if hist1 is not None:
        # fused topk pass 1: exact [B,256] ordered-top-byte counts of the
        # written score columns (caller zeroes it before the first
        # segment and hands it to topk(ext_hist1=...))
        assert hist1.dtype == torch.int32 and hist1.numel() >= B * 256
    norm_a = k1 * (1.0 - b)
    norm_b = k1 * b / max(avgdl, 1e-9)
    _ext.lib().infomesh_bm25_block(
        doc_ids.data_ptr(), tfdl.data_ptr(), qt_off.data_ptr(),
        qt_ut.data_ptr(), qt_idf.data_ptr(), u_begin.data_ptr(),
        u_end.data_ptr(), bounds.data_ptr(), scores.data_ptr(),
        0 if hist1 is None else hist1.data_ptr(),
        B, U, N, doc_base, nseg, bd,
        norm_a, norm_b, k1 + 1.0, _ext.stream_ptr())
    return scores


def score_combine(a: torch.Tensor, b: torch.Tensor, wa: float,
                  wb: float) -> torch.Tensor:
    out = torch.empty_like(a)
    _ext.lib().infomesh_score_combine(a.data_ptr(), b.data_ptr(),
                                      out.data_ptr(), wa, wb, a.numel(),
                                      _ext.stream_ptr())
    return out


def simhash_fingerprint(offsets: torch.Tensor,
                        hashes: torch.Tensor) -> torch.Tensor:
    """CSR shingle hashes (int64-as-u64) -> int64 fingerprints [D]."""
    ndocs = offsets.numel() - 1
    out = torch.empty(ndocs, device=offsets.device, dtype=torch.int64)
    _ext.lib().infomesh_simhash_fingerprint(
        offsets.data_ptr(), hashes.data_ptr(), out.data_ptr(), ndocs,
        _ext.stream_ptr())
    return out


def hamming_scan(queries: torch.Tensor, table: torch.Tensor,
                 radius: int = 3, cap: int = 65536
                 ) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Matches (q_idx, table_idx, dist) with hamming(q, t) <= radius."""
    M, N = queries.numel(), table.numel()
    dev = table.device
    out_q = torch.empty(cap, device=dev, dtype=torch.int32)
    out_n = torch.empty(cap, device=dev, dtype=torch.int32)
    out_d = torch.empty(cap, device=dev, dtype=torch.int32)
    cnt = torch.zeros(1, device=dev, dtype=torch.int32)
    _ext.lib().infomesh_hamming_scan(
        queries.data_ptr(), table.data_ptr(), out_q.data_ptr(),
        out_n.data_ptr(), out_d.data_ptr(), cnt.data_ptr(), M, N, radius,
        cap, _ext.stream_ptr())
    n = min(int(cnt.item()), cap)
    return out_q[:n], out_n[:n], out_d[:n]


def attn_fused(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
               valid_len: torch.Tensor | None = None,
               causal: bool = False, scale: float | None = None
               ) -> torch.Tensor:
    """Fused flash attention: q [B,nh,Sq,d], k/v [B,nhk,Sk,d] — any
    batch/head/row strides (element stride must be 1, d in {32,64,96,
    128}). GQA via nh % nhk == 0. valid_len: [B] i32 key limits.
    Returns O [B*nh, Sq, d] bf16 contiguous."""
    assert q.dim() == 4 and k.dim() == 4 and v.dim() == 4
    B, nh, Sq, d = q.shape
    _, nhk, Sk, dk = k.shape
    assert d == dk and d in (32, 64, 96, 128) and nh % nhk == 0
    for t in (q, k, v):
        assert t.dtype == torch.bfloat16 and t.stride(3) == 1
    if scale is None:
        scale = d ** -0.5
    out = torch.empty(B * nh, Sq, d, device=q.device, dtype=torch.bfloat16)
    if valid_len is not None:
        _check(valid_len, torch.int32, "valid_len")
        assert valid_len.numel() == B
    _ext.lib().infomesh_attn_fused(
        q.data_ptr(), k.data_ptr(), v.data_ptr(), _ptr(valid_len),
        out.data_ptr(), B, nh, nhk, Sq, Sk, d,
        q.stride(0), q.stride(1), q.stride(2),
        k.stride(0), k.stride(1), k.stride(2),
        v.stride(0), v.stride(1), v.stride(2),
        int(causal), scale, _ext.stream_ptr())
    return out


def attn_decode(q: torch.Tensor, k_cache: torch.Tensor,
                v_cache: torch.Tensor, lens: torch.Tensor,
                scale: float) -> torch.Tensor:
    """q [B,H,D], caches [B,Hkv,Smax,D], lens [B] i32 -> out [B,H,D].

    Uses flash-decoding sequence splitting when B*H alone cannot fill
    the chip (partials per chunk + combine kernel)."""
    import math
    B, H, D = q.shape
    _, Hkv, Smax, _ = k_cache.shape
    _check(q, torch.bfloat16, "q")
    assert D % 8 == 0 and D <= 128
    out = torch.empty_like(q)
    nc_min = max(1, math.ceil(Smax / 2048))
    nc = max(nc_min, min(64, math.ceil(512 / max(1, B * H))))
    if nc <= 1:
        _ext.lib().infomesh_attn_decode(
            q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
            lens.data_ptr(), out.data_ptr(), B, H, Hkv, Smax, D, scale,
            _ext.stream_ptr())
    else:
        part = torch.empty(B, H, nc, D + 2, device=q.device,
                           dtype=torch.float32)
        _ext.lib().infomesh_attn_decode_split(
            q.data_ptr(), k_cache.data_ptr(), v_cache.data_ptr(),
            lens.data_ptr(), part.data_ptr(), out.data_ptr(),
            B, H, Hkv, Smax, D, nc, scale, _ext.stream_ptr())
    return out


def qkv_split(qkv: torch.Tensor, B: int, S: int, nh: int, nkv: int,
              d: int, cos_t: torch.Tensor | None = None,
              sin_t: torch.Tensor | None = None,
              pos: torch.Tensor | None = None,
              rot: int | None = None, want_vt: bool = True
              ) -> tuple[torch.Tensor, torch.Tensor, torch.Tensor | None]:
    """Fused head split (+RoPE on q/k when tables given):
    qkv [B*S, (nh+2nkv)*d] -> q [B*nh,S,d], k [B*nkv,S,d], and (when
    want_vt) the PV-GEMM-ready transposed vt [B*nkv,d,S]; the fused
    attention path reads V strided from qkv instead (want_vt=False)."""
    _check(qkv, torch.bfloat16, "qkv")
    dev = qkv.device
    q = torch.empty(B * nh, S, d, device=dev, dtype=torch.bfloat16)
    k = torch.empty(B * nkv, S, d, device=dev, dtype=torch.bfloat16)
    vt = torch.empty(B * nkv, d, S, device=dev, dtype=torch.bfloat16) \
        if want_vt else None
    _ext.lib().infomesh_qkv_split(
        qkv.data_ptr(), q.data_ptr(), k.data_ptr(), _ptr(vt),
        _ptr(cos_t), _ptr(sin_t), _ptr(pos),
        B, S, nh, nkv, d, rot or d, _ext.stream_ptr())
    return q, k, vt


def merge_heads(ctx: torch.Tensor, B: int, S: int, nh: int,
                d: int) -> torch.Tensor:
    """[B*nh, S, d] -> [B*S, nh*d]."""
    _check(ctx, torch.bfloat16, "ctx")
    out = torch.empty(B * S, nh * d, device=ctx.device, dtype=torch.bfloat16)
    _ext.lib().infomesh_merge_heads(ctx.data_ptr(), out.data_ptr(),
                                    B, S, nh, d, _ext.stream_ptr())
    return out


def silu_mul_fused(gu: torch.Tensor, F: int) -> torch.Tensor:
    """gu [rows, 2F] -> silu(gu[:, :F]) * gu[:, F:] without slicing."""
    _check(gu, torch.bfloat16, "gu")
    rows = gu.shape[0]
    assert gu.shape[1] == 2 * F and F % 8 == 0
    out = torch.empty(rows, F, device=gu.device, dtype=torch.bfloat16)
    _ext.lib().infomesh_silu_mul_fused(gu.data_ptr(), out.data_ptr(),
                                       rows, F, _ext.stream_ptr())
    return out


def kv_append(k_new: torch.Tensor, v_new: torch.Tensor,
              k_cache: torch.Tensor, v_cache: torch.Tensor,
              pos: torch.Tensor) -> None:
    B, Hkv, D = k_new.shape
    _, _, Smax, _ = k_cache.shape
    _ext.lib().infomesh_kv_append(
        k_new.data_ptr(), v_new.data_ptr(), k_cache.data_ptr(),
        v_cache.data_ptr(), pos.data_ptr(), B, Hkv, Smax, D,
        _ext.stream_ptr())
