// Fused attention forward (flash-style) for encoder / reranker /
// causal prefill: O = softmax(Q K^T * scale + mask) V without
// materializing the S x S score matrix.
//
// Replaces the decomposed QK^T-GEMM -> f32 softmax -> PV-GEMM path
// (which writes/reads G*S*S f32 scores to HBM and wastes MFMA tiles at
// S=160-class shapes). Structure per guide Appendix B "fused attention
// prefill": per block one 64-row Q tile of one (batch*head), 4 waves x
// 16 q-rows; K/V tiles of 64 keys staged in LDS (V transposed at stage
// time so PV's B-fragments read contiguous keys); online softmax with
// per-row running (m, l); P routed through LDS to become MFMA A-frags.
//
// Template D in {32, 64, 96, 128} covers bge-small (32), reranker (64),
// phi-3 (96) and common 128-dim heads.
#include "common.h"

#define QTILE 64
#define KTILE 64
#define VPAD 72   // V^T LDS row stride (elems): 16-B-aligned rows + bank spread

// Strided operands: element address = base + b*bs + h*hs + row*rs + j.
// g = b*nhq + hq; kv head = hq / (nhq/nhk) — GQA without expansion.
struct TensorView {
  const bf16* ptr;
  long bs, hs, rs;
};

namespace {

template <int D>
__global__ __launch_bounds__(256) void attn_fused_kernel(
    TensorView Q, TensorView K, TensorView V,
    const int* __restrict__ valid_len,
    bf16* __restrict__ O, int nhq, int nhk,
    int Sq, int Sk, int causal, float scale) {
  __shared__ bf16 k_lds[KTILE][D];          // [key][d] linear
  __shared__ bf16 vt_lds[D][VPAD];          // [d][key] transposed (+pad)
  __shared__ bf16 p_lds[4][16][KTILE];      // per-wave P tiles
  const int g = blockIdx.y;
  const int q0 = blockIdx.x * QTILE;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int b = g / nhq, hq = g % nhq;
  const int hk = hq / (nhq / nhk);
  const int limit_all = valid_len ? min(Sk, valid_len[b]) : Sk;

  const bf16* Qg = Q.ptr + (long)b * Q.bs + (long)hq * Q.hs;
  const bf16* Kg = K.ptr + (long)b * K.bs + (long)hk * K.hs;
  const bf16* Vg = V.ptr + (long)b * V.bs + (long)hk * V.hs;

  // ---- Q fragments in registers: wave owns rows q0+wid*16 .. +16
  const int qrow_f = lane & 15;             // A-frag row within 16
  const int kslice = lane >> 4;             // 0..3 -> k-offset *8
  bf16x8 qfrag[D / 32];
  {
    const int qr = min(q0 + wid * 16 + qrow_f, Sq - 1);
#pragma unroll
    for (int f = 0; f < D / 32; ++f)
      qfrag[f] = *reinterpret_cast<const bf16x8*>(
          Qg + (long)qr * Q.rs + f * 32 + kslice * 8);
  }

  // ---- accumulators: O C-frags [16 x 16] x (D/16); row stats
  f32x4 o_acc[D / 16];
#pragma unroll
  for (int f = 0; f < D / 16; ++f) o_acc[f] = f32x4{0.f, 0.f, 0.f, 0.f};
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m_run[r] = -INFINITY; l_run[r] = 0.f; }
  const int crow0 = (lane >> 4) * 4;        // C-frag rows this lane owns
  const int ccol = lane & 15;

  // causal upper bound for this q-tile: max key index needed + 1
  int kv_limit = limit_all;
  if (causal) {
    const int max_q = min(q0 + QTILE - 1, Sq - 1);
    kv_limit = min(kv_limit, max_q + (Sk - Sq) + 1);
  }

  for (int kv0 = 0; kv0 < kv_limit; kv0 += KTILE) {
    // ---- cooperative staging (256 threads)
    {
      // K: rows of 64 keys x D; thread -> (key, 8-elem d chunk)
      const int perrow = D / 8;
      for (int t = threadIdx.x; t < KTILE * perrow; t += 256) {
        const int key = t / perrow, dc = (t % perrow) * 8;
        const int src = min(kv0 + key, Sk - 1);
        *reinterpret_cast<bf16x8*>(&k_lds[key][dc]) =
            *reinterpret_cast<const bf16x8*>(Kg + (long)src * K.rs + dc);
        const bf16x8 vv = *reinterpret_cast<const bf16x8*>(
            Vg + (long)src * V.rs + dc);
#pragma unroll
        for (int j = 0; j < 8; ++j) vt_lds[dc + j][key] = vv[j];
      }
    }
    __syncthreads();

    // ---- S tile: 4 chunks of 16 keys
    // mfma(A=qfrag, B=kfrag): C row=(lane>>4)*4+r = Q row,
    // col=lane&15 = key -> row stats reduce within 16-lane groups.
    f32x4 s_chunk[KTILE / 16];
#pragma unroll
    for (int kc = 0; kc < KTILE / 16; ++kc) {
      f32x4 acc = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int f = 0; f < D / 32; ++f) {
        const bf16x8 kf = *reinterpret_cast<const bf16x8*>(
            &k_lds[kc * 16 + qrow_f][f * 32 + kslice * 8]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qfrag[f], kf, acc, 0, 0, 0);
      }
      s_chunk[kc] = acc;
    }

    // ---- mask + per-row max over this tile
    float tile_max[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) tile_max[r] = -INFINITY;
#pragma unroll
    for (int kc = 0; kc < KTILE / 16; ++kc) {
      const int key = kv0 + kc * 16 + ccol;
      const bool key_ok = key < limit_all;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = q0 + wid * 16 + crow0 + r;
        bool ok = key_ok && qrow < Sq;
        if (causal) ok = ok && key <= qrow + (Sk - Sq);
        float v = ok ? s_chunk[kc][r] * scale : -INFINITY;
        s_chunk[kc][r] = v;
        tile_max[r] = fmaxf(tile_max[r], v);
      }
    }
    // cross-lane max within the 16-lane col group
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        tile_max[r] = fmaxf(tile_max[r],
                            __shfl_xor(tile_max[r], off, 64));
    }

    // ---- online rescale + exp + row sums
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float m_new = fmaxf(m_run[r], tile_max[r]);
      alpha[r] = (m_run[r] > -INFINITY) ? __expf(m_run[r] - m_new) : 0.f;
      m_run[r] = m_new;
    }
    float row_sum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kc = 0; kc < KTILE / 16; ++kc) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = (s_chunk[kc][r] > -INFINITY && m_run[r] > -INFINITY)
            ? __expf(s_chunk[kc][r] - m_run[r]) : 0.f;
        s_chunk[kc][r] = p;
        row_sum[r] += p;
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        row_sum[r] += __shfl_xor(row_sum[r], off, 64);
      l_run[r] = l_run[r] * alpha[r] + row_sum[r];
    }

    // ---- rescale O accumulators by alpha (per row)
#pragma unroll
    for (int f = 0; f < D / 16; ++f)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[f][r] *= alpha[r];

    // ---- P -> LDS (C layout -> A-frag layout roundtrip)
#pragma unroll
    for (int kc = 0; kc < KTILE / 16; ++kc)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        p_lds[wid][crow0 + r][kc * 16 + ccol] = f2bf(s_chunk[kc][r]);
    // wave-local LDS write/read: no cross-wave sharing, so no barrier —
    // but LDS ops within a wave complete in order; lgkmcnt handled by
    // the compiler before the reads below.

    // ---- PV: o_acc[dchunk] += P[16q x 32k] @ V^T fragments
#pragma unroll
    for (int f = 0; f < D / 16; ++f) {
#pragma unroll
      for (int ks = 0; ks < KTILE / 32; ++ks) {
        const bf16x8 pf = *reinterpret_cast<const bf16x8*>(
            &p_lds[wid][qrow_f][ks * 32 + kslice * 8]);
        const bf16x8 vf = *reinterpret_cast<const bf16x8*>(
            &vt_lds[f * 16 + qrow_f][ks * 32 + kslice * 8]);
        o_acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pf, vf, o_acc[f], 0, 0, 0);
      }
    }
    __syncthreads();   // protect k_lds/vt_lds before next tile's staging
  }

  // ---- epilogue: divide by l, store 16 rows x D
#pragma unroll
  for (int f = 0; f < D / 16; ++f) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int qrow = q0 + wid * 16 + crow0 + r;
      if (qrow >= Sq) continue;
      const float inv = l_run[r] > 0.f ? 1.0f / l_run[r] : 0.f;
      O[(long)g * Sq * D + (long)qrow * D + f * 16 + ccol] =
          f2bf(o_acc[f][r] * inv);
    }
  }
}

}  // namespace

extern "C" void infomesh_attn_fused(
    const void* Q, const void* K, const void* V, const void* valid_len,
    void* O, int B, int nhq, int nhk, int Sq, int Sk, int D,
    long qb, long qh, long qr, long kb, long kh, long kr,
    long vb, long vh, long vr,
    int causal, float scale, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  dim3 grid((Sq + QTILE - 1) / QTILE, B * nhq), block(256);
  TensorView qv{(const bf16*)Q, qb, qh, qr};
  TensorView kv{(const bf16*)K, kb, kh, kr};
  TensorView vv{(const bf16*)V, vb, vh, vr};
#define CASE(DV)                                                          \
  case DV:                                                                \
    hipLaunchKernelGGL(attn_fused_kernel<DV>, grid, block, 0, s,          \
                       qv, kv, vv, (const int*)valid_len, (bf16*)O,       \
                       nhq, nhk, Sq, Sk, causal, scale);                  \
    break;
  switch (D) {
    CASE(32) CASE(64) CASE(96) CASE(128)
    default:
      break;  // unsupported D: wrapper validates
  }
#undef CASE
}
