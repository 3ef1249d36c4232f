// Skinny-M GEMV-class kernel for decode projections (M <= 16):
// C[g,m,n] = act(alpha * sum_k A[g,m,k] * B[g,n,k] + bias[n]).
//
// Guide Appendix "GEMV / M <= 16 decode weights": no LDS round trip —
// each WAVE owns one output column n, lanes stride K with 16-B loads
// (B rows coalesce per-wave; A rows are tiny and L1/L2-resident),
// wave-reduce per m, lane 0 stores. Weights stream once at HBM rate;
// the 128x128 MFMA tile kernel wastes 127/128 of its A tile here and
// runs ~10x slower at M=1 (measured in profiles/r01_decode.md).
// Replaces: the decode-time projection matmuls of the external LLM
// the reference calls over HTTP (infomesh/summarizer/engine.py:111-318)
// -- in-process Phi-3-shaped decode runs these at HBM rate.
#include "common.h"

namespace {

template <bool OUT_F32>
__global__ __launch_bounds__(256) void gemv_bf16_nt_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    void* __restrict__ C, const float* __restrict__ bias,
    int M, int N, int K, long strideA, long strideB, long strideC,
    int act, float alpha) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int n = blockIdx.x * 4 + wid;
  if (n >= N) return;
  const int g = blockIdx.y;
  const bf16* Ag = A + (long)g * strideA;
  const bf16* Brow = B + (long)g * strideB + (long)n * K;

  float acc[16];
#pragma unroll
  for (int m = 0; m < 16; ++m) acc[m] = 0.f;

#pragma unroll 2
  for (int k0 = lane * 8; k0 < K; k0 += 64 * 8) {
    const bf16x8 b8 = *reinterpret_cast<const bf16x8*>(Brow + k0);
    float bf[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) bf[j] = bf2f(b8[j]);
    for (int m = 0; m < M; ++m) {
      const bf16x8 a8 = *reinterpret_cast<const bf16x8*>(Ag + (long)m * K + k0);
      float d = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) d += bf2f(a8[j]) * bf[j];
      acc[m] += d;
    }
  }
  const float bv = bias ? bias[n] : 0.0f;
  for (int m = 0; m < M; ++m) {
    float v = wave_reduce_sum(acc[m]);
    if (lane == 0) {
      v = apply_act(alpha * v + bv, act);
      if (OUT_F32)
        reinterpret_cast<float*>(C)[(long)g * strideC + (long)m * N + n] = v;
      else
        reinterpret_cast<bf16*>(C)[(long)g * strideC + (long)m * N + n] =
            f2bf(v);
    }
  }
}

}  // namespace

extern "C" void infomesh_gemv_bf16_nt(
    const void* A, const void* B, void* C, const void* bias,
    int M, int N, int K, int batch,
    long strideA, long strideB, long strideC,
    int act, float alpha, int out_f32, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  dim3 grid((N + 3) / 4, batch), block(256);
  if (out_f32)
    hipLaunchKernelGGL(gemv_bf16_nt_kernel<true>, grid, block, 0, s,
                       (const bf16*)A, (const bf16*)B, C,
                       (const float*)bias, M, N, K, strideA, strideB,
                       strideC, act, alpha);
  else
    hipLaunchKernelGGL(gemv_bf16_nt_kernel<false>, grid, block, 0, s,
                       (const bf16*)A, (const bf16*)B, C,
                       (const float*)bias, M, N, K, strideA, strideB,
                       strideC, act, alpha);
}
