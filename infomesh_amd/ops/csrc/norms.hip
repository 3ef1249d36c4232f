// Fused normalization kernels (gfx950): LayerNorm and RMSNorm over the
// last dim, with optional fused residual-add, bf16 I/O vectorized as
// short8 (guide G13: scalar bf16 loads are ~2x slower).
// Replaces: the reference's torch layernorms inside sentence-transformers
// (infomesh/index/vector_store.py:104-118) — encoder/reranker/summarizer
// norm layers run through these.
#include "common.h"

namespace {

// One block per row; H multiple of 8 (wrapper asserts).
template <bool RMS, bool RESIDUAL>
__global__ __launch_bounds__(256) void norm_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ res,
    bf16* __restrict__ out, bf16* __restrict__ res_out,
    const bf16* __restrict__ gamma, const bf16* __restrict__ beta,
    int H, float eps) {
  __shared__ float scratch[16];
  const long row = blockIdx.x;
  const bf16* xr = x + row * H;
  const bf16* rr = RESIDUAL ? res + row * H : nullptr;

  // Pass 1: accumulate sums, keeping the (residual-added) values in regs
  // for rows up to 8*256*MAXV elements; fall back to re-read for huge H.
  constexpr int MAXV = 8;  // supports H <= 16384 fully register-resident
  f32x4 keep[2 * MAXV];
  const int nvec = H / 8;
  float sum = 0.f, sumsq = 0.f;
  for (int v = threadIdx.x, it = 0; v < nvec; v += blockDim.x, ++it) {
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(xr + v * 8);
    float f[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) f[j] = bf2f(xv[j]);
    if (RESIDUAL) {
      bf16x8 rv = *reinterpret_cast<const bf16x8*>(rr + v * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) f[j] += bf2f(rv[j]);
      if (res_out) {
        bf16x8 ov;
#pragma unroll
        for (int j = 0; j < 8; ++j) ov[j] = f2bf(f[j]);
        *reinterpret_cast<bf16x8*>(res_out + row * H + v * 8) = ov;
      }
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) { sum += f[j]; sumsq += f[j] * f[j]; }
    if (it < MAXV) {
      keep[2 * it] = f32x4{f[0], f[1], f[2], f[3]};
      keep[2 * it + 1] = f32x4{f[4], f[5], f[6], f[7]};
    }
  }
  float mean = 0.f;
  if (!RMS) {
    mean = block_reduce_sum(sum, scratch) / H;
    __syncthreads();
  }
  const float ms = block_reduce_sum(sumsq, scratch) / H;
  const float inv = rsqrtf((RMS ? ms : ms - mean * mean) + eps);

  for (int v = threadIdx.x, it = 0; v < nvec; v += blockDim.x, ++it) {
    float f[8];
    if (it < MAXV) {
#pragma unroll
      for (int j = 0; j < 4; ++j) f[j] = keep[2 * it][j];
#pragma unroll
      for (int j = 0; j < 4; ++j) f[4 + j] = keep[2 * it + 1][j];
    } else {
      bf16x8 xv = *reinterpret_cast<const bf16x8*>(xr + v * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) f[j] = bf2f(xv[j]);
      if (RESIDUAL) {
        bf16x8 rv = *reinterpret_cast<const bf16x8*>(rr + v * 8);
#pragma unroll
        for (int j = 0; j < 8; ++j) f[j] += bf2f(rv[j]);
      }
    }
    bf16x8 gv = *reinterpret_cast<const bf16x8*>(gamma + v * 8);
    bf16x8 ov;
    if (!RMS && beta) {
      bf16x8 bv = *reinterpret_cast<const bf16x8*>(beta + v * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        ov[j] = f2bf((f[j] - mean) * inv * bf2f(gv[j]) + bf2f(bv[j]));
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        ov[j] = f2bf((f[j] - (RMS ? 0.f : mean)) * inv * bf2f(gv[j]));
    }
    *reinterpret_cast<bf16x8*>(out + row * H + v * 8) = ov;
  }
}

// Wave-per-row variant for small H (<= 2048): 4 rows per 256-thread
// block, reductions stay inside each wave (no LDS round trip). The
// one-row-per-block kernel above idles (256 - H/8) threads at H=384.
template <bool RMS, bool RESIDUAL>
__global__ __launch_bounds__(256) void norm_rowwave_kernel(
    const bf16* __restrict__ x, const bf16* __restrict__ res,
    bf16* __restrict__ out, bf16* __restrict__ res_out,
    const bf16* __restrict__ gamma, const bf16* __restrict__ beta,
    long rows, int H, float eps) {
  const long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= rows) return;
  const int lane = threadIdx.x & 63;
  const bf16* xr = x + row * H;
  const bf16* rr = RESIDUAL ? res + row * H : nullptr;
  const int nvec = H / 8;
  float f[8 * 4];  // up to H=2048: nvec<=256 -> <=4 vecs/lane
  float sum = 0.f, sumsq = 0.f;
  int it = 0;
  for (int v = lane; v < nvec; v += 64, ++it) {
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(xr + v * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) f[it * 8 + j] = bf2f(xv[j]);
    if (RESIDUAL) {
      bf16x8 rv = *reinterpret_cast<const bf16x8*>(rr + v * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) f[it * 8 + j] += bf2f(rv[j]);
      if (res_out) {
        bf16x8 ov;
#pragma unroll
        for (int j = 0; j < 8; ++j) ov[j] = f2bf(f[it * 8 + j]);
        *reinterpret_cast<bf16x8*>(res_out + row * H + v * 8) = ov;
      }
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      sum += f[it * 8 + j];
      sumsq += f[it * 8 + j] * f[it * 8 + j];
    }
  }
  float mean = 0.f;
  if (!RMS) mean = wave_reduce_sum(sum) / H;
  const float ms = wave_reduce_sum(sumsq) / H;
  const float inv = rsqrtf((RMS ? ms : ms - mean * mean) + eps);
  it = 0;
  for (int v = lane; v < nvec; v += 64, ++it) {
    bf16x8 gv = *reinterpret_cast<const bf16x8*>(gamma + v * 8);
    bf16x8 ov;
    if (!RMS && beta) {
      bf16x8 bv = *reinterpret_cast<const bf16x8*>(beta + v * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        ov[j] = f2bf((f[it * 8 + j] - mean) * inv * bf2f(gv[j])
                     + bf2f(bv[j]));
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j)
        ov[j] = f2bf((f[it * 8 + j] - (RMS ? 0.f : mean)) * inv
                     * bf2f(gv[j]));
    }
    *reinterpret_cast<bf16x8*>(out + row * H + v * 8) = ov;
  }
}

}  // namespace

#define NORM_LAUNCH(RMS, RES, ...)                                        \
  do {                                                                    \
    if (H <= 2048) {                                                      \
      dim3 g((unsigned)((rows + 3) / 4));                                 \
      hipLaunchKernelGGL((norm_rowwave_kernel<RMS, RES>), g, dim3(256),   \
                         0, s, __VA_ARGS__, rows, H, eps);                \
    } else {                                                              \
      dim3 g((unsigned)rows);                                             \
      hipLaunchKernelGGL((norm_kernel<RMS, RES>), g, dim3(256), 0, s,     \
                         __VA_ARGS__, H, eps);                            \
    }                                                                     \
  } while (0)

extern "C" void infomesh_layernorm(
    const void* x, const void* residual, void* out, void* res_out,
    const void* gamma, const void* beta,
    long rows, int H, float eps, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  if (residual)
    NORM_LAUNCH(false, true, (const bf16*)x, (const bf16*)residual,
                (bf16*)out, (bf16*)res_out, (const bf16*)gamma,
                (const bf16*)beta);
  else
    NORM_LAUNCH(false, false, (const bf16*)x, nullptr, (bf16*)out,
                nullptr, (const bf16*)gamma, (const bf16*)beta);
}

extern "C" void infomesh_rmsnorm(
    const void* x, const void* residual, void* out, void* res_out,
    const void* gamma, long rows, int H, float eps, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  if (residual)
    NORM_LAUNCH(true, true, (const bf16*)x, (const bf16*)residual,
                (bf16*)out, (bf16*)res_out, (const bf16*)gamma, nullptr);
  else
    NORM_LAUNCH(true, false, (const bf16*)x, nullptr, (bf16*)out,
                nullptr, (const bf16*)gamma, nullptr);
}
