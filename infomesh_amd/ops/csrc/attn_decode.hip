// Single-token decode attention with KV cache (summarizer decode path).
// Memory-bound KV read (guide Appendix B "Attention decode"): one block
// per (batch, head); waves stripe the sequence for QK^T + online scores
// in LDS, then stripe again for the P·V accumulation with lane=dim
// coalesced V reads. GQA via kv_head = head / (H / Hkv).
// Replaces: the reference's external llama.cpp/vLLM decode
// (infomesh/summarizer/engine.py:186-318).
#include "common.h"

#define DECODE_SMAX 8192

namespace {

// Q [B, H, D], Kc [B, Hkv, Smax, D], Vc same, lens [B], out [B, H, D].
__global__ __launch_bounds__(256) void attn_decode_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ Kc,
    const bf16* __restrict__ Vc, const int* __restrict__ lens,
    bf16* __restrict__ out, int H, int Hkv, int Smax, int D, float scale) {
  __shared__ float scratch[16];
  __shared__ float p_lds[DECODE_SMAX];
  __shared__ float o_lds[4][128];
  const int b = blockIdx.y, h = blockIdx.x;
  const int kvh = h / (H / Hkv);
  const int len = lens[b];
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  const bf16* q = Q + ((long)b * H + h) * D;
  const bf16* K = Kc + ((long)b * Hkv + kvh) * (long)Smax * D;
  const bf16* V = Vc + ((long)b * Hkv + kvh) * (long)Smax * D;

  // Phase 1: scores. Each lane owns one s per step (4 waves * 64 lanes).
  float local_max = -INFINITY;
  for (int s0 = 0; s0 < len; s0 += 256) {
    const int s = s0 + wid * 64 + lane;
    if (s < len) {
      float dot = 0.f;
      const bf16* kr = K + (long)s * D;
      for (int d = 0; d < D; d += 8) {
        bf16x8 kq = *reinterpret_cast<const bf16x8*>(kr + d);
        bf16x8 qq = *reinterpret_cast<const bf16x8*>(q + d);
#pragma unroll
        for (int j = 0; j < 8; ++j) dot += bf2f(kq[j]) * bf2f(qq[j]);
      }
      const float sc = dot * scale;
      p_lds[s] = sc;
      local_max = fmaxf(local_max, sc);
    }
  }
  const float mx = block_reduce_max(local_max, scratch);
  __syncthreads();
  float local_sum = 0.f;
  for (int s = threadIdx.x; s < len; s += 256) {
    const float p = __expf(p_lds[s] - mx);
    p_lds[s] = p;
    local_sum += p;
  }
  const float denom = block_reduce_sum(local_sum, scratch);
  const float inv = denom > 0.f ? 1.0f / denom : 0.f;

  // Phase 2: O = Σ p·V. Wave w stripes s; lane d and d+64 accumulate.
  float o0 = 0.f, o1 = 0.f;
  for (int s = wid; s < len; s += 4) {
    const float p = p_lds[s];
    const bf16* vr = V + (long)s * D;
    if (lane < D) o0 += p * bf2f(vr[lane]);
    if (lane + 64 < D) o1 += p * bf2f(vr[lane + 64]);
  }
  if (lane < D) o_lds[wid][lane] = o0;
  if (lane + 64 < D) o_lds[wid][lane + 64] = o1;
  __syncthreads();
  for (int d = threadIdx.x; d < D; d += 256)
    out[((long)b * H + h) * D + d] = f2bf(
        (o_lds[0][d] + o_lds[1][d] + o_lds[2][d] + o_lds[3][d]) * inv);
}

// Append one token's K/V [B, Hkv, D] into the cache at position pos[b].
__global__ void kv_append_kernel(const bf16* __restrict__ knew,
                                 const bf16* __restrict__ vnew,
                                 bf16* __restrict__ Kc, bf16* __restrict__ Vc,
                                 const int* __restrict__ pos,
                                 int Hkv, int Smax, int D) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = (long)gridDim.y * Hkv * D;  // gridDim.y == B
  (void)total;
  const int b = blockIdx.y;
  const long hd = idx;
  if (hd >= (long)Hkv * D) return;
  const int h = (int)(hd / D), d = (int)(hd % D);
  const int p = pos[b];
  Kc[(((long)b * Hkv + h) * Smax + p) * D + d] = knew[((long)b * Hkv + h) * D + d];
  Vc[(((long)b * Hkv + h) * Smax + p) * D + d] = vnew[((long)b * Hkv + h) * D + d];
}

// ---- flash-decoding split: sequence chunks scored by separate blocks
// (fills the chip at B=1), partials (m, l, o) combined by a 2nd kernel.

// partials: [B, H, NC, D+2] f32  (o..., m, l)
__global__ __launch_bounds__(256) void attn_decode_split_kernel(
    const bf16* __restrict__ Q, const bf16* __restrict__ Kc,
    const bf16* __restrict__ Vc, const int* __restrict__ lens,
    float* __restrict__ part, int H, int Hkv, int Smax, int D, int NC,
    float scale) {
  __shared__ float scratch[16];
  __shared__ float p_lds[2048];
  __shared__ float o_lds[4][128];
  const int c = blockIdx.x, h = blockIdx.y, b = blockIdx.z;
  const int kvh = h / (H / Hkv);
  const int len = lens[b];
  const int chunk = (len + NC - 1) / NC;
  const int s_begin = c * chunk;
  const int s_end = min(len, s_begin + chunk);
  float* pout = part + (((long)b * H + h) * NC + c) * (D + 2);
  const int lane = threadIdx.x & 63, wid = threadIdx.x >> 6;
  if (s_begin >= s_end) {
    for (int i = threadIdx.x; i < D; i += 256) pout[i] = 0.f;
    if (threadIdx.x == 0) { pout[D] = -INFINITY; pout[D + 1] = 0.f; }
    return;
  }
  const bf16* q = Q + ((long)b * H + h) * D;
  const bf16* K = Kc + ((long)b * Hkv + kvh) * (long)Smax * D;
  const bf16* V = Vc + ((long)b * Hkv + kvh) * (long)Smax * D;
  const int n = s_end - s_begin;     // <= 2048 (wrapper guarantees)
  float local_max = -INFINITY;
  for (int i = threadIdx.x; i < n; i += 256) {
    const int s = s_begin + i;
    float dot = 0.f;
    const bf16* kr = K + (long)s * D;
    for (int d = 0; d < D; d += 8) {
      bf16x8 kq = *reinterpret_cast<const bf16x8*>(kr + d);
      bf16x8 qq = *reinterpret_cast<const bf16x8*>(q + d);
#pragma unroll
      for (int j = 0; j < 8; ++j) dot += bf2f(kq[j]) * bf2f(qq[j]);
    }
    const float sc = dot * scale;
    p_lds[i] = sc;
    local_max = fmaxf(local_max, sc);
  }
  const float mx = block_reduce_max(local_max, scratch);
  __syncthreads();
  float local_sum = 0.f;
  for (int i = threadIdx.x; i < n; i += 256) {
    const float p = __expf(p_lds[i] - mx);
    p_lds[i] = p;
    local_sum += p;
  }
  const float lsum = block_reduce_sum(local_sum, scratch);
  float o0 = 0.f, o1 = 0.f;
  for (int i = wid; i < n; i += 4) {
    const float p = p_lds[i];
    const bf16* vr = V + (long)(s_begin + i) * D;
    if (lane < D) o0 += p * bf2f(vr[lane]);
    if (lane + 64 < D) o1 += p * bf2f(vr[lane + 64]);
  }
  if (lane < D) o_lds[wid][lane] = o0;
  if (lane + 64 < D) o_lds[wid][lane + 64] = o1;
  __syncthreads();
  for (int d = threadIdx.x; d < D; d += 256)
    pout[d] = o_lds[0][d] + o_lds[1][d] + o_lds[2][d] + o_lds[3][d];
  if (threadIdx.x == 0) { pout[D] = mx; pout[D + 1] = lsum; }
}

__global__ void attn_decode_combine_kernel(
    const float* __restrict__ part, bf16* __restrict__ out,
    int H, int D, int NC) {
  const int h = blockIdx.x, b = blockIdx.y;
  const float* p = part + (((long)b * H + h) * NC) * (D + 2);
  __shared__ float w[64];
  float m = -INFINITY;
  for (int c = 0; c < NC; ++c) m = fmaxf(m, p[(long)c * (D + 2) + D]);
  float l = 0.f;
  for (int c = threadIdx.x; c < NC; c += blockDim.x) {
    const float mi = p[(long)c * (D + 2) + D];
    w[c] = (mi > -INFINITY) ? __expf(mi - m) : 0.f;
    l += p[(long)c * (D + 2) + D + 1] * w[c];
  }
  __syncthreads();
  // reduce l across threads via lds (NC <= 64)
  __shared__ float lred[64];
  lred[threadIdx.x] = l;
  __syncthreads();
  if (threadIdx.x == 0) {
    float t = 0.f;
    for (int i = 0; i < blockDim.x; ++i) t += lred[i];
    lred[0] = t;
  }
  __syncthreads();
  const float inv = lred[0] > 0.f ? 1.0f / lred[0] : 0.f;
  for (int d = threadIdx.x; d < D; d += blockDim.x) {
    float acc = 0.f;
    for (int c = 0; c < NC; ++c)
      acc += p[(long)c * (D + 2) + d] * w[c];
    out[((long)b * H + h) * D + d] = f2bf(acc * inv);
  }
}

}  // namespace

extern "C" void infomesh_attn_decode(
    const void* Q, const void* Kc, const void* Vc, const void* lens,
    void* out, int B, int H, int Hkv, int Smax, int D, float scale,
    void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  dim3 grid(H, B), block(256);
  hipLaunchKernelGGL(attn_decode_kernel, grid, block, 0, s,
                     (const bf16*)Q, (const bf16*)Kc, (const bf16*)Vc,
                     (const int*)lens, (bf16*)out, H, Hkv, Smax, D, scale);
}

extern "C" void infomesh_attn_decode_split(
    const void* Q, const void* Kc, const void* Vc, const void* lens,
    void* part, void* out, int B, int H, int Hkv, int Smax, int D,
    int NC, float scale, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  dim3 g1(NC, H, B), b1(256);
  hipLaunchKernelGGL(attn_decode_split_kernel, g1, b1, 0, s,
                     (const bf16*)Q, (const bf16*)Kc, (const bf16*)Vc,
                     (const int*)lens, (float*)part, H, Hkv, Smax, D, NC,
                     scale);
  dim3 g2(H, B), b2(64);
  hipLaunchKernelGGL(attn_decode_combine_kernel, g2, b2, 0, s,
                     (const float*)part, (bf16*)out, H, D, NC);
}

extern "C" void infomesh_kv_append(
    const void* knew, const void* vnew, void* Kc, void* Vc, const void* pos,
    int B, int Hkv, int Smax, int D, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  const long hd = (long)Hkv * D;
  dim3 grid((unsigned)((hd + 255) / 256), B), block(256);
  hipLaunchKernelGGL(kv_append_kernel, grid, block, 0, s,
                     (const bf16*)knew, (const bf16*)vnew,
                     (bf16*)Kc, (bf16*)Vc, (const int*)pos, Hkv, Smax, D);
}
