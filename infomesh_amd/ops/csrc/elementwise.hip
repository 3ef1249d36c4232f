// Fused elementwise kernels: bias+activation, SwiGLU (silu(gate)*up),
// residual add, RoPE apply (host-precomputed cos/sin tables — guide App. B),
// embedding gather (+scaled add), CLS/mean pooling with L2 norm, greedy
// argmax. All bf16 I/O vectorized 8-wide.
// Replaces: torch elementwise glue in the reference's model stacks
// (sentence-transformers internals behind infomesh/index/vector_store.py
// :104-118 and the external-LLM serving the reference calls over HTTP,
// infomesh/summarizer/engine.py:111-318).
#include "common.h"

namespace {

__global__ void bias_act_kernel(bf16* __restrict__ x,
                                const float* __restrict__ bias,
                                long rows, int N, int act) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long total = rows * (long)(N / 8);
  for (long v = idx; v < total; v += (long)gridDim.x * blockDim.x) {
    const long r = v / (N / 8);
    const int c = (int)(v % (N / 8)) * 8;
    bf16x8 xv = *reinterpret_cast<bf16x8*>(x + r * N + c);
#pragma unroll
    for (int j = 0; j < 8; ++j)
      xv[j] = f2bf(apply_act(bf2f(xv[j]) + (bias ? bias[c + j] : 0.f), act));
    *reinterpret_cast<bf16x8*>(x + r * N + c) = xv;
  }
}

// out = silu(gate) * up   (phi-3 MLP); gate/up [rows, N].
__global__ void silu_mul_kernel(const bf16* __restrict__ gate,
                                const bf16* __restrict__ up,
                                bf16* __restrict__ out, long n8) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (long v = idx; v < n8; v += (long)gridDim.x * blockDim.x) {
    bf16x8 g = *reinterpret_cast<const bf16x8*>(gate + v * 8);
    bf16x8 u = *reinterpret_cast<const bf16x8*>(up + v * 8);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gf = bf2f(g[j]);
      o[j] = f2bf(gf / (1.0f + __expf(-gf)) * bf2f(u[j]));
    }
    *reinterpret_cast<bf16x8*>(out + v * 8) = o;
  }
}

__global__ void add_kernel(const bf16* __restrict__ a,
                           const bf16* __restrict__ b,
                           bf16* __restrict__ out, long n8) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (long v = idx; v < n8; v += (long)gridDim.x * blockDim.x) {
    bf16x8 av = *reinterpret_cast<const bf16x8*>(a + v * 8);
    bf16x8 bv = *reinterpret_cast<const bf16x8*>(b + v * 8);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(bf2f(av[j]) + bf2f(bv[j]));
    *reinterpret_cast<bf16x8*>(out + v * 8) = o;
  }
}

// RoPE (NeoX/phi-3 half-rotation style): x [rows, H, D], rotate first
// rot_dim dims; cos/sin [max_pos, rot_dim/2] f32; pos[rows] int32.
__global__ void rope_kernel(bf16* __restrict__ x,
                            const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t,
                            const int* __restrict__ pos,
                            long rows, int H, int D, int rot) {
  const int half = rot / 2;
  const long total = rows * (long)H * half;
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (long v = idx; v < total; v += (long)gridDim.x * blockDim.x) {
    const long rh = v / half;
    const int d = (int)(v % half);
    const long r = rh / H;
    const int p = pos[r];
    bf16* base = x + rh * D;
    const float c = cos_t[(long)p * half + d], s = sin_t[(long)p * half + d];
    const float x0 = bf2f(base[d]), x1 = bf2f(base[d + half]);
    base[d] = f2bf(x0 * c - x1 * s);
    base[d + half] = f2bf(x0 * s + x1 * c);
  }
}

// Embedding gather: out[r,:] = table[ids[r],:] * scale  (bf16 table).
__global__ void gather_kernel(const bf16* __restrict__ table,
                              const int* __restrict__ ids,
                              bf16* __restrict__ out,
                              long rows, int H, float scale) {
  const long total = rows * (long)(H / 8);
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (long v = idx; v < total; v += (long)gridDim.x * blockDim.x) {
    const long r = v / (H / 8);
    const int c = (int)(v % (H / 8)) * 8;
    bf16x8 t = *reinterpret_cast<const bf16x8*>(
        table + (long)ids[r] * H + c);
    if (scale != 1.0f) {
#pragma unroll
      for (int j = 0; j < 8; ++j) t[j] = f2bf(bf2f(t[j]) * scale);
    }
    *reinterpret_cast<bf16x8*>(out + r * H + c) = t;
  }
}

// Pooling: mode 0 = CLS token (bge-style), 1 = masked mean. Input
// x [B,S,H] bf16, lens [B]; output f32 [B,H], L2-normalized when l2 != 0.
__global__ __launch_bounds__(256) void pool_kernel(
    const bf16* __restrict__ x, const int* __restrict__ lens,
    float* __restrict__ out, int S, int H, int mode, int l2) {
  __shared__ float scratch[16];
  const int b = blockIdx.x;
  const int len = lens ? lens[b] : S;
  float ssq = 0.f;
  for (int h = threadIdx.x; h < H; h += blockDim.x) {
    float v;
    if (mode == 0) {
      v = bf2f(x[((long)b * S) * H + h]);
    } else {
      float acc = 0.f;
      for (int t = 0; t < len; ++t)
        acc += bf2f(x[((long)b * S + t) * H + h]);
      v = acc / max(1, len);
    }
    out[(long)b * H + h] = v;
    ssq += v * v;
  }
  if (l2) {
    ssq = block_reduce_sum(ssq, scratch);
    const float inv = rsqrtf(fmaxf(ssq, 1e-12f));
    for (int h = threadIdx.x; h < H; h += blockDim.x)
      out[(long)b * H + h] *= inv;
  }
}

// Greedy argmax over logits [rows, V] f32 -> int32 [rows].
__global__ __launch_bounds__(256) void argmax_kernel(
    const float* __restrict__ logits, int* __restrict__ out, int V) {
  __shared__ float scratch[16];
  __shared__ int sidx;
  const long row = blockIdx.x;
  const float* lr = logits + row * V;
  float best = -INFINITY;
  int bi = 0;
  for (int j = threadIdx.x; j < V; j += blockDim.x)
    if (lr[j] > best) { best = lr[j]; bi = j; }
  const float gmax = block_reduce_max(best, scratch);
  if (threadIdx.x == 0) sidx = V;
  __syncthreads();
  if (best == gmax) atomicMin(&sidx, bi);  // lowest index on ties
  __syncthreads();
  if (threadIdx.x == 0) out[row] = sidx;
}

}  // namespace

static inline dim3 gs_grid(long work) {
  long blocks = (work + 255) / 256;
  if (blocks > 2048) blocks = 2048;  // grid-stride (guide G11)
  if (blocks < 1) blocks = 1;
  return dim3((unsigned)blocks);
}

extern "C" void infomesh_bias_act(void* x, const void* bias, long rows,
                                  int N, int act, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(bias_act_kernel, gs_grid(rows * (N / 8)), dim3(256), 0, s,
                     (bf16*)x, (const float*)bias, rows, N, act);
}

extern "C" void infomesh_silu_mul(const void* gate, const void* up, void* out,
                                  long n, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(silu_mul_kernel, gs_grid(n / 8), dim3(256), 0, s,
                     (const bf16*)gate, (const bf16*)up, (bf16*)out, n / 8);
}

extern "C" void infomesh_add(const void* a, const void* b, void* out, long n,
                             void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(add_kernel, gs_grid(n / 8), dim3(256), 0, s,
                     (const bf16*)a, (const bf16*)b, (bf16*)out, n / 8);
}

extern "C" void infomesh_rope(void* x, const void* cos_t, const void* sin_t,
                              const void* pos, long rows, int H, int D,
                              int rot, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(rope_kernel, gs_grid(rows * H * (rot / 2)), dim3(256), 0,
                     s, (bf16*)x, (const float*)cos_t, (const float*)sin_t,
                     (const int*)pos, rows, H, D, rot);
}

extern "C" void infomesh_gather(const void* table, const void* ids, void* out,
                                long rows, int H, float scale, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(gather_kernel, gs_grid(rows * (H / 8)), dim3(256), 0, s,
                     (const bf16*)table, (const int*)ids, (bf16*)out,
                     rows, H, scale);
}

extern "C" void infomesh_pool(const void* x, const void* lens, void* out,
                              int B, int S, int H, int mode, int l2,
                              void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(pool_kernel, dim3(B), dim3(256), 0, s,
                     (const bf16*)x, (const int*)lens, (float*)out,
                     S, H, mode, l2);
}

extern "C" void infomesh_argmax(const void* logits, void* out, long rows,
                                int V, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(argmax_kernel, dim3((unsigned)rows), dim3(256), 0, s,
                     (const float*)logits, (int*)out, V);
}
