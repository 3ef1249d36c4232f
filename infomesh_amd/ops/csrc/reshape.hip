// Head-layout shuffle kernels: fused QKV split (+optional RoPE) and
// head merge. These replace per-layer torch permute+contiguous copies
// in the encoder/reranker/decoder forward paths (4-6 HBM round trips
// per layer become 2 fused passes), plus a strided SwiGLU that reads
// gate/up halves in place.
// Replaces: torch permute/contiguous head-reshape glue inside the
// reference's model stacks (sentence-transformers internals behind
// infomesh/index/vector_store.py:104-157).
#include "common.h"

namespace {

// qkv [B, S, (nh+2*nkv)*d] bf16 (row-major fused projection output)
//  -> q [B*nh, S, d], k [B*nkv, S, d], vt [B*nkv, d, S]   (all bf16)
// RoPE (NeoX half-rotation) applied to q and k when cos/sin != null;
// pos[b*S + s] gives the rotary position of each token row.
__global__ void qkv_split_kernel(
    const bf16* __restrict__ qkv, bf16* __restrict__ q,
    bf16* __restrict__ k, bf16* __restrict__ vt,
    const float* __restrict__ cos_t, const float* __restrict__ sin_t,
    const int* __restrict__ pos,
    int B, int S, int nh, int nkv, int d, int rot) {
  const int half = rot / 2;
  const long total = (long)B * S * (nh + nkv) * (d / 2);
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const int dqkv = (nh + 2 * nkv) * d;
  for (long v = idx; v < total; v += (long)gridDim.x * blockDim.x) {
    // each work item handles a (row, head, d-pair): pair j = (j, j+half)
    // for roped heads, plain (2j, 2j+1) copies otherwise.
    const int j = (int)(v % (d / 2));
    long rest = v / (d / 2);
    const int h = (int)(rest % (nh + nkv));
    const long row = rest / (nh + nkv);           // b*S + s
    const long b = row / S, s = row % S;
    const bf16* src = qkv + row * dqkv + (long)h * d;
    {
      // Q or K head: apply rope on the (j, j+half) pair
      float x0, x1;
      if (cos_t != nullptr && j < half) {
        const int p = pos[row];
        const float c = cos_t[(long)p * half + j];
        const float sn = sin_t[(long)p * half + j];
        const float a0 = bf2f(src[j]), a1 = bf2f(src[j + half]);
        x0 = a0 * c - a1 * sn;
        x1 = a0 * sn + a1 * c;
      } else {
        // no rope (encoder) -> copy two adjacent elems for coalescing
        x0 = bf2f(src[2 * j]);
        x1 = bf2f(src[2 * j + 1]);
      }
      const int j0 = (cos_t != nullptr && j < half) ? j : 2 * j;
      const int j1 = (cos_t != nullptr && j < half) ? j + half : 2 * j + 1;
      if (h < nh) {
        bf16* dst = q + ((b * nh + h) * (long)S + s) * d;
        dst[j0] = f2bf(x0);
        dst[j1] = f2bf(x1);
      } else {
        bf16* dst = k + ((b * nkv + (h - nh)) * (long)S + s) * d;
        dst[j0] = f2bf(x0);
        dst[j1] = f2bf(x1);
      }
    }
    // V heads are handled by the LDS-tiled vt_transpose_kernel below
    // (scatter stores here would be fully uncoalesced).
  }
}

// V slice of qkv [B,S,(nh+2nkv)*d] -> vt [B*nkv, d, S] via a classic
// 32x32 LDS-tiled transpose (coalesced along d on load, along s on
// store). Grid: (ceil(S/32), ceil(d/32), B*nkv); block 32x8.
__global__ __launch_bounds__(256) void vt_transpose_kernel(
    const bf16* __restrict__ qkv, bf16* __restrict__ vt,
    int B, int S, int nh, int nkv, int d) {
  __shared__ bf16 tile[32][33];
  const int g = blockIdx.z;                  // b*nkv + hv
  const int b = g / nkv, hv = g % nkv;
  const int s0 = blockIdx.x * 32, d0 = blockIdx.y * 32;
  const int dqkv = (nh + 2 * nkv) * d;
  const long voff = (long)(nh + nkv + hv) * d;
  const int tx = threadIdx.x & 31, ty = threadIdx.x >> 5;   // 32x8
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int s = s0 + ty + 8 * i;
    const int dd = d0 + tx;
    if (s < S && dd < d)
      tile[ty + 8 * i][tx] =
          qkv[((long)b * S + s) * dqkv + voff + dd];
  }
  __syncthreads();
  bf16* out = vt + (long)g * d * S;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int dd = d0 + ty + 8 * i;
    const int s = s0 + tx;
    if (s < S && dd < d)
      out[(long)dd * S + s] = tile[tx][ty + 8 * i];
  }
}

// ctx [B*nh, S, d] -> merged [B*S, nh*d]
__global__ void merge_heads_kernel(
    const bf16* __restrict__ ctx, bf16* __restrict__ out,
    int B, int S, int nh, int d) {
  const long total = (long)B * nh * S * (d / 8);
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (long v = idx; v < total; v += (long)gridDim.x * blockDim.x) {
    const int j = (int)(v % (d / 8)) * 8;
    long rest = v / (d / 8);
    const long s = rest % S;
    rest /= S;
    const int h = (int)(rest % nh);
    const long b = rest / nh;
    bf16x8 val = *reinterpret_cast<const bf16x8*>(
        ctx + ((b * nh + h) * (long)S + s) * d + j);
    *reinterpret_cast<bf16x8*>(
        out + (b * S + s) * (long)(nh * d) + (long)h * d + j) = val;
  }
}

// gu [rows, 2F] -> out [rows, F] = silu(gu[:, :F]) * gu[:, F:]
__global__ void silu_mul_fused_kernel(const bf16* __restrict__ gu,
                                      bf16* __restrict__ out,
                                      long rows, int F) {
  const long total = rows * (long)(F / 8);
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (long v = idx; v < total; v += (long)gridDim.x * blockDim.x) {
    const long r = v / (F / 8);
    const int c = (int)(v % (F / 8)) * 8;
    bf16x8 g = *reinterpret_cast<const bf16x8*>(gu + r * 2 * F + c);
    bf16x8 u = *reinterpret_cast<const bf16x8*>(gu + r * 2 * F + F + c);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float gf = bf2f(g[j]);
      o[j] = f2bf(gf / (1.0f + __expf(-gf)) * bf2f(u[j]));
    }
    *reinterpret_cast<bf16x8*>(out + r * F + c) = o;
  }
}

}  // namespace

static inline dim3 gs(long work) {
  long blocks = (work + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  if (blocks < 1) blocks = 1;
  return dim3((unsigned)blocks);
}

extern "C" void infomesh_qkv_split(
    const void* qkv, void* q, void* k, void* vt,
    const void* cos_t, const void* sin_t, const void* pos,
    int B, int S, int nh, int nkv, int d, int rot, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  const long total = (long)B * S * (nh + nkv) * (d / 2);
  hipLaunchKernelGGL(qkv_split_kernel, gs(total), dim3(256), 0, s,
                     (const bf16*)qkv, (bf16*)q, (bf16*)k, (bf16*)vt,
                     (const float*)cos_t, (const float*)sin_t,
                     (const int*)pos, B, S, nh, nkv, d, rot);
  if (vt != nullptr) {
    dim3 grid((S + 31) / 32, (d + 31) / 32, B * nkv);
    hipLaunchKernelGGL(vt_transpose_kernel, grid, dim3(256), 0, s,
                       (const bf16*)qkv, (bf16*)vt, B, S, nh, nkv, d);
  }
}

extern "C" void infomesh_merge_heads(const void* ctx, void* out,
                                     int B, int S, int nh, int d,
                                     void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(merge_heads_kernel,
                     gs((long)B * nh * S * (d / 8)), dim3(256), 0, s,
                     (const bf16*)ctx, (bf16*)out, B, S, nh, d);
}

extern "C" void infomesh_silu_mul_fused(const void* gu, void* out,
                                        long rows, int F, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(silu_mul_fused_kernel, gs(rows * (F / 8)), dim3(256),
                     0, s, (const bf16*)gu, (bf16*)out, rows, F);
}
