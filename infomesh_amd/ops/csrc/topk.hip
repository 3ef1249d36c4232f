// Per-row top-k selection over [B, N] f32 score arrays (N up to tens of
// millions, K <= 1024) via 2-level radix select + compaction + one-block
// bitonic sort. Replaces: ChromaDB HNSW top-k (reference
// infomesh/index/vector_store.py:216-220) and the FTS5 ORDER BY bm25()
// LIMIT path (index/local_store.py:316-332) on the GPU shards.
//
// Passes (all streaming, ~3 reads of the score array):
//   1. hist1: 256-bin histogram of ordered-float top byte (LDS-staged).
//   2. select1: find byte bin containing the Kth value.
//   3. hist2: 256-bin histogram of byte 2 within that bin.
//   4. select2: 16-bit threshold prefix.
//   5. compact: gather (value, idx) — strictly-above candidates into a
//      reserved region (provably < K of them), threshold-prefix-equal
//      candidates until the cap (excess ties dropped: sound, they are
//      interchangeable at rank K).
//   6. sort: one block per row bitonic-sorts candidates, emits top-K.
#include "common.h"

#define TOPK_CAP 8192

namespace {

// 8 padded LDS histogram copies: a single 256-bin LDS histogram is
// atomic-serialization bound (~2.1 TB/s measured). Copy c starts at
// c*264 ints — 264 % 64 = 8, so the 8 copies of any bin land in 8
// DISTINCT LDS banks, and lane L writing copy L&7 spreads a wave's
// atomics across banks even when bins collide.
__global__ __launch_bounds__(256) void hist1_kernel(
    const float* __restrict__ scores, unsigned* __restrict__ hist,
    long N) {
  constexpr int NC = 8, STR = 264;
  __shared__ unsigned lh[NC * STR];
  const int b = blockIdx.y;
  for (int i = threadIdx.x; i < NC * STR; i += blockDim.x) lh[i] = 0;
  __syncthreads();
  unsigned* my = lh + (threadIdx.x & (NC - 1)) * STR;
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long step = (long)gridDim.x * blockDim.x;
  const float* row = scores + (long)b * N;
  const long n4 = N / 4;
  for (long i = start; i < n4; i += step) {
    const float4 v = reinterpret_cast<const float4*>(row)[i];
    atomicAdd(&my[float_to_ordered(v.x) >> 24], 1u);
    atomicAdd(&my[float_to_ordered(v.y) >> 24], 1u);
    atomicAdd(&my[float_to_ordered(v.z) >> 24], 1u);
    atomicAdd(&my[float_to_ordered(v.w) >> 24], 1u);
  }
  for (long i = n4 * 4 + start; i < N; i += step)
    atomicAdd(&my[float_to_ordered(row[i]) >> 24], 1u);
  __syncthreads();
  for (int i = threadIdx.x; i < 256; i += blockDim.x) {
    unsigned sum = 0;
#pragma unroll
    for (int c = 0; c < NC; ++c) sum += lh[c * STR + i];
    if (sum) atomicAdd(&hist[(long)b * 256 + i], sum);
  }
}

__global__ void select1_kernel(const unsigned* __restrict__ hist,
                               unsigned* __restrict__ bin1,
                               unsigned* __restrict__ chi1, int K) {
  const int b = blockIdx.x;
  if (threadIdx.x != 0) return;
  unsigned cum = 0;
  for (int i = 255; i >= 0; --i) {
    unsigned c = hist[(long)b * 256 + i];
    if (cum + c >= (unsigned)K || i == 0) {
      bin1[b] = (unsigned)i;
      chi1[b] = cum;
      return;
    }
    cum += c;
  }
}

__global__ __launch_bounds__(256) void hist2_kernel(
    const float* __restrict__ scores, const unsigned* __restrict__ bin1,
    unsigned* __restrict__ hist2, long N) {
  __shared__ unsigned lh[256];
  const int b = blockIdx.y;
  const unsigned b1 = bin1[b];
  for (int i = threadIdx.x; i < 256; i += blockDim.x) lh[i] = 0;
  __syncthreads();
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long step = (long)gridDim.x * blockDim.x;
  const float* row = scores + (long)b * N;
  const long n4 = N / 4;
  for (long i = start; i < n4; i += step) {
    const float4 v = reinterpret_cast<const float4*>(row)[i];
    const float f[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const unsigned o = float_to_ordered(f[j]);
      if ((o >> 24) == b1) atomicAdd(&lh[(o >> 16) & 255], 1u);
    }
  }
  for (long i = n4 * 4 + start; i < N; i += step) {
    const unsigned o = float_to_ordered(row[i]);
    if ((o >> 24) == b1) atomicAdd(&lh[(o >> 16) & 255], 1u);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < 256; i += blockDim.x)
    if (lh[i]) atomicAdd(&hist2[(long)b * 256 + i], lh[i]);
}

__global__ void select2_kernel(const unsigned* __restrict__ hist2,
                               const unsigned* __restrict__ bin1,
                               const unsigned* __restrict__ chi1,
                               unsigned* __restrict__ thresh16, int K) {
  const int b = blockIdx.x;
  if (threadIdx.x != 0) return;
  unsigned cum = chi1[b];
  for (int i = 255; i >= 0; --i) {
    unsigned c = hist2[(long)b * 256 + i];
    if (cum + c >= (unsigned)K || i == 0) {
      thresh16[b] = (bin1[b] << 8) | (unsigned)i;
      return;
    }
    cum += c;
  }
}

// Candidate layout per row (TOPK_CAP slots):
//   [0, HI_RES)        strictly-above-threshold candidates (the exact
//                      select guarantees < K <= 1024 of them; the
//                      sampled threshold targets ~3K with tail slack,
//                      hence the 4096 reserve)
//   [HI_RES, TOPK_CAP) threshold-PREFIX-EQUAL candidates, kept until
//                      the region fills. Prefix-equal candidates are
//                      interchangeable at rank K (any K of them is a
//                      valid top-k), so DROPPING the excess is sound:
//                      massively tied planes (BM25 over tiny-vocab
//                      corpora) no longer overflow. The final bitonic
//                      sorts by the full 32-bit ordered value, so
//                      prefix-equal candidates still order exactly.
//                      The overflow flag now only fires when the
//                      strictly-above region overflows — impossible
//                      for the exact path, possible for the sampled
//                      threshold (checked loudly by the wrapper).
#define HI_RES 4096

__global__ __launch_bounds__(256) void compact_kernel(
    const float* __restrict__ scores, const unsigned* __restrict__ thresh16,
    unsigned long long* __restrict__ cand, unsigned* __restrict__ cnt,
    unsigned* __restrict__ cnt_eq, unsigned* __restrict__ overflow,
    long N) {
  const int b = blockIdx.y;
  const unsigned t16 = thresh16[b];
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long step = (long)gridDim.x * blockDim.x;
  const float* row = scores + (long)b * N;
  unsigned long long* crow = cand + (long)b * TOPK_CAP;
  const long n4 = N / 4;
  for (long i = start; i < n4; i += step) {
    const float4 v = reinterpret_cast<const float4*>(row)[i];
    const float f[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const unsigned o = float_to_ordered(f[j]);
      const unsigned p16 = o >> 16;
      // Ascending sort key: ~ordered in high bits (desc value),
      // raw idx in low bits (ties -> smaller idx first).
      const unsigned long long key =
          ((unsigned long long)(~o) << 32) | (unsigned)(i * 4 + j);
      if (p16 > t16) {
        const unsigned pos = atomicAdd(&cnt[b], 1u);
        if (pos < HI_RES) crow[pos] = key;
        else *overflow = 1u;
      } else if (p16 == t16) {
        const unsigned pos = atomicAdd(&cnt_eq[b], 1u);
        if (pos < TOPK_CAP - HI_RES) crow[HI_RES + pos] = key;
      }
    }
  }
  for (long i = n4 * 4 + start; i < N; i += step) {
    const unsigned o = float_to_ordered(row[i]);
    const unsigned p16 = o >> 16;
    const unsigned long long key =
        ((unsigned long long)(~o) << 32) | (unsigned)i;
    if (p16 > t16) {
      const unsigned pos = atomicAdd(&cnt[b], 1u);
      if (pos < HI_RES) crow[pos] = key;
      else *overflow = 1u;
    } else if (p16 == t16) {
      const unsigned pos = atomicAdd(&cnt_eq[b], 1u);
      if (pos < TOPK_CAP - HI_RES) crow[HI_RES + pos] = key;
    }
  }
}

// ---- sampling select: gather a strided sample into the (pre-compact)
// candidate buffer, run the cheap exact 2-level radix select ON THE
// SAMPLE (3 passes over ~32 KB/row instead of the full array), then
// threshold the full array with the resulting 16-bit prefix. Saves two
// full passes; the wrapper verifies K <= cnt <= CAP afterwards (loud,
// never silent). Assumes a near-continuous distribution — callers keep
// heavily-tied planes (BM25 tails) on the exact path.
__global__ void sample_gather_kernel(const float* __restrict__ scores,
                                     float* __restrict__ sample,
                                     long N, int stride) {
  const int b = blockIdx.y;
  const float* row = scores + (long)b * N;
  float* out = sample + (long)b * TOPK_CAP;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < TOPK_CAP;
       i += gridDim.x * blockDim.x) {
    const long idx = (long)i * stride + (b % stride);
    out[i] = idx < N ? row[idx] : -INFINITY;
  }
}

__global__ __launch_bounds__(256) void sort_emit_kernel(
    unsigned long long* __restrict__ cand, const unsigned* __restrict__ cnt,
    const unsigned* __restrict__ cnt_eq,
    float* __restrict__ out_vals, int* __restrict__ out_idx, int K) {
  __shared__ unsigned long long d[TOPK_CAP];
  const int b = blockIdx.x;
  const int nhi = min(cnt[b], (unsigned)HI_RES);
  const int neq = min(cnt_eq[b], (unsigned)(TOPK_CAP - HI_RES));
  const int n = nhi + neq;
  // Bitonic network size: next pow2 of the actual candidate count —
  // typical rows carry ~K+epsilon candidates, so this cuts the sort
  // from CAP=8192 to 128/256 most of the time.
  int n2 = 64;
  while (n2 < n) n2 <<= 1;
  const unsigned long long* crow = cand + (long)b * TOPK_CAP;
  for (int i = threadIdx.x; i < n2; i += blockDim.x)
    d[i] = (i < nhi) ? crow[i]
         : (i < n) ? crow[HI_RES + (i - nhi)]
         : ~0ULL;  // pad = worst
  __syncthreads();
  for (int k = 2; k <= n2; k <<= 1) {
    for (int j = k >> 1; j > 0; j >>= 1) {
      for (int i = threadIdx.x; i < n2; i += blockDim.x) {
        const int ixj = i ^ j;
        if (ixj > i) {
          const bool up = ((i & k) == 0);
          const unsigned long long a = d[i], c = d[ixj];
          if ((a > c) == up) { d[i] = c; d[ixj] = a; }
        }
      }
      __syncthreads();
    }
  }
  for (int i = threadIdx.x; i < K; i += blockDim.x) {
    if (i < n) {
      const unsigned long long v = d[i];
      out_vals[(long)b * K + i] = ordered_to_float(~(unsigned)(v >> 32));
      out_idx[(long)b * K + i] = (int)(v & 0xffffffffu);
    } else {
      out_vals[(long)b * K + i] = -INFINITY;
      out_idx[(long)b * K + i] = -1;
    }
  }
}

}  // namespace

// Workspace layout (u32 units), provided zero-initialized by the wrapper:
//   hist1 [B*256] | hist2 [B*256] | bin1 [B] | chi1 [B] | thresh16 [B]
//   | cnt [B] | cnt_eq [B] | overflow [1] | cand (u64) [B*TOPK_CAP]
extern "C" long infomesh_topk_workspace_u32(int B) {
  long u = (long)B * 256 * 2 + (long)B * 5 + 1;
  u = (u + 1) & ~1L;  // align cand to 8 bytes
  return u + (long)B * TOPK_CAP * 2;
}

// ext_hist1: non-null = a producer-fused [B*256] top-byte histogram of
// the FULL score rows (exact counts, zeros included); pass 1 is skipped
// and select1 reads it directly. Incompatible with sampled (the sampled
// path histograms its own gathered sample).
extern "C" void infomesh_topk(const void* scores, void* workspace,
                              void* out_vals, void* out_idx,
                              int B, long N, int K, int sampled,
                              const void* ext_hist1, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  unsigned* ws = reinterpret_cast<unsigned*>(workspace);
  unsigned* hist1 = ws;
  unsigned* hist2 = hist1 + (long)B * 256;
  unsigned* bin1 = hist2 + (long)B * 256;
  unsigned* chi1 = bin1 + B;
  unsigned* thresh16 = chi1 + B;
  unsigned* cnt = thresh16 + B;
  unsigned* cnt_eq = cnt + B;
  unsigned* overflow = cnt_eq + B;
  long off = (long)B * 256 * 2 + (long)B * 5 + 1;
  off = (off + 1) & ~1L;
  auto* cand = reinterpret_cast<unsigned long long*>(ws + off);

  int chunks = (int)min((N + 256 * 64 - 1) / (256 * 64), (long)1024);
  if (chunks < 1) chunks = 1;
  dim3 g1(chunks, B), blk(256);
  if (sampled) {
    const int stride = (int)((N + TOPK_CAP - 1) / TOPK_CAP) < 256
        ? 256 : (int)((N + TOPK_CAP - 1) / TOPK_CAP);
    // Sample rank: the count above the Kp-th largest of the sample is
    // ~Gamma(Kp) x stride. Kp >= 8 (and mean >= 3K) keeps the
    // undershoot tail (< K) negligible for continuous scores while the
    // mean stays far from the CAP (the extra candidates inflate the
    // final bitonic sort, which is why sampling only pays at large N).
    int Kp = (int)((3L * K + stride - 1) / stride);
    if (Kp < 8) Kp = 8;
    float* sample = reinterpret_cast<float*>(cand);  // pre-compact reuse
    hipLaunchKernelGGL(sample_gather_kernel, dim3(32, B), blk, 0, s,
                       (const float*)scores, sample, N, stride);
    dim3 gs(1, B);
    hipLaunchKernelGGL(hist1_kernel, gs, blk, 0, s,
                       sample, hist1, (long)TOPK_CAP);
    hipLaunchKernelGGL(select1_kernel, dim3(B), dim3(64), 0, s,
                       hist1, bin1, chi1, Kp);
    hipLaunchKernelGGL(hist2_kernel, gs, blk, 0, s,
                       sample, bin1, hist2, (long)TOPK_CAP);
    hipLaunchKernelGGL(select2_kernel, dim3(B), dim3(64), 0, s,
                       hist2, bin1, chi1, thresh16, Kp);
    hipLaunchKernelGGL(compact_kernel, g1, blk, 0, s,
                       (const float*)scores, thresh16, cand, cnt,
                       cnt_eq, overflow, N);
  } else {
    const unsigned* h1 = hist1;
    if (ext_hist1) {
      h1 = (const unsigned*)ext_hist1;   // producer-fused pass 1
    } else {
      hipLaunchKernelGGL(hist1_kernel, g1, blk, 0, s,
                         (const float*)scores, hist1, N);
    }
    hipLaunchKernelGGL(select1_kernel, dim3(B), dim3(64), 0, s,
                       h1, bin1, chi1, K);
    hipLaunchKernelGGL(hist2_kernel, g1, blk, 0, s,
                       (const float*)scores, bin1, hist2, N);
    hipLaunchKernelGGL(select2_kernel, dim3(B), dim3(64), 0, s,
                       hist2, bin1, chi1, thresh16, K);
    hipLaunchKernelGGL(compact_kernel, g1, blk, 0, s,
                       (const float*)scores, thresh16, cand, cnt,
                       cnt_eq, overflow, N);
  }
  hipLaunchKernelGGL(sort_emit_kernel, dim3(B), blk, 0, s,
                     cand, cnt, cnt_eq, (float*)out_vals, (int*)out_idx,
                     K);
}
