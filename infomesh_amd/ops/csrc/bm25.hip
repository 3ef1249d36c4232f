// GPU BM25 scorer over per-segment CSR posting lists in HBM — v3:
// doc-block LDS accumulation FUSED with per-block top-k selection.
//
// Replaces: SQLite FTS5 `MATCH ... ORDER BY bm25()` (reference
// infomesh/index/local_store.py:316-332) for the GPU shards; the CPU
// FTS5 path remains for the CPU-plumbing config.
//
// score(q, d) = Σ_t idf(t) · tf·(k1+1) / (tf + k1·(1−b+b·dl/avgdl))
//
// Evolution (profiles/):
//   v1  global atomic scatter-add into a [B, N] score matrix: ~982 us
//       kernel + ~0.6 ms zero-fill at 1.25M docs / B=128.
//   v2  doc-block LDS accumulation + bounds pre-pass + query-major
//       grid (L2 posting reuse): 405 us kernel — but the [B, N] score
//       write (5.1 GB at 10M docs) plus the 3-pass global top-k
//       (15 GB reads) still bounded the leg (6.4 ms at 10M).
//   v3  the block's scores never leave LDS: a 2-level in-LDS radix
//       select emits each (query, doc-block)'s top-k candidates
//       (~100 x 8 B instead of 32 KB of scores), and one small global
//       top-k over [B, nblocks*k] candidates finishes. Global traffic
//       drops to postings + O(B * nblocks * k) — the [B, N] matrix no
//       longer exists.
//
// Tie semantics match ops/csrc/topk.hip: candidates sharing the
// 16-bit ordered-float threshold prefix are interchangeable at rank k;
// any k of them is a valid top-k.
//
// Per-posting doc length travels packed with tf (tf | dl<<16) so the
// norm is computed in-kernel from the CURRENT global avgdl — this is
// what makes O(new) segment appends exact: BM25 stats shift with the
// corpus while installed segments stay immutable.
#include "common.h"

namespace {

DEVINL long lower_bound_i32(const int* __restrict__ a, long lo, long hi,
                            int v) {
  while (lo < hi) {
    const long mid = (lo + hi) >> 1;
    if (a[mid] < v) lo = mid + 1; else hi = mid;
  }
  return lo;
}

// Pre-pass: for every (unique-term, doc-block) pair, binary-search the
// posting sub-range once into a bounds table (i32 offsets relative to
// the term's posting begin). The main kernel then has ZERO serial
// binary searches — per (q,term) they were ~34 dependent HBM loads,
// poorly hidden at a few workgroups/CU; here 1 thread per pair with
// tens of thousands in flight hides them completely, and queries
// sharing a term (Zipf-common) reuse the same entry.
__global__ __launch_bounds__(256) void bm25_bounds_kernel(
    const int* __restrict__ doc_ids,
    const long* __restrict__ u_begin,   // [U] per unique term
    const long* __restrict__ u_end,
    int* __restrict__ bounds,           // [U * nblocks * 2]
    int U, int nblocks, int BD) {
  const int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= U * nblocks) return;
  const int ut = i / nblocks, blk = i % nblocks;
  const long b = u_begin[ut], e = u_end[ut];
  const long d0 = (long)blk * BD;
  const long lo = lower_bound_i32(doc_ids, b, e, (int)d0);
  const long hi = lower_bound_i32(doc_ids, lo, e, (int)(d0 + BD));
  bounds[2 * i] = (int)(lo - b);
  bounds[2 * i + 1] = (int)(hi - b);
}

__global__ __launch_bounds__(256) void bm25_block_kernel(
    const int* __restrict__ doc_ids,        // [P] segment-local, asc per term
    const unsigned int* __restrict__ tfdl,  // [P] tf | dl<<16
    const int* __restrict__ qt_off,         // [B+1] per-query tuple CSR
    const int* __restrict__ qt_ut,          // [T] unique-term index
    const float* __restrict__ qt_idf,       // [T]
    const long* __restrict__ u_begin,       // [U]
    const int* __restrict__ bounds,         // [U * nblocks * 2]
    float* __restrict__ out_vals,           // [B, total_blocks * k]
    int* __restrict__ out_idx,              // [B, total_blocks * k]
    long doc_base, long nseg, int BD, int nblocks,
    int blk_base, int total_blocks, int k_sel,
    float norm_a, float norm_b, float k1p1) {
  // dynamic LDS: scores [BD] f32 | hist1 [8 x 264 padded copies] |
  // hist2 [256] | ctl [8]. 8 bank-padded histogram copies (copy c at
  // c*264 u32 — 264%64=8, so a bin's copies land in distinct banks)
  // break the atomic serialization of concentrated score
  // distributions (same trick as topk.hip hist1).
  extern __shared__ float lds_scores[];
  unsigned* hist1 = reinterpret_cast<unsigned*>(lds_scores + BD);
  unsigned* hist2 = hist1 + 8 * 264;
  unsigned* ctl = hist2 + 256;   // [0]=b1 [1]=chi [2]=t16 [3]=strict [4]=eq
  // grid: x = query (fast), y = doc-block — adjacent workgroups are
  // the SAME posting sub-range for different queries, so the XCD's L2
  // serves the repeat reads instead of HBM
  const int q = blockIdx.x;
  const int blk = blockIdx.y;
  const long d0 = (long)blk * BD;
  const int nd = (int)min((long)BD, nseg - d0);
  for (int i = threadIdx.x; i < nd; i += blockDim.x) lds_scores[i] = 0.0f;
  for (int i = threadIdx.x; i < 8 * 264; i += blockDim.x) hist1[i] = 0;
  for (int i = threadIdx.x; i < 256; i += blockDim.x) hist2[i] = 0;
  if (threadIdx.x < 8) ctl[threadIdx.x] = 0;
  __syncthreads();
  const int t0 = qt_off[q], t1 = qt_off[q + 1];
  const int wave = threadIdx.x / WAVE, lane = threadIdx.x % WAVE;
  const int nwaves = blockDim.x / WAVE;
  // one wave per query-term tuple, round-robin; lanes stride the
  // block's posting sub-range
  for (int ti = t0 + wave; ti < t1; ti += nwaves) {
    const int ut = qt_ut[ti];
    const int* bp = bounds + 2 * ((long)ut * nblocks + blk);
    const long base = u_begin[ut];
    const long lo = base + bp[0], hi = base + bp[1];
    const float w = qt_idf[ti] * k1p1;
    for (long p = lo + lane; p < hi; p += WAVE) {
      const int d = doc_ids[p] - (int)d0;
      const unsigned int td = tfdl[p];
      const float tf = (float)(td & 0xffffu);
      const float dl = (float)(td >> 16);
      atomicAdd(&lds_scores[d], w * tf / (tf + norm_a + norm_b * dl));
    }
  }
  __syncthreads();
  if (k_sel == 0) {   // profiling mode: accumulate only, emit nothing
    if (lds_scores[threadIdx.x] > 1e30f)   // never true; defeat DCE
      out_vals[0] = lds_scores[threadIdx.x];
    return;
  }

  // ---- in-LDS 2-level radix select of the block's top-k ------------
  // BM25 scores are >= 0; ZEROS (unmatched docs, the vast majority of
  // a block) are skipped everywhere — histogramming them would
  // serialize thousands of atomics on one bin. The zero prefix is
  // 0x8000, so "score > 0" == "prefix > 0x8000".
  const int kq = nd < k_sel ? nd : k_sel;
  unsigned* my_h1 = hist1 + (threadIdx.x & 7) * 264;
  for (int i = threadIdx.x; i < nd; i += blockDim.x) {
    const float v = lds_scores[i];
    if (v != 0.0f)
      atomicAdd(&my_h1[float_to_ordered(v) >> 24], 1u);
  }
  __syncthreads();
  if (threadIdx.x < 64) {   // merge the 8 copies (wave 0)
#pragma unroll
    for (int b = threadIdx.x; b < 256; b += 64) {
      unsigned s = hist1[b];
#pragma unroll
      for (int c = 1; c < 8; ++c) s += hist1[c * 264 + b];
      hist1[b] = s;
    }
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned cum = 0;
    int b1 = -1;
    for (int i = 255; i > 128; --i) {   // positive floats only
      const unsigned c = hist1[i];
      if (cum + c >= (unsigned)kq) { b1 = i; break; }
      cum += c;
    }
    if (b1 < 0) {
      // fewer than kq nonzero scores: every nonzero is strict, zeros
      // fill the quota (t16 = the zero prefix)
      ctl[0] = 0; ctl[1] = cum; ctl[2] = 0x8000u;
    } else {
      ctl[0] = (unsigned)b1; ctl[1] = cum; ctl[2] = 0;
    }
  }
  __syncthreads();
  const unsigned b1 = ctl[0];
  if (b1) {
    for (int i = threadIdx.x; i < nd; i += blockDim.x) {
      const float v = lds_scores[i];
      if (v == 0.0f) continue;
      const unsigned o = float_to_ordered(v);
      if ((o >> 24) == b1) atomicAdd(&hist2[(o >> 16) & 255], 1u);
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      unsigned cum = ctl[1];
      for (int i = 255; i >= 0; --i) {
        const unsigned c = hist2[i];
        if (cum + c >= (unsigned)kq || i == 0) {
          ctl[2] = (b1 << 8) | (unsigned)i; break;
        }
        cum += c;
      }
    }
    __syncthreads();
  }
  const unsigned t16 = ctl[2];
  const long obase = ((long)q * total_blocks + blk_base + blk)
      * (long)k_sel;
  // strictly-above candidates first (provably < kq of them) ...
  for (int i = threadIdx.x; i < nd; i += blockDim.x) {
    const float v = lds_scores[i];
    if (v == 0.0f) continue;
    const unsigned o = float_to_ordered(v);
    if ((o >> 16) > t16) {
      const unsigned pos = atomicAdd(&ctl[3], 1u);
      if (pos < (unsigned)k_sel) {   // bound guard (denormal edge)
        out_vals[obase + pos] = v;
        out_idx[obase + pos] = (int)(doc_base + d0) + i;
      }
    }
  }
  __syncthreads();
  const unsigned strict = min(ctl[3], (unsigned)kq);
  const unsigned need_eq = (unsigned)kq - strict;
  // ... then threshold-prefix-equal until the quota (interchangeable
  // at rank k — matching the global selector's tie semantics). The
  // ctl[4] pre-check gates the atomic once the quota is filled: a
  // massively tied bin (or the zero-fill case) would otherwise
  // serialize thousands of increments on one LDS counter.
  for (int i = threadIdx.x; i < nd; i += blockDim.x) {
    if (ctl[4] >= need_eq) break;    // racy fast-exit; exact via pos
    const float v = lds_scores[i];
    const unsigned o = float_to_ordered(v);
    if ((o >> 16) == t16) {
      const unsigned pos = atomicAdd(&ctl[4], 1u);
      if (pos < need_eq) {
        out_vals[obase + strict + pos] = v;
        out_idx[obase + strict + pos] = (int)(doc_base + d0) + i;
      }
    }
  }
  __syncthreads();
  const unsigned eq = ctl[4] < need_eq ? ctl[4] : need_eq;
  const int filled = (int)(strict + eq);
  for (int j = filled + threadIdx.x; j < k_sel; j += blockDim.x) {
    out_vals[obase + j] = -INFINITY;
    out_idx[obase + j] = -1;
  }
}

// Fused score-combine: out = wa * a + wb * b (optional linear hybrid;
// RRF merge of top-k lists happens host-side).
__global__ void combine_kernel(const float* __restrict__ a,
                               const float* __restrict__ b,
                               float* __restrict__ out,
                               float wa, float wb, long n) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (long i = idx; i < n; i += (long)gridDim.x * blockDim.x)
    out[i] = wa * a[i] + wb * b[i];
}

}  // namespace

extern "C" void infomesh_bm25_block(
    const void* doc_ids, const void* tfdl, const void* qt_off,
    const void* qt_ut, const void* qt_idf, const void* u_begin,
    const void* u_end, void* bounds, void* out_vals, void* out_idx,
    int B, int U, long doc_base, long nseg,
    int BD, int blk_base, int total_blocks, int k_sel,
    float norm_a, float norm_b, float k1p1, void* stream) {
  if (nseg <= 0 || B <= 0) return;
  auto s = reinterpret_cast<hipStream_t>(stream);
  const int nblocks = (int)((nseg + BD - 1) / BD);
  if (U > 0) {
    const int total = U * nblocks;
    hipLaunchKernelGGL(bm25_bounds_kernel,
                       dim3((unsigned)((total + 255) / 256)), dim3(256), 0, s,
                       (const int*)doc_ids, (const long*)u_begin,
                       (const long*)u_end, (int*)bounds, U, nblocks, BD);
  }
  dim3 grid((unsigned)B, (unsigned)nblocks);
  const size_t lds = (size_t)BD * sizeof(float)
      + (8 * 264 + 256 + 8) * 4;
  hipLaunchKernelGGL(bm25_block_kernel, grid, dim3(256), lds, s,
                     (const int*)doc_ids, (const unsigned int*)tfdl,
                     (const int*)qt_off, (const int*)qt_ut,
                     (const float*)qt_idf, (const long*)u_begin,
                     (const int*)bounds, (float*)out_vals, (int*)out_idx,
                     doc_base, nseg, BD, nblocks, blk_base, total_blocks,
                     k_sel, norm_a, norm_b, k1p1);
}

extern "C" void infomesh_score_combine(const void* a, const void* b,
                                       void* out, float wa, float wb,
                                       long n, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  long blocks = min((n + 255) / 256, (long)2048);
  hipLaunchKernelGGL(combine_kernel, dim3((unsigned)blocks), dim3(256), 0, s,
                     (const float*)a, (const float*)b, (float*)out, wa, wb, n);
}
