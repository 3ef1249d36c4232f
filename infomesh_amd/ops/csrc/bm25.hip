// GPU BM25 scorer over per-segment CSR posting lists in HBM — v2:
// doc-block LDS accumulation instead of global atomic scatter-add.
//
// Replaces: SQLite FTS5 `MATCH ... ORDER BY bm25()` (reference
// infomesh/index/local_store.py:316-332) for the GPU shards; the CPU
// FTS5 path remains for the CPU-plumbing config.
//
// score(q, d) = Σ_t idf(t) · tf·(k1+1) / (tf + k1·(1−b+b·dl/avgdl))
//
// Design (MI355X): the v1 kernel did one global atomicAdd per posting
// into scores[q*N+d] — random read-modify-write lines over a multi-GB
// working set, measured HBM-bound at ~1 ms/batch (profiles/r01_*).
// v2 tiles the doc axis: each workgroup owns one (query, doc-block)
// tile, accumulates its block's scores in LDS (ds_add_f32 atomics are
// conflict-cheap), then streams the block out once. Each term's
// postings are doc-sorted, so the block's sub-range is two binary
// searches. Global traffic drops to: postings read once + scores
// written exactly once (which also removes the separate zero-fill —
// every column of the output is written by exactly one workgroup,
// because segments partition the doc axis and blocks partition each
// segment).
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
// poorly hidden at 2 workgroups/CU; here 1 thread per pair with tens
// of thousands in flight hides them completely, and queries sharing a
// term (Zipf-common) reuse the same entry.
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

// Fused hist1: every written score is also counted into a per-query
// 256-bin ordered-top-byte histogram (8 padded LDS copies, zeros
// register-aggregated per wave since empty blocks dominate), merged by
// global atomics. infomesh_topk(ext_hist1=...) then SKIPS its first
// full read of the score array — at 10M docs that pass alone is
// ~0.9 ms/batch of HBM streaming. Counts are exact (each column is
// written and histogrammed exactly once across segments x blocks), so
// downstream select/compact semantics are bit-identical.
template <bool HIST>
__global__ __launch_bounds__(256) void bm25_block_kernel(
    const int* __restrict__ doc_ids,        // [P] segment-local, asc per term
    const unsigned int* __restrict__ tfdl,  // [P] tf | dl<<16
    const int* __restrict__ qt_off,         // [B+1] per-query tuple CSR
    const int* __restrict__ qt_ut,          // [T] unique-term index
    const float* __restrict__ qt_idf,       // [T]
    const long* __restrict__ u_begin,       // [U]
    const int* __restrict__ bounds,         // [U * nblocks * 2]
    float* __restrict__ scores,             // [B, rowN]
    unsigned* __restrict__ hist1,           // [B * 256] or null
    long rowN, long doc_base, long nseg, int BD, int nblocks,
    float norm_a, float norm_b, float k1p1) {
  extern __shared__ float lds_scores[];     // [BD] (+ hist copies)
  // grid: x = query (fast), y = doc-block — adjacent workgroups are
  // the SAME posting sub-range for different queries, so the XCD's L2
  // serves the repeat reads instead of HBM
  const int q = blockIdx.x;
  const int blk = blockIdx.y;
  const long d0 = (long)blk * BD;
  const int nd = (int)min((long)BD, nseg - d0);
  // 8 padded histogram copies after the score tile (same bank spread
  // as topk.hip hist1_kernel: 264-int stride)
  unsigned* lh = reinterpret_cast<unsigned*>(lds_scores + BD);
  for (int i = threadIdx.x; i < nd; i += blockDim.x) lds_scores[i] = 0.0f;
  if (HIST)
    for (int i = threadIdx.x; i < 8 * 264; i += blockDim.x) lh[i] = 0;
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
  // scores are written once and only re-read by the top-k streaming
  // pass — nontemporal keeps them out of L2, which the posting reads
  // (shared across adjacent same-block workgroups) actually want
  float* __restrict__ srow = scores + (long)q * rowN + doc_base + d0;
  if (!HIST) {
    for (int i = threadIdx.x; i < nd; i += blockDim.x)
      __builtin_nontemporal_store(lds_scores[i], srow + i);
    return;
  }
  unsigned* my = lh + (threadIdx.x & 7) * 264;
  unsigned zc = 0;  // zero scores dominate: aggregate in a register
  for (int i = threadIdx.x; i < nd; i += blockDim.x) {
    const float v = lds_scores[i];
    __builtin_nontemporal_store(v, srow + i);
    if (v == 0.0f) ++zc;
    else atomicAdd(&my[float_to_ordered(v) >> 24], 1u);
  }
  // wave-reduce the zero count; one LDS atomic per wave
  for (int off = WAVE / 2; off; off >>= 1)
    zc += __shfl_xor(zc, off, WAVE);
  constexpr unsigned ZBIN = 0x80000000u >> 24;  // ordered(+0.0f) byte
  if ((threadIdx.x % WAVE) == 0 && zc)
    atomicAdd(&my[ZBIN], zc);
  __syncthreads();
  unsigned* hrow = hist1 + (long)q * 256;
  for (int i = threadIdx.x; i < 256; i += blockDim.x) {
    unsigned sum = 0;
#pragma unroll
    for (int c = 0; c < 8; ++c) sum += lh[c * 264 + i];
    if (sum) atomicAdd(&hrow[i], sum);
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
    const void* u_end, void* bounds, void* scores, void* hist1,
    int B, int U, long rowN, long doc_base, long nseg,
    int BD, float norm_a, float norm_b, float k1p1, void* stream) {
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
  const size_t shm = (size_t)BD * sizeof(float)
      + (hist1 ? 8 * 264 * sizeof(unsigned) : 0);
  if (hist1)
    hipLaunchKernelGGL(bm25_block_kernel<true>, grid, dim3(256), shm, s,
                       (const int*)doc_ids, (const unsigned int*)tfdl,
                       (const int*)qt_off, (const int*)qt_ut,
                       (const float*)qt_idf, (const long*)u_begin,
                       (const int*)bounds, (float*)scores,
                       (unsigned*)hist1,
                       rowN, doc_base, nseg, BD, nblocks,
                       norm_a, norm_b, k1p1);
  else
    hipLaunchKernelGGL(bm25_block_kernel<false>, grid, dim3(256), shm, s,
                       (const int*)doc_ids, (const unsigned int*)tfdl,
                       (const int*)qt_off, (const int*)qt_ut,
                       (const float*)qt_idf, (const long*)u_begin,
                       (const int*)bounds, (float*)scores, nullptr,
                       rowN, doc_base, nseg, BD, nblocks,
                       norm_a, norm_b, k1p1);
}

extern "C" void infomesh_score_combine(const void* a, const void* b,
                                       void* out, float wa, float wb,
                                       long n, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  long blocks = min((n + 255) / 256, (long)2048);
  hipLaunchKernelGGL(combine_kernel, dim3((unsigned)blocks), dim3(256), 0, s,
                     (const float*)a, (const float*)b, (float*)out, wa, wb, n);
}
