// GPU BM25 term-at-a-time scorer over CSR posting lists in HBM.
// Replaces: SQLite FTS5 `MATCH ... ORDER BY bm25()` (reference
// infomesh/index/local_store.py:316-332) for the GPU shards; the CPU
// FTS5 path remains for the CPU-plumbing config.
//
// score(q, d) = Σ_t idf(t) · tf·(k1+1) / (tf + k1·(1−b+b·dl/avgdl))
// with the per-doc denominator part precomputed as doc_norm[d].
// Work is pre-chunked host-side into (query-row, term, posting-offset)
// triples so Zipf-skewed posting lists spread over many blocks.
#include "common.h"

namespace {

__global__ __launch_bounds__(256) void bm25_kernel(
    const long* __restrict__ offsets, const int* __restrict__ doc_ids,
    const unsigned short* __restrict__ tfs,
    const float* __restrict__ doc_norm,
    const int* __restrict__ chunk_qrow, const int* __restrict__ chunk_term,
    const long* __restrict__ chunk_off, const float* __restrict__ chunk_idf,
    float* __restrict__ scores, long N, int chunk_size, float k1p1) {
  const int c = blockIdx.x;
  const int q = chunk_qrow[c];
  const int t = chunk_term[c];
  const long begin = chunk_off[c];
  const long end = min(offsets[t + 1], begin + (long)chunk_size);
  const float idf = chunk_idf[c];
  float* srow = scores + (long)q * N;
  for (long p = begin + threadIdx.x; p < end; p += blockDim.x) {
    const int d = doc_ids[p];
    const float tf = (float)tfs[p];
    atomicAdd(&srow[d], idf * tf * k1p1 / (tf + doc_norm[d]));
  }
}

// Fused score-combine: out = w_bm25 * bm25 / (bm25 + sat) + w_dense * dense
// (optional linear hybrid; RRF merge of top-k lists happens host-side).
__global__ void combine_kernel(const float* __restrict__ a,
                               const float* __restrict__ b,
                               float* __restrict__ out,
                               float wa, float wb, long n) {
  const long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
  for (long i = idx; i < n; i += (long)gridDim.x * blockDim.x)
    out[i] = wa * a[i] + wb * b[i];
}

}  // namespace

extern "C" void infomesh_bm25_score(
    const void* offsets, const void* doc_ids, const void* tfs,
    const void* doc_norm, const void* chunk_qrow, const void* chunk_term,
    const void* chunk_off, const void* chunk_idf, void* scores,
    int nchunks, long N, int chunk_size, float k1, void* stream) {
  if (nchunks <= 0) return;
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(bm25_kernel, dim3(nchunks), dim3(256), 0, s,
                     (const long*)offsets, (const int*)doc_ids,
                     (const unsigned short*)tfs, (const float*)doc_norm,
                     (const int*)chunk_qrow, (const int*)chunk_term,
                     (const long*)chunk_off, (const float*)chunk_idf,
                     (float*)scores, N, chunk_size, k1 + 1.0f);
}

extern "C" void infomesh_score_combine(const void* a, const void* b,
                                       void* out, float wa, float wb,
                                       long n, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  long blocks = min((n + 255) / 256, (long)2048);
  hipLaunchKernelGGL(combine_kernel, dim3((unsigned)blocks), dim3(256), 0, s,
                     (const float*)a, (const float*)b, (float*)out, wa, wb, n);
}
