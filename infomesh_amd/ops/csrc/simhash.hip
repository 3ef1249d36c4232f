// SimHash on CDNA4: batched 64-bit fingerprint (bit-vote over shingle
// hashes) and XOR+popcount near-duplicate scan.
// Replaces: the reference's Python Charikar simhash + linear scan
// (infomesh/crawler/simhash.py:43-96,131-213) — SURVEY.md §2.9.
//
// Fingerprint uses the wave64 idiom directly: lane b owns bit b, and
// __ballot(vote > 0) IS the 64-bit fingerprint.
#include "common.h"

namespace {

// Docs as CSR shingle-hash arrays: offsets [D+1] i64, hashes [T] u64.
// One wave per doc (4 docs per 256-thread block).
__global__ __launch_bounds__(256) void fingerprint_kernel(
    const long* __restrict__ offsets,
    const unsigned long long* __restrict__ hashes,
    unsigned long long* __restrict__ out, long ndocs) {
  const long doc = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (doc >= ndocs) return;
  const int bit = threadIdx.x & 63;
  const long begin = offsets[doc], end = offsets[doc + 1];
  int vote = 0;
  for (long i = begin; i < end; ++i)  // all lanes read the same word: broadcast
    vote += (int)((hashes[i] >> bit) & 1ULL) * 2 - 1;
  const unsigned long long fp = __ballot(vote > 0);
  if (bit == 0) out[doc] = fp;
}

// Scan M query fingerprints against a table of N; emit (q, n, dist)
// matches with hamming distance <= radius via an atomic cursor.
__global__ __launch_bounds__(256) void hamming_scan_kernel(
    const unsigned long long* __restrict__ queries,
    const unsigned long long* __restrict__ table,
    int* __restrict__ out_q, int* __restrict__ out_n, int* __restrict__ out_d,
    unsigned* __restrict__ cnt, int M, long N, int radius, int cap) {
  extern __shared__ __attribute__((aligned(16))) unsigned long long qs[];
  for (int i = threadIdx.x; i < M; i += blockDim.x) qs[i] = queries[i];
  __syncthreads();
  const long start = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long step = (long)gridDim.x * blockDim.x;
  for (long n = start; n < N; n += step) {
    const unsigned long long t = table[n];
    for (int m = 0; m < M; ++m) {
      const int d = __popcll(t ^ qs[m]);
      if (d <= radius) {
        const unsigned pos = atomicAdd(cnt, 1u);
        if (pos < (unsigned)cap) {
          out_q[pos] = m;
          out_n[pos] = (int)n;
          out_d[pos] = d;
        }
      }
    }
  }
}

}  // namespace

extern "C" void infomesh_simhash_fingerprint(
    const void* offsets, const void* hashes, void* out, long ndocs,
    void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  const long blocks = (ndocs + 3) / 4;
  hipLaunchKernelGGL(fingerprint_kernel, dim3((unsigned)blocks), dim3(256), 0,
                     s, (const long*)offsets,
                     (const unsigned long long*)hashes,
                     (unsigned long long*)out, ndocs);
}

extern "C" void infomesh_hamming_scan(
    const void* queries, const void* table, void* out_q, void* out_n,
    void* out_d, void* cnt, int M, long N, int radius, int cap,
    void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  long blocks = min((N + 255) / 256, (long)2048);
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(hamming_scan_kernel, dim3((unsigned)blocks), dim3(256),
                     M * sizeof(unsigned long long), s,
                     (const unsigned long long*)queries,
                     (const unsigned long long*)table,
                     (int*)out_q, (int*)out_n, (int*)out_d,
                     (unsigned*)cnt, M, N, radius, cap);
}
