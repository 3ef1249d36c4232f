// Streaming dense-scoring GEMM for the cosine plane (gfx950):
// C[M,N] f32 = A[M,K] bf16 @ B[N,K]^T, specialized for the query-batch
// shape M <= 128 (queries), N ~ millions (docs), K <= ~512 (embedding).
//
// The generic 128x128-tile GEMM re-stages the tiny A tile and pays a
// full vmcnt(0)+barrier pipeline per 64-wide K step — measured 1.9 TB/s
// effective on 128 x 1.25M x 384. Here the WHOLE query block lives in
// LDS (M*(K+8)*2 bytes, padded +8 bf16 per row so the 16 rows of an
// MFMA A-fragment read from 16 distinct banks), and doc embeddings
// stream HBM->registers through a 4-slot prefetch ring — no barriers
// and no LDS traffic for B in the K loop.
//
// Occupancy note: one wave covering all 8 M-fragments needs 128 acc
// AGPRs -> 1 wave/SIMD, and the ds_read->MFMA dependency then stalls
// every fragment (measured 1.7 TB/s). So blocks run WGROUPS=2 x 4
// waves: each M-half needs only FMW<=4 fragments (64 acc regs -> 2
// waves/SIMD), the two halves share the same 256 docs (B loads of the
// sibling wave hit L1), and one wave's MFMA hides the other's LDS
// reads. A re-reads are served from L2 (one block's A is ~100 KB,
// shared by every block on the XCD).
// Replaces the generic-GEMM path of index/gpu_index.py::search_dense
// (reference: ChromaDB cosine queries, infomesh/index/vector_store.py:
// 216-220).
#include "common.h"

namespace {

// WGROUPS M-halves x 4 doc-waves; each wave: FMW m-frags x 64 docs.
template <int FMW, int WGROUPS>
__global__ __launch_bounds__(WGROUPS * 256, 1) void dense_score_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    float* __restrict__ C, int M, long N, int K, float alpha) {
  extern __shared__ bf16 sA[];  // [16*FMW*WGROUPS][K+8]
  const int KP = K + 8;

  // ---- stage A (M*K, once) --------------------------------------------
  {
    const int elems = M * K;
    for (int i = threadIdx.x * 8; i < elems; i += blockDim.x * 8) {
      const int m = i / K, k = i % K;  // K % 8 == 0 keeps rows vector-whole
      *reinterpret_cast<bf16x8*>(&sA[m * KP + k]) =
          *reinterpret_cast<const bf16x8*>(&A[(long)m * K + k]);
    }
  }
  __syncthreads();

  const int wave = threadIdx.x >> 6;
  const int mg = wave >> 2;              // M-half (0 when WGROUPS==1)
  const int m0 = mg * FMW * 16;
  const long tile = xcd_swizzle(blockIdx.x, gridDim.x);
  const long n0 = tile * 256 + (wave & 3) * 64;  // this wave's docs
  const int lane = threadIdx.x & 63;
  const int fr = lane & 15;          // fragment row/col within 16
  const int fk = (lane >> 4) * 8;    // fragment k base (32-wide chunk)

  f32x4 acc[FMW][4];
#pragma unroll
  for (int i = 0; i < FMW; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // B fragment addresses for this lane: doc row n0 + j*16 + fr.
  const bf16* bp[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    long n = n0 + j * 16 + fr;
    if (n >= N) n = N - 1;  // clamp: duplicate loads, stores guarded
    bp[j] = B + n * K;
  }

  // 4-slot register ring, 3 K-steps of prefetch (12 KB in flight per
  // wave). Static slot indices via a manual 4-step unroll (dynamic
  // indexing would spill).
  const int steps = K / 32;  // wrapper guarantees K % 128 == 0 -> %4==0
  bf16x8 bq[4][4];
#pragma unroll
  for (int p = 0; p < 3; ++p)
#pragma unroll
    for (int j = 0; j < 4; ++j)
      bq[p][j] = *reinterpret_cast<const bf16x8*>(bp[j] + p * 32 + fk);

#define DS_BODY(PH)                                                       \
  do {                                                                    \
    const int t = tb + (PH);                                              \
    if (t + 3 < steps) {                                                  \
      _Pragma("unroll")                                                   \
      for (int j = 0; j < 4; ++j)                                         \
        bq[((PH) + 3) & 3][j] = *reinterpret_cast<const bf16x8*>(         \
            bp[j] + (t + 3) * 32 + fk);                                   \
    }                                                                     \
    const int kc = t * 32;                                                \
    _Pragma("unroll")                                                     \
    for (int i = 0; i < FMW; ++i) {                                       \
      const bf16x8 a = *reinterpret_cast<const bf16x8*>(                  \
          &sA[(m0 + i * 16 + fr) * KP + kc + fk]);                        \
      _Pragma("unroll")                                                   \
      for (int j = 0; j < 4; ++j)                                         \
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(              \
            a, bq[(PH) & 3][j], acc[i][j], 0, 0, 0);                      \
    }                                                                     \
  } while (0)

  for (int tb = 0; tb < steps; tb += 4) {
    DS_BODY(0);
    DS_BODY(1);
    DS_BODY(2);
    DS_BODY(3);
  }
#undef DS_BODY

  // ---- epilogue: C row m (query), col n (doc) --------------------------
  const int crow = (lane >> 4) * 4;
  const int ccol = lane & 15;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const long n = n0 + j * 16 + ccol;
    if (n >= N) continue;
#pragma unroll
    for (int i = 0; i < FMW; ++i) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + i * 16 + crow + r;
        if (m < M) C[(long)m * N + n] = alpha * acc[i][j][r];
      }
    }
  }
}

}  // namespace

extern "C" int infomesh_dense_scores(
    const void* A, const void* B, void* C,
    int M, long N, int K, float alpha, void* stream) {
  if (M < 1 || M > 128 || K % 128 != 0 || K < 128) return -1;
  const int FM = (M + 15) / 16;
  const int wg = FM > 4 ? 2 : 1;              // M-halves
  const int fmw = wg == 2 ? (FM + 1) / 2 : FM;  // frags per wave
  const int lds = 16 * FM * (K + 8) * 2;
  if (lds > 160 * 1024) return -1;
  auto s = reinterpret_cast<hipStream_t>(stream);
  const long blocks = (N + 255) / 256;
  if (blocks > 0x7fffffffL) return -1;
  dim3 grid((unsigned)blocks), blk(wg * 256);
  // dynamic LDS above 64 KB requires the explicit opt-in (once per
  // template instantiation); cfg index = (wg-1)*4 + fmw-1
  static bool attr_set[8] = {};
  const int cfg = (wg - 1) * 4 + fmw - 1;
  const void* fns[8] = {
      (const void*)&dense_score_kernel<1, 1>,
      (const void*)&dense_score_kernel<2, 1>,
      (const void*)&dense_score_kernel<3, 1>,
      (const void*)&dense_score_kernel<4, 1>,
      (const void*)&dense_score_kernel<1, 2>,
      (const void*)&dense_score_kernel<2, 2>,
      (const void*)&dense_score_kernel<3, 2>,
      (const void*)&dense_score_kernel<4, 2>};
  if (!attr_set[cfg]) {
    if (hipFuncSetAttribute(fns[cfg],
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024) != hipSuccess)
      return -1;
    attr_set[cfg] = true;
  }
#define DS_LAUNCH(FMV, WGV)                                               \
  hipLaunchKernelGGL((dense_score_kernel<FMV, WGV>), grid, blk, lds, s,   \
                     (const bf16*)A, (const bf16*)B, (float*)C, M, N, K,  \
                     alpha)
  switch (cfg) {
    case 0: DS_LAUNCH(1, 1); break;
    case 1: DS_LAUNCH(2, 1); break;
    case 2: DS_LAUNCH(3, 1); break;
    case 3: DS_LAUNCH(4, 1); break;
    case 4: DS_LAUNCH(1, 2); break;
    case 5: DS_LAUNCH(2, 2); break;
    case 6: DS_LAUNCH(3, 2); break;
    case 7: DS_LAUNCH(4, 2); break;
  }
#undef DS_LAUNCH
  return 0;
}
