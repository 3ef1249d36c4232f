// Streaming dense-scoring GEMM for the cosine plane (gfx950):
// C[M,N] f32 = A[M,K] bf16 @ B[N,K]^T, specialized for the query-batch
// shape M <= 128 (queries), N ~ millions (docs), K <= ~512 (embedding).
//
// The generic 128x128-tile GEMM re-stages the tiny A tile and pays a
// full vmcnt(0)+barrier pipeline per 64-wide K step — measured 1.9 TB/s
// effective on 128 x 1.25M x 384. Here the WHOLE query block lives in
// LDS (M*(K+8)*2 bytes, padded +8 bf16 per row so the 16 rows of an
// MFMA A-fragment read from 16 distinct banks), each wave owns 64 docs
// per block-tile and streams their embeddings HBM->registers with a
// one-step prefetch — no barriers and no LDS traffic for B at all.
// A re-reads are served from L2 (one block's A load is ~100 KB, shared
// by every block on the XCD). Replaces the generic-GEMM path of
// index/gpu_index.py::search_dense (reference: ChromaDB cosine queries,
// infomesh/index/vector_store.py:216-220).
#include "common.h"

namespace {

// 4 waves x 64 docs = 256 docs per block; M rows (<=128) shared via LDS.
template <int FM>  // M fragments: M = FM*16
__global__ __launch_bounds__(256, 1) void dense_score_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    float* __restrict__ C, int M, long N, int K, float alpha) {
  extern __shared__ bf16 sA[];  // [M][K+8]
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

  const long tile = xcd_swizzle(blockIdx.x, gridDim.x);
  const long n0 = tile * 256 + (threadIdx.x >> 6) * 64;  // this wave's docs
  const int lane = threadIdx.x & 63;
  const int fr = lane & 15;          // fragment row/col within 16
  const int fk = (lane >> 4) * 8;    // fragment k base (32-wide chunk)

  f32x4 acc[FM][4];
#pragma unroll
  for (int i = 0; i < FM; ++i)
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

  const int steps = K / 32;
  bf16x8 bcur[4], bnxt[4];
#pragma unroll
  for (int j = 0; j < 4; ++j)
    bcur[j] = *reinterpret_cast<const bf16x8*>(bp[j] + fk);

  for (int t = 0; t < steps; ++t) {
    const int kc = t * 32;
    if (t + 1 < steps) {
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bnxt[j] = *reinterpret_cast<const bf16x8*>(bp[j] + kc + 32 + fk);
    }
#pragma unroll
    for (int i = 0; i < FM; ++i) {
      const bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &sA[(i * 16 + fr) * KP + kc + fk]);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, bcur[j], acc[i][j], 0, 0, 0);
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) bcur[j] = bnxt[j];
  }

  // ---- epilogue: C row m (query), col n (doc) --------------------------
  const int crow = (lane >> 4) * 4;
  const int ccol = lane & 15;
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const long n = n0 + j * 16 + ccol;
    if (n >= N) continue;
#pragma unroll
    for (int i = 0; i < FM; ++i) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = i * 16 + crow + r;
        if (m < M) C[(long)m * N + n] = alpha * acc[i][j][r];
      }
    }
  }
}

}  // namespace

extern "C" int infomesh_dense_scores(
    const void* A, const void* B, void* C,
    int M, long N, int K, float alpha, void* stream) {
  if (M < 1 || M > 128 || K % 32 != 0 || K < 32) return -1;
  const int FM = (M + 15) / 16;
  const int lds = 16 * FM * (K + 8) * 2;
  if (lds > 160 * 1024) return -1;
  auto s = reinterpret_cast<hipStream_t>(stream);
  const long blocks = (N + 255) / 256;
  if (blocks > 0x7fffffffL) return -1;
  dim3 grid((unsigned)blocks), blk(256);
  // dynamic LDS above 64 KB requires the explicit opt-in (once per
  // template instantiation)
  static bool attr_set[9] = {};
  if (!attr_set[FM]) {
    const void* fns[9] = {nullptr,
        (const void*)&dense_score_kernel<1>, (const void*)&dense_score_kernel<2>,
        (const void*)&dense_score_kernel<3>, (const void*)&dense_score_kernel<4>,
        (const void*)&dense_score_kernel<5>, (const void*)&dense_score_kernel<6>,
        (const void*)&dense_score_kernel<7>, (const void*)&dense_score_kernel<8>};
    if (hipFuncSetAttribute(fns[FM],
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            160 * 1024) != hipSuccess)
      return -1;
    attr_set[FM] = true;
  }
#define DS_LAUNCH(FMV)                                                    \
  hipLaunchKernelGGL((dense_score_kernel<FMV>), grid, blk, lds, s,        \
                     (const bf16*)A, (const bf16*)B, (float*)C, M, N, K,  \
                     alpha)
  switch (FM) {
    case 1: DS_LAUNCH(1); break;
    case 2: DS_LAUNCH(2); break;
    case 3: DS_LAUNCH(3); break;
    case 4: DS_LAUNCH(4); break;
    case 5: DS_LAUNCH(5); break;
    case 6: DS_LAUNCH(6); break;
    case 7: DS_LAUNCH(7); break;
    case 8: DS_LAUNCH(8); break;
    default: return -1;
  }
#undef DS_LAUNCH
  return 0;
}
