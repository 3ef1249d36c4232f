// Batched-strided bf16 MFMA GEMM for gfx950: C[g] = act(A[g] @ B[g]^T + bias).
//
// Layout: A [M,K] row-major, B [N,K] row-major (both K-contiguous — "NT"),
// C [M,N]. f32 accumulation on mfma_f32_16x16x32_bf16; output bf16 or f32.
//
// Structure (guide cdna_hip_programming.md §5 "step-3"): 128x128 tile,
// BK∈{32,64}, 4 waves (2x2) each owning a 64x64 sub-tile as 4x4 fragments
// of 16x16, double-buffered LDS staged by __builtin_amdgcn_global_load_lds
// width 16 (lane-linear LDS image), one vmcnt(0)+barrier per K-step.
// Replaces: the reference's torch/sentence-transformers matmuls
// (infomesh/index/vector_store.py:120-157) and external-LLM HTTP calls
// (infomesh/summarizer/engine.py:111-318) — see SURVEY.md §2.9.
#include "common.h"
#include <cstdlib>

namespace {

// LDS swizzle for BK=64 tiles (involution, 16 B granules), applied to
// the glds SOURCE address and the ds_read address (rule 21: the LDS
// destination stays lane-linear). BK=32 rows are 64 B and keep the
// linear image. Rows are
// 128 B so bank_start = 32*(row&1) + 4*chunk collapses rows r and r+2
// onto the same banks; xoring the 16 B-chunk index (bits 4-6) with
// h(row) = (row ^ (row>>3)) & 7 (bits 7-12 of the offset) gives all 16
// rows of a quarter-wave MFMA operand read distinct 4-bank windows.
// Measured: the SQ_LDS_BANK_CONFLICT counter on the pipelined tile
// (4.16e8, profiles/) is UNCHANGED by this — those conflicts come from
// the epilogue f32 staging (4-way ds_write folding, ~192 cycles/wave-
// tile), and wall time is glds-latency-bound either way; the swizzle is
// kept because conflict-free operand reads cost nothing. Key uses only
// bits >=7, which the xor never touches, so swz64(swz64(x)) == x and
// region bases (bit 13+) pass through unchanged.
__device__ __forceinline__ int swz64(int byte_off) {
  return byte_off ^ ((((byte_off >> 7) ^ (byte_off >> 10)) & 7) << 4);
}

// Tile configurations (4 waves each):
//   (128,128): 2x2 wave grid, 64x64 per wave (4x4 fragments) — default.
//   ( 64,128): 1x4 grid, 64x32 per wave — skinny-M (query scoring).
//   ( 64, 64): 2x2 grid, 32x32 per wave — latency-bound small-K shapes
//              (encoder projections) where 128^2 tiles underfill the
//              chip and the 6-step K loop's stage latency dominates.
template <int BM, int BN, int BK, bool OUT_F32>
__global__ __launch_bounds__(256) void gemm_bf16_nt_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    void* __restrict__ C, const float* __restrict__ bias,
    int M, int N, int K,
    long strideA, long strideB, long strideC,
    int act, float alpha) {
  // One __shared__ object only (glds pipeline rule, guide §5 item 4a).
  // Layout per buffer: A tile [BM*BK] then B tile [BN*BK].
  __shared__ bf16 smem[2][(BM + BN) * BK];

  const int tiles_n = (N + BN - 1) / BN;
  const int tiles_m = (M + BM - 1) / BM;
  const int tile = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int tm = tile / tiles_n, tn = tile % tiles_n;
  const int m0 = tm * BM, n0 = tn * BN;
  const int g = blockIdx.y;

  const bf16* Ag = A + (long)g * strideA;
  const bf16* Bg = B + (long)g * strideB;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;          // 4 waves
  constexpr int WM = (BM == 128 || BN == 64) ? 2 : 1;   // wave grid M
  constexpr int WN = 4 / WM;                // wave grid N
  constexpr int FM = (BM / WM) / 16;        // m-frags per wave
  constexpr int FN = (BN / WN) / 16;        // n-frags per wave
  const int wm = wid / WN, wn = wid % WN;

  // ---- glds staging geometry -------------------------------------------
  // One glds instruction: 64 lanes x 16 B = 1 KiB = 8 rows of BK=64 bf16
  // (or 4 rows-pairs when BK=32: 1 KiB = 16 rows of 64 B).
  // A tile is BM*BK*2 bytes = BM*BK/512 KiB -> chunks of 1 KiB each.
  constexpr int ROWS_PER_CHUNK = 1024 / (BK * 2);      // 8 (BK=64) / 16 (BK=32)
  // Chunk count follows the larger (B) tile; A stages only its first
  // BM*BK*2/1024 chunks (guarded below) when BM < BN.
  constexpr int CHUNKS = ((BM > BN ? BM : BN) * BK * 2) / 1024;
  constexpr int CHUNKS_PER_WAVE = CHUNKS / 4;
  constexpr int ACHUNKS = (BM * BK * 2) / 1024;
  constexpr int BCHUNKS2 = (BN * BK * 2) / 1024;
  const int lanes_per_row = BK / 8;                     // 8 elements per lane
  const int lrow = lane / lanes_per_row;
  const int lcol = (lane % lanes_per_row) * 8;

  const int n_ksteps = K / BK;

  constexpr int BCHUNKS = (BN * BK * 2) / 1024;      // B half count
  auto stage = [&](int buf, int kstep) {
    const long k0 = (long)kstep * BK;
#pragma unroll
    for (int c = 0; c < CHUNKS_PER_WAVE; ++c) {
      const int chunk = wid * CHUNKS_PER_WAVE + c;
      // region-relative source byte under the swizzle involution; the
      // LDS destination stays lane-linear (glds writes base + lane*16)
      const int lin = chunk * 1024 + lane * 16;
      const int src = (BK == 64) ? swz64(lin) : lin;
      const int row = src / (BK * 2);
      const int colb = src % (BK * 2);
      if (chunk < BCHUNKS2) {
        int brow = n0 + row; brow = brow < N ? brow : N - 1;
        const bf16* gb = Bg + (long)brow * K + k0 + colb / 2;
        auto* lb = (__attribute__((address_space(3))) unsigned int*)
            &smem[buf][BM * BK + chunk * 512];
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)gb, lb, 16, 0, 0);
      }
      if (chunk < ACHUNKS) {
        int arow = m0 + row; arow = arow < M ? arow : M - 1;
        const bf16* ga = Ag + (long)arow * K + k0 + colb / 2;
        auto* la = (__attribute__((address_space(3))) unsigned int*)
            &smem[buf][chunk * 512];
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)ga, la, 16, 0, 0);
      }
    }
  };

  f32x4 acc[FM][FN];
#pragma unroll
  for (int i = 0; i < FM; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  stage(0, 0);
  __syncthreads();

  const int fr = lane & 15;          // fragment row (A) / col (B)
  const int fk = (lane >> 4) * 8;    // fragment k base

  int cur = 0;
  for (int t = 0; t < n_ksteps; ++t) {
    if (t + 1 < n_ksteps) stage(cur ^ 1, t + 1);
#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      bf16x8 a[FM], b[FN];
#pragma unroll
      for (int i = 0; i < FM; ++i) {
        const int row = wm * (BM / WM) + i * 16 + fr;
        const int off = (row * BK + ks * 32 + fk) * 2;
        a[i] = *reinterpret_cast<const bf16x8*>(
            (const char*)&smem[cur][0]
            + ((BK == 64) ? swz64(off) : off));
      }
#pragma unroll
      for (int j = 0; j < FN; ++j) {
        const int row = wn * (BN / WN) + j * 16 + fr;
        const int off = (row * BK + ks * 32 + fk) * 2;
        b[j] = *reinterpret_cast<const bf16x8*>(
            (const char*)&smem[cur][BM * BK]
            + ((BK == 64) ? swz64(off) : off));
      }
#pragma unroll
      for (int i = 0; i < FM; ++i)
#pragma unroll
        for (int j = 0; j < FN; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i], b[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();   // drains the in-flight glds (vmcnt(0)) + barrier
    cur ^= 1;
  }

  // ---- epilogue ---------------------------------------------------------
  // Huge-N f32 output (the dense scoring plane): stage the wave's
  // 64-col row segments through LDS and emit 16-B dwordx4 stores,
  // NONTEMPORAL — C is written once and next read by the streaming
  // top-k, so allocating it in L2 only evicts the B rows being
  // streamed. Smaller/bf16 outputs keep the direct scalar path.
  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
  if (OUT_F32 && (BN == 128 || BN == 64) && n0 + BN <= N
      && m0 + BM <= M) {
    // reuse the (drained) double buffer as an f32 staging tile:
    // per wave a [BM/WM][BN/WN] block = 64x64 (or 64x32/32x32) f32
    float* stage_f32 = reinterpret_cast<float*>(&smem[0][0])
        + wid * (BM / WM) * (BN / WN);
    __syncthreads();   // all MFMA reads of smem are done
#pragma unroll
    for (int i = 0; i < FM; ++i)
#pragma unroll
      for (int j = 0; j < FN; ++j)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int lm = i * 16 + crow_base + r;      // wave-local row
          const int ln = j * 16 + ccol;               // wave-local col
          const int n = n0 + wn * (BN / WN) + ln;
          const float bv = bias ? bias[n] : 0.0f;
          stage_f32[lm * (BN / WN) + ln] =
              apply_act(alpha * acc[i][j][r] + bv, act);
        }
    __builtin_amdgcn_s_waitcnt(0);  // lgkmcnt: LDS writes visible
    __builtin_amdgcn_wave_barrier();
    // each lane streams 16 B of a row segment: WAVEN cols * 4 B / 16
    constexpr int SEGN = BN / WN;                  // cols per wave block
    constexpr int LPR = SEGN * 4 / 16;             // lanes per row
    constexpr int ROWS_PER_IT = 64 / LPR;
    const int srow = lane / LPR, scol4 = (lane % LPR) * 4;
#pragma unroll
    for (int base = 0; base < BM / WM; base += ROWS_PER_IT) {
      const int lm = base + srow;
      const int m = m0 + wm * (BM / WM) + lm;
      const int n = n0 + wn * SEGN + scol4;
      f32x4 v = *reinterpret_cast<const f32x4*>(
          &stage_f32[lm * SEGN + scol4]);
      __builtin_nontemporal_store(
          v, reinterpret_cast<f32x4*>(
              reinterpret_cast<float*>(C)
              + (long)g * strideC + (long)m * N + n));
    }
    return;
  }
#pragma unroll
  for (int i = 0; i < FM; ++i) {
#pragma unroll
    for (int j = 0; j < FN; ++j) {
      const int n = n0 + wn * (BN / WN) + j * 16 + ccol;
      if (n >= N) continue;
      const float bv = bias ? bias[n] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + wm * (BM / WM) + i * 16 + crow_base + r;
        if (m >= M) continue;
        float v = apply_act(alpha * acc[i][j][r] + bv, act);
        if (OUT_F32)
          reinterpret_cast<float*>(C)[(long)g * strideC + (long)m * N + n] = v;
        else
          reinterpret_cast<bf16*>(C)[(long)g * strideC + (long)m * N + n] = f2bf(v);
      }
    }
  }
}

// ---- 3-buffer glds-span pipelined 64x64 tile ---------------------------
// For the latency-bound regime (encoder projections: M~4096, small N/K,
// few K-steps): the serial stage->vmcnt(0)->barrier->compute structure
// of the generic tile pays a full LDS-fill latency every K-step. Here
// stages t+1 AND t+2 stay in flight across RAW s_barriers with counted
// vmcnt waits (guide "pipelining across barriers": 3-buf span +83% over
// serial at 1-block/CU), and 16 KiB/buffer keeps 3 workgroups resident
// per CU on top. Round-1 BACKLOG item "counted-vmcnt small-tile
// pipeline (gemm8-style scheduling at 64^2)".
template <int BNT, bool OUT_F32>
__global__ __launch_bounds__(256) void gemm_pipe64_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    void* __restrict__ C, const float* __restrict__ bias,
    int M, int N, int K,
    long strideA, long strideB, long strideC,
    int act, float alpha) {
  constexpr int BM = 64, BN = BNT, BK = 64;
  __shared__ bf16 smem[3][(BM + BN) * BK];   // 3 x 16 KiB

  const int tiles_n = (N + BN - 1) / BN;
  const int tiles_m = (M + BM - 1) / BM;
  const int tile = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int m0 = (tile / tiles_n) * BM, n0 = (tile % tiles_n) * BN;
  const int g = blockIdx.y;
  const bf16* Ag = A + (long)g * strideA;
  const bf16* Bg = B + (long)g * strideB;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wm = wid >> 1, wn = wid & 1;    // 2x2 waves, 32 x BN/2 each
  constexpr int FN = BN / 2 / 16;           // n-frags per wave
  const int n_ksteps = K / BK;

  // tile = (BM+BN)*BK*2 bytes in 1 KiB chunks (A chunks then B);
  // CPW glds per wave per stage
  constexpr int CPW = (BM + BN) * BK * 2 / 1024 / 4;
  auto stage = [&](int buf, int t) {
    const int ks = t < n_ksteps ? t : n_ksteps - 1;  // clamp (static cnt)
    const long k0 = (long)ks * BK;
#pragma unroll
    for (int c = 0; c < CPW; ++c) {
      const int chunk = wid * CPW + c;
      const int lin = chunk * 1024 + lane * 16;
      const int src = swz64(lin);
      const int row = src / (BK * 2);
      const int colb = src % (BK * 2);
      const bool is_b = chunk >= (BM * BK * 2 / 1024);
      int grow = is_b ? (n0 + row - BM) : (m0 + row);
      const int lim = is_b ? N : M;
      grow = grow < lim ? grow : lim - 1;
      const bf16* gsrc = (is_b ? Bg : Ag) + (long)grow * K + k0 + colb / 2;
      auto* dst = (__attribute__((address_space(3))) unsigned int*)
          ((char*)&smem[buf][0] + (long)chunk * 1024);
      // NT on the A loads was tried here (gemm8-style): encoder
      // 993 -> 1135 us. At these shapes BOTH operands fit the XCD L2
      // together (A 3.1 MB x 18-way n-reuse + B 0.9 MB), so the NT
      // hint only destroyed A's own reuse. Keep default caching.
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gsrc,
          dst, 16, 0, 0);
    }
  };

  const int fr = lane & 15;
  const int fk = (lane >> 4) * 8;
  f32x4 acc[2][FN];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < FN; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  stage(0, 0);
  stage(1, 1);
  asm volatile("s_waitcnt vmcnt(%0)" :: "n"(CPW) : "memory");  // t0 in
  __builtin_amdgcn_s_barrier();

  for (int t = 0; t < n_ksteps; ++t) {
    const char* abase = (const char*)&smem[t % 3][0];
    const char* bbase = abase + BM * BK * 2;
    bf16x8 a[2][2], b[FN][2];
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int row = wm * 32 + i * 16 + fr;
        a[i][ks] = *reinterpret_cast<const bf16x8*>(
            abase + swz64((row * BK + ks * 32 + fk) * 2));
      }
#pragma unroll
    for (int j = 0; j < FN; ++j)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int brow = wn * (BN / 2) + j * 16 + fr;
        b[j][ks] = *reinterpret_cast<const bf16x8*>(
            bbase + swz64((brow * BK + ks * 32 + fk) * 2));
      }
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int j = 0; j < FN; ++j)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[i][ks], b[j][ks], acc[i][j], 0, 0, 0);
    stage((t + 2) % 3, t + 2);
    // own t+1 chunks landed (t+2's CPW stay in flight); cross-wave
    // visibility via the raw barrier — __syncthreads() would emit
    // vmcnt(0) and drain the span (guide pitfall)
    asm volatile("s_waitcnt vmcnt(%0)" :: "n"(CPW) : "memory");
    __builtin_amdgcn_s_barrier();
  }

  // drain the clamped tail prefetches before reusing smem
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
  constexpr int SEGN = BN / 2;
  if (OUT_F32 && n0 + BN <= N && m0 + BM <= M) {
    // staged dwordx4 nontemporal epilogue (see the generic tile)
    float* stage_f32 = reinterpret_cast<float*>(&smem[0][0])
        + wid * 32 * SEGN;
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int j = 0; j < FN; ++j)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int lm = i * 16 + crow_base + r;
          const int ln = j * 16 + ccol;
          const float bv = bias ? bias[n0 + wn * SEGN + ln] : 0.0f;
          stage_f32[lm * SEGN + ln] =
              apply_act(alpha * acc[i][j][r] + bv, act);
        }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();
    constexpr int LPR = SEGN / 4;
    constexpr int RPI = 64 / LPR;
    const int srow = lane / LPR, scol4 = (lane % LPR) * 4;
#pragma unroll
    for (int base = 0; base < 32; base += RPI) {
      const int lm = base + srow;
      const int m = m0 + wm * 32 + lm;
      const int n = n0 + wn * SEGN + scol4;
      f32x4 v = *reinterpret_cast<const f32x4*>(
          &stage_f32[lm * SEGN + scol4]);
      __builtin_nontemporal_store(
          v, reinterpret_cast<f32x4*>(
              reinterpret_cast<float*>(C)
              + (long)g * strideC + (long)m * N + n));
    }
    return;
  }
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < FN; ++j) {
      const int n = n0 + wn * SEGN + j * 16 + ccol;
      if (n >= N) continue;
      const float bv = bias ? bias[n] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + wm * 32 + i * 16 + crow_base + r;
        if (m >= M) continue;
        float v = apply_act(alpha * acc[i][j][r] + bv, act);
        if (OUT_F32)
          reinterpret_cast<float*>(C)[(long)g * strideC + (long)m * N + n] = v;
        else
          reinterpret_cast<bf16*>(C)[(long)g * strideC + (long)m * N + n] =
              f2bf(v);
      }
    }
  }
}

}  // namespace

extern "C" void infomesh_gemm_bf16_nt(
    const void* A, const void* B, void* C, const void* bias,
    int M, int N, int K, int batch,
    long strideA, long strideB, long strideC,
    int act, float alpha, int out_f32, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  const bool bk64 = (K % 64 == 0);
  int bm = (M <= 64) ? 64 : 128, bn = 128;
  // Latency-bound regime: few K-steps and not enough 128^2 tiles to
  // fill the chip -> quarter tiles for 4x the block-level overlap.
  const long blocks128 =
      (long)((M + 127) / 128) * ((N + 127) / 128) * batch;
  if (bm == 128 && blocks128 < 512 && K <= 2048) bm = bn = 64;
  // Tuning override (read once): INFOMESH_GEMM_TILE=64|128 forces the
  // tile; scripts/gemm_tile_probe.py uses it to validate the heuristic.
  static const int ov = [] {
    const char* e = getenv("INFOMESH_GEMM_TILE");
    return e ? atoi(e) : -1;
  }();
  if (ov == 64 && M > 64) bm = bn = 64;
  else if (ov == 128) { bm = 128; bn = 128; }
  const int tiles = ((M + bm - 1) / bm) * ((N + bn - 1) / bn);
  dim3 grid(tiles, batch);
  dim3 block(256);
#define LAUNCH(BMV, BNV, BKV, OF)                                            \
  hipLaunchKernelGGL((gemm_bf16_nt_kernel<BMV, BNV, BKV, OF>), grid, block,  \
                     0, s, (const bf16*)A, (const bf16*)B, C,                \
                     (const float*)bias, M, N, K, strideA, strideB,          \
                     strideC, act, alpha)
#define PICK(BMV, BNV)                                                       \
  do {                                                                       \
    if (bk64) { if (out_f32) LAUNCH(BMV, BNV, 64, true);                     \
                else LAUNCH(BMV, BNV, 64, false); }                          \
    else      { if (out_f32) LAUNCH(BMV, BNV, 32, true);                     \
                else LAUNCH(BMV, BNV, 32, false); }                          \
  } while (0)
  // (BK=128 for the 64^2 tile was probe-tested and is ~60% SLOWER:
  // doubling LDS to 64 KB halves resident blocks per CU, which costs
  // more latency hiding than the halved K-step drains save.)
  static const int pipe_ov = [] {
    const char* e = getenv("INFOMESH_GEMM_PIPE");
    return e ? atoi(e) : 1;
  }();
  if (bm == 64 && bn == 64 && bk64 && pipe_ov) {
    // latency-bound regime: 3-buffer glds-span pipeline; BN=128 when N
    // has whole 128-tiles (pipe_ov=2 forces BN=64, =3 forces BN=128)
    // BN=128 measured SLOWER on the encoder shapes (1278 vs 990 us
    // full-encoder: 2 WGs/CU vs 3 loses more than wider MFMA gains);
    // kept behind =3 for future shapes
    const bool wide = (pipe_ov == 3);
    if (wide) {
      const int tiles_w = ((M + 63) / 64) * ((N + 127) / 128);
      dim3 gw(tiles_w, batch);
      if (out_f32)
        hipLaunchKernelGGL((gemm_pipe64_kernel<128, true>), gw, block,
                           0, s, (const bf16*)A, (const bf16*)B, C,
                           (const float*)bias, M, N, K, strideA,
                           strideB, strideC, act, alpha);
      else
        hipLaunchKernelGGL((gemm_pipe64_kernel<128, false>), gw, block,
                           0, s, (const bf16*)A, (const bf16*)B, C,
                           (const float*)bias, M, N, K, strideA,
                           strideB, strideC, act, alpha);
    } else if (out_f32)
      hipLaunchKernelGGL((gemm_pipe64_kernel<64, true>), grid, block, 0, s,
                         (const bf16*)A, (const bf16*)B, C,
                         (const float*)bias, M, N, K, strideA, strideB,
                         strideC, act, alpha);
    else
      hipLaunchKernelGGL((gemm_pipe64_kernel<64, false>), grid, block, 0, s,
                         (const bf16*)A, (const bf16*)B, C,
                         (const float*)bias, M, N, K, strideA, strideB,
                         strideC, act, alpha);
  }
  else if (bm == 64 && bn == 64) PICK(64, 64);
  else if (bm == 64) PICK(64, 128);
  else PICK(128, 128);
#undef PICK
#undef LAUNCH
}
