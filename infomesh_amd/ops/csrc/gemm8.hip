// Deep-pipelined 256x256 bf16 MFMA GEMM (gfx950) for large projection
// shapes — the guide's "256² 8-phase" class structure (§5 template):
// 8 waves (2M x 4N), BK=64, double-buffered LDS staged by
// global_load_lds, COUNTED vmcnt across tile boundaries (loads stay in
// flight through barriers), st_16x32 LDS swizzle on the staging SOURCE
// address + ds_read offsets (rule 21), setprio around MFMA clusters.
//
// Half-tile schedule (2 glds per wave per half; halves h0,h1 = A rows
// 0-127 / 128-255; h2,h3 = B rows 0-127 / 128-255; all four halves of
// tile T land in buf[T&1]):
//   prologue: stage t0.h0..h3, t1.h0,h1  -> vmcnt(4), barrier
//   per tile t (reading buf[cur]):
//     P1: ds_read frag set 1 | stage t+1.h2 -> buf^1 | bar | MFMA Q0,Q1
//     P2: ds_read frag set 2 | stage t+1.h3 -> buf^1 | bar | MFMA Q2,Q3
//         (after P2's barrier every wave has consumed buf[cur])
//     P3: stage t+2.h0 -> buf[cur] | MFMA Q4,Q5   (frags in registers)
//     P4: stage t+2.h1 -> buf[cur] | MFMA Q6,Q7
//     vmcnt(4) + barrier   (own t+1 halves landed; cross-wave via bar)
// Out-of-range prefetch steps clamp to the last K-tile (harmless
// redundant loads) so the vmcnt counts stay static.
// Replaces (with gemm.hip): the reference's large matmuls inside
// sentence-transformers/cross-encoder models (infomesh/index/
// vector_store.py:120-157) at the shapes where the 128^2 tile is
// pipeline-bound.
#include "common.h"
#include <cstdlib>

#define G8_BM 256
#define G8_BN 256
#define G8_BK 64

namespace {

// LDS swizzle on a [rows][64 bf16] row-major image (involution, 16 B
// granules): xor the 16 B-chunk index (bits 4-6) with
// h(row) = (row ^ (row>>3)) & 7 so the 16 rows of a quarter-wave MFMA
// operand read land on 16 distinct 4-bank windows (the old single-bit
// flip left rows r/r+2 colliding — see gemm.hip swz64 for the bank
// arithmetic and the measured outcome: conflict counters/wall time are
// epilogue- and latency-dominated, so this is hygiene, not a win).
// Key uses only bits >=7, untouched by the xor, so swz(swz(x)) == x
// and region bases (bit 13+) pass through.
DEVINL int swz(int byte_off) {
  return byte_off ^ ((((byte_off >> 7) ^ (byte_off >> 10)) & 7) << 4);
}

template <bool OUT_F32>
__global__ __launch_bounds__(512, 1) void gemm8_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    void* __restrict__ C, const float* __restrict__ bias,
    int M, int N, int K, int ldc,
    long strideA, long strideB, long strideC,
    int act, float alpha) {
  // one buffer = A tile (256x64) + B tile (256x64) = 64 KiB
  __shared__ bf16 smem[2][(G8_BM + G8_BN) * G8_BK];

  const int tiles_n = (N + G8_BN - 1) / G8_BN;
  const int tiles_m = (M + G8_BM - 1) / G8_BM;
  const int tile_id = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int m0 = (tile_id / tiles_n) * G8_BM;
  const int n0 = (tile_id % tiles_n) * G8_BN;
  const int g = blockIdx.y;
  const bf16* Ag = A + (long)g * strideA;
  const bf16* Bg = B + (long)g * strideB;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;        // 8 waves
  const int wm = wid >> 2, wn = wid & 3;   // 2 x 4 wave grid

  const int n_ksteps = K / G8_BK;

  // ---- staging: one half (128 rows x 64 cols bf16 = 16 KiB) is 16
  // glds instructions; per wave 2. glds writes lane-linear 16 B: lane's
  // LDS bytes = chunkbase + lane*16. Swizzle goes on the SOURCE column.
  // half-chunk c (0..15): rows c*8 .. c*8+8; within: lane/8 row,
  // (lane%8)*16 byte col (8 bf16).
  auto stage_half = [&](int buf, int kstep, int half) {
    const int ks = kstep < n_ksteps ? kstep : n_ksteps - 1;  // clamp
    const long k0 = (long)ks * G8_BK;
    const bool is_b = half >= 2;
    const int row_base = (half & 1) * 128;
    // region base in BYTES: A at 0, B at BM*BK*2; half 1 at +16 KiB
    const long region = (is_b ? (long)G8_BM * G8_BK * 2 : 0) +
                        (long)row_base * 128;
#pragma unroll
    for (int c2 = 0; c2 < 2; ++c2) {
      const int chunk = wid * 2 + c2;               // 0..15
      // lane-linear LDS byte within the half; glds writes base+lane*16
      const int lin_byte = chunk * 1024 + lane * 16;
      // source position under the involution (rule 21: swizzle the
      // per-lane GLOBAL address, keep the LDS destination linear)
      const int s_byte = swz(lin_byte);
      const int lrow = s_byte / 128;                // 0..127 in half
      const int colb = s_byte % 128;                // byte col in row
      int grow = (is_b ? n0 : m0) + row_base + lrow;
      const int lim = is_b ? N : M;
      grow = grow < lim ? grow : lim - 1;
      const bf16* gsrc = (is_b ? Bg : Ag) + (long)grow * K + k0 + colb / 2;
      auto* dst = (__attribute__((address_space(3))) unsigned int*)
          ((char*)&smem[buf][0] + region + (long)chunk * 1024);
      // A (M up to ~512k rows) streams through exactly once; B
      // (N*K*2 <= a few MB at the shapes this kernel serves) wants to
      // stay in the XCD's 4 MB L2 across m-tiles. Marking the A loads
      // non-temporal (CPol nt, aux=2) stops the A stream from evicting
      // B — without it B thrashes and the kernel runs at the
      // read-everything-from-HBM roofline (~700 GF/s at the reranker
      // shapes; torch/rocBLAS was 2x faster purely on this reuse).
      if (is_b)
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)gsrc,
            dst, 16, 0, 0);
      else
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)gsrc,
            dst, 16, 0, /*nt*/ 2);
    }
  };

  // ---- fragment reads (swizzled): A frag (mi-quadrant row r, ks):
  // row = wm*128 + fi*16 + (lane&15); byte = row*128 + (ks*32 +
  // (lane>>4)*8)*2, swizzled.
  const int fr = lane & 15;
  const int fkb = (lane >> 4) * 16;        // byte offset of k slice

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // prologue: t0 all 4 halves + t1 h0,h1
  stage_half(0, 0, 0); stage_half(0, 0, 1);
  stage_half(0, 0, 2); stage_half(0, 0, 3);
  stage_half(1, 1, 0); stage_half(1, 1, 1);
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  int cur = 0;
  for (int t = 0; t < n_ksteps; ++t) {
    const char* abase = (const char*)&smem[cur][0];
    const char* bbase = abase + (long)G8_BM * G8_BK * 2;
    // ---- P1: read afr (A mi 0..3) + bfr (B ni 0..1); stage t+1.h2;
    //          MFMA Q0 = afr x bfr
    bf16x8 afr[4][2], bfr[2][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int row = wm * 128 + i * 16 + fr;
        afr[i][ks] = *reinterpret_cast<const bf16x8*>(
            abase + swz(row * 128 + ks * 64 + fkb));
      }
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int row = wn * 64 + j * 16 + fr;
        bfr[j][ks] = *reinterpret_cast<const bf16x8*>(
            bbase + swz(row * 128 + ks * 64 + fkb));
      }
    stage_half(cur ^ 1, t + 1, 2);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i][ks], bfr[j][ks], acc[i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // ---- P2: read bfr2 (B ni 2..3); stage t+1.h3; MFMA Q1 = afr x bfr2
    bf16x8 bfr2[2][2];
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int row = wn * 64 + (j + 2) * 16 + fr;
        bfr2[j][ks] = *reinterpret_cast<const bf16x8*>(
            bbase + swz(row * 128 + ks * 64 + fkb));
      }
    stage_half(cur ^ 1, t + 1, 3);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[i][j + 2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i][ks], bfr2[j][ks], acc[i][j + 2], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // ---- P3: read afr2 (A mi 4..7, reusing afr's registers — afr is
    //          dead); MFMA Q2 = afr2 x bfr
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int row = wm * 128 + (i + 4) * 16 + fr;
        afr[i][ks] = *reinterpret_cast<const bf16x8*>(
            abase + swz(row * 128 + ks * 64 + fkb));
      }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[i + 4][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i][ks], bfr[j][ks], acc[i + 4][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // ---- P4: ALL buf[cur] reads done -> barrier kills the buffer,
    //          stage t+2.h0+h1 into it; MFMA Q3 = afr2 x bfr2
    __builtin_amdgcn_s_barrier();
    stage_half(cur, t + 2, 0);
    stage_half(cur, t + 2, 1);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[i + 4][j + 2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i][ks], bfr2[j][ks], acc[i + 4][j + 2], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // boundary: own t+1 halves (all issued before the last 4 glds)
    // landed; cross-wave visibility via the barrier.
    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    cur ^= 1;
  }

  // ---- epilogue
  // bf16 full tiles: stage each wave's 128x64 slab through LDS and
  // emit 16-B dwordx4 NONTEMPORAL stores. At short K (encoder /
  // reranker projections, K<=3072) the C write is a large share of
  // the kernel's traffic, and the scalar 2-B stores ran it at ~1 TB/s
  // — measured 2.2x behind the vendor library at M=512k,K=768 shapes
  // before this (profiles/r02_summary.md). The in-flight tail
  // prefetches must drain first (they write the smem we re-use).
  const int crow0 = (lane >> 4) * 4;
  const int ccol = lane & 15;
  if (!OUT_F32 && m0 + G8_BM <= M && n0 + G8_BN <= N) {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    bf16* stage_bf = reinterpret_cast<bf16*>(&smem[0][0])
        + (long)wid * 128 * 64;
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int ln = j * 16 + ccol;
        const float bv = bias ? bias[n0 + wn * 64 + ln] : 0.0f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int lm = i * 16 + crow0 + r;
          stage_bf[lm * 64 + ln] =
              f2bf(apply_act(alpha * acc[i][j][r] + bv, act));
        }
      }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_wave_barrier();
    const int srow = lane >> 3, scol8 = (lane & 7) * 8;  // 8 lanes/row
#pragma unroll
    for (int base = 0; base < 128; base += 8) {
      const int lm = base + srow;
      const int m = m0 + wm * 128 + lm;
      const int n = n0 + wn * 64 + scol8;
      bf16x8 v = *reinterpret_cast<const bf16x8*>(
          &stage_bf[lm * 64 + scol8]);
      __builtin_nontemporal_store(
          v, reinterpret_cast<bf16x8*>(
              reinterpret_cast<bf16*>(C)
              + (long)g * strideC + (long)m * ldc + n));
    }
    return;
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int n = n0 + wn * 64 + j * 16 + ccol;
      if (n >= N) continue;
      const float bv = bias ? bias[n] : 0.0f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int m = m0 + wm * 128 + i * 16 + crow0 + r;
        if (m >= M) continue;
        float v = apply_act(alpha * acc[i][j][r] + bv, act);
        if (OUT_F32)
          reinterpret_cast<float*>(C)[(long)g * strideC + (long)m * ldc + n] = v;
        else
          reinterpret_cast<bf16*>(C)[(long)g * strideC + (long)m * ldc + n] =
              f2bf(v);
      }
    }
  }
}

// ---- persistent-tile variant -------------------------------------------
// At short K (encoder/reranker projections: 6-48 K-tiles) the 8-phase
// schedule's prologue (6 half-stages) + drain serialize per tile — and
// at 1 block/CU there is no second block to overlap them, costing ~2x
// vs the vendor library at K=768 (profiles/r02_summary.md). Here ONE
// resident block per CU walks a contiguous range of tiles and the
// (tile, kstep) sequence is flattened into one absolute step stream:
// the double-buffer rotation, counted vmcnt waits and raw barriers
// carry straight across tile boundaries, so the pipeline ramps once
// per RANGE instead of once per tile. The epilogue stays scalar (the
// LDS staging would force a vmcnt(0) drain mid-stream); its stores
// ride the same vm counter and retire during the next tile's MFMAs.
template <bool OUT_F32>
__global__ __launch_bounds__(512, 1) void gemm8p_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    void* __restrict__ C, const float* __restrict__ bias,
    int M, int N, int K, int ldc,
    long strideA, long strideB, long strideC,
    int act, float alpha) {
  __shared__ bf16 smem[2][(G8_BM + G8_BN) * G8_BK];

  const int tiles_n = (N + G8_BN - 1) / G8_BN;
  const int tiles_m = (M + G8_BM - 1) / G8_BM;
  const int tiles_total = tiles_m * tiles_n;
  const int per = (tiles_total + gridDim.x - 1) / gridDim.x;
  const int t_begin = blockIdx.x * per;
  const int t_end = min(t_begin + per, tiles_total);
  if (t_begin >= tiles_total) return;
  const int g = blockIdx.y;
  const bf16* Ag = A + (long)g * strideA;
  const bf16* Bg = B + (long)g * strideB;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wm = wid >> 2, wn = wid & 3;
  const int n_ksteps = K / G8_BK;
  const long S_total = (long)(t_end - t_begin) * n_ksteps;

  // absolute step s -> (m0, n0, kstep); prefetch overruns clamp to the
  // final step so the vmcnt counts stay static
  auto stage_half = [&](int buf, long s, int half) {
    if (s >= S_total) s = S_total - 1;
    const int t = t_begin + (int)(s / n_ksteps);
    const long k0 = (long)(s % n_ksteps) * G8_BK;
    const int m0 = (t / tiles_n) * G8_BM;
    const int n0 = (t % tiles_n) * G8_BN;
    const bool is_b = half >= 2;
    const int row_base = (half & 1) * 128;
    const long region = (is_b ? (long)G8_BM * G8_BK * 2 : 0) +
                        (long)row_base * 128;
#pragma unroll
    for (int c2 = 0; c2 < 2; ++c2) {
      const int chunk = wid * 2 + c2;
      const int lin_byte = chunk * 1024 + lane * 16;
      const int s_byte = swz(lin_byte);
      const int lrow = s_byte / 128;
      const int colb = s_byte % 128;
      int grow = (is_b ? n0 : m0) + row_base + lrow;
      const int lim = is_b ? N : M;
      grow = grow < lim ? grow : lim - 1;
      const bf16* gsrc = (is_b ? Bg : Ag) + (long)grow * K + k0 + colb / 2;
      auto* dst = (__attribute__((address_space(3))) unsigned int*)
          ((char*)&smem[buf][0] + region + (long)chunk * 1024);
      // A (M up to ~512k rows) streams through exactly once; B
      // (N*K*2 <= a few MB at the shapes this kernel serves) wants to
      // stay in the XCD's 4 MB L2 across m-tiles. Marking the A loads
      // non-temporal (CPol nt, aux=2) stops the A stream from evicting
      // B — without it B thrashes and the kernel runs at the
      // read-everything-from-HBM roofline (~700 GF/s at the reranker
      // shapes; torch/rocBLAS was 2x faster purely on this reuse).
      if (is_b)
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)gsrc,
            dst, 16, 0, 0);
      else
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) unsigned int*)gsrc,
            dst, 16, 0, /*nt*/ 2);
    }
  };

  const int fr = lane & 15;
  const int fkb = (lane >> 4) * 16;
  const int crow0 = (lane >> 4) * 4;
  const int ccol = lane & 15;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  stage_half(0, 0, 0); stage_half(0, 0, 1);
  stage_half(0, 0, 2); stage_half(0, 0, 3);
  stage_half(1, 1, 0); stage_half(1, 1, 1);
  asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
  __builtin_amdgcn_s_barrier();

  int cur = 0;
  for (long s = 0; s < S_total; ++s) {
    const char* abase = (const char*)&smem[cur][0];
    const char* bbase = abase + (long)G8_BM * G8_BK * 2;
    bf16x8 afr[4][2], bfr[2][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int row = wm * 128 + i * 16 + fr;
        afr[i][ks] = *reinterpret_cast<const bf16x8*>(
            abase + swz(row * 128 + ks * 64 + fkb));
      }
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int row = wn * 64 + j * 16 + fr;
        bfr[j][ks] = *reinterpret_cast<const bf16x8*>(
            bbase + swz(row * 128 + ks * 64 + fkb));
      }
    stage_half(cur ^ 1, s + 1, 2);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i][ks], bfr[j][ks], acc[i][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    bf16x8 bfr2[2][2];
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int row = wn * 64 + (j + 2) * 16 + fr;
        bfr2[j][ks] = *reinterpret_cast<const bf16x8*>(
            bbase + swz(row * 128 + ks * 64 + fkb));
      }
    stage_half(cur ^ 1, s + 1, 3);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[i][j + 2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i][ks], bfr2[j][ks], acc[i][j + 2], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        const int row = wm * 128 + (i + 4) * 16 + fr;
        afr[i][ks] = *reinterpret_cast<const bf16x8*>(
            abase + swz(row * 128 + ks * 64 + fkb));
      }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[i + 4][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i][ks], bfr[j][ks], acc[i + 4][j], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    __builtin_amdgcn_s_barrier();
    stage_half(cur, s + 2, 0);
    stage_half(cur, s + 2, 1);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int ks = 0; ks < 2; ++ks)
          acc[i + 4][j + 2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i][ks], bfr2[j][ks], acc[i + 4][j + 2], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    cur ^= 1;

    if ((s + 1) % n_ksteps == 0) {
      // tile finished: scalar epilogue (no smem use — the pipeline's
      // prefetches for the NEXT tile stay in flight), reset acc
      const int t = t_begin + (int)(s / n_ksteps);
      const int m0 = (t / tiles_n) * G8_BM;
      const int n0 = (t % tiles_n) * G8_BN;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int n = n0 + wn * 64 + j * 16 + ccol;
          if (n >= N) continue;
          const float bv = bias ? bias[n] : 0.0f;
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int m = m0 + wm * 128 + i * 16 + crow0 + r;
            if (m >= M) continue;
            float v = apply_act(alpha * acc[i][j][r] + bv, act);
            if (OUT_F32)
              reinterpret_cast<float*>(C)[
                  (long)g * strideC + (long)m * ldc + n] = v;
            else
              reinterpret_cast<bf16*>(C)[
                  (long)g * strideC + (long)m * ldc + n] = f2bf(v);
          }
          acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
        }
      }
    }
  }
}

}  // namespace

extern "C" void infomesh_gemm8_bf16_nt(
    const void* A, const void* B, void* C, const void* bias,
    int M, int N, int K, int batch,
    long strideA, long strideB, long strideC,
    int act, float alpha, int out_f32, void* stream) {
  const int tiles = ((M + G8_BM - 1) / G8_BM) * ((N + G8_BN - 1) / G8_BN);
  auto s = reinterpret_cast<hipStream_t>(stream);
  // Short-K regime: per-tile prologue/drain dominates at 1 block/CU —
  // the persistent variant carries the pipeline across tiles.
  // INFOMESH_GEMM8P=0 disables, =1 forces for all K.
  static const int p_ov = [] {
    const char* e = getenv("INFOMESH_GEMM8P");
    return e ? atoi(e) : -1;
  }();
  const int n_ksteps = K / G8_BK;
  // Measured WORSE than per-tile blocks at M=512k,K=768 (495 vs
  // 564 GF/s): the scalar epilogue's stores ride the same vm counter
  // as the glds span, so every post-boundary vmcnt(4) drains them —
  // stalling the pipeline it was meant to keep alive — and the
  // contiguous chunk assignment loses xcd_swizzle's L2 locality.
  // Kept behind INFOMESH_GEMM8P=1 for future reruns; default OFF.
  const bool persistent = (p_ov == 1) && n_ksteps >= 1 && tiles >= 1;
  if (persistent) {
    dim3 grid((unsigned)(tiles < 256 ? tiles : 256), batch);
    if (out_f32)
      hipLaunchKernelGGL(gemm8p_kernel<true>, grid, dim3(512), 0, s,
                         (const bf16*)A, (const bf16*)B, C,
                         (const float*)bias, M, N, K, N, strideA, strideB,
                         strideC, act, alpha);
    else
      hipLaunchKernelGGL(gemm8p_kernel<false>, grid, dim3(512), 0, s,
                         (const bf16*)A, (const bf16*)B, C,
                         (const float*)bias, M, N, K, N, strideA, strideB,
                         strideC, act, alpha);
    return;
  }
  // N-chunking for over-L2 B panels (default OFF — INFOMESH_GEMM8_CHUNK=1;
  // designed/unvalidated this round, see BACKLOG): when the B panel
  // (N*K*2 bytes) exceeds the 4 MB per-XCD L2 even the NT-A stream
  // cannot keep it resident; splitting N into chunks whose slice fits
  // ~2.5 MB trades (n_chunks-1) extra reads of A for eliminating the
  // B thrash. Only applied when that trade is favorable.
  // read per call (a getenv is ~100 ns vs the 100s-of-us GEMMs it
  // gates): a first-call-latched static made the flag unusable from
  // code that builds the extension before configuring the env
  const char* chunk_env = getenv("INFOMESH_GEMM8_CHUNK");
  const int chunk_ov = chunk_env ? atoi(chunk_env) : 0;
  const long b_bytes = (long)N * K * 2;
  if (chunk_ov && b_bytes > 3 * 1024 * 1024 && batch == 1) {
    int cols = (int)(2.5 * 1024 * 1024 / (K * 2)) / G8_BN * G8_BN;
    if (cols >= G8_BN) {
      const int nch = (N + cols - 1) / cols;
      const long a_bytes = (long)M * K * 2;
      // extra A reads must undercut the avoided B re-reads (one per
      // 512 rows of m-tiles is a conservative thrash estimate)
      if ((long)(nch - 1) * a_bytes < b_bytes * (M / 512)) {
        for (int c0 = 0; c0 < N; c0 += cols) {
          const int nc = (N - c0) < cols ? (N - c0) : cols;
          const int t = ((M + G8_BM - 1) / G8_BM) * ((nc + G8_BN - 1) / G8_BN);
          dim3 gc(t, 1);
          const bf16* Bc = (const bf16*)B + (long)c0 * K;
          const float* bc = bias ? (const float*)bias + c0 : nullptr;
          if (out_f32)
            hipLaunchKernelGGL(gemm8_kernel<true>, gc, dim3(512), 0, s,
                               (const bf16*)A, Bc,
                               (char*)C + (long)c0 * 4, bc,
                               M, nc, K, N, strideA, strideB, strideC,
                               act, alpha);
          else
            hipLaunchKernelGGL(gemm8_kernel<false>, gc, dim3(512), 0, s,
                               (const bf16*)A, Bc,
                               (char*)C + (long)c0 * 2, bc,
                               M, nc, K, N, strideA, strideB, strideC,
                               act, alpha);
        }
        return;
      }
    }
  }
  dim3 grid(tiles, batch), block(512);
  if (out_f32)
    hipLaunchKernelGGL(gemm8_kernel<true>, grid, block, 0, s,
                       (const bf16*)A, (const bf16*)B, C,
                       (const float*)bias, M, N, K, N, strideA, strideB,
                       strideC, act, alpha);
  else
    hipLaunchKernelGGL(gemm8_kernel<false>, grid, block, 0, s,
                       (const bf16*)A, (const bf16*)B, C,
                       (const float*)bias, M, N, K, N, strideA, strideB,
                       strideC, act, alpha);
}
