// Common helpers for infomesh-amd CDNA4 (gfx950) kernels.
// Hand-written HIP for MI355X only: wave64, MFMA, 160 KiB LDS, 8 XCDs.
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

#define WAVE 64
#define DEVINL __device__ __forceinline__

typedef __bf16 bf16;
typedef __bf16 bf16x2 __attribute__((ext_vector_type(2)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x2 __attribute__((ext_vector_type(2)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef short s16x8 __attribute__((ext_vector_type(8)));
typedef unsigned int u32x4 __attribute__((ext_vector_type(4)));

DEVINL float bf2f(bf16 v) { return __bfloat162float(*reinterpret_cast<__hip_bfloat16*>(&v)); }
DEVINL bf16 f2bf(float v) {
  __hip_bfloat16 h = __float2bfloat16(v);
  return *reinterpret_cast<bf16*>(&h);
}

// ------------------------------------------------------------ wave reduce
DEVINL float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}
DEVINL float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// Block reduce via LDS (callers supply a __shared__ float[16] scratch).
DEVINL float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  const int nw = (blockDim.x + WAVE - 1) / WAVE;
  v = (threadIdx.x < nw) ? scratch[threadIdx.x] : 0.0f;
  if (wid == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
    if (lane == 0) scratch[0] = v;
  }
  __syncthreads();
  return scratch[0];
}
DEVINL float block_reduce_max(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_reduce_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  const int nw = (blockDim.x + WAVE - 1) / WAVE;
  v = (threadIdx.x < nw) ? scratch[threadIdx.x] : -INFINITY;
  if (wid == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
    if (lane == 0) scratch[0] = v;
  }
  __syncthreads();
  return scratch[0];
}

// -------------------------------------------------------------- epilogue
// Activation codes shared with the Python wrapper (ops/_ext.py).
#define ACT_NONE 0
#define ACT_GELU 1
#define ACT_SILU 2
#define ACT_RELU 3
#define ACT_TANH 4

DEVINL float apply_act(float x, int act) {
  switch (act) {
    case ACT_GELU: {  // tanh approximation (BERT/GPT standard)
      const float c = 0.7978845608028654f;  // sqrt(2/pi)
      return 0.5f * x * (1.0f + tanhf(c * (x + 0.044715f * x * x * x)));
    }
    case ACT_SILU: return x / (1.0f + __expf(-x));
    case ACT_RELU: return fmaxf(x, 0.0f);
    case ACT_TANH: return tanhf(x);
    default: return x;
  }
}

// ---------------------------------------------------------- misc helpers
DEVINL int cdiv_i(int a, int b) { return (a + b - 1) / b; }
constexpr int cdiv_c(int a, int b) { return (a + b - 1) / b; }

// Ordered-float mapping: monotone bijection f32 -> u32 (for radix select).
DEVINL uint32_t float_to_ordered(float f) {
  uint32_t u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}
DEVINL float ordered_to_float(uint32_t u) {
  uint32_t v = (u & 0x80000000u) ? (u & 0x7fffffffu) : ~u;
  return __uint_as_float(v);
}

// XCD-aware bijective blockIdx remap (guide §5.5 T1): contiguous chunks
// per XCD so neighbouring tiles share an L2.
DEVINL int xcd_swizzle(int bid, int nwg) {
  const int NXCD = 8;
  if (nwg < NXCD) return bid;
  int q = nwg / NXCD, r = nwg % NXCD;
  int xcd = bid % NXCD, idx = bid / NXCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

#define HIP_CHECK_LAUNCH() do { } while (0)
