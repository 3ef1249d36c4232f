// Masked row softmax for attention scores: f32 in -> bf16 out (PV input).
// Supports causal masking and per-group valid-length masking.
// One block per row; online not needed (row fits in a grid-stride pass).
// Replaces: torch softmax in the reference's model stacks (attention
// softmax inside sentence-transformers, infomesh/index/vector_store.py
// :104-118); the fused attention path subsumes it -- this kernel serves
// the decomposed fallback and tests.
#include "common.h"

namespace {

// scores: [G, Sq, Sk] f32, out bf16 same shape.
// causal: mask j > i + (Sk - Sq)  (standard causal offset for prefill).
// valid_len: per-g valid key length (j >= valid -> masked), or nullptr.
__global__ __launch_bounds__(256) void softmax_kernel(
    const float* __restrict__ scores, bf16* __restrict__ out,
    const int* __restrict__ valid_len,
    int Sq, int Sk, int causal, float scale) {
  __shared__ float scratch[16];
  const long row = blockIdx.x;           // g * Sq + i
  const int g = (int)(row / Sq);
  const int i = (int)(row % Sq);
  const float* sr = scores + row * Sk;
  bf16* orow = out + row * Sk;
  int limit = Sk;
  if (causal) limit = min(limit, i + (Sk - Sq) + 1);
  if (valid_len) limit = min(limit, valid_len[g]);

  float mx = -INFINITY;
  for (int j = threadIdx.x; j < limit; j += blockDim.x)
    mx = fmaxf(mx, sr[j] * scale);
  mx = block_reduce_max(mx, scratch);
  __syncthreads();
  float sum = 0.f;
  for (int j = threadIdx.x; j < limit; j += blockDim.x)
    sum += __expf(sr[j] * scale - mx);
  sum = block_reduce_sum(sum, scratch);
  const float inv = (sum > 0.f) ? 1.0f / sum : 0.f;
  for (int j = threadIdx.x; j < Sk; j += blockDim.x)
    orow[j] = f2bf(j < limit ? __expf(sr[j] * scale - mx) * inv : 0.f);
}

}  // namespace

extern "C" void infomesh_softmax(
    const void* scores, void* out, const void* valid_len,
    long G, int Sq, int Sk, int causal, float scale, void* stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  dim3 grid((unsigned)(G * Sq)), block(256);
  hipLaunchKernelGGL(softmax_kernel, grid, block, 0, s,
                     (const float*)scores, (bf16*)out, (const int*)valid_len,
                     Sq, Sk, causal, scale);
}
