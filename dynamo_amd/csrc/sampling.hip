// Token sampling kernels, MI355X-native.
//
// greedy: fused argmax over the vocab.
// gumbel: exact softmax(logits/T) sampling via the Gumbel-max trick —
//   argmax(logits/T + G_i), G_i = -log(-log(U_i)) with a counter-based
//   in-kernel hash RNG (deterministic given seed; no host noise tensor).
// Both are single-pass, vectorized, one workgroup per sequence.
//
// Capability parity: the reference delegates sampling to its engines; this
// is the native path (top-p/top-k currently handled at the Python layer).
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr int kBlock = 512;

DEVINL uint64_t hash_u64(uint64_t x) {
  // splitmix64 finalizer
  x += 0x9e3779b97f4a7c15ull;
  x = (x ^ (x >> 30)) * 0xbf58476d1ce4e5b9ull;
  x = (x ^ (x >> 27)) * 0x94d049bb133111ebull;
  return x ^ (x >> 31);
}

template <bool GUMBEL>
__global__ void sample_kernel(int32_t* __restrict__ out,          // [B]
                              const float* __restrict__ logits,   // [B, V]
                              const float* __restrict__ inv_temp, // [B] or null
                              uint64_t seed, int V) {
  const int b = blockIdx.x;
  const float* row = logits + (int64_t)b * V;
  const float it = GUMBEL ? inv_temp[b] : 1.f;

  float best = -1e38f;
  int besti = -1;
  for (int v = threadIdx.x; v < V; v += kBlock) {
    float x = row[v] * it;
    if constexpr (GUMBEL) {
      const uint64_t h = hash_u64(seed ^ ((uint64_t)b << 32) ^ (uint64_t)v);
      // uniform in (0,1): use top 53 bits
      const float u = (float)((h >> 11) + 1) * 4.8828125e-4f * 2.2737367544323206e-13f;
      x += -__logf(-__logf(u));
    }
    if (x > best || (x == best && v < besti)) { best = x; besti = v; }
  }
  // wave reduce (value, index)
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ob = __shfl_xor(best, off, WAVE_SIZE);
    const int oi = __shfl_xor(besti, off, WAVE_SIZE);
    if (ob > best || (ob == best && oi < besti)) { best = ob; besti = oi; }
  }
  __shared__ float sb[kBlock / WAVE_SIZE];
  __shared__ int si[kBlock / WAVE_SIZE];
  const int wid = threadIdx.x / WAVE_SIZE;
  if ((threadIdx.x & 63) == 0) { sb[wid] = best; si[wid] = besti; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < kBlock / WAVE_SIZE; w++) {
      if (sb[w] > best || (sb[w] == best && si[w] < besti)) { best = sb[w]; besti = si[w]; }
    }
    out[b] = besti;
  }
}

}  // namespace

void greedy_sample(torch::Tensor out, torch::Tensor logits) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kFloat32);
  TORCH_CHECK(out.dtype() == torch::kInt32);
  const int B = logits.size(0);
  const int V = logits.size(1);
  if (B == 0) return;
  auto stream = at::cuda::getCurrentHIPStream();
  sample_kernel<false><<<B, kBlock, 0, stream>>>(
      out.data_ptr<int32_t>(), logits.data_ptr<float>(), nullptr, 0, V);
  HIP_CHECK_KERNEL();
}

void gumbel_sample(torch::Tensor out, torch::Tensor logits, torch::Tensor inv_temp,
                   int64_t seed) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kFloat32);
  TORCH_CHECK(inv_temp.dtype() == torch::kFloat32);
  const int B = logits.size(0);
  const int V = logits.size(1);
  if (B == 0) return;
  auto stream = at::cuda::getCurrentHIPStream();
  sample_kernel<true><<<B, kBlock, 0, stream>>>(
      out.data_ptr<int32_t>(), logits.data_ptr<float>(),
      inv_temp.data_ptr<float>(), (uint64_t)seed, V);
  HIP_CHECK_KERNEL();
}
