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
                              uint64_t seed,
                              const uint64_t* __restrict__ row_seeds,  // [B] or null
                              int V) {
  const int b = blockIdx.x;
  const float* row = logits + (int64_t)b * V;
  const float it = GUMBEL ? inv_temp[b] : 1.f;
  // per-row seed (client-supplied sampling seed, already host-mixed with
  // the output position) or the scalar engine-step stream
  const uint64_t rs = row_seeds ? row_seeds[b] : (seed ^ ((uint64_t)b << 32));

  float best = -1e38f;
  int besti = -1;
  for (int v = threadIdx.x; v < V; v += kBlock) {
    float x = row[v] * it;
    if constexpr (GUMBEL) {
      const uint64_t h = hash_u64(rs ^ (uint64_t)v);
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

// Fused top-k / top-p (nucleus) sampling: one block per row, no sort.
//
// The k-th-largest-prob threshold (top-k) and the nucleus mass threshold
// (top-p) are found by a FUSED bisection on the probability scale: each
// iteration makes one pass over the vocab computing count(p_i >= t_k) and
// mass(p_i >= t_p) for the two candidate thresholds, 24 iterations
// (resolution 6e-8 on [0,1] — tokens below that probability are sampling
// noise). The final pass Gumbel-argmaxes logits/T over the surviving set,
// which samples the renormalized filtered distribution exactly. Nucleus
// semantics match the torch reference in engine/sampling.py: the keep-set
// is defined on the T=1 softmax; temperature only shapes sampling inside
// the set. Single launch, no host sync, hipGraph-capturable.
__global__ __launch_bounds__(kBlock) void topkp_sample_kernel(
    int32_t* __restrict__ out,          // [B]
    const float* __restrict__ logits,   // [B, V]
    const float* __restrict__ inv_temp, // [B]
    const int32_t* __restrict__ top_k,  // [B] (<=0: off)
    const float* __restrict__ top_p,    // [B] (>=1: off)
    uint64_t seed,
    const uint64_t* __restrict__ row_seeds,  // [B] or null
    int V) {
  const int b = blockIdx.x;
  const uint64_t rs = row_seeds ? row_seeds[b] : (seed ^ ((uint64_t)b << 32));
  const float* row = logits + (int64_t)b * V;
  const int k = top_k[b];
  const float p = top_p[b];
  const bool use_k = (k > 0 && k < V);
  const bool use_p = (p < 1.0f);
  constexpr int kWaves = kBlock / WAVE_SIZE;
  __shared__ float red[kWaves];
  __shared__ float red2[kWaves];
  __shared__ float bcast[2];

  // pass 1: row max (softmax stability)
  float m = -1e38f;
  for (int v = threadIdx.x; v < V; v += kBlock) m = fmaxf(m, row[v]);
  m = wave_reduce_max(m);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x / WAVE_SIZE] = m;
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < kWaves; w++) m = fmaxf(m, red[w]);
    bcast[0] = m;
  }
  __syncthreads();
  m = bcast[0];

  // pass 2: partition sum Z
  float z = 0.f;
  for (int v = threadIdx.x; v < V; v += kBlock) z += __expf(row[v] - m);
  z = wave_reduce_sum(z);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x / WAVE_SIZE] = z;
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < kWaves; w++) z += red[w];
    bcast[0] = z;
  }
  __syncthreads();
  const float inv_z = 1.f / bcast[0];

  // fused bisection: t_k (count >= k) and t_p (mass >= p), both on [0,1]
  float lo_k = 0.f, hi_k = 1.f, lo_p = 0.f, hi_p = 1.f;
  if (use_k || use_p) {
    for (int it = 0; it < 24; it++) {
      const float mid_k = 0.5f * (lo_k + hi_k);
      const float mid_p = 0.5f * (lo_p + hi_p);
      float cnt = 0.f, mass = 0.f;
      for (int v = threadIdx.x; v < V; v += kBlock) {
        const float pv = __expf(row[v] - m) * inv_z;
        if (use_k && pv >= mid_k) cnt += 1.f;
        if (use_p && pv >= mid_p) mass += pv;
      }
      cnt = wave_reduce_sum(cnt);
      mass = wave_reduce_sum(mass);
      const int wid = threadIdx.x / WAVE_SIZE;
      if ((threadIdx.x & 63) == 0) { red[wid] = cnt; red2[wid] = mass; }
      __syncthreads();
      if (threadIdx.x == 0) {
        for (int w = 1; w < kWaves; w++) { cnt += red[w]; mass += red2[w]; }
        bcast[0] = cnt;
        bcast[1] = mass;
      }
      __syncthreads();
      cnt = bcast[0];
      mass = bcast[1];
      if (use_k) { if (cnt >= (float)k) lo_k = mid_k; else hi_k = mid_k; }
      if (use_p) { if (mass >= p) lo_p = mid_p; else hi_p = mid_p; }
      __syncthreads();
    }
  }
  const float thr = fmaxf(use_k ? lo_k : 0.f, use_p ? lo_p : 0.f);

  // final pass: Gumbel-argmax of logits/T over {p_i >= thr}
  const float it_ = inv_temp[b];
  float best = -1e38f;
  int besti = -1;
  for (int v = threadIdx.x; v < V; v += kBlock) {
    const float pv = __expf(row[v] - m) * inv_z;
    if (pv < thr) continue;
    const uint64_t h = hash_u64(rs ^ (uint64_t)v);
    const float u = (float)((h >> 11) + 1) * 4.8828125e-4f * 2.2737367544323206e-13f;
    const float x = row[v] * it_ + -__logf(-__logf(u));
    if (x > best || (x == best && v < besti)) { best = x; besti = v; }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ob = __shfl_xor(best, off, WAVE_SIZE);
    const int oi = __shfl_xor(besti, off, WAVE_SIZE);
    if (ob > best || (ob == best && oi != -1 && (besti == -1 || oi < besti))) {
      best = ob; besti = oi;
    }
  }
  __shared__ float sb2[kWaves];
  __shared__ int si2[kWaves];
  const int wid = threadIdx.x / WAVE_SIZE;
  if ((threadIdx.x & 63) == 0) { sb2[wid] = best; si2[wid] = besti; }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < kWaves; w++) {
      if (sb2[w] > best || (sb2[w] == best && si2[w] != -1 &&
                            (besti == -1 || si2[w] < besti))) {
        best = sb2[w]; besti = si2[w];
      }
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
      out.data_ptr<int32_t>(), logits.data_ptr<float>(), nullptr, 0, nullptr,
      V);
  HIP_CHECK_KERNEL();
}

static const uint64_t* opt_seeds(const c10::optional<torch::Tensor>& t,
                                 int B) {
  if (!t.has_value()) return nullptr;
  TORCH_CHECK(t->dtype() == torch::kInt64 && t->numel() == B &&
              t->is_cuda() && t->is_contiguous());
  return reinterpret_cast<const uint64_t*>(t->data_ptr<int64_t>());
}

void gumbel_sample(torch::Tensor out, torch::Tensor logits, torch::Tensor inv_temp,
                   int64_t seed, c10::optional<torch::Tensor> row_seeds) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kFloat32);
  TORCH_CHECK(inv_temp.dtype() == torch::kFloat32);
  const int B = logits.size(0);
  const int V = logits.size(1);
  if (B == 0) return;
  auto stream = at::cuda::getCurrentHIPStream();
  sample_kernel<true><<<B, kBlock, 0, stream>>>(
      out.data_ptr<int32_t>(), logits.data_ptr<float>(),
      inv_temp.data_ptr<float>(), (uint64_t)seed, opt_seeds(row_seeds, B), V);
  HIP_CHECK_KERNEL();
}

void topkp_sample(torch::Tensor out, torch::Tensor logits,
                  torch::Tensor inv_temp, torch::Tensor top_k,
                  torch::Tensor top_p, int64_t seed,
                  c10::optional<torch::Tensor> row_seeds) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kFloat32);
  TORCH_CHECK(inv_temp.dtype() == torch::kFloat32);
  TORCH_CHECK(top_k.dtype() == torch::kInt32);
  TORCH_CHECK(top_p.dtype() == torch::kFloat32);
  const int B = logits.size(0);
  const int V = logits.size(1);
  TORCH_CHECK(top_k.numel() == B && top_p.numel() == B && inv_temp.numel() == B);
  if (B == 0) return;
  auto stream = at::cuda::getCurrentHIPStream();
  topkp_sample_kernel<<<B, kBlock, 0, stream>>>(
      out.data_ptr<int32_t>(), logits.data_ptr<float>(),
      inv_temp.data_ptr<float>(), top_k.data_ptr<int32_t>(),
      top_p.data_ptr<float>(), (uint64_t)seed, opt_seeds(row_seeds, B), V);
  HIP_CHECK_KERNEL();
}
