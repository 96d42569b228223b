// Rotary position embedding (NeoX / Llama style), in-place on q and k.
//
// cos/sin are precomputed on host into a [max_pos, rot_dim] fp32 table
// (first half cos, second half sin) per guide Appendix B: on-device trig
// turns a memory-bound op VALU-bound. Vectorized 8-wide bf16 loads.
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr int kBlock = 256;

__global__ void rope_kernel(short* __restrict__ q,    // [T, Hq*hd]
                            short* __restrict__ k,    // [T, Hk*hd]
                            const int32_t* __restrict__ positions,  // [T]
                            const float* __restrict__ cos_sin,      // [max_pos, hd]
                            int T, int Hq, int Hk, int hd) {
  const int t = blockIdx.x;
  if (t >= T) return;
  const int half = hd / 2;
  const int pos = positions[t];
  const float* cs = cos_sin + (int64_t)pos * hd;

  // each thread handles 8 consecutive rotary indices of one head
  const int total = (Hq + Hk) * (half / 8);
  for (int idx = threadIdx.x; idx < total; idx += kBlock) {
    const int h = idx / (half / 8);
    const int i0 = (idx % (half / 8)) * 8;
    short* base = (h < Hq) ? q + ((int64_t)t * Hq + h) * hd
                           : k + ((int64_t)t * Hk + (h - Hq)) * hd;
    short8 x1 = *reinterpret_cast<const short8*>(base + i0);
    short8 x2 = *reinterpret_cast<const short8*>(base + half + i0);
    float4v c0 = *reinterpret_cast<const float4v*>(cs + i0);
    float4v c1 = *reinterpret_cast<const float4v*>(cs + i0 + 4);
    float4v s0 = *reinterpret_cast<const float4v*>(cs + half + i0);
    float4v s1 = *reinterpret_cast<const float4v*>(cs + half + i0 + 4);
    short8 o1, o2;
#pragma unroll
    for (int i = 0; i < 8; i++) {
      float c = (i < 4) ? c0[i] : c1[i - 4];
      float s = (i < 4) ? s0[i] : s1[i - 4];
      float a = bf16_to_f32(x1[i]);
      float b = bf16_to_f32(x2[i]);
      o1[i] = f32_to_bf16(a * c - b * s);
      o2[i] = f32_to_bf16(b * c + a * s);
    }
    *reinterpret_cast<short8*>(base + i0) = o1;
    *reinterpret_cast<short8*>(base + half + i0) = o2;
  }
}

}  // namespace

void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
                  torch::Tensor cos_sin_cache, int64_t num_q_heads,
                  int64_t num_k_heads, int64_t head_dim) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  TORCH_CHECK(positions.dtype() == torch::kInt32);
  TORCH_CHECK(cos_sin_cache.dtype() == torch::kFloat32);
  TORCH_CHECK(head_dim % 16 == 0, "head_dim must be a multiple of 16");
  const int T = q.numel() / (num_q_heads * head_dim);
  if (T == 0) return;
  auto stream = at::cuda::getCurrentHIPStream();
  rope_kernel<<<T, kBlock, 0, stream>>>(
      (short*)q.data_ptr(), (short*)k.data_ptr(),
      positions.data_ptr<int32_t>(), cos_sin_cache.data_ptr<float>(),
      T, (int)num_q_heads, (int)num_k_heads, (int)head_dim);
  HIP_CHECK_KERNEL();
}
