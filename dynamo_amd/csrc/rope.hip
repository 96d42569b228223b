// Rotary position embedding (NeoX / Llama style), in-place on q and k.
//
// cos/sin are precomputed on host into a [max_pos, rot_dim] fp32 table
// (first half cos, second half sin) per guide Appendix B: on-device trig
// turns a memory-bound op VALU-bound. Vectorized 8-wide bf16 loads.
#include "common.h"
#include <hip/hip_fp8.h>
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

// Fused QKV epilogue: reads the QKV-GEMM output [T, (Hq+2Hkv)*hd] STRIDED
// (no q/k .contiguous() copies), optionally adds the qkv bias, applies
// NeoX rope to q and k, writes q to a contiguous [T, Hq*hd] buffer and
// scatters k/v straight into the paged cache. Replaces 4 kernels
// (q copy, k copy, rope, kv_append) with one launch — on Llama-70B decode
// that is 3 fewer hipGraph nodes per layer x 80 layers.
// VT: d-major (transposed) V pages — vcache [P, Hkv, hd, ps]; the decode
// kernel then reads PV A-fragments as contiguous b128 token runs. The
// per-token scatter below touches hd cache lines (~32x DRAM write
// amplification at decode: ~10 MB/step, irrelevant next to the 20+ GB/step
// of V reads it speeds up; prefill batches cover whole lines in L2).
template <int FP8 = 0, int VT = 0>
__global__ void rope_append_kernel(
    short* __restrict__ q_out,          // [T, Hq*hd] contiguous
    short* __restrict__ kcache,         // [P, Hkv, ps, hd]
    short* __restrict__ vcache,
    const short* __restrict__ qkv,      // [T, row_stride] (q|k|v packed)
    const short* __restrict__ bias,     // [(Hq+2Hkv)*hd] or null
    const int32_t* __restrict__ positions,  // [T]
    const int64_t* __restrict__ slots,      // [T]
    const float* __restrict__ cos_sin,      // [max_pos, hd]
    int T, int Hq, int Hkv, int page_size, int hd, int row_stride) {
  const int t = blockIdx.x;
  if (t >= T) return;
  const int half = hd / 2;
  const int pos = positions[t];
  const float* cs = cos_sin + (int64_t)pos * hd;
  const short* row = qkv + (int64_t)t * row_stride;
  const int64_t slot = slots[t];
  const int64_t page = slot >= 0 ? slot / page_size : 0;
  const int off = slot >= 0 ? (int)(slot % page_size) : 0;

  // part 1: rope heads (q then k), 8 rotary indices per thread
  const int rope_units = (Hq + Hkv) * (half / 8);
  for (int idx = threadIdx.x; idx < rope_units; idx += kBlock) {
    const int h = idx / (half / 8);
    const int i0 = (idx % (half / 8)) * 8;
    const int src_off = h * hd;  // q heads first, then k heads
    short8 x1 = *reinterpret_cast<const short8*>(row + src_off + i0);
    short8 x2 = *reinterpret_cast<const short8*>(row + src_off + half + i0);
    float4v c0 = *reinterpret_cast<const float4v*>(cs + i0);
    float4v c1 = *reinterpret_cast<const float4v*>(cs + i0 + 4);
    float4v s0 = *reinterpret_cast<const float4v*>(cs + half + i0);
    float4v s1 = *reinterpret_cast<const float4v*>(cs + half + i0 + 4);
    short8 o1, o2;
#pragma unroll
    for (int i = 0; i < 8; i++) {
      const float c = (i < 4) ? c0[i] : c1[i - 4];
      const float s = (i < 4) ? s0[i] : s1[i - 4];
      float a = bf16_to_f32(x1[i]);
      float b = bf16_to_f32(x2[i]);
      if (bias) {
        a += bf16_to_f32(bias[src_off + i0 + i]);
        b += bf16_to_f32(bias[src_off + half + i0 + i]);
      }
      o1[i] = f32_to_bf16(a * c - b * s);
      o2[i] = f32_to_bf16(b * c + a * s);
    }
    if (h < Hq) {
      short* qb = q_out + ((int64_t)t * Hq + h) * hd;
      *reinterpret_cast<short8*>(qb + i0) = o1;
      *reinterpret_cast<short8*>(qb + half + i0) = o2;
    } else if (slot >= 0) {
      const int kh = h - Hq;
      const int64_t kb = ((page * Hkv + kh) * page_size + off) * hd;
      if constexpr (FP8) {
        unsigned char* kc8 = reinterpret_cast<unsigned char*>(kcache);
        uchar8 p1, p2;
#pragma unroll
        for (int e = 0; e < 8; e++) {
          p1[e] = __hip_cvt_float_to_fp8(bf16_to_f32(o1[e]),
                                         __HIP_SATFINITE, __HIP_E4M3);
          p2[e] = __hip_cvt_float_to_fp8(bf16_to_f32(o2[e]),
                                         __HIP_SATFINITE, __HIP_E4M3);
        }
        *reinterpret_cast<uchar8*>(kc8 + kb + i0) = p1;
        *reinterpret_cast<uchar8*>(kc8 + kb + half + i0) = p2;
      } else {
        *reinterpret_cast<short8*>(kcache + kb + i0) = o1;
        *reinterpret_cast<short8*>(kcache + kb + half + i0) = o2;
      }
    }
  }
  // part 2: v heads (plain copy into the cache)
  if (slot >= 0) {
    const int voff = (Hq + Hkv) * hd;
    const int nvec = Hkv * hd / 8;
    for (int i = threadIdx.x; i < nvec; i += kBlock) {
      const int h = (i * 8) / hd;
      const int d = (i * 8) % hd;
      short8 v = *reinterpret_cast<const short8*>(row + voff + h * hd + d);
      if (bias) {
#pragma unroll
        for (int e = 0; e < 8; e++)
          v[e] = f32_to_bf16(bf16_to_f32(v[e]) +
                             bf16_to_f32(bias[voff + h * hd + d + e]));
      }
      if constexpr (VT) {
        // transposed page: elem offset d * ps + token_in_page
        const int64_t vb0 = ((page * Hkv + h) * hd + d) * page_size + off;
#pragma unroll
        for (int e = 0; e < 8; e++) {
          if constexpr (FP8) {
            reinterpret_cast<unsigned char*>(vcache)[vb0 + (int64_t)e *
                                                     page_size] =
                __hip_cvt_float_to_fp8(bf16_to_f32(v[e]), __HIP_SATFINITE,
                                       __HIP_E4M3);
          } else {
            vcache[vb0 + (int64_t)e * page_size] = v[e];
          }
        }
        continue;
      }
      const int64_t vb = ((page * Hkv + h) * page_size + off) * hd + d;
      if constexpr (FP8) {
        uchar8 pv;
#pragma unroll
        for (int e = 0; e < 8; e++)
          pv[e] = __hip_cvt_float_to_fp8(bf16_to_f32(v[e]),
                                         __HIP_SATFINITE, __HIP_E4M3);
        *reinterpret_cast<uchar8*>(
            reinterpret_cast<unsigned char*>(vcache) + vb) = pv;
      } else {
        *reinterpret_cast<short8*>(vcache + vb) = v;
      }
    }
  }
}

}  // namespace

void rope_append_qkv(torch::Tensor q_out, torch::Tensor kcache,
                     torch::Tensor vcache, torch::Tensor qkv,
                     c10::optional<torch::Tensor> bias,
                     torch::Tensor positions, torch::Tensor slot_mapping,
                     torch::Tensor cos_sin_cache, bool v_transposed) {
  TORCH_CHECK(qkv.is_cuda() && qkv.dtype() == torch::kBFloat16);
  TORCH_CHECK(qkv.dim() == 2 && qkv.stride(1) == 1);
  TORCH_CHECK(positions.dtype() == torch::kInt32);
  TORCH_CHECK(slot_mapping.dtype() == torch::kInt64);
  TORCH_CHECK(cos_sin_cache.dtype() == torch::kFloat32);
  const int T = qkv.size(0);
  const int Hkv = kcache.size(1);
  const int page_size = kcache.size(2);
  const int hd = kcache.size(3);
  const int Hq = q_out.numel() / (T ? (int64_t)T * hd : hd);
  TORCH_CHECK(hd % 16 == 0 && (hd / 2) % 8 == 0);
  TORCH_CHECK(qkv.size(1) >= (Hq + 2 * Hkv) * hd);
  const short* bp = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->numel() == (Hq + 2 * Hkv) * hd &&
                bias->dtype() == torch::kBFloat16 && bias->is_contiguous());
    bp = (const short*)bias->data_ptr();
  }
  if (T == 0) return;
  const bool fp8 = kcache.dtype() == torch::kFloat8_e4m3fn;
  TORCH_CHECK(fp8 || kcache.dtype() == torch::kBFloat16,
              "kv cache must be bf16 or float8_e4m3fn");
  auto stream = at::cuda::getCurrentHIPStream();
  auto launch = [&](auto kern) {
    kern<<<T, kBlock, 0, stream>>>(
        (short*)q_out.data_ptr(), (short*)kcache.data_ptr(),
        (short*)vcache.data_ptr(), (const short*)qkv.data_ptr(), bp,
        positions.data_ptr<int32_t>(), slot_mapping.data_ptr<int64_t>(),
        cos_sin_cache.data_ptr<float>(), T, Hq, Hkv, page_size, hd,
        (int)qkv.stride(0));
  };
  if (v_transposed)
    TORCH_CHECK(vcache.size(2) == hd && vcache.size(3) == page_size,
                "v_transposed expects vcache [P, Hkv, hd, ps]");
  if (fp8 && v_transposed) launch(rope_append_kernel<1, 1>);
  else if (fp8) launch(rope_append_kernel<1, 0>);
  else if (v_transposed) launch(rope_append_kernel<0, 1>);
  else launch(rope_append_kernel<0, 0>);
  HIP_CHECK_KERNEL();
}

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
