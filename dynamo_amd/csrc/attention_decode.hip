// Paged-attention decode (single query token per sequence), MI355X-native.
//
// Flash-decode structure: phase 1 fans (seq, kv_head, context-chunk) over
// workgroups, each computing an online-softmax partial (m, l, acc) for the
// GQA group of query heads sharing that kv head; phase 2 merges chunks.
// Decode attention is HBM-bound (streaming the KV cache once); the kernel is
// laid out for coalesced KV reads: a wave covers (64/DP) tokens x DP
// dim-slices, so consecutive lanes read consecutive 16 B chunks of a page
// row. DP=8 for GQA group <= 4; DP=16 for group 8 (register pressure).
//
// Capability parity: the reference (ai-dynamo/dynamo) delegates paged
// attention to vLLM/TRT-LLM; this is the native CDNA4 engine kernel.
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr int kBlock = 256;        // 4 waves
constexpr int kChunk = 512;        // context tokens per workgroup
constexpr int kSlab = kChunk / 4;  // tokens per wave (128)
constexpr float kNegInf = -1e30f;

// G = GQA group size (Hq / Hkv); DP = lanes per token.
// HEADSPLIT (G >= 4): the 4 waves split the GQA group (G/4 heads each) and
// every wave streams the whole chunk — shrinks the per-lane accumulator 4x
// (occupancy: 2 -> 6+ waves/SIMD for G=8) and removes the LDS merge; the 4
// waves read the same KV stream, so 3 of 4 passes hit L1/L2.
// !HEADSPLIT (G < 4): waves split the chunk into 128-token slabs and merge
// partials through LDS.
template <int G, int DP, bool HEADSPLIT>
__global__ __launch_bounds__(kBlock) void paged_decode_phase1(
    float* __restrict__ partial,        // [B, Hq, C, hd] fp32
    float* __restrict__ ml,             // [B, Hq, C, 2] fp32 (m, l)
    short* __restrict__ out,            // [B, Hq, hd] bf16 (used when C==1)
    const short* __restrict__ q,        // [B, Hq, hd]
    const short* __restrict__ kcache,   // [P, Hkv, ps, hd]
    const short* __restrict__ vcache,
    const int32_t* __restrict__ page_table,  // [B, max_pages]
    const int32_t* __restrict__ ctx_lens,    // [B]
    float scale, int B, int Hkv, int C, int max_pages, int log2_ps, int hd) {
  constexpr int ND = 128 / DP;       // dims per lane
  constexpr int NV8 = ND / 8;        // short8 loads per row slice
  constexpr int TS = 64 / DP;        // tokens per wave step
  const int b = blockIdx.x;
  const int h = blockIdx.y;   // kv head
  const int c = blockIdx.z;   // context chunk
  const int Hq = Hkv * G;
  const int ctx = ctx_lens[b];
  const int chunk_start = c * kChunk;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int ts = lane / DP;       // token slot within step
  const int dp = lane % DP;       // dim slice: dims [dp*ND, dp*ND+ND)
  const int ps = 1 << log2_ps;

  // LDS: q tile [G][hd] fp32 (pre-scaled) + merge scratch [4][G][hd+2]
  extern __shared__ float lds[];
  float* q_lds = lds;                       // G * hd
  float* merge = lds + G * hd;              // 4 * G * (hd + 2)

  if (chunk_start >= ctx) {
    if (C > 1) {
      for (int i = threadIdx.x; i < G; i += kBlock) {
        const int qh = h * G + i;
        float* mlp = ml + (((int64_t)b * Hq + qh) * C + c) * 2;
        mlp[0] = kNegInf; mlp[1] = 0.f;
      }
    }
    return;
  }

  // stage q as raw bf16 (scale is applied to the dot result)
  short* q_lds_s = reinterpret_cast<short*>(q_lds);
  for (int i = threadIdx.x; i < G * hd; i += kBlock) {
    const int g = i / hd;
    q_lds_s[i] = ((const short*)q)[((int64_t)b * Hq + h * G + g) * hd + i % hd];
  }
  __syncthreads();

  // heads this wave computes
  constexpr int GW = HEADSPLIT ? G / 4 : G;
  const int hoff = HEADSPLIT ? wid * GW : 0;

  // this lane's q slice per group head, kept as packed bf16 (VGPR budget)
  short8 qreg[GW][NV8];
#pragma unroll
  for (int g = 0; g < GW; g++)
#pragma unroll
    for (int i = 0; i < NV8; i++)
      qreg[g][i] = *reinterpret_cast<const short8*>(
          q_lds_s + (hoff + g) * hd + dp * ND + i * 8);

  float m[GW], l[GW], acc[GW][ND];
#pragma unroll
  for (int g = 0; g < GW; g++) {
    m[g] = kNegInf; l[g] = 0.f;
#pragma unroll
    for (int i = 0; i < ND; i++) acc[g][i] = 0.f;
  }

  const int slab_start = HEADSPLIT ? chunk_start : chunk_start + wid * kSlab;
  const int slab_end = min(HEADSPLIT ? chunk_start + kChunk : slab_start + kSlab,
                           ctx);
  const int32_t* pt = page_table + (int64_t)b * max_pages;

  // iterate page-aligned windows (page base lookup is wave-uniform and
  // hoisted); within a window, K/V loads for step+TS are issued before
  // processing step (register double-buffer) so HBM latency overlaps the
  // softmax VALU work.
  for (int w0 = slab_start; w0 < slab_end; w0 += ps) {
    const int wend = min(w0 + ps, slab_end);
    const int64_t pbase = (((int64_t)pt[w0 >> log2_ps] * Hkv + h) * ps) * hd;
    short8 kcur[NV8], vcur[NV8], knxt[NV8], vnxt[NV8];
    {
      const int t = w0 + ts;
      const bool v = t < wend;
      const short* kp = kcache + pbase + (int64_t)(t & (ps - 1)) * hd + dp * ND;
      const short* vp = vcache + pbase + (int64_t)(t & (ps - 1)) * hd + dp * ND;
#pragma unroll
      for (int i = 0; i < NV8; i++) {
        kcur[i] = v ? *reinterpret_cast<const short8*>(kp + i * 8) : short8{};
        vcur[i] = v ? *reinterpret_cast<const short8*>(vp + i * 8) : short8{};
      }
    }
    for (int step = w0; step < wend; step += TS) {
      // prefetch next token group
      if (step + TS < wend) {
        const int t = step + TS + ts;
        const bool v = t < wend;
        const short* kp = kcache + pbase + (int64_t)(t & (ps - 1)) * hd + dp * ND;
        const short* vp = vcache + pbase + (int64_t)(t & (ps - 1)) * hd + dp * ND;
#pragma unroll
        for (int i = 0; i < NV8; i++) {
          knxt[i] = v ? *reinterpret_cast<const short8*>(kp + i * 8) : short8{};
          vnxt[i] = v ? *reinterpret_cast<const short8*>(vp + i * 8) : short8{};
        }
      }
      const bool valid = (step + ts) < wend;

      float s[GW];
#pragma unroll
      for (int g = 0; g < GW; g++) {
        float d = 0.f;
#pragma unroll
        for (int i = 0; i < NV8; i++)
#pragma unroll
          for (int e = 0; e < 8; e++)
            d += bf16_to_f32(kcur[i][e]) * bf16_to_f32(qreg[g][i][e]);
#pragma unroll
        for (int off = 1; off < DP; off <<= 1) d += __shfl_xor(d, off, WAVE_SIZE);
        s[g] = valid ? d * scale : kNegInf;
      }

      float vf[ND];
#pragma unroll
      for (int i = 0; i < NV8; i++)
#pragma unroll
        for (int e = 0; e < 8; e++) vf[i * 8 + e] = bf16_to_f32(vcur[i][e]);

#pragma unroll
      for (int g = 0; g < GW; g++) {
        float ms = s[g];
#pragma unroll
        for (int off = DP; off < 64; off <<= 1)
          ms = fmaxf(ms, __shfl_xor(ms, off, WAVE_SIZE));
        if (ms > m[g]) {
          const float corr = __expf(m[g] - ms);
          l[g] *= corr;
#pragma unroll
          for (int i = 0; i < ND; i++) acc[g][i] *= corr;
          m[g] = ms;
        }
        const float p = (s[g] > kNegInf * 0.5f) ? __expf(s[g] - m[g]) : 0.f;
        float psum = p;
#pragma unroll
        for (int off = DP; off < 64; off <<= 1)
          psum += __shfl_xor(psum, off, WAVE_SIZE);
        l[g] += psum;
#pragma unroll
        for (int i = 0; i < ND; i++) acc[g][i] = fmaf(p, vf[i], acc[g][i]);
      }
#pragma unroll
      for (int i = 0; i < NV8; i++) { kcur[i] = knxt[i]; vcur[i] = vnxt[i]; }
    }
  }

  // fold the token-slot partials: acc holds per-(ts) sums
#pragma unroll
  for (int g = 0; g < GW; g++)
#pragma unroll
    for (int i = 0; i < ND; i++) {
      float a = acc[g][i];
#pragma unroll
      for (int off = DP; off < 64; off <<= 1)
        a += __shfl_xor(a, off, WAVE_SIZE);
      acc[g][i] = a;
    }

  if (HEADSPLIT) {
    // each wave covered the whole chunk for its own heads: write directly
    if (ts == 0) {  // lanes 0..DP-1 cover the DP dim slices exactly once
#pragma unroll
      for (int g = 0; g < GW; g++) {
        const int qh = h * G + hoff + g;
#pragma unroll
        for (int i = 0; i < ND; i++) {
          const int d = dp * ND + i;
          if (C == 1) {
            out[((int64_t)b * Hq + qh) * hd + d] =
                f32_to_bf16(l[g] > 0.f ? acc[g][i] / l[g] : 0.f);
          } else {
            partial[(((int64_t)b * Hq + qh) * C + c) * hd + d] = acc[g][i];
          }
        }
        if (C > 1 && dp == 0) {
          float* mlp = ml + (((int64_t)b * Hq + qh) * C + c) * 2;
          mlp[0] = m[g]; mlp[1] = l[g];
        }
      }
    }
    return;
  }

  // cross-wave merge via LDS. Wave w writes [G][hd] acc + m,l.
  __syncthreads();  // q_lds no longer needed
  float* my = merge + wid * G * (hd + 2);
  if (ts == 0) {  // lanes 0..DP-1 cover the DP dim slices exactly once
#pragma unroll
    for (int g = 0; g < GW; g++) {
#pragma unroll
      for (int i = 0; i < ND; i++) my[g * (hd + 2) + dp * ND + i] = acc[g][i];
      if (dp == 0) { my[g * (hd + 2) + hd] = m[g]; my[g * (hd + 2) + hd + 1] = l[g]; }
    }
  }
  __syncthreads();

  // threads [0, G*hd) each merge one (g, d) across the 4 waves
  for (int i = threadIdx.x; i < G * hd; i += kBlock) {
    const int g = i / hd;
    const int d = i % hd;
    float mw[4], lw[4];
    float mstar = kNegInf;
#pragma unroll
    for (int w = 0; w < 4; w++) {
      mw[w] = merge[w * G * (hd + 2) + g * (hd + 2) + hd];
      lw[w] = merge[w * G * (hd + 2) + g * (hd + 2) + hd + 1];
      mstar = fmaxf(mstar, mw[w]);
    }
    float lsum = 0.f, asum = 0.f;
#pragma unroll
    for (int w = 0; w < 4; w++) {
      const float corr = (lw[w] > 0.f) ? __expf(mw[w] - mstar) : 0.f;
      lsum += lw[w] * corr;
      asum += merge[w * G * (hd + 2) + g * (hd + 2) + d] * corr;
    }
    const int qh = h * G + g;
    if (C == 1) {
      out[((int64_t)b * Hq + qh) * hd + d] =
          f32_to_bf16(lsum > 0.f ? asum / lsum : 0.f);
    } else {
      partial[(((int64_t)b * Hq + qh) * C + c) * hd + d] = asum;
      if (d == 0) {
        float* mlp = ml + (((int64_t)b * Hq + qh) * C + c) * 2;
        mlp[0] = mstar; mlp[1] = lsum;
      }
    }
  }
}

// Phase 2: merge chunk partials. grid (B, Hq), block = 128.
__global__ void paged_decode_phase2(short* __restrict__ out,  // [B, Hq, hd]
                                    const float* __restrict__ partial,
                                    const float* __restrict__ ml,
                                    const int32_t* __restrict__ ctx_lens,
                                    int Hq, int C, int hd) {
  const int b = blockIdx.x;
  const int qh = blockIdx.y;
  const int nc = min(C, (ctx_lens[b] + kChunk - 1) / kChunk);
  const float* mlp = ml + (((int64_t)b * Hq + qh) * C) * 2;

  __shared__ float smax[1];
  float mstar = kNegInf;
  for (int c = threadIdx.x; c < nc; c += blockDim.x) mstar = fmaxf(mstar, mlp[2 * c]);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) mstar = fmaxf(mstar, __shfl_xor(mstar, off, WAVE_SIZE));
  if (threadIdx.x == 0) smax[0] = mstar;
  __syncthreads();
  if (threadIdx.x == 64) smax[0] = fmaxf(smax[0], mstar);
  __syncthreads();
  mstar = smax[0];

  for (int d = threadIdx.x; d < hd; d += blockDim.x) {
    float asum = 0.f, lsum = 0.f;
    for (int c = 0; c < nc; c++) {
      const float lc = mlp[2 * c + 1];
      if (lc <= 0.f) continue;
      const float corr = __expf(mlp[2 * c] - mstar);
      asum += partial[(((int64_t)b * Hq + qh) * C + c) * hd + d] * corr;
      lsum += lc * corr;
    }
    out[((int64_t)b * Hq + qh) * hd + d] = f32_to_bf16(lsum > 0.f ? asum / lsum : 0.f);
  }
}

}  // namespace

int64_t paged_decode_num_chunks(int64_t max_ctx) {
  return std::max<int64_t>(1, (max_ctx + kChunk - 1) / kChunk);
}

void paged_attention_decode(torch::Tensor out, torch::Tensor q,
                            torch::Tensor kcache, torch::Tensor vcache,
                            torch::Tensor page_table, torch::Tensor ctx_lens,
                            torch::Tensor partial, torch::Tensor ml,
                            double scale) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  TORCH_CHECK(page_table.dtype() == torch::kInt32 && ctx_lens.dtype() == torch::kInt32);
  const int B = q.size(0);
  const int Hq = q.size(1);
  const int hd = q.size(2);
  const int Hkv = kcache.size(1);
  const int ps = kcache.size(2);
  const int G = Hq / Hkv;
  const int max_pages = page_table.size(1);
  const int C = ml.size(2);  // chunk capacity allocated by caller
  TORCH_CHECK(hd == 128, "only head_dim=128 supported natively");
  TORCH_CHECK((ps & (ps - 1)) == 0, "page_size must be a power of 2");
  int log2_ps = 0; while ((1 << log2_ps) < ps) log2_ps++;
  if (B == 0) return;
  auto stream = at::cuda::getCurrentHIPStream();

  const int lds_bytes = (G * hd + 4 * G * (hd + 2)) * sizeof(float);
  dim3 grid(B, Hkv, C);
#define LAUNCH_G(GG, DP, HS)                                                  \
  paged_decode_phase1<GG, DP, HS><<<grid, kBlock, lds_bytes, stream>>>(       \
      partial.data_ptr<float>(), ml.data_ptr<float>(), (short*)out.data_ptr(),\
      (const short*)q.data_ptr(), (const short*)kcache.data_ptr(),            \
      (const short*)vcache.data_ptr(), page_table.data_ptr<int32_t>(),        \
      ctx_lens.data_ptr<int32_t>(), (float)scale, B, Hkv, C, max_pages,       \
      log2_ps, hd)
  switch (G) {
    case 1: LAUNCH_G(1, 8, false); break;
    case 2: LAUNCH_G(2, 8, false); break;
    case 4: LAUNCH_G(4, 16, false); break;
    case 8: LAUNCH_G(8, 16, true); break;
    default: TORCH_CHECK(false, "unsupported GQA group size ", G);
  }
#undef LAUNCH_G
  HIP_CHECK_KERNEL();
  if (C > 1) {
    dim3 grid2(B, Hq);
    paged_decode_phase2<<<grid2, 128, 0, stream>>>(
        (short*)out.data_ptr(), partial.data_ptr<float>(), ml.data_ptr<float>(),
        ctx_lens.data_ptr<int32_t>(), Hq, C, hd);
    HIP_CHECK_KERNEL();
  }
}
