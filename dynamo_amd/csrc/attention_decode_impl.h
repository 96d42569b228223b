// Paged-attention decode kernels (torch-free; included by the extension and
// by the standalone sweep tool benchmarks/decode_sweep.hip).
//
// Template space:
//   G     : GQA group size (Hq / Hkv)
//   DP    : lanes per token (8 or 16); dims per lane = 128/DP
//   HS    : head-split factor (1, 2 or 4): the 4 waves form HS head-groups
//           x (4/HS) token-groups. More head-split => smaller accumulators
//           (occupancy) but each KV token is streamed by HS waves.
//   DEPTH : software-pipeline depth (token groups in flight per wave).
//           Static buffer indices only (runtime-indexed register arrays
//           spill to scratch).
#pragma once
#include "common.h"
#include <cstdlib>

namespace decode_attn {

constexpr int kBlock = 256;        // 4 waves

// Context tokens per workgroup ("chunk"). RUNTIME-chosen per engine from
// the max context: big chunks amortize per-tile overhead and shrink the
// phase-2 merge (sweep: G8 MFMA 2982 -> 3344 GB/s from 512 -> 2048 at
// ctx 8192), but the grid is B*Hkv*C blocks, so short contexts need small
// chunks to fill 256 CUs. Python mirrors this via the binding.
inline int decode_chunk_tokens(int max_ctx) {
#ifndef __HIP_DEVICE_COMPILE__
  if (const char* e = getenv("DYNAMO_DECODE_CHUNK")) {
    const int v = atoi(e);
    if (v >= 128 && v % 128 == 0) return v;   // A/B sweeps on real serving
  }
#endif
  if (max_ctx >= 8192) return 2048;
  if (max_ctx >= 2048) return 1024;
  return 512;
}
#ifndef DECODE_KCHUNK
#define DECODE_KCHUNK 512
#endif
constexpr int kChunk = DECODE_KCHUNK;  // sweep-tool default (see above)
constexpr float kNegInf = -1e30f;

template <int G, int DP, int HS, int DEPTH>
__global__ __launch_bounds__(kBlock) void paged_decode_phase1(
    float* __restrict__ partial,        // [B, Hq, C, hd] fp32
    float* __restrict__ ml,             // [B, Hq, C, 2] fp32 (m, l)
    short* __restrict__ out,            // [B, Hq, hd] bf16 (used when C==1)
    const short* __restrict__ q,        // [B, Hq, hd]
    const short* __restrict__ kcache,   // [P, Hkv, ps, hd]
    const short* __restrict__ vcache,
    const int32_t* __restrict__ page_table,  // [B, max_pages]
    const int32_t* __restrict__ ctx_lens,    // [B]
    float scale, int chunk, int B, int Hkv, int C, int max_pages,
    int log2_ps, int hd) {
  constexpr int ND = 128 / DP;       // dims per lane
  constexpr int NV8 = ND / 8;        // short8 loads per row slice
  constexpr int TS = 64 / DP;        // tokens per wave step
  constexpr int GW = G / HS;         // heads per wave
  constexpr int TG = 4 / HS;         // token-groups (waves splitting tokens)
  const int kSlab = chunk / TG;
  const int b = blockIdx.x;
  const int h = blockIdx.y;   // kv head
  const int c = blockIdx.z;   // context chunk
  const int Hq = Hkv * G;
  const int ctx = ctx_lens[b];
  const int chunk_start = c * chunk;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int tg = wid / HS;        // token-group index
  const int hg = wid % HS;        // head-group index
  const int ts = lane / DP;       // token slot within step
  const int dp = lane % DP;       // dim slice: dims [dp*ND, dp*ND+ND)
  const int ps = 1 << log2_ps;
  const int hoff = hg * GW;

  // LDS: q tile [G][hd] bf16 + merge scratch [TG][G][hd+2] fp32
  extern __shared__ float lds[];
  float* merge = lds;                          // TG * G * (hd+2)
  short* q_lds_s = reinterpret_cast<short*>(lds + (TG > 1 ? TG * G * (hd + 2) : 0));

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

  for (int i = threadIdx.x; i < G * hd; i += kBlock) {
    const int g = i / hd;
    q_lds_s[i] = q[((int64_t)b * Hq + h * G + g) * hd + i % hd];
  }
  __syncthreads();

  // q kept packed bf16; QK dot uses v_dot2_f32_bf16 (2 MACs/inst, no
  // converts — 3x fewer VALU ops than cvt+fma per element)
  typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2_t;
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

  const int slab_start = chunk_start + tg * kSlab;
  const int slab_end = min(slab_start + kSlab, ctx);
  const int32_t* pt = page_table + (int64_t)b * max_pages;

  for (int w0 = slab_start; w0 < slab_end; w0 += ps) {
    const int wend = min(w0 + ps, slab_end);
    const int64_t pbase = (((int64_t)pt[w0 >> log2_ps] * Hkv + h) * ps) * hd;
    const int ngroups = (wend - w0 + TS - 1) / TS;

    short8 kbuf[DEPTH][NV8], vbuf[DEPTH][NV8];

    auto load_group = [&](int gi, short8 (&kb)[NV8], short8 (&vb)[NV8]) {
      const int t = w0 + gi * TS + ts;
      const bool v = t < wend;
      const short* kp = kcache + pbase + (int64_t)(t & (ps - 1)) * hd + dp * ND;
      const short* vp = vcache + pbase + (int64_t)(t & (ps - 1)) * hd + dp * ND;
#pragma unroll
      for (int i = 0; i < NV8; i++) {
        kb[i] = v ? *reinterpret_cast<const short8*>(kp + i * 8) : short8{};
        vb[i] = v ? *reinterpret_cast<const short8*>(vp + i * 8) : short8{};
      }
    };

    auto process_group = [&](int gi, short8 (&kb)[NV8], short8 (&vb)[NV8]) {
      const bool valid = (w0 + gi * TS + ts) < wend;
      float s[GW];
#pragma unroll
      for (int g = 0; g < GW; g++) {
        float d = 0.f;
#pragma unroll
        for (int i = 0; i < NV8; i++) {
          const bf16x2_t* k2 = reinterpret_cast<const bf16x2_t*>(&kb[i]);
          const bf16x2_t* q2 = reinterpret_cast<const bf16x2_t*>(&qreg[g][i]);
#pragma unroll
          for (int e = 0; e < 4; e++)
            d = __builtin_amdgcn_fdot2_f32_bf16(k2[e], q2[e], d, false);
        }
        // DP-lane dot fold: pure-DPP butterflies (ds_bpermute chains
        // serialize on the LDS unit; measured lever on the MFMA variant)
        if constexpr (DP == 8) d = group8_reduce_sum(d);
        else d = row16_reduce_sum(d);
        s[g] = valid ? d * scale : kNegInf;
      }
      float vf[ND];
#pragma unroll
      for (int i = 0; i < NV8; i++)
#pragma unroll
        for (int e = 0; e < 8; e++) vf[i * 8 + e] = bf16_to_f32(vb[i][e]);
#pragma unroll
      for (int g = 0; g < GW; g++) {
        float ms = s[g];
        if constexpr (DP == 8)   // fold the two 8-groups of the row first
          ms = fmaxf(ms, dpp_movf<0x128>(ms));
        ms = fmaxf(ms, __shfl_xor(ms, 16, WAVE_SIZE));
        ms = fmaxf(ms, __shfl_xor(ms, 32, WAVE_SIZE));
        if (ms > m[g]) {
          const float corr = __expf(m[g] - ms);
          l[g] *= corr;
#pragma unroll
          for (int i = 0; i < ND; i++) acc[g][i] *= corr;
          m[g] = ms;
        }
        const float p = (s[g] > kNegInf * 0.5f) ? __expf(s[g] - m[g]) : 0.f;
        float psum = p;
        if constexpr (DP == 8)
          psum += dpp_movf<0x128>(psum);
        psum += __shfl_xor(psum, 16, WAVE_SIZE);
        psum += __shfl_xor(psum, 32, WAVE_SIZE);
        l[g] += psum;
#pragma unroll
        for (int i = 0; i < ND; i++) acc[g][i] = fmaf(p, vf[i], acc[g][i]);
      }
    };

#pragma unroll
    for (int j = 0; j < DEPTH; j++)
      if (j < ngroups) load_group(j, kbuf[j], vbuf[j]);

    int gi = 0;
    while (gi + DEPTH <= ngroups) {
#pragma unroll
      for (int j = 0; j < DEPTH; j++) {
        process_group(gi + j, kbuf[j], vbuf[j]);
        if (gi + j + DEPTH < ngroups)
          load_group(gi + j + DEPTH, kbuf[j], vbuf[j]);
      }
      gi += DEPTH;
    }
#pragma unroll
    for (int j = 0; j < DEPTH; j++)
      if (gi + j < ngroups) process_group(gi + j, kbuf[j], vbuf[j]);
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

  if (TG == 1) {
    // each wave covered the whole chunk for its heads: write directly
    if (ts == 0) {
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

  // cross-token-group merge via LDS: wave (tg, hg) writes its heads' slice.
  __syncthreads();
  float* my = merge + tg * G * (hd + 2);
  if (ts == 0) {
#pragma unroll
    for (int g = 0; g < GW; g++) {
      const int gq = hoff + g;
#pragma unroll
      for (int i = 0; i < ND; i++) my[gq * (hd + 2) + dp * ND + i] = acc[g][i];
      if (dp == 0) { my[gq * (hd + 2) + hd] = m[g]; my[gq * (hd + 2) + hd + 1] = l[g]; }
    }
  }
  __syncthreads();

  for (int i = threadIdx.x; i < G * hd; i += kBlock) {
    const int g = i / hd;
    const int d = i % hd;
    float mw[TG], lw[TG];
    float mstar = kNegInf;
#pragma unroll
    for (int w = 0; w < TG; w++) {
      mw[w] = merge[w * G * (hd + 2) + g * (hd + 2) + hd];
      lw[w] = merge[w * G * (hd + 2) + g * (hd + 2) + hd + 1];
      mstar = fmaxf(mstar, mw[w]);
    }
    float lsum = 0.f, asum = 0.f;
#pragma unroll
    for (int w = 0; w < TG; w++) {
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

// ---------------------------------------------------------------------------
// MFMA decode variant: the GQA group rides the MFMA M dimension, so the
// per-head VALU cost of the dot/accumulate path disappears (measured: the
// VALU kernel scales 4978 -> 2467 GB/s from G=1 to G=8; this variant keeps
// the G=1 rate for G=8).
//   S   = Q[G<=16 pad, 128] x K^T[128, 16 toks]   (K B-frags read DIRECTLY
//         from the paged cache — no LDS staging for K)
//   P   row-major via a small per-wave LDS tile (C/D -> A-frag re-layout)
//   PV  = P[16, 32 toks] x V[32 toks, 128]; V staged per-wave in an LDS
//         layout matching ds_read_b64_tr_b16's native transpose pattern
//         (slab-permuted 4x16 blocks), so B-frags cost 2 tr-reads each.
// Softmax runs on the C/D layout: row = head (4 per lane), col = token
// (16 lanes), reductions are 4 shfl per row per 16 tokens.
// Requires page_size % 32 == 0 and hd == 128. G (<= 16) is RUNTIME — the
// group only appears in bounds/guards, never in register shapes, so one
// kernel serves every GQA group (e.g. qwen2's G=7).
//   DEFER: defer-max rescale skip (guide T13): while the tile max stays
//          within 8 of the running max, keep m_old and skip the
//          8xf32x4 acc rescale (P bounded by e^8 — fp32 accum headroom).
//   PRIO:  s_setprio(1) around the MFMA clusters (guide T5).
template <int DEFER = 0, int PRIO = 0>
__global__ __launch_bounds__(kBlock) void paged_decode_mfma(
    float* __restrict__ partial, float* __restrict__ ml,
    short* __restrict__ out, const short* __restrict__ q,
    const short* __restrict__ kcache, const short* __restrict__ vcache,
    const int32_t* __restrict__ page_table, const int32_t* __restrict__ ctx_lens,
    float scale, int chunk, int G, int B, int Hkv, int C, int max_pages,
    int log2_ps, int hd) {
  typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
  const int kSlab = chunk / 4;        // tokens per wave
  const int b = blockIdx.x;
  const int h = blockIdx.y;
  const int c = blockIdx.z;
  const int Hq = Hkv * G;
  const int ctx = ctx_lens[b];
  const int chunk_start = c * chunk;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int lr = lane & 15;       // col lane (token / dim-subtile index)
  const int lg = lane >> 4;       // 4 k-groups
  const int ps = 1 << log2_ps;

  // LDS: [4 waves][V^T tile 10KB] + [4 waves][P 16x32 bf16 1KB] + q G*128
  // bf16 + merge scratch (4*G*(hd+2) fp32). V is staged TRANSPOSED
  // [128 dims][32 toks] with 80-byte row stride (16B pad) so PV B-frags
  // are single b128 row reads (the 80B stride spreads the 16 lanes' rows
  // across banks; was 8 scalar gathers per frag in the row-major layout).
  extern __shared__ float lds[];
  float* merge = lds;                                   // 4*G*(hd+2)
  short* q_lds_s = reinterpret_cast<short*>(merge + 4 * G * (hd + 2));
  short* v_lds = q_lds_s + G * hd + wid * 5120;         // per-wave 10KB
  short* p_lds = q_lds_s + G * hd + 4 * 5120 + wid * 512;  // per-wave 1KB

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

  for (int i = threadIdx.x; i < G * hd; i += kBlock) {
    const int g = i / hd;
    q_lds_s[i] = q[((int64_t)b * Hq + h * G + g) * hd + i % hd];
  }
  __syncthreads();

  // A-frags: row = lr (head, zero-padded beyond G), k = lg*8+j (+32*kc)
  bf16x8_t q_frag[4];
#pragma unroll
  for (int kc = 0; kc < 4; kc++) {
    short8 v{};
    if (lr < G)
      v = *reinterpret_cast<const short8*>(q_lds_s + lr * hd + kc * 32 + lg * 8);
    q_frag[kc] = *reinterpret_cast<bf16x8_t*>(&v);
  }

  float m[4], l[4];
  f32x4 acc[8];
#pragma unroll
  for (int r = 0; r < 4; r++) { m[r] = kNegInf; l[r] = 0.f; }
#pragma unroll
  for (int d = 0; d < 8; d++) acc[d] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int slab_start = chunk_start + wid * kSlab;
  const int slab_end = min(slab_start + kSlab, ctx);
  const int32_t* pt = page_table + (int64_t)b * max_pages;

  // NOTE on ds_read_b64_tr_b16 (HW-probed, tests test_tr16_probe_mapping):
  // the instruction is CROSS-LANE cooperative — per 16-lane group it reads
  // four 64-bit rows at the addresses of subgroup-leader lanes {0,4,8,12}
  // and hands every lane column (lane&3) of that 4x4 tile. Only 4 distinct
  // columns reach a 16-lane group, so it cannot feed a 16-column MFMA
  // B-fragment; the PV path therefore reads the transposed V fragments as
  // swizzled scalar LDS loads instead.

  // Uniform tile count across ALL waves (inactive waves still hit the
  // barriers — __syncthreads inside a loop with per-wave iteration counts
  // is barrier divergence / UB).
  // (A register K-prefetch across tiles was measured SLOWER — 2559 vs
  // 3000 GB/s — VGPR pressure beats the latency win on this kernel.)
  for (int ti = 0; ti < kSlab / 32; ti++) {
    const int t0 = slab_start + ti * 32;
    const bool active = t0 < slab_end;
    const int64_t pbase = active
        ? (((int64_t)pt[t0 >> log2_ps] * Hkv + h) * ps) * hd : 0;
    f32x4 sA{0.f, 0.f, 0.f, 0.f}, sB{0.f, 0.f, 0.f, 0.f};
    if (active) {
      const int tA = t0 + lr, tB = t0 + 16 + lr;
      const short* krA = kcache + pbase + (int64_t)(tA & (ps - 1)) * hd;
      const short* krB = kcache + pbase + (int64_t)(tB & (ps - 1)) * hd;
      const bool vA = tA < slab_end, vB = tB < slab_end;
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kc = 0; kc < 4; kc++) {
        short8 ka = vA ? *reinterpret_cast<const short8*>(krA + kc * 32 + lg * 8)
                       : short8{};
        sA = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            q_frag[kc], *reinterpret_cast<bf16x8_t*>(&ka), sA, 0, 0, 0);
      }
      if (t0 + 16 < slab_end) {
#pragma unroll
        for (int kc = 0; kc < 4; kc++) {
          short8 kb2 = vB ? *reinterpret_cast<const short8*>(krB + kc * 32 + lg * 8)
                          : short8{};
          sB = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              q_frag[kc], *reinterpret_cast<bf16x8_t*>(&kb2), sB, 0, 0, 0);
        }
      }
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
    }
    // ---- stage V^T tile (128 dims x 32 toks, 80B row stride) ----
    // unit = (8-dim chunk, 4-token group): 4 b128 global loads, then 8
    // b64 packed transpose writes (4 toks each) — 16 b64 writes/lane/tile
    // and NO scalar LDS traffic on either side of the transpose.
    if (active) {
      typedef __attribute__((ext_vector_type(4))) short short4v;
#pragma unroll
      for (int u = 0; u < 2; u++) {
        const int unit = lane + u * 64;  // 128 units = 16 d8 x 8 tokgroups
        const int d8 = unit & 15;
        const int tg = unit >> 4;
        short8 rows[4];
#pragma unroll
        for (int j = 0; j < 4; j++) {
          const int t = t0 + tg * 4 + j;
          rows[j] = (t < slab_end)
              ? *reinterpret_cast<const short8*>(
                    vcache + pbase + (int64_t)(t & (ps - 1)) * hd + d8 * 8)
              : short8{};
        }
#pragma unroll
        for (int i = 0; i < 8; i++) {
          const int d = d8 * 8 + i;
          short4v pk = {rows[0][i], rows[1][i], rows[2][i], rows[3][i]};
          *reinterpret_cast<short4v*>((char*)v_lds + d * 80 + tg * 8) = pk;
        }
      }
    }

    // ---- online softmax on the C/D layout ----
    const bool vA = (t0 + lr) < slab_end, vB = (t0 + 16 + lr) < slab_end;
    float pA[4], pB[4];
    if (active)
#pragma unroll
    for (int r = 0; r < 4; r++) {
      float a = vA ? sA[r] * scale : kNegInf;
      float bb = vB ? sB[r] * scale : kNegInf;
      float mt = row16_reduce_max(fmaxf(a, bb));
      // defer-max: mt is row-uniform after the reduce, so the skip
      // condition is uniform across the row's 16 lanes (no vote needed)
      const float thr = DEFER ? 8.0f : 0.0f;
      if (mt > m[r] + thr) {
        const float corr = (m[r] <= kNegInf * 0.5f) ? 0.f : __expf(m[r] - mt);
        l[r] *= corr;
#pragma unroll
        for (int d = 0; d < 8; d++) acc[d][r] *= corr;
        m[r] = mt;
      }
      pA[r] = (a > kNegInf * 0.5f) ? __expf(a - m[r]) : 0.f;
      pB[r] = (bb > kNegInf * 0.5f) ? __expf(bb - m[r]) : 0.f;
      l[r] += row16_reduce_sum(pA[r] + pB[r]);
    }
    // write P tile [16 heads][32 toks] (bank-spread via row XOR)
    if (active)
#pragma unroll
    for (int r = 0; r < 4; r++) {
      const int row = lg * 4 + r;
      const int x = (row & 3) << 4;
      *(short*)((char*)p_lds + row * 64 + ((lr * 2) ^ x)) = f32_to_bf16(pA[r]);
      *(short*)((char*)p_lds + row * 64 + (((16 + lr) * 2) ^ x)) = f32_to_bf16(pB[r]);
    }
    // per-wave buffers: a wave-level scheduling fence + in-order LDS
    // suffice (ds ops of one wave complete in issue order); sweep-verified
    // against the __syncthreads variant for correctness and speed
    __builtin_amdgcn_wave_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    __builtin_amdgcn_sched_barrier(0);

    // ---- PV: A = P (re-layout via LDS), B = V transposed in LDS ----
    if (active) {
    short8 pa_s;
    {
      const int row = lr;
      const int x = (row & 3) << 4;
      pa_s = *reinterpret_cast<const short8*>(
          (char*)p_lds + row * 64 + ((lg * 16) ^ x));
    }
    bf16x8_t pa = *reinterpret_cast<bf16x8_t*>(&pa_s);
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int db = 0; db < 8; db++) {
      // B-frag: lane holds V[tok = 8*lg + j][dim = db*16 + lr], j = 0..7 —
      // ONE b128 read of the transposed V^T row (80B stride spreads banks)
      short8 vb_s = *reinterpret_cast<const short8*>(
          (const char*)v_lds + (db * 16 + lr) * 80 + lg * 16);
      bf16x8_t vbf = *reinterpret_cast<bf16x8_t*>(&vb_s);
      acc[db] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vbf, acc[db],
                                                        0, 0, 0);
    }
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
    }  // active
    __builtin_amdgcn_wave_barrier();
  }

  // ---- cross-wave merge (same scheme as the VALU kernel) ----
  __syncthreads();
  float* my = merge + wid * G * (hd + 2);
#pragma unroll
  for (int r = 0; r < 4; r++) {
    const int row = lg * 4 + r;
    if (row < G) {
#pragma unroll
      for (int db = 0; db < 8; db++)
        my[row * (hd + 2) + db * 16 + lr] = acc[db][r];
      if (lr == 0) {
        my[row * (hd + 2) + hd] = m[r];
        my[row * (hd + 2) + hd + 1] = l[r];
      }
    }
  }
  __syncthreads();

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

inline int mfma_lds_bytes(int G, int hd) {
  return 4 * G * (hd + 2) * 4 + G * hd * 2 + 4 * 5120 * 2 + 4 * 512 * 2;
}

// ---------------------------------------------------------------------------
// SWAPPED-operand MFMA decode: S^T = mfma(K, Q) puts TOKENS on the MFMA M
// rows and HEADS on the columns, so each lane's softmax state is ONE
// head's (m, l) scalar and — crucially — the P -> PV-B-fragment
// re-layout becomes 4 v_cvt_pk_bf16_f32 + 4 permlane swaps IN REGISTERS
// (verified swap semantics: permlane32_swap(v0,v1) -> x = {lo: v0(l),
// hi: v1(l-32)}, y = {lo: v0(l+32), hi: v1(l)}; permlane16_swap the same
// within each 32-half). This deletes the per-tile P LDS roundtrip and its
// wave-level waitcnt fence of paged_decode_mfma, and the O rescale is one
// scalar multiply (no per-row alpha shuffles). PV computes
// O^T = mfma(V^T, P^T); K loads, V^T staging and the cross-wave merge
// keep the same shapes. Derivation of the swap network:
//   after s1 = permlane32_swap(A0, B0):
//     x1@lg{0,1} = A0(lg),   x1@lg{2,3} = B0(lg-2)
//     y1@lg{0,1} = A0(lg+2), y1@lg{2,3} = B0(lg)
//   after s2 = permlane16_swap(x1, y1):
//     x2@lg = (correct tile)(2*(lg&1)),  y2@lg = (correct tile)(2*(lg&1)+1)
// which is exactly B-frag words {tok 8lg+0..1} and {tok 8lg+4..5}; the
// (A1, B1) pair gives {8lg+2..3} and {8lg+6..7}.
// KPF: cross-tile register prefetch (T14 async-stage) — MEASURED 2.5x
// WORSE (1.4 vs 3.5 TB/s, k2048 sweep): the `kfr[cur]` runtime-indexed
// register double-buffer goes to scratch (guide rule #20), exactly the
// round-1 K-prefetch regression repeated. Kept compiled for the sweep's
// record; production uses KPF=0.
// FP8: the paged cache stores OCP e4m3 bytes (halves decode HBM traffic);
// fragments convert to bf16 in-kernel via packed v_cvt (guide: decode is
// HBM-bound, VALU converts are free under the MFMA/mem overlap).
// VS: V^T LDS row stride in BYTES. 80 (the original) makes the write
// pattern collapse: rows step 8 apart start at the same bank
// (8*20 banks % 32 == 0), so each b64 staging write is ~16-way
// conflicted (PMC: 2.3e9 SQ_LDS_BANK_CONFLICT). An 88B stride gives
// rows-step-8 a 16-bank offset (8*22 % 32 == 16), halving write
// conflicts; 72 ditto with less LDS.
// XK2: XOR the token-group slot with bit d>>4 (<<5 bytes): per write
// instruction the (d step 8) row-groups collapse to 2 bank starts; the
// extra key doubles the distinct byte-slots (8 -> 16, i.e. ~4-way
// writes). Reads shift whole 16B pairs (key constant per row), so the
// b128 PV reads stay aligned.
// VT: the V cache pages are stored TRANSPOSED (d-major: elem offset
// d * page_size + token_in_page instead of token_in_page * hd + d). The
// PV A-fragment (V^T[dim][8 consecutive tokens]) is then ONE contiguous
// b128 global load per da - the whole load_v/store_v/ds_read staging
// pipeline (16 ds_writes + ~64 VALU packs + 8 ds_reads per tile, the
// kernel's VALU bottleneck: PMC VALU:MFMA was 28:1) disappears. The
// write side pays ~32x DRAM amplification ONLY for single-token decode
// appends (~10 MB/step vs 22 GB/step of V reads); prefill appends cover
// whole 64B lines in L2 before eviction.
// LG2: softmax in the log2 domain + interior-tile mask elision. The
// per-element chain mul(scale)+sub+mul(log2e)+exp collapses to
// fma(s, scale*log2e, -m') + v_exp (saves 16 VALU/tile), and interior
// tiles (t0+32 <= slab_end) skip the 8 cmp+sel masks. m is converted
// back to the natural-log domain at the merge so phase2 is unchanged.
// MINW: minimum waves/SIMD the register allocator must honor (the VT2
// double-buffer costs 285 VGPR+AGPR = 1 wave/SIMD; forcing 2 trades
// registers for occupancy - decide by measurement, watch ScratchSize)
template <int DEFER = 1, int PRIO = 1, int KPF = 0, int FP8 = 0,
          int VS = 80, int XK2 = 0, int VT = 0, int LG2 = 0, int MINW = 1>
__global__ __launch_bounds__(kBlock, MINW) void paged_decode_mfma_swapped(
    float* __restrict__ partial, float* __restrict__ ml,
    short* __restrict__ out, const short* __restrict__ q,
    const short* __restrict__ kcache, const short* __restrict__ vcache,
    const int32_t* __restrict__ page_table, const int32_t* __restrict__ ctx_lens,
    float scale, int chunk, int G, int B, int Hkv, int C, int max_pages,
    int log2_ps, int hd, int32_t* __restrict__ chunk_cnt) {
  // chunk_cnt (optional, [B*Hkv] zero-initialized): fused chunk merge.
  // When non-null and C > 1, every block of a (b, h) column counts down
  // via atomicAdd after publishing its partial/ml; the LAST block inlines
  // the phase2 merge and resets the counter - one kernel launch instead
  // of two per decode layer (~0.45 ms/step on the 80-layer flagship).
  typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
  typedef __attribute__((ext_vector_type(2))) unsigned int uint2_t;
  const int kSlab = chunk / 4;
  const int b = blockIdx.x;
  const int h = blockIdx.y;
  const int c = blockIdx.z;
  const int Hq = Hkv * G;
  const int ctx = ctx_lens[b];
  const int chunk_start = c * chunk;
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int lr = lane & 15;
  const int lg = lane >> 4;
  const int ps = 1 << log2_ps;

  extern __shared__ float lds[];
  float* merge = lds;                                   // 4*G*(hd+2)
  short* q_lds_s = reinterpret_cast<short*>(merge + 4 * G * (hd + 2));
  short* v_lds = q_lds_s + G * hd + wid * (hd * VS / 2);  // per-wave

  auto fused_merge = [&]() {
    // countdown; the last-arriving block of this (b, h) column merges
    __threadfence();
    __shared__ int lastflag;
    if (threadIdx.x == 0) {
      const int prev = atomicAdd(&chunk_cnt[b * Hkv + h], 1);
      lastflag = (prev == (int)gridDim.z - 1) ? 1 : 0;
    }
    __syncthreads();
    if (!lastflag) return;
    __threadfence();  // acquire the other blocks' partial/ml stores
    if (threadIdx.x == 0) chunk_cnt[b * Hkv + h] = 0;  // next-launch reset
    const int nc = min(C, (ctx + chunk - 1) / chunk);
    for (int i = threadIdx.x; i < G * hd; i += kBlock) {
      const int g = i / hd;
      const int d = i % hd;
      const int qh = h * G + g;
      const float* mlp = ml + (((int64_t)b * Hq + qh) * C) * 2;
      float mstar = kNegInf;
      for (int cc = 0; cc < nc; cc++) mstar = fmaxf(mstar, mlp[2 * cc]);
      float asum = 0.f, lsum2 = 0.f;
      for (int cc = 0; cc < nc; cc++) {
        const float lc = mlp[2 * cc + 1];
        if (lc <= 0.f) continue;
        const float corr = __expf(mlp[2 * cc] - mstar);
        asum += partial[(((int64_t)b * Hq + qh) * C + cc) * hd + d] * corr;
        lsum2 += lc * corr;
      }
      out[((int64_t)b * Hq + qh) * hd + d] =
          f32_to_bf16(lsum2 > 0.f ? asum / lsum2 : 0.f);
    }
  };

  if (chunk_start >= ctx) {
    if (C > 1) {
      for (int i = threadIdx.x; i < G; i += kBlock) {
        const int qh = h * G + i;
        float* mlp = ml + (((int64_t)b * Hq + qh) * C + c) * 2;
        mlp[0] = kNegInf; mlp[1] = 0.f;
      }
      if (chunk_cnt != nullptr) fused_merge();
    }
    return;
  }

  for (int i = threadIdx.x; i < G * hd; i += kBlock) {
    const int g = i / hd;
    q_lds_s[i] = q[((int64_t)b * Hq + h * G + g) * hd + i % hd];
  }
  __syncthreads();

  // Q as the QK B-operand: same fragment registers as the A-frag variant
  bf16x8_t q_frag[4];
#pragma unroll
  for (int kc = 0; kc < 4; kc++) {
    short8 v{};
    if (lr < G)
      v = *reinterpret_cast<const short8*>(q_lds_s + lr * hd + kc * 32 + lg * 8);
    q_frag[kc] = *reinterpret_cast<bf16x8_t*>(&v);
  }

  // per-lane softmax state for head `lr` (replicated across the 4 lg
  // groups); acc[da][r] = O^T[dim da*16 + lg*4 + r][head lr]
  float m_run = kNegInf, l_run = 0.f;
  f32x4 acc[8];
#pragma unroll
  for (int d = 0; d < 8; d++) acc[d] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int slab_start = chunk_start + wid * kSlab;
  const int slab_end = min(slab_start + kSlab, ctx);
  const int32_t* pt = page_table + (int64_t)b * max_pages;

  // prefetchable loads (KPF): K fragments (8x short8) + V staging rows
  // (8x short8) per tile, double-buffered across iterations
  typedef __attribute__((ext_vector_type(4))) short short4v;
  auto ld8 = [&](const short* base, int64_t elem_off) -> short8 {
    // one 8-element K/V slice as a bf16x8-compatible short8; FP8 caches
    // hold bytes at the same ELEMENT offsets
    if constexpr (FP8) {
      const unsigned char* b =
          reinterpret_cast<const unsigned char*>(base) + elem_off;
      uchar8 raw = *reinterpret_cast<const uchar8*>(b);
      bf16x8 cv = fp8x8_to_bf16x8(raw);
      return *reinterpret_cast<short8*>(&cv);
    } else {
      return *reinterpret_cast<const short8*>(base + elem_off);
    }
  };
  auto load_k = [&](int t0_, short8 (&kf)[8]) {
    if (t0_ >= slab_end) return;
    const int64_t pb = (((int64_t)pt[t0_ >> log2_ps] * Hkv + h) * ps) * hd;
    const int tA = t0_ + lr, tB = t0_ + 16 + lr;
    const int64_t oA = pb + (int64_t)(tA & (ps - 1)) * hd;
    const int64_t oB = pb + (int64_t)(tB & (ps - 1)) * hd;
    const bool vA = tA < slab_end, vB = tB < slab_end;
#pragma unroll
    for (int kc = 0; kc < 4; kc++) {
      kf[kc] = vA ? ld8(kcache, oA + kc * 32 + lg * 8) : short8{};
      kf[4 + kc] = vB ? ld8(kcache, oB + kc * 32 + lg * 8) : short8{};
    }
  };
  auto load_v = [&](int t0_, short8 (&vf)[8]) {
    if (t0_ >= slab_end) return;
    const int64_t pb = (((int64_t)pt[t0_ >> log2_ps] * Hkv + h) * ps) * hd;
#pragma unroll
    for (int u = 0; u < 2; u++) {
      const int unit = lane + u * 64;
      const int d8 = unit & 15;
      const int tg = unit >> 4;
#pragma unroll
      for (int j = 0; j < 4; j++) {
        const int t = t0_ + tg * 4 + j;
        vf[u * 4 + j] = (t < slab_end)
            ? ld8(vcache, pb + (int64_t)(t & (ps - 1)) * hd + d8 * 8)
            : short8{};
      }
    }
  };
  auto load_vt = [&](int t0_, short8 (&vf)[8]) {
    const int64_t pb = (((int64_t)pt[t0_ >> log2_ps] * Hkv + h) * ps) * hd;
    const int tin = t0_ & (ps - 1);
#pragma unroll
    for (int da = 0; da < 8; da++) {
      const int vrow = da * 16 + lr;
      vf[da] = ld8(vcache, pb + (int64_t)vrow * ps + tin + lg * 8);
    }
    // boundary tile: zero tokens past slab_end (page tails can hold
    // stale NaN/Inf bits; P=0 alone does not mask NaN * 0)
    const int rem = slab_end - t0_;
    if (rem < 32) {
#pragma unroll
      for (int da = 0; da < 8; da++)
#pragma unroll
        for (int j = 0; j < 8; j++)
          if (lg * 8 + j >= rem) vf[da][j] = 0;
    }
  };
  auto store_v = [&](const short8 (&vf)[8]) {
#pragma unroll
    for (int u = 0; u < 2; u++) {
      const int unit = lane + u * 64;
      const int d8 = unit & 15;
      const int tg = unit >> 4;
#pragma unroll
      for (int i = 0; i < 8; i++) {
        const int d = d8 * 8 + i;
        short4v pk = {vf[u * 4 + 0][i], vf[u * 4 + 1][i],
                      vf[u * 4 + 2][i], vf[u * 4 + 3][i]};
        const int toff = XK2 ? ((tg * 8) ^ (((d >> 4) & 1) << 5)) : (tg * 8);
        *reinterpret_cast<short4v*>((char*)v_lds + d * VS + toff) = pk;
      }
    }
  };

  short8 kfr[2][8], vfr[2][8];
  if constexpr (KPF) {
    load_k(slab_start, kfr[0]);
    load_v(slab_start, vfr[0]);
  }

  // VT-only tile compute (QK -> softmax -> P pack -> PV from registers):
  // a specialization of the main-loop body below; keep the two in sync.
  const float scale2 = scale * 1.44269504f;  // scale * log2(e)
  auto tile_vt = [&](int t0, const short8 (&kf)[8], const short8 (&vf)[8]) {
    f32x4 sA{0.f, 0.f, 0.f, 0.f}, sB{0.f, 0.f, 0.f, 0.f};
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kc = 0; kc < 4; kc++) {
      short8 ka = kf[kc];
      sA = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          *reinterpret_cast<bf16x8_t*>(&ka), q_frag[kc], sA, 0, 0, 0);
    }
    if (t0 + 16 < slab_end) {
#pragma unroll
      for (int kc = 0; kc < 4; kc++) {
        short8 kb2 = kf[4 + kc];
        sB = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            *reinterpret_cast<bf16x8_t*>(&kb2), q_frag[kc], sB, 0, 0, 0);
      }
    }
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
    float pA[4], pB[4];
    float mt = kNegInf;
    if constexpr (LG2) {
      // raw (unscaled) values; max commutes with the positive scale
      const bool tail = t0 + 32 > slab_end;
      if (!tail) {
#pragma unroll
        for (int r = 0; r < 4; r++) {
          pA[r] = sA[r];
          pB[r] = sB[r];
          mt = fmaxf(mt, fmaxf(pA[r], pB[r]));
        }
      } else {
#pragma unroll
        for (int r = 0; r < 4; r++) {
          const int tokA = t0 + lg * 4 + r, tokB = t0 + 16 + lg * 4 + r;
          pA[r] = (tokA < slab_end) ? sA[r] : kNegInf;
          pB[r] = (tokB < slab_end) ? sB[r] : kNegInf;
          mt = fmaxf(mt, fmaxf(pA[r], pB[r]));
        }
      }
      mt = fmaxf(mt, __shfl_xor(mt, 16, WAVE_SIZE));
      mt = fmaxf(mt, __shfl_xor(mt, 32, WAVE_SIZE));
      const float mts = (mt <= kNegInf * 0.5f) ? kNegInf : mt * scale2;
      const float thr = DEFER ? 8.0f * 1.44269504f : 0.0f;
      if (mts > m_run + thr) {
        const float corr =
            (m_run <= kNegInf * 0.5f) ? 0.f : __builtin_amdgcn_exp2f(m_run - mts);
        l_run *= corr;
#pragma unroll
        for (int d = 0; d < 8; d++) acc[d] *= corr;
        m_run = mts;
      }
      float psum = 0.f;
#pragma unroll
      for (int r = 0; r < 4; r++) {
        pA[r] = (pA[r] > kNegInf * 0.5f)
                    ? __builtin_amdgcn_exp2f(fmaf(pA[r], scale2, -m_run)) : 0.f;
        pB[r] = (pB[r] > kNegInf * 0.5f)
                    ? __builtin_amdgcn_exp2f(fmaf(pB[r], scale2, -m_run)) : 0.f;
        psum += pA[r] + pB[r];
      }
      psum += __shfl_xor(psum, 16, WAVE_SIZE);
      psum += __shfl_xor(psum, 32, WAVE_SIZE);
      l_run += psum;
    } else {
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const int tokA = t0 + lg * 4 + r, tokB = t0 + 16 + lg * 4 + r;
        pA[r] = (tokA < slab_end) ? sA[r] * scale : kNegInf;
        pB[r] = (tokB < slab_end) ? sB[r] * scale : kNegInf;
        mt = fmaxf(mt, fmaxf(pA[r], pB[r]));
      }
      mt = fmaxf(mt, __shfl_xor(mt, 16, WAVE_SIZE));
      mt = fmaxf(mt, __shfl_xor(mt, 32, WAVE_SIZE));
      const float thr = DEFER ? 8.0f : 0.0f;
      if (mt > m_run + thr) {
        const float corr = (m_run <= kNegInf * 0.5f) ? 0.f : __expf(m_run - mt);
        l_run *= corr;
#pragma unroll
        for (int d = 0; d < 8; d++) acc[d] *= corr;
        m_run = mt;
      }
      float psum = 0.f;
#pragma unroll
      for (int r = 0; r < 4; r++) {
        pA[r] = (pA[r] > kNegInf * 0.5f) ? __expf(pA[r] - m_run) : 0.f;
        pB[r] = (pB[r] > kNegInf * 0.5f) ? __expf(pB[r] - m_run) : 0.f;
        psum += pA[r] + pB[r];
      }
      psum += __shfl_xor(psum, 16, WAVE_SIZE);
      psum += __shfl_xor(psum, 32, WAVE_SIZE);
      l_run += psum;
    }
    auto cvtpk = [](float a, float bb) {
      unsigned int r;
      asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(a), "v"(bb));
      return r;
    };
    const unsigned int A0 = cvtpk(pA[0], pA[1]), A1 = cvtpk(pA[2], pA[3]);
    const unsigned int B0 = cvtpk(pB[0], pB[1]), B1 = cvtpk(pB[2], pB[3]);
    uint2_t s1 = __builtin_amdgcn_permlane32_swap(A0, B0, false, false);
    uint2_t s2 = __builtin_amdgcn_permlane32_swap(A1, B1, false, false);
    uint2_t f02 = __builtin_amdgcn_permlane16_swap(s1.x, s1.y, false, false);
    uint2_t f13 = __builtin_amdgcn_permlane16_swap(s2.x, s2.y, false, false);
    unsigned int w[4] = {f02.x, f13.x, f02.y, f13.y};
    bf16x8_t p_frag = *reinterpret_cast<bf16x8_t*>(w);
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int da = 0; da < 8; da++) {
      short8 va_s = vf[da];
      acc[da] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          *reinterpret_cast<bf16x8_t*>(&va_s), p_frag, acc[da], 0, 0, 0);
    }
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
  };

  // ---- VT3: 64-token tiles ----------------------------------------
  // Rationale (PMC): at 32-token tiles the wave issues only 16 b128
  // loads per compute batch, so load-issue duty caps effective BW at
  // ~77% of the pattern ceiling (WAIT_ANY-dominated). 64-token tiles
  // issue 32 loads back-to-back (2 KB in flight per wave) before the
  // QK/softmax/PV batch. Single-buffered; MINW=2 keeps 2 waves/SIMD.
  auto softmax_pack64 = [&](int t0, f32x4 (&sT)[4], bf16x8_t (&p_frag)[2]) {
    float pv[4][4];
    float mt = kNegInf;
    const bool tail = t0 + 64 > slab_end;
    if (!tail) {
#pragma unroll
      for (int tg = 0; tg < 4; tg++)
#pragma unroll
        for (int r = 0; r < 4; r++) {
          pv[tg][r] = sT[tg][r] * scale;
          mt = fmaxf(mt, pv[tg][r]);
        }
    } else {
#pragma unroll
      for (int tg = 0; tg < 4; tg++)
#pragma unroll
        for (int r = 0; r < 4; r++) {
          const int tok = t0 + tg * 16 + lg * 4 + r;
          pv[tg][r] = (tok < slab_end) ? sT[tg][r] * scale : kNegInf;
          mt = fmaxf(mt, pv[tg][r]);
        }
    }
    mt = fmaxf(mt, __shfl_xor(mt, 16, WAVE_SIZE));
    mt = fmaxf(mt, __shfl_xor(mt, 32, WAVE_SIZE));
    const float thr = DEFER ? 8.0f : 0.0f;
    if (mt > m_run + thr) {
      const float corr = (m_run <= kNegInf * 0.5f) ? 0.f : __expf(m_run - mt);
      l_run *= corr;
#pragma unroll
      for (int d = 0; d < 8; d++) acc[d] *= corr;
      m_run = mt;
    }
    float psum = 0.f;
#pragma unroll
    for (int tg = 0; tg < 4; tg++)
#pragma unroll
      for (int r = 0; r < 4; r++) {
        pv[tg][r] = (pv[tg][r] > kNegInf * 0.5f)
                        ? __expf(pv[tg][r] - m_run) : 0.f;
        psum += pv[tg][r];
      }
    psum += __shfl_xor(psum, 16, WAVE_SIZE);
    psum += __shfl_xor(psum, 32, WAVE_SIZE);
    l_run += psum;
    auto cvtpk = [](float a, float bb) {
      unsigned int r;
      asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(a), "v"(bb));
      return r;
    };
    // two 32-token PV B-frags, each combining two 16-token S tiles via
    // the same cvt_pk + permlane swap network as the 32-token path
#pragma unroll
    for (int tc = 0; tc < 2; tc++) {
      const float* pa = pv[tc * 2];
      const float* pb = pv[tc * 2 + 1];
      const unsigned int A0 = cvtpk(pa[0], pa[1]), A1 = cvtpk(pa[2], pa[3]);
      const unsigned int B0 = cvtpk(pb[0], pb[1]), B1 = cvtpk(pb[2], pb[3]);
      uint2_t s1 = __builtin_amdgcn_permlane32_swap(A0, B0, false, false);
      uint2_t s2 = __builtin_amdgcn_permlane32_swap(A1, B1, false, false);
      uint2_t f02 = __builtin_amdgcn_permlane16_swap(s1.x, s1.y, false, false);
      uint2_t f13 = __builtin_amdgcn_permlane16_swap(s2.x, s2.y, false, false);
      unsigned int w[4] = {f02.x, f13.x, f02.y, f13.y};
      p_frag[tc] = *reinterpret_cast<bf16x8_t*>(w);
    }
  };
  auto tile_vt64_post = [&](int t0, f32x4 (&sT)[4],
                            const short8 (&vf)[16]) {
    bf16x8_t p_frag[2];
    softmax_pack64(t0, sT, p_frag);
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int da = 0; da < 8; da++)
#pragma unroll
      for (int tc = 0; tc < 2; tc++) {
        short8 va_s = vf[da * 2 + tc];
        acc[da] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            *reinterpret_cast<bf16x8_t*>(&va_s), p_frag[tc], acc[da], 0, 0, 0);
      }
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
  };
  auto tile_vt64 = [&](int t0, const short8 (&kf)[16],
                       const short8 (&vf)[16]) {
    f32x4 sT[4];
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int tg = 0; tg < 4; tg++) {
      f32x4 acc_s{0.f, 0.f, 0.f, 0.f};
      if (t0 + tg * 16 < slab_end) {
#pragma unroll
        for (int kc = 0; kc < 4; kc++) {
          short8 ka = kf[tg * 4 + kc];
          acc_s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              *reinterpret_cast<bf16x8_t*>(&ka), q_frag[kc], acc_s, 0, 0, 0);
        }
      }
      sT[tg] = acc_s;
    }
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
    tile_vt64_post(t0, sT, vf);
  };
  auto load_k64 = [&](int t0_, short8 (&kf)[16]) {
    // uniform (no break-in-unroll): groups past slab_end clamp their page
    // lookup to the last valid token and mask values via vA
#pragma unroll
    for (int tg = 0; tg < 4; tg++) {
      const int tb = t0_ + tg * 16;
      const int tsafe = tb < slab_end ? tb : slab_end - 1;
      const int64_t pb =
          (((int64_t)pt[tsafe >> log2_ps] * Hkv + h) * ps) * hd;
      const int tA = tb + lr;
      const int64_t oA = pb + (int64_t)(tA & (ps - 1)) * hd;
      const bool vA = tA < slab_end;
#pragma unroll
      for (int kc = 0; kc < 4; kc++)
        kf[tg * 4 + kc] = vA ? ld8(kcache, oA + kc * 32 + lg * 8) : short8{};
    }
  };
  auto load_v64 = [&](int t0_, short8 (&vf)[16]) {
    // uniform: a 32-token group fully past slab_end clamps its page to the
    // last valid 32-block (safe memory) and rem <= 0 zeroes every element
#pragma unroll
    for (int tc = 0; tc < 2; tc++) {
      const int tb = t0_ + tc * 32;
      const int tlast = ((slab_end - 1) >> 5) << 5;  // last valid 32-block
      const int tsafe = tb < slab_end ? tb : tlast;
      const int64_t pb =
          (((int64_t)pt[tsafe >> log2_ps] * Hkv + h) * ps) * hd;
      const int tin = tsafe & (ps - 1);
      const int rem = slab_end - tb;
#pragma unroll
      for (int da = 0; da < 8; da++) {
        const int vrow = da * 16 + lr;
        short8 v8 = ld8(vcache, pb + (int64_t)vrow * ps + tin + lg * 8);
        if (rem < 32) {
#pragma unroll
          for (int j = 0; j < 8; j++)
            if (lg * 8 + j >= rem) v8[j] = 0;
        }
        vf[da * 2 + tc] = v8;
      }
    }
  };

  // ---- VT4: VT3 + K staged through per-wave LDS --------------------
  // K's direct B-frag loads touch 16 rows x 64B per instruction (the
  // 5.74 TB/s pattern); staging K cooperatively IN-WAVE with 4-row x
  // 256B fully-contiguous 1KB bursts targets the 6.3 linear ceiling for
  // half the kernel's bytes. Two 32-token halves reuse one 8 KB buffer
  // per wave (block LDS stays under 2-resident at MINW=2); V loads stay
  // direct (their rows are already 128B-contiguous).
  auto stage_k32 = [&](int tb32, short* klds_w, short8 (&sreg)[8]) {
    // 8 bursts: burst i covers rows {i*4 .. i*4+3} fully (lane>>4 = row,
    // lane&15 = 16B chunk)
#pragma unroll
    for (int i = 0; i < 8; i++) {
      const int tok = tb32 + i * 4 + (lane >> 4);
      const int tsafe = tok < slab_end ? tok : slab_end - 1;
      const int64_t pb =
          (((int64_t)pt[tsafe >> log2_ps] * Hkv + h) * ps) * hd;
      sreg[i] = ld8(kcache,
                    pb + (int64_t)(tsafe & (ps - 1)) * hd + (lane & 15) * 8);
    }
#pragma unroll
    for (int i = 0; i < 8; i++) {
      const int row = i * 4 + (lane >> 4);
      *reinterpret_cast<short8*>(
          (char*)klds_w + row * 256 +
          (((lane & 15) * 16) ^ ((row & 7) << 4))) = sreg[i];
    }
    __builtin_amdgcn_wave_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  };
  auto qk_half = [&](int t0, int half, const short* klds_w, f32x4 (&sT)[4]) {
#pragma unroll
    for (int tg2 = 0; tg2 < 2; tg2++) {
      const int tg = half * 2 + tg2;
      f32x4 acc_s{0.f, 0.f, 0.f, 0.f};
      if (t0 + tg * 16 < slab_end) {
#pragma unroll
        for (int kc = 0; kc < 4; kc++) {
          const int row = tg2 * 16 + lr;
          short8 ka = *reinterpret_cast<const short8*>(
              (const char*)klds_w + row * 256 +
              ((kc * 64 + lg * 16) ^ ((row & 7) << 4)));
          const bool valid = t0 + tg * 16 + lr < slab_end;
          if (!valid) ka = short8{};
          acc_s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              *reinterpret_cast<bf16x8_t*>(&ka), q_frag[kc], acc_s, 0, 0, 0);
        }
      }
      sT[tg] = acc_s;
    }
  };

  // ---- VT5: VT4 + V through the same LDS buffer, fully linear -------
  // With ps == 64 every 64-token tile is page-aligned, so a (page, head)
  // d-major V plane is 16 KB CONTIGUOUS. Stage it in two 8 KB dim-halves
  // (dims [0,64) then [64,128)) with 1KB linear bursts, consuming each
  // half with the matching PV da range (da 0-3 reads rows < 64).
  auto stage_v_half = [&](int tb64, int dlo, short* klds_w) {
    const int64_t pb =
        (((int64_t)pt[tb64 >> log2_ps] * Hkv + h) * ps) * hd;
    const int rem = slab_end - tb64;
    short8 vreg[8];
#pragma unroll
    for (int i = 0; i < 8; i++)
      vreg[i] = ld8(vcache, pb + (int64_t)dlo * ps + (i * 64 + lane) * 8);
    if (rem < 64) {
      const int tok0 = (lane & 7) * 8;
#pragma unroll
      for (int i = 0; i < 8; i++)
#pragma unroll
        for (int j = 0; j < 8; j++)
          if (tok0 + j >= rem) vreg[i][j] = 0;
    }
#pragma unroll
    for (int i = 0; i < 8; i++) {
      const int row = i * 8 + (lane >> 3);
      *reinterpret_cast<short8*>(
          (char*)klds_w + row * 128 +
          ((((lane & 7)) * 16) ^ (((row ^ (row >> 3)) & 7) << 4))) = vreg[i];
    }
    __builtin_amdgcn_wave_barrier();
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  };
  auto pv_half = [&](int dabase, const short* klds_w,
                     const bf16x8_t (&p_frag)[2]) {
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int dl = 0; dl < 4; dl++)
#pragma unroll
      for (int tc = 0; tc < 2; tc++) {
        const int row = dl * 16 + lr;
        short8 va_s = *reinterpret_cast<const short8*>(
            (const char*)klds_w + row * 128 +
            ((tc * 64 + lg * 16) ^ (((row ^ (row >> 3)) & 7) << 4)));
        acc[dabase + dl] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            *reinterpret_cast<bf16x8_t*>(&va_s), p_frag[tc],
            acc[dabase + dl], 0, 0, 0);
      }
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
  };

  if constexpr (VT == 5) {
    short* klds_w = v_lds;
    short8 sreg[8];
    for (int t0 = slab_start; t0 < slab_end; t0 += 64) {
      f32x4 sT[4];
      stage_k32(t0, klds_w, sreg);
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
      qk_half(t0, 0, klds_w, sT);
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_wave_barrier();
      stage_k32(t0 + 32, klds_w, sreg);
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
      qk_half(t0, 1, klds_w, sT);
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
      bf16x8_t p_frag[2];
      softmax_pack64(t0, sT, p_frag);
      __builtin_amdgcn_wave_barrier();
      stage_v_half(t0, 0, klds_w);
      pv_half(0, klds_w, p_frag);
      __builtin_amdgcn_wave_barrier();
      stage_v_half(t0, 64, klds_w);
      pv_half(4, klds_w, p_frag);
      // back-edge fence: the NEXT tile's staging ds_writes must not be
      // scheduled above this tile's outstanding LDS frag reads
      __builtin_amdgcn_wave_barrier();
    }
  } else if constexpr (VT == 4) {
    short* klds_w = v_lds;  // reuse the (otherwise unused) V-staging LDS
    short8 v64[16], sreg[8];
    for (int t0 = slab_start; t0 < slab_end; t0 += 64) {
      f32x4 sT[4];
      stage_k32(t0, klds_w, sreg);          // K half A (tokens t0..t0+31)
      load_v64(t0, v64);                    // V loads overlap QK below
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
      qk_half(t0, 0, klds_w, sT);
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_wave_barrier();
      stage_k32(t0 + 32, klds_w, sreg);     // K half B overwrites A
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
      qk_half(t0, 1, klds_w, sT);
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
      tile_vt64_post(t0, sT, v64);
      __builtin_amdgcn_wave_barrier();  // back-edge fence (see VT5)
    }
  } else if constexpr (VT == 3) {
    short8 k64[16], v64[16];
    for (int t0 = slab_start; t0 < slab_end; t0 += 64) {
      load_k64(t0, k64);
      load_v64(t0, v64);
      tile_vt64(t0, k64, v64);
    }
  } else if constexpr (VT == 2) {
    // software-pipelined: tile i+1's K/V global loads are in flight while
    // tile i computes; two compile-time register sets (no runtime
    // indexing -> no scratch spill, unlike KPF)
    short8 kA[8], vA[8], kB[8], vB[8];
    const int ntiles = kSlab / 32;
    if (slab_start < slab_end) {
      load_k(slab_start, kA);
      load_vt(slab_start, vA);
    }
    for (int ti = 0; ti < ntiles; ti += 2) {
      const int t0 = slab_start + ti * 32;
      const int t1 = t0 + 32;
      if (t1 < slab_end) {
        load_k(t1, kB);
        load_vt(t1, vB);
      }
      if (t0 < slab_end) tile_vt(t0, kA, vA);
      const int t2 = t0 + 64;
      if (t2 < slab_end) {
        load_k(t2, kA);
        load_vt(t2, vA);
      }
      if (t1 < slab_end) tile_vt(t1, kB, vB);
    }
  } else {
  for (int ti = 0; ti < kSlab / 32; ti++) {
    const int t0 = slab_start + ti * 32;
    const bool active = t0 < slab_end;
    const int cur = ti & 1;
    f32x4 sA{0.f, 0.f, 0.f, 0.f}, sB{0.f, 0.f, 0.f, 0.f};
    if constexpr (!KPF) {
      if (active) {
        load_k(t0, kfr[0]);
        if constexpr (VT) load_vt(t0, vfr[0]);
        else load_v(t0, vfr[0]);
      }
    }
    const short8(&kf)[8] = KPF ? kfr[cur] : kfr[0];
    const short8(&vf)[8] = KPF ? vfr[cur] : vfr[0];
    if (active) {
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kc = 0; kc < 4; kc++) {
        short8 ka = kf[kc];
        sA = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            *reinterpret_cast<bf16x8_t*>(&ka), q_frag[kc], sA, 0, 0, 0);
      }
      if (t0 + 16 < slab_end) {
#pragma unroll
        for (int kc = 0; kc < 4; kc++) {
          short8 kb2 = kf[4 + kc];
          sB = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              *reinterpret_cast<bf16x8_t*>(&kb2), q_frag[kc], sB, 0, 0, 0);
        }
      }
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
      // stage THIS tile's V rows (loaded last iteration under KPF);
      // VT reads V^T fragments straight from global - nothing to stage
      if constexpr (!VT) store_v(vf);
      if constexpr (KPF) {
        // issue next tile's K/V global loads now: their latency hides
        // under this tile's softmax + PV
        load_k(t0 + 32, kfr[cur ^ 1]);
        load_v(t0 + 32, vfr[cur ^ 1]);
      }
    }

    if (active) {
      // ---- softmax: rows are TOKENS (lg*4+r, +16 for the B tile), the
      // lane's column is head lr; token-reduction = 8-reg fold + 2 shfls
      float pA[4], pB[4];
      float mt = kNegInf;
#pragma unroll
      for (int r = 0; r < 4; r++) {
        const int tokA = t0 + lg * 4 + r, tokB = t0 + 16 + lg * 4 + r;
        pA[r] = (tokA < slab_end) ? sA[r] * scale : kNegInf;
        pB[r] = (tokB < slab_end) ? sB[r] * scale : kNegInf;
        mt = fmaxf(mt, fmaxf(pA[r], pB[r]));
      }
      mt = fmaxf(mt, __shfl_xor(mt, 16, WAVE_SIZE));
      mt = fmaxf(mt, __shfl_xor(mt, 32, WAVE_SIZE));
      const float thr = DEFER ? 8.0f : 0.0f;
      if (mt > m_run + thr) {
        const float corr = (m_run <= kNegInf * 0.5f) ? 0.f
                                                     : __expf(m_run - mt);
        l_run *= corr;
#pragma unroll
        for (int d = 0; d < 8; d++) acc[d] *= corr;
        m_run = mt;
      }
      float psum = 0.f;
#pragma unroll
      for (int r = 0; r < 4; r++) {
        pA[r] = (pA[r] > kNegInf * 0.5f) ? __expf(pA[r] - m_run) : 0.f;
        pB[r] = (pB[r] > kNegInf * 0.5f) ? __expf(pB[r] - m_run) : 0.f;
        psum += pA[r] + pB[r];
      }
      psum += __shfl_xor(psum, 16, WAVE_SIZE);
      psum += __shfl_xor(psum, 32, WAVE_SIZE);
      l_run += psum;

      // ---- P^T -> PV B-fragment in registers (4 cvt_pk + 4 swaps) ----
      auto cvtpk = [](float a, float bb) {
        unsigned int r;
        asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(a), "v"(bb));
        return r;
      };
      const unsigned int A0 = cvtpk(pA[0], pA[1]), A1 = cvtpk(pA[2], pA[3]);
      const unsigned int B0 = cvtpk(pB[0], pB[1]), B1 = cvtpk(pB[2], pB[3]);
      uint2_t s1 = __builtin_amdgcn_permlane32_swap(A0, B0, false, false);
      uint2_t s2 = __builtin_amdgcn_permlane32_swap(A1, B1, false, false);
      uint2_t f02 = __builtin_amdgcn_permlane16_swap(s1.x, s1.y, false, false);
      uint2_t f13 = __builtin_amdgcn_permlane16_swap(s2.x, s2.y, false, false);
      unsigned int w[4] = {f02.x, f13.x, f02.y, f13.y};
      bf16x8_t p_frag = *reinterpret_cast<bf16x8_t*>(w);

      if constexpr (!VT) {
        // V staging for THIS tile must be visible (same wave, in-order LDS)
        __builtin_amdgcn_wave_barrier();
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
      }

      // ---- PV: O^T[dim][head] += V^T[dim][tok] P^T[tok][head] ----
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int da = 0; da < 8; da++) {
        short8 va_s;
        if constexpr (VT) {
          va_s = vf[da];
        } else {
          const int vrow = da * 16 + lr;
          const int vtoff = XK2 ? ((lg * 16) ^ (((vrow >> 4) & 1) << 5))
                                : (lg * 16);
          va_s = *reinterpret_cast<const short8*>(
              (const char*)v_lds + vrow * VS + vtoff);
        }
        acc[da] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            *reinterpret_cast<bf16x8_t*>(&va_s), p_frag, acc[da], 0, 0, 0);
      }
      if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
    }
    __builtin_amdgcn_wave_barrier();
  }
  }  // VT != 2

  // ---- cross-wave merge: acc[da][r] is O^T[dim da*16+lg*4+r][head lr]
  __syncthreads();
  float* my = merge + wid * G * (hd + 2);
  if (lr < G) {
#pragma unroll
    for (int da = 0; da < 8; da++)
#pragma unroll
      for (int r = 0; r < 4; r++)
        my[lr * (hd + 2) + da * 16 + lg * 4 + r] = acc[da][r];
    if (lg == 0) {
      my[lr * (hd + 2) + hd] = m_run;
      my[lr * (hd + 2) + hd + 1] = l_run;
    }
  }
  __syncthreads();

  for (int i = threadIdx.x; i < G * hd; i += kBlock) {
    const int g = i / hd;
    const int d = i % hd;
    float mw[4], lw[4];
    float mstar = kNegInf;
#pragma unroll
    for (int w2 = 0; w2 < 4; w2++) {
      mw[w2] = merge[w2 * G * (hd + 2) + g * (hd + 2) + hd];
      lw[w2] = merge[w2 * G * (hd + 2) + g * (hd + 2) + hd + 1];
      mstar = fmaxf(mstar, mw[w2]);
    }
    float lsum = 0.f, asum = 0.f;
#pragma unroll
    for (int w2 = 0; w2 < 4; w2++) {
      const float corr = (lw[w2] > 0.f) ? __expf(mw[w2] - mstar) : 0.f;
      lsum += lw[w2] * corr;
      asum += merge[w2 * G * (hd + 2) + g * (hd + 2) + d] * corr;
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
  if (C > 1 && chunk_cnt != nullptr) {
    __syncthreads();  // all merge-loop writes of this block done
    fused_merge();
  }
}

// VT layout needs no V staging LDS: merge + q only
inline int mfma_swapped_vt_lds_bytes(int G, int hd) {
  return 4 * G * (hd + 2) * 4 + G * hd * 2;
}

inline int mfma_swapped_lds_bytes(int G, int hd, int vs = 80) {
  return 4 * G * (hd + 2) * 4 + G * hd * 2 + 4 * hd * vs;  // no P tile
}

// Phase 2: merge chunk partials. grid (B, Hq), block = 128.
__global__ inline void paged_decode_phase2(short* __restrict__ out,
                                           const float* __restrict__ partial,
                                           const float* __restrict__ ml,
                                           const int32_t* __restrict__ ctx_lens,
                                           int chunk, int Hq, int C, int hd) {
  const int b = blockIdx.x;
  const int qh = blockIdx.y;
  const int nc = min(C, (ctx_lens[b] + chunk - 1) / chunk);
  const float* mlp = ml + (((int64_t)b * Hq + qh) * C) * 2;

  __shared__ float smax[1];
  float mstar = kNegInf;
  for (int c = threadIdx.x; c < nc; c += blockDim.x) mstar = fmaxf(mstar, mlp[2 * c]);
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    mstar = fmaxf(mstar, __shfl_xor(mstar, off, WAVE_SIZE));
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
    out[((int64_t)b * Hq + qh) * hd + d] =
        f32_to_bf16(lsum > 0.f ? asum / lsum : 0.f);
  }
}

// LDS bytes for phase 1
inline int phase1_lds_bytes(int G, int HS, int hd) {
  const int TG = 4 / HS;
  const int merge_f = (TG > 1) ? TG * G * (hd + 2) : 0;
  return merge_f * 4 + G * hd * 2;
}

}  // namespace decode_attn
