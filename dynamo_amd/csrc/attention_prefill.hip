// Paged flash-attention prefill (causal, GQA, chunked-prefill-aware),
// MI355X-native, MFMA mfma_f32_16x16x32_bf16 with LDS-staged KV tiles.
//
// Structure (guide §5/§B): workgroup = 4 waves; each workgroup owns a 64-row
// query tile of one query head; waves own 16 rows each. K/V tiles of 64
// tokens are gathered from the paged cache into LDS (XOR-swizzled layouts,
// guide T2/G4: row-major [*][128] bf16 is otherwise a 16-32-way bank
// conflict on ds_read_b128). Online softmax with per-row running (m, l).
// The P tile round-trips through LDS to re-shape the S-layout (C/D frag)
// into the PV A-operand layout.
//
// MFMA fragment mappings used (verified on HW by tests/test_gpu_mfma.py):
//   A: lane l holds A[row = l%16][k = 8*(l/16) + i]      (i = 0..7)
//   B: lane l holds B[k = 8*(l/16) + i][col = l%16]
//   C/D: lane l, reg r holds D[row = (l/16)*4 + r][col = l%16]
//
// Capability parity: the reference (ai-dynamo/dynamo) delegates prefill
// attention to vLLM/TRT-LLM; this is the native CDNA4 engine kernel.
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;

constexpr int kBlock = 256;  // 4 waves
constexpr int kQTile = 64;   // query rows per workgroup
constexpr int kKvTile = 64;  // kv tokens per LDS tile
constexpr float kNegInf = -1e30f;

DEVINL int swz(int byte_in_row, int row) { return byte_in_row ^ ((row & 7) << 4); }
// V^T variant: fold row>>3 into the key (write instructions hold row&7
// constant across their 16 row-slices — see the XK note in the 32x32
// kernel below)
DEVINL int swzv(int byte_in_row, int row) {
  return byte_in_row ^ (((row ^ (row >> 3)) & 7) << 4);
}

template <int VT = 0>  // VT: vcache is d-major [P, Hkv, hd, ps]
__global__ __launch_bounds__(kBlock) void prefill_kernel(
    short* __restrict__ out,            // [Tq, Hq, 128]
    const short* __restrict__ q,        // [Tq, Hq, 128]
    const short* __restrict__ kcache,   // [P, Hkv, ps, 128]
    const short* __restrict__ vcache,
    const int32_t* __restrict__ page_table,   // [nseq, max_pages]
    const int32_t* __restrict__ tile_seq,     // [ntiles]
    const int32_t* __restrict__ tile_q0,      // [ntiles] local q row of tile
    const int32_t* __restrict__ seq_q_start,  // [nseq] offset into Tq
    const int32_t* __restrict__ seq_q_len,    // [nseq]
    const int32_t* __restrict__ seq_ctx_len,  // [nseq] total kv len
    float scale, int Hq, int Hkv, int max_pages, int log2_ps) {
  constexpr int HD = 128;
  const int tile = blockIdx.x;
  const int qh = blockIdx.y;
  const int kvh = qh / (Hq / Hkv);
  const int seq = tile_seq[tile];
  const int q0 = tile_q0[tile];
  const int qlen = seq_q_len[seq];
  const int ctx = seq_ctx_len[seq];
  const int qstart = seq_q_start[seq];
  const int ps = 1 << log2_ps;
  const int32_t* pt = page_table + (int64_t)seq * max_pages;
  const float scale2 = scale * 1.44269504f;  // scale * log2(e)

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int lr = lane & 15;   // row-or-col lane index
  const int lg = lane >> 4;   // 4 k-groups

  // LDS: K [64][128] (16 KB) + V^T [128][64] (16 KB) + P [4][16][64] (8 KB)
  __shared__ short k_lds[kKvTile * HD];
  __shared__ short vt_lds[HD * kKvTile];
  __shared__ short p_lds[4][16 * kKvTile];

  // ---- load Q fragments (once; reused across all kv tiles) ----
  // wave's rows: q0 + wid*16 + lr ; A-frag kc covers dims [kc*32, kc*32+32)
  bf16x8_t q_frag[4];
  const int my_qrow = q0 + wid * 16 + lr;
  const bool row_valid = my_qrow < qlen;
  {
    const short* qrow_ptr = q + ((int64_t)(qstart + (row_valid ? my_qrow : 0)) * Hq + qh) * HD;
#pragma unroll
    for (int kc = 0; kc < 4; kc++) {
      short8 v = row_valid ? *reinterpret_cast<const short8*>(qrow_ptr + kc * 32 + lg * 8)
                           : short8{};
      q_frag[kc] = *reinterpret_cast<bf16x8_t*>(&v);
    }
  }
  const int my_qpos = ctx - qlen + my_qrow;  // absolute kv position of this row

  float m[4], lsum[4];
  f32x4 acc_o[8];
#pragma unroll
  for (int r = 0; r < 4; r++) { m[r] = kNegInf; lsum[r] = 0.f; }
#pragma unroll
  for (int d = 0; d < 8; d++) acc_o[d] = f32x4{0.f, 0.f, 0.f, 0.f};

  // last kv position any row of this tile may attend to
  const int tile_qpos_max = ctx - qlen + min(q0 + kQTile - 1, qlen - 1);
  const int kv_end = min(ctx, tile_qpos_max + 1);

  for (int t0 = 0; t0 < kv_end; t0 += kKvTile) {
    __syncthreads();
    // ---- stage K tile + V^T tile ----
#pragma unroll
    for (int i = 0; i < 4; i++) {
      const int slot = i * kBlock + threadIdx.x;  // 1024 short8 slots
      const int row = slot >> 4;
      const int col8 = slot & 15;
      const int t = t0 + row;
      short8 kv_k{}, kv_v{};
      if (t < ctx) {
        const int64_t page = pt[t >> log2_ps];
        const int64_t base = ((page * Hkv + kvh) * ps + (t & (ps - 1))) * HD + col8 * 8;
        kv_k = *reinterpret_cast<const short8*>(kcache + base);
        if constexpr (!VT)
          kv_v = *reinterpret_cast<const short8*>(vcache + base);
      }
      *reinterpret_cast<short8*>((char*)k_lds + row * 256 + swz(col8 * 16, row)) = kv_k;
      if constexpr (!VT) {
#pragma unroll
        for (int e = 0; e < 8; e++) {
          const int dim = col8 * 8 + e;
          *(short*)((char*)vt_lds + dim * 128 + swzv(row * 2, dim)) = kv_v[e];
        }
      }
    }
    if constexpr (VT) {
      // d-major pages: each slot stages one 8-token run of one dim row —
      // contiguous global b128, b128 LDS store (swzv key is 16B-granular)
#pragma unroll
      for (int i = 0; i < 4; i++) {
        const int slot = i * kBlock + threadIdx.x;  // 128 dims x 8 chunks
        const int dim = slot >> 3, tc = slot & 7;
        const int tcs = t0 + tc * 8;
        const int tsafe = min(tcs, (ctx - 1) & ~7);
        const int64_t page = pt[tsafe >> log2_ps];
        short8 vv = *reinterpret_cast<const short8*>(
            vcache + ((page * Hkv + kvh) * (int64_t)HD + dim) * ps +
            (tsafe & (ps - 1)));
        if (tsafe + 7 >= ctx) {
#pragma unroll
          for (int e = 0; e < 8; e++)
            if (tsafe + e >= ctx) vv[e] = 0;
        }
        *reinterpret_cast<short8*>(
            (char*)vt_lds + dim * 128 + swzv(tc * 16, dim)) = vv;
      }
    }
    __syncthreads();

    // ---- S = Q K^T for the wave's 16 rows x 64 tokens ----
    f32x4 s[4];
#pragma unroll
    for (int n = 0; n < 4; n++) {
      f32x4 accs{0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kc = 0; kc < 4; kc++) {
        // B-frag: K[tok = n*16 + lr][d = kc*32 + lg*8 + j]
        const int tok = n * 16 + lr;
        short8 kv = *reinterpret_cast<const short8*>(
            (char*)k_lds + tok * 256 + swz(kc * 64 + lg * 16, tok));
        accs = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            q_frag[kc], *reinterpret_cast<bf16x8_t*>(&kv), accs, 0, 0, 0);
      }
      s[n] = accs;
    }

    // ---- online softmax, log2 domain (rows live in (lg, r); cols lr) ----
    // Same two VALU cuts as prefill32 (log2 domain + full-visible-tile
    // mask elision); m/lsum stay kernel-internal so the domain is too.
    const bool fullt =
        (t0 + kKvTile - 1) <= (ctx - qlen + q0 + wid * 16);
    float mt[4];
#pragma unroll
    for (int r = 0; r < 4; r++) {
      float mx = kNegInf;
      if (fullt) {
#pragma unroll
        for (int n = 0; n < 4; n++) mx = fmaxf(mx, s[n][r]);
      } else {
        const int qrow = q0 + wid * 16 + lg * 4 + r;
        const int qpos = ctx - qlen + qrow;
#pragma unroll
        for (int n = 0; n < 4; n++) {
          const int kvpos = t0 + n * 16 + lr;
          float sv = s[n][r];
          sv = (qrow < qlen && kvpos <= qpos && kvpos < ctx) ? sv : kNegInf;
          s[n][r] = sv;
          mx = fmaxf(mx, sv);
        }
      }
      mx = fmaxf(mx, __shfl_xor(mx, 1, WAVE_SIZE));
      mx = fmaxf(mx, __shfl_xor(mx, 2, WAVE_SIZE));
      mx = fmaxf(mx, __shfl_xor(mx, 4, WAVE_SIZE));
      mx = fmaxf(mx, __shfl_xor(mx, 8, WAVE_SIZE));
      mt[r] = (mx <= kNegInf * 0.5f) ? kNegInf : mx * scale2;
    }

    // NOTE: rows of S (and acc_o) map to (lg, r): row = lg*4 + r. The running
    // m/l state for a given physical row is therefore kept by all 16 lanes
    // with that lg, redundantly — shuffles above keep them consistent.
    float p[4][4];  // [n][r]
#pragma unroll
    for (int r = 0; r < 4; r++) {
      float mnew = m[r];
      if (mt[r] > mnew) {
        const float corr = (mnew <= kNegInf * 0.5f)
                               ? 0.f
                               : __builtin_amdgcn_exp2f(mnew - mt[r]);
        lsum[r] *= corr;
#pragma unroll
        for (int d = 0; d < 8; d++) acc_o[d][r] *= corr;
        mnew = mt[r];
        m[r] = mnew;
      }
      float rowsum = 0.f;
      if (fullt) {
#pragma unroll
        for (int n = 0; n < 4; n++) {
          const float pv =
              __builtin_amdgcn_exp2f(fmaf(s[n][r], scale2, -mnew));
          p[n][r] = pv;
          rowsum += pv;
        }
      } else {
#pragma unroll
        for (int n = 0; n < 4; n++) {
          const float pv =
              (s[n][r] <= kNegInf * 0.5f || mnew <= kNegInf * 0.5f)
                  ? 0.f
                  : __builtin_amdgcn_exp2f(fmaf(s[n][r], scale2, -mnew));
          p[n][r] = pv;
          rowsum += pv;
        }
      }
      rowsum += __shfl_xor(rowsum, 1, WAVE_SIZE);
      rowsum += __shfl_xor(rowsum, 2, WAVE_SIZE);
      rowsum += __shfl_xor(rowsum, 4, WAVE_SIZE);
      rowsum += __shfl_xor(rowsum, 8, WAVE_SIZE);
      lsum[r] += rowsum;
    }

    // ---- write P tile to LDS (re-layout for the PV A-operand) ----
    __syncthreads();  // all waves done reading k_lds-dependent S
#pragma unroll
    for (int r = 0; r < 4; r++) {
#pragma unroll
      for (int n = 0; n < 4; n++) {
        const int row = lg * 4 + r;       // q row within wave tile
        const int tok = n * 16 + lr;
        *(short*)((char*)p_lds[wid] + row * 128 + swz(tok * 2, row)) =
            f32_to_bf16(p[n][r]);
      }
    }
    __syncthreads();

    // ---- O += P V ----
#pragma unroll
    for (int kt = 0; kt < 2; kt++) {
      // A-frag: P[row = lr][tok = kt*32 + lg*8 + j]
      short8 pa = *reinterpret_cast<const short8*>(
          (char*)p_lds[wid] + lr * 128 + swz(kt * 64 + lg * 16, lr));
#pragma unroll
      for (int d = 0; d < 8; d++) {
        // B-frag: V[tok = kt*32 + lg*8 + j][dim = d*16 + lr] = V^T[dim][tok]
        const int dim = d * 16 + lr;
        short8 vb = *reinterpret_cast<const short8*>(
            (char*)vt_lds + dim * 128 + swzv(kt * 64 + lg * 16, dim));
        acc_o[d] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            *reinterpret_cast<bf16x8_t*>(&pa), *reinterpret_cast<bf16x8_t*>(&vb),
            acc_o[d], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: normalize and store ----
#pragma unroll
  for (int r = 0; r < 4; r++) {
    const int qrow = q0 + wid * 16 + lg * 4 + r;
    if (qrow >= qlen) continue;
    const float inv = lsum[r] > 0.f ? 1.f / lsum[r] : 0.f;
    short* orow = out + ((int64_t)(qstart + qrow) * Hq + qh) * HD;
#pragma unroll
    for (int d = 0; d < 8; d++) orow[d * 16 + lr] = f32_to_bf16(acc_o[d][r] * inv);
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// 8-wave 32x32 variant (GQA group == 8, e.g. Llama-70B 64q/8kv): one block
// covers ALL 8 q-heads of one kv head on a 32-row q tile, so each K/V tile
// is staged once and read by 8 warps. Swapped QK^T (mfma(K,Q), 32x32x16)
// keeps every lane's scores on ONE q-row -> softmax fully in-register (fmax
// chain + one cross-half shfl); P converts to PV A-fragments by a symmetric
// half-exchange, no LDS round-trip. K row-major + V TRANSPOSED in LDS, both
// XOR-swizzled; double-buffered with register prefetch (one barrier/tile);
// role-split staging (256 threads K b128 / 256 threads V^T b64); defer-max
// rescale skip. Measured 260 TF vs 128 TF for the 4-wave 16x16 kernel above
// (benchmarks/prefill32_sweep.hip ladder; guide "8-warp 32x32 ladder").
//
// 32x32x16 fragment mappings (verified by the sweep's probe32 on HW):
//   A: lane l holds A[row=l%32][k=8*(l/32)+i]
//   B: lane l holds B[k=8*(l/32)+i][col=l%32]
//   D: lane l, reg r holds D[row=(r&3)+8*(r>>2)+4*(l/32)][col=l%32]
namespace {

typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) short short4_t;


// cross-half (lane^32) exchange via ONE v_permlane32_swap instead of a
// ds_bpermute shfl_xor: swap(x,x) gives the partner's x in r.y (lo half)
// / r.x (hi half)
__device__ __forceinline__ float xor32_swap(float x, int hi) {
  typedef __attribute__((ext_vector_type(2))) unsigned int uint2_sw;
  unsigned int u = __builtin_bit_cast(unsigned int, x);
  uint2_sw r = __builtin_amdgcn_permlane32_swap(u, u, false, false);
  return __builtin_bit_cast(float, hi ? r.x : r.y);
}

constexpr int kQB32 = 32;   // q rows per block
constexpr int kKB32 = 64;   // kv tokens per tile
constexpr int kLdsHalf32 = kKB32 * 256 + 128 * 128;  // K + V^T = 32 KB

// GSPLIT generalizes to GQA groups G = 8/GSPLIT: the 8 warps are G heads
// x GSPLIT 32-row q-subtiles, all sharing one staged K/V tile (block tile
// is GSPLIT*32 q rows). G=8 -> 1 subtile; G=4 -> 2; G=2 -> 4.
// FP8: paged cache stores OCP e4m3 (converted to bf16 while staging to
// LDS; compute unchanged)
template <int GSPLIT, int FP8 = 0, int VT = 0>
__global__ __launch_bounds__(512) void prefill32_kernel(
    short* __restrict__ out,            // [Tq, Hq, 128]
    const short* __restrict__ q,        // [Tq, Hq, 128]
    const short* __restrict__ kcache,   // [P, Hkv, ps, 128]
    const short* __restrict__ vcache,
    const int32_t* __restrict__ page_table,   // [nseq, max_pages]
    const int32_t* __restrict__ tile_seq,     // [ntiles] (GSPLIT*32-row tiles)
    const int32_t* __restrict__ tile_q0,
    const int32_t* __restrict__ seq_q_start,
    const int32_t* __restrict__ seq_q_len,
    const int32_t* __restrict__ seq_ctx_len,
    float scale, int Hq, int Hkv, int max_pages, int log2_ps) {
  constexpr int HD = 128;
  constexpr int G = 8 / GSPLIT;           // q heads per kv head
  const int tile = blockIdx.x;
  const int kvh = blockIdx.y;
  const int w = threadIdx.x >> 6;
  const int l = threadIdx.x & 63;
  const int h = kvh * G + (w % G);
  const int seq = tile_seq[tile];
  const int q0blk = tile_q0[tile];
  const int q0 = q0blk + (w / G) * kQB32; // this warp's q-subtile
  const int qlen = seq_q_len[seq];
  const int ctx = seq_ctx_len[seq];
  const int qstart = seq_q_start[seq];
  const int ps = 1 << log2_ps;
  const int32_t* pt = page_table + (int64_t)seq * max_pages;
  const int lo = l & 31, hi = l >> 5;
  const float scale2 = scale * 1.44269504f;  // scale * log2(e)

  extern __shared__ char lds32[];

  // Q B-fragments for this warp's head, q row q0+lo (clamped if past qlen)
  const int my_qrow = q0 + lo;
  const bool row_valid = my_qrow < qlen;
  const short* qrow_p =
      q + ((int64_t)(qstart + (row_valid ? my_qrow : qlen - 1)) * Hq + h) * HD;
  bf16x8_t qreg[8];
#pragma unroll
  for (int ds = 0; ds < 8; ds++) {
    short8 v = *reinterpret_cast<const short8*>(qrow_p + ds * 16 + 8 * hi);
    qreg[ds] = *reinterpret_cast<bf16x8_t*>(&v);
  }

  f32x16 o[4] = {};
  float m_run = kNegInf, l_run = 0.f;
  const int my_qpos = ctx - qlen + my_qrow;
  // block-uniform kv range: last q row of the whole block tile
  const int tile_qpos_max =
      ctx - qlen + min(q0blk + GSPLIT * kQB32 - 1, qlen - 1);
  const int kv_end = min(ctx, tile_qpos_max + 1);

  // role-split staging: threads 0-255 stage V^T (b64 writes, 4 toks/write),
  // threads 256-511 stage K (b128 writes); register prefetch double-buffer
  const int vrole = threadIdx.x < 256;
  short8 sreg[4];
  auto cache8 = [&](const short* base, int64_t elem_off) -> short8 {
    if constexpr (FP8) {
      uchar8 raw = *reinterpret_cast<const uchar8*>(
          reinterpret_cast<const unsigned char*>(base) + elem_off);
      bf16x8 cv = fp8x8_to_bf16x8(raw);
      return *reinterpret_cast<short8*>(&cv);
    } else {
      return *reinterpret_cast<const short8*>(base + elem_off);
    }
  };
  auto load_tile = [&](int t0) {
    if (vrole) {
      if constexpr (VT) {
        // d-major pages: 256 v-threads x 4 slots = 128 dims x 8 chunks;
        // each slot is one contiguous 8-token b128 run of one dim row
#pragma unroll
        for (int j = 0; j < 4; j++) {
          const int slot = threadIdx.x + j * 256;
          const int d = slot >> 3, tc = slot & 7;
          const int tcs = t0 + tc * 8;
          const int tsafe = min(tcs, (ctx - 1) & ~7);
          const int64_t page = pt[tsafe >> log2_ps];
          sreg[j] = cache8(vcache,
              ((page * Hkv + kvh) * (int64_t)HD + d) * ps + (tsafe & (ps - 1)));
          if (tsafe + 7 >= ctx) {
#pragma unroll
            for (int e = 0; e < 8; e++)
              if (tsafe + e >= ctx) sreg[j][e] = 0;
          }
        }
        return;
      }
      const int unit = threadIdx.x;
      const int row0 = (unit >> 4) * 4, d0 = (unit & 15) * 8;
#pragma unroll
      for (int j = 0; j < 4; j++) {
        const int t = min(t0 + row0 + j, ctx - 1);
        const int64_t page = pt[t >> log2_ps];
        sreg[j] = cache8(vcache,
            ((page * Hkv + kvh) * ps + (t & (ps - 1))) * HD + d0);
      }
    } else {
      const int idx = threadIdx.x - 256;
#pragma unroll
      for (int u = 0; u < 4; u++) {
        const int c = idx + u * 256;
        const int row = c >> 4, col8 = (c & 15) * 8;
        const int t = min(t0 + row, ctx - 1);
        const int64_t page = pt[t >> log2_ps];
        sreg[u] = cache8(kcache,
            ((page * Hkv + kvh) * ps + (t & (ps - 1))) * HD + col8);
      }
    }
  };
  auto store_tile = [&](int buf) {
    char* kl = lds32 + buf * kLdsHalf32;
    char* vl = kl + kKB32 * 256;
    if (vrole) {
      if constexpr (VT) {
#pragma unroll
        for (int j = 0; j < 4; j++) {
          const int slot = threadIdx.x + j * 256;
          const int d = slot >> 3, tc = slot & 7;
          const int key = (d ^ (d >> 3)) & 7;
          *reinterpret_cast<short8*>(
              vl + d * 128 + ((tc * 16) ^ (key << 4))) = sreg[j];
        }
        return;
      }
      const int unit = threadIdx.x;
      const int row0 = (unit >> 4) * 4, d0 = (unit & 15) * 8;
#pragma unroll
      for (int i = 0; i < 8; i++) {
        const int d = d0 + i;
        // swizzle key folds d>>3: per-write (d & 7) is constant across
        // the wave's 16 d-slices (16-way bank collapse; PMC 4.1e9
        // conflicts). (d ^ d>>3) & 7 spreads writes 8-wide at zero LDS
        // cost; reads keep 8-distinct keys per 8 consecutive rows.
        // Measured 570 -> 686 TF (prefill32_sweep XK ladder).
        const int key = (d ^ (d >> 3)) & 7;
        short4_t pk = {sreg[0][i], sreg[1][i], sreg[2][i], sreg[3][i]};
        *(short4_t*)(vl + d * 128 + ((row0 * 2) ^ (key << 4))) = pk;
      }
    } else {
      const int idx = threadIdx.x - 256;
#pragma unroll
      for (int u = 0; u < 4; u++) {
        const int c = idx + u * 256;
        const int row = c >> 4, col8 = (c & 15) * 8;
        *reinterpret_cast<short8*>(
            kl + row * 256 + ((col8 * 2) ^ ((row & 7) << 4))) = sreg[u];
      }
    }
  };

  load_tile(0);
  store_tile(0);
  __syncthreads();
  if (kKB32 < kv_end) load_tile(kKB32);

  for (int t0 = 0; t0 < kv_end; t0 += kKB32) {
    const int cur = (t0 / kKB32) & 1;
    const char* kl = lds32 + cur * kLdsHalf32;
    const char* vl = kl + kKB32 * 256;

    // ---- QK^T: S^T tiles [32tok x 32q] via mfma(K, Q) ----
    f32x16 s0 = {}, s1 = {};
#pragma unroll
    for (int ds = 0; ds < 8; ds++) {
      const int koff = ds * 32 + hi * 16;
      short8 a0 = *reinterpret_cast<const short8*>(
          kl + lo * 256 + (koff ^ ((lo & 7) << 4)));
      short8 a1 = *reinterpret_cast<const short8*>(
          kl + (lo + 32) * 256 + (koff ^ ((lo & 7) << 4)));
      s0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          *reinterpret_cast<bf16x8_t*>(&a0), qreg[ds], s0, 0, 0, 0);
      s1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          *reinterpret_cast<bf16x8_t*>(&a1), qreg[ds], s1, 0, 0, 0);
    }

    // ---- mask + softmax, log2 domain (lane owns q-row lo) ----
    // PMC post-XK: VALU:MFMA 16.6:1 issue-bound, conflicts gone. Two
    // VALU cuts (null on the wait-bound decode, real here):
    //  - log2 domain: raw S max (scale folds into one mul of the max),
    //    exp chain mul+sub+mul+exp -> fma+exp2. m/l never leave the
    //    kernel, so the domain is fully internal.
    //  - full-tile fast path: a kv tile entirely at/below every q row of
    //    this wave (t0+63 <= qpos of row q0) skips all 64 causal
    //    cmp+sels; ~half of all tiles in a long causal prefill qualify.
    float p[32];
    float mt = kNegInf;
    const bool full = (t0 + kKB32 - 1) <= (ctx - qlen + q0);
    if (full) {
#pragma unroll
      for (int r = 0; r < 16; r++) {
        p[r] = s0[r];
        p[16 + r] = s1[r];
        mt = fmaxf(mt, fmaxf(p[r], p[16 + r]));
      }
    } else {
#pragma unroll
      for (int r = 0; r < 16; r++) {
        const int trow = (r & 3) + 8 * (r >> 2) + 4 * hi;
        const bool ok0 = row_valid && t0 + trow <= my_qpos;
        const bool ok1 = row_valid && t0 + 32 + trow <= my_qpos;
        p[r] = ok0 ? s0[r] : kNegInf;
        p[16 + r] = ok1 ? s1[r] : kNegInf;
        mt = fmaxf(mt, fmaxf(p[r], p[16 + r]));
      }
    }
    mt = fmaxf(mt, xor32_swap(mt, hi));
    const float mts = (mt <= kNegInf * 0.5f) ? kNegInf : mt * scale2;
    float m_new = fmaxf(m_run, mts);
    bool skip_rescale = false;
    // defer-max: skip the O-rescale while the tile max stays within 8
    // (8 nats = 11.54 in the log2 domain)
    if (__all(mts - m_run <= 11.5417f)) { m_new = m_run; skip_rescale = true; }
    const float alpha =
        (skip_rescale || m_run <= kNegInf * 0.5f)
            ? 1.f : __builtin_amdgcn_exp2f(m_run - m_new);
    float ls = 0.f;
    if (full) {
#pragma unroll
      for (int r = 0; r < 32; r++) {
        p[r] = __builtin_amdgcn_exp2f(fmaf(p[r], scale2, -m_new));
        ls += p[r];
      }
    } else {
#pragma unroll
      for (int r = 0; r < 32; r++) {
        p[r] = (p[r] <= kNegInf * 0.5f || m_new <= kNegInf * 0.5f)
                   ? 0.f
                   : __builtin_amdgcn_exp2f(fmaf(p[r], scale2, -m_new));
        ls += p[r];
      }
    }
    ls += xor32_swap(ls, hi);
    l_run = l_run * alpha + ls;
    m_run = m_new;
    if (!skip_rescale) {
      // O is in D-layout: reg r belongs to q-row (r&3)+8*(r>>2)+4*hi, so
      // rescale with THAT row's alpha (alpha is half-replicated per q-row)
      float arow[16];
#pragma unroll
      for (int r = 0; r < 16; r++)
        arow[r] = __shfl(alpha, (r & 3) + 8 * (r >> 2) + 4 * hi, WAVE_SIZE);
#pragma unroll
      for (int dt = 0; dt < 4; dt++)
#pragma unroll
        for (int r = 0; r < 16; r++) o[dt][r] *= arow[r];
    }

    // ---- P -> PV A-fragments via cvt_pk + permlane32_swap ----
    // Pack P pairs to bf16x2 with v_cvt_pk_bf16_f32, then one
    // permlane32_swap yields (word0, word2) on both lane halves and a
    // second (word1, word3): 16 cvt_pk + 8 permlane replaces 16 shfl +
    // 32 scalar converts. Slot ks's own values sit at regs b..b+7,
    // b = 8*(ks&1) + 16*(ks>>1). Measured 260 -> 569 TF standalone (the
    // shfl/ds_bpermute chain was the VALU critical path;
    // benchmarks/prefill32_sweep.hip).
    typedef __attribute__((ext_vector_type(2))) unsigned int uint2_t;
    auto cvtpk = [](float a, float b) {
      unsigned int r;
      asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
      return r;
    };
    bf16x8_t pa[4];
#pragma unroll
    for (int ks = 0; ks < 4; ks++) {
      const int b0 = 8 * (ks & 1) + 16 * (ks >> 1);
      uint2_t rA = __builtin_amdgcn_permlane32_swap(
          cvtpk(p[b0], p[b0 + 1]), cvtpk(p[b0 + 4], p[b0 + 5]), false, false);
      uint2_t rB = __builtin_amdgcn_permlane32_swap(
          cvtpk(p[b0 + 2], p[b0 + 3]), cvtpk(p[b0 + 6], p[b0 + 7]),
          false, false);
      unsigned int w[4] = {rA.x, rB.x, rA.y, rB.y};
      pa[ks] = *reinterpret_cast<bf16x8_t*>(w);
    }

    // ---- O += P V ----
#pragma unroll
    for (int dt = 0; dt < 4; dt++) {
      const int drow = dt * 32 + lo;
#pragma unroll
      for (int ks = 0; ks < 4; ks++) {
        const int toff = (ks * 16 + 8 * hi) * 2;
        const int rkey = (drow ^ (drow >> 3)) & 7;
        short8 vb = *reinterpret_cast<const short8*>(
            vl + drow * 128 + (toff ^ (rkey << 4)));
        o[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            pa[ks], *reinterpret_cast<bf16x8_t*>(&vb), o[dt], 0, 0, 0);
      }
    }

    // stage next tile into the other buffer; prefetch tile t+2
    if (t0 + kKB32 < kv_end) {
      store_tile(cur ^ 1);
      if (t0 + 2 * kKB32 < kv_end) load_tile(t0 + 2 * kKB32);
    }
    __syncthreads();
  }

  // ---- epilogue: divide by the q-ROW's l (D-layout), store bf16 ----
  float lrow[16];
#pragma unroll
  for (int r = 0; r < 16; r++)
    lrow[r] = __shfl(l_run, (r & 3) + 8 * (r >> 2) + 4 * hi, WAVE_SIZE);
#pragma unroll
  for (int dt = 0; dt < 4; dt++)
#pragma unroll
    for (int r = 0; r < 16; r++) {
      const int qrow_i = q0 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      if (qrow_i >= qlen) continue;
      const float inv = lrow[r] > 0.f ? 1.f / lrow[r] : 0.f;
      out[((int64_t)(qstart + qrow_i) * Hq + h) * HD + dt * 32 + lo] =
          f32_to_bf16(o[dt][r] * inv);
    }
}

}  // namespace

void attention_prefill_paged(torch::Tensor out, torch::Tensor q,
                             torch::Tensor kcache, torch::Tensor vcache,
                             torch::Tensor page_table, torch::Tensor tile_seq,
                             torch::Tensor tile_q0, torch::Tensor seq_q_start,
                             torch::Tensor seq_q_len, torch::Tensor seq_ctx_len,
                             double scale, bool v_transposed) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  TORCH_CHECK(q.size(-1) == 128, "only head_dim=128 supported natively");
  TORCH_CHECK(page_table.dtype() == torch::kInt32);
  const bool fp8 = kcache.dtype() == torch::kFloat8_e4m3fn;
  TORCH_CHECK(fp8 || kcache.dtype() == torch::kBFloat16,
              "kv cache must be bf16 or float8_e4m3fn");
  const int Hq = q.size(1);
  const int Hkv = kcache.size(1);
  const int ps = kcache.size(2);
  TORCH_CHECK((ps & (ps - 1)) == 0);
  int log2_ps = 0; while ((1 << log2_ps) < ps) log2_ps++;
  const int ntiles = tile_seq.size(0);
  const int max_pages = page_table.size(1);
  if (ntiles == 0) return;
  auto stream = at::cuda::getCurrentHIPStream();
  // GQA group G in {2,4,8} -> 8-wave 32x32 kernel with GSPLIT=8/G q-subtiles
  // (tile rows = (8/G)*32, built by the Python layer via prefill_tile_rows —
  // keep the predicates in sync)
  const int G = (Hq % Hkv == 0) ? Hq / Hkv : 0;
  if (v_transposed) {
    TORCH_CHECK(vcache.dim() == 4 && vcache.size(2) == q.size(-1) &&
                    vcache.size(3) == ps,
                "v_transposed expects vcache [P, Hkv, hd, ps]");
    TORCH_CHECK(ps % 8 == 0, "v_transposed prefill needs page_size%8==0");
  }
  if (G == 8 || G == 4 || G == 2) {
    dim3 grid(ntiles, Hkv);
    auto launch = [&](auto kern) {
      kern<<<grid, 512, 2 * kLdsHalf32, stream>>>(
          (short*)out.data_ptr(), (const short*)q.data_ptr(),
          (const short*)kcache.data_ptr(), (const short*)vcache.data_ptr(),
          page_table.data_ptr<int32_t>(), tile_seq.data_ptr<int32_t>(),
          tile_q0.data_ptr<int32_t>(), seq_q_start.data_ptr<int32_t>(),
          seq_q_len.data_ptr<int32_t>(), seq_ctx_len.data_ptr<int32_t>(),
          (float)scale, Hq, Hkv, max_pages, log2_ps);
    };
    if (v_transposed) {
      if (fp8) {
        if (G == 8) launch(prefill32_kernel<1, 1, 1>);
        else if (G == 4) launch(prefill32_kernel<2, 1, 1>);
        else launch(prefill32_kernel<4, 1, 1>);
      } else if (G == 8) launch(prefill32_kernel<1, 0, 1>);
      else if (G == 4) launch(prefill32_kernel<2, 0, 1>);
      else launch(prefill32_kernel<4, 0, 1>);
    } else if (fp8) {
      if (G == 8) launch(prefill32_kernel<1, 1>);
      else if (G == 4) launch(prefill32_kernel<2, 1>);
      else launch(prefill32_kernel<4, 1>);
    } else if (G == 8) launch(prefill32_kernel<1>);
    else if (G == 4) launch(prefill32_kernel<2>);
    else launch(prefill32_kernel<4>);
    HIP_CHECK_KERNEL();
    return;
  }
  TORCH_CHECK(!fp8, "fp8 KV prefill needs GQA group in {2,4,8}");
  dim3 grid(ntiles, Hq);
  auto launch16 = [&](auto kern) {
    kern<<<grid, kBlock, 0, stream>>>(
        (short*)out.data_ptr(), (const short*)q.data_ptr(),
        (const short*)kcache.data_ptr(), (const short*)vcache.data_ptr(),
        page_table.data_ptr<int32_t>(), tile_seq.data_ptr<int32_t>(),
        tile_q0.data_ptr<int32_t>(), seq_q_start.data_ptr<int32_t>(),
        seq_q_len.data_ptr<int32_t>(), seq_ctx_len.data_ptr<int32_t>(),
        (float)scale, Hq, Hkv, max_pages, log2_ps);
  };
  if (v_transposed) launch16(prefill_kernel<1>);
  else launch16(prefill_kernel<0>);
  HIP_CHECK_KERNEL();
}
