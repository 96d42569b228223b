// Standalone development harness for the 8-wave 32x32 prefill attention
// kernel (swapped-QK in-register softmax structure). Build:
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 benchmarks/prefill32_sweep.hip \
//     -o benchmarks/prefill32_sweep
// Run on the GPU box: ./benchmarks/prefill32_sweep
//
// Stages:
//   1. probe: verify the assumed A/B/D fragment layouts of
//      v_mfma_f32_32x32x16_bf16 against a CPU reference product.
//   2. check: flash-attention kernel vs CPU O(S^2 D) reference (S=256).
//   3. bench: causal prefill at (B=2, Hq=64, Hkv=8, S=4096, D=128) and
//      (B=1, ..., S=8192), report TF (2*B*Hq*S^2*D causal flops).
#include <hip/hip_runtime.h>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define CK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  printf("HIP error %s @%d\n", hipGetErrorString(e), __LINE__); exit(1); } } while (0)

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(2))) unsigned int uint2_t;
typedef __attribute__((ext_vector_type(8))) short short8;

__host__ __device__ static inline float bf2f(short u) {
  union { float f; unsigned int i; } c; c.i = ((unsigned int)(unsigned short)u) << 16; return c.f;
}
__host__ __device__ static inline short f2bf(float f) {
  union { float f; unsigned int i; } c; c.f = f;
  unsigned int r = c.i + 0x7fff + ((c.i >> 16) & 1);
  return (short)(r >> 16);
}

// ---------------------------------------------------------------------------
// 1. fragment layout probe
// Assumed layouts (generalizing the HW-verified 16x16x32 mappings):
//   A [32r x 16k]: lane l holds A[row=l%32][k=8*(l/32)+i], i=0..7
//   B [16k x 32c]: lane l holds B[k=8*(l/32)+i][col=l%32]
//   D [32r x 32c]: lane l, reg r: D[row=(r&3)+8*(r>>2)+4*(l>>5)][col=l%32]
__global__ void probe32_kernel(float* d_out, const short* a, const short* b) {
  const int l = threadIdx.x;
  bf16x8 av, bv;
  for (int i = 0; i < 8; i++) {
    av[i] = (__bf16)bf2f(a[(l % 32) * 16 + 8 * (l / 32) + i]);
    bv[i] = (__bf16)bf2f(b[(8 * (l / 32) + i) * 32 + (l % 32)]);
  }
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(av, bv, acc, 0, 0, 0);
  for (int r = 0; r < 16; r++) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * (l >> 5);
    const int col = l % 32;
    d_out[row * 32 + col] = acc[r];
  }
}

static bool probe32() {
  std::vector<short> a(32 * 16), b(16 * 32);
  for (auto& x : a) x = f2bf((float)(rand() % 17 - 8));
  for (auto& x : b) x = f2bf((float)(rand() % 17 - 8));
  std::vector<float> ref(32 * 32, 0.f);
  for (int r = 0; r < 32; r++)
    for (int c = 0; c < 32; c++) {
      float s = 0;
      for (int k = 0; k < 16; k++) s += bf2f(a[r * 16 + k]) * bf2f(b[k * 32 + c]);
      ref[r * 32 + c] = s;
    }
  short *da, *db; float* dd;
  CK(hipMalloc(&da, a.size() * 2)); CK(hipMalloc(&db, b.size() * 2));
  CK(hipMalloc(&dd, ref.size() * 4));
  CK(hipMemcpy(da, a.data(), a.size() * 2, hipMemcpyHostToDevice));
  CK(hipMemcpy(db, b.data(), b.size() * 2, hipMemcpyHostToDevice));
  probe32_kernel<<<1, 64>>>(dd, da, db);
  CK(hipDeviceSynchronize());
  std::vector<float> got(ref.size());
  CK(hipMemcpy(got.data(), dd, ref.size() * 4, hipMemcpyDeviceToHost));
  int bad = 0;
  for (size_t i = 0; i < ref.size(); i++)
    if (fabsf(got[i] - ref[i]) > 1e-3f && bad++ < 5)
      printf("  probe mismatch [%zu] got %f want %f\n", i, got[i], ref[i]);
  printf("probe32 %s\n", bad ? "FAIL" : "PASS");
  hipFree(da); hipFree(db); hipFree(dd);
  return bad == 0;
}

// permlane32_swap semantics probe: establish which half gets what.
__global__ void permprobe_kernel(unsigned int* out) {
  unsigned int v0 = threadIdx.x;            // "old" operand
  unsigned int v1 = 1000 + threadIdx.x;     // "src" operand
  uint2_t r = __builtin_amdgcn_permlane32_swap(v0, v1, false, false);
  out[threadIdx.x * 2] = r.x;
  out[threadIdx.x * 2 + 1] = r.y;
}

static void permprobe() {
  unsigned int* d; CK(hipMalloc(&d, 128 * 4));
  permprobe_kernel<<<1, 64>>>(d);
  CK(hipDeviceSynchronize());
  std::vector<unsigned int> h(128);
  CK(hipMemcpy(h.data(), d, 128 * 4, hipMemcpyDeviceToHost));
  printf("permlane32_swap lane0:(%u,%u) lane32:(%u,%u) lane5:(%u,%u) lane37:(%u,%u)\n",
         h[0], h[1], h[64], h[65], h[10], h[11], h[74], h[75]);
  hipFree(d);
}

// ---------------------------------------------------------------------------
// 2+3. the flash prefill kernel
//
// Structure (guide "8-warp 32x32 ladder"): GW warps per block, one warp per
// q-head of a GQA group; all warps share one kv-head's K/V tiles. Per warp:
// 32 q-rows in registers (8x bf16x8 B-fragments), KVBLK=64 token tiles.
// K staged row-major [64][128] XOR-swizzled; V staged TRANSPOSED [128][64]
// XOR-swizzled so PV B-fragments are conflict-free b128 row reads.
// Swapped QK^T (mfma(K,Q)) keeps each lane's 32 scores on ONE q-row:
// softmax is in-register (fmax chain + one cross-half exchange).
#define KVBLK 64
#define QBLK 32

// cross-half (lane^32) exchange via ONE v_permlane32_swap instead of a
// ds_bpermute shfl_xor: swap(x,x) gives the partner's x in r.y (lo half)
// / r.x (hi half)
__device__ __forceinline__ float xor32_swap(float x, int hi) {
  typedef __attribute__((ext_vector_type(2))) unsigned int uint2_sw;
  unsigned int u = __builtin_bit_cast(unsigned int, x);
  uint2_sw r = __builtin_amdgcn_permlane32_swap(u, u, false, false);
  return __builtin_bit_cast(float, hi ? r.x : r.y);
}


// VSTAGE: role-split staging (requires GW=8): threads 256-511 stage K with
//   b128 writes; threads 0-255 stage V^T with b64 writes (4 tokens packed
//   per write) — 4x fewer LDS write ops than scalar V^T stores.
// DEFER: skip the O-rescale when the tile max is within 8 of the running
//   max (P bounded by e^8; guide "defer-max RESCALE_THRESHOLD").
// XK=1: V^T swizzle key ((d ^ (d>>3)) & 7) instead of (d & 7) — per
// staging-write instruction (d & 7) is CONSTANT across the wave's 16
// d-slices (16-way bank collapse; PMC 4.1e9 conflicts); folding d>>3
// into the key spreads writes 8-wide at ZERO LDS cost while reads keep
// their 8-distinct spread per 8 consecutive rows (16B-aligned b128).
template <int GW, int ASTAGE, int VSTAGE = 0, int DEFER = 0, int PRIO = 0, int VST = 128, int XK = 0>
__global__ __launch_bounds__(GW * 64) void prefill32_kernel(
    short* __restrict__ out,        // [B, S, Hq, 128]
    const short* __restrict__ q,    // [B, S, Hq, 128]
    const short* __restrict__ kc,   // [B, Hkv, S, 128]
    const short* __restrict__ vc,   // [B, Hkv, S, 128]
    int S, int Hq, int Hkv, float scale) {
  const int qt = blockIdx.x;              // q tile (32 rows)
  const int kvh = blockIdx.y;
  const int bb = blockIdx.z;
  const int w = threadIdx.x / 64;         // warp = q-head within group
  const int l = threadIdx.x % 64;
  const int h = kvh * GW + w;             // this warp's q head
  const int q0 = qt * QBLK;
  const int lo = l & 31, hi = l >> 5;

  extern __shared__ char lds[];
  constexpr int kHalfB = KVBLK * 256 + 128 * VST;
  short* k_lds = (short*)lds;                    // [64][128] swizzled
  short* v_lds = (short*)(lds + KVBLK * 256);    // [128][VST/2] (V^T)

  // Q B-fragments: qreg[ds] holds Q[q0+lo][ds*16 + 8*hi + i]
  const short* qrow = q + (((int64_t)bb * S + q0 + lo) * Hq + h) * 128;
  bf16x8 qreg[8];
#pragma unroll
  for (int ds = 0; ds < 8; ds++)
    qreg[ds] = *reinterpret_cast<const bf16x8*>(qrow + ds * 16 + 8 * hi);

  f32x16 o[4] = {};
  float m_run = -1e30f, l_run = 0.f;

  const int kv_end = q0 + QBLK;           // causal: tiles with t0 < kv_end
  const short* kbase = kc + (((int64_t)bb * Hkv + kvh) * S) * 128;
  const short* vbase = vc + (((int64_t)bb * Hkv + kvh) * S) * 128;

  // per-thread staging chunks (each chunk = one bf16x8 of K and of V)
  constexpr int NCH = VSTAGE ? 4 : (KVBLK * 16) / (GW * 64);
  short8 kreg[NCH], vreg[VSTAGE ? 1 : NCH];
  const int vrole = VSTAGE && (threadIdx.x < 256);
  auto load_tile = [&](int t0) {
    if constexpr (VSTAGE) {
      if (vrole) {
        // V unit: 4 consecutive token rows, one 8-wide d chunk
        const int unit = threadIdx.x;
        const int row0 = (unit >> 4) * 4, d0 = (unit & 15) * 8;
#pragma unroll
        for (int j = 0; j < 4; j++)
          kreg[j] = *reinterpret_cast<const short8*>(
              vbase + (int64_t)(t0 + row0 + j) * 128 + d0);
      } else {
        const int idx = threadIdx.x - 256;
#pragma unroll
        for (int u = 0; u < 4; u++) {
          const int c = idx + u * 256;
          const int row = c >> 4, col8 = (c & 15) * 8;
          kreg[u] = *reinterpret_cast<const short8*>(
              kbase + (int64_t)(t0 + row) * 128 + col8);
        }
      }
      return;
    }
#pragma unroll
    for (int u = 0; u < NCH; u++) {
      const int c = threadIdx.x + u * GW * 64;
      const int row = c >> 4, col8 = (c & 15) * 8;
      kreg[u] = *reinterpret_cast<const short8*>(
          kbase + (int64_t)(t0 + row) * 128 + col8);
      vreg[u] = *reinterpret_cast<const short8*>(
          vbase + (int64_t)(t0 + row) * 128 + col8);
    }
  };
  auto store_tile = [&](int buf) {
    char* kl = (char*)lds + buf * kHalfB;
    char* vl = kl + KVBLK * 256;
    if constexpr (VSTAGE) {
      if (vrole) {
        typedef __attribute__((ext_vector_type(4))) short short4_t;
        const int unit = threadIdx.x;
        const int row0 = (unit >> 4) * 4, d0 = (unit & 15) * 8;
#pragma unroll
        for (int i = 0; i < 8; i++) {
          const int d = d0 + i;
          const int key = XK ? ((d ^ (d >> 3)) & 7) : (d & 7);
          short4_t pk = {kreg[0][i], kreg[1][i], kreg[2][i], kreg[3][i]};
          *(short4_t*)(vl + d * VST + ((row0 * 2) ^ (key << 4))) = pk;
        }
      } else {
        const int idx = threadIdx.x - 256;
#pragma unroll
        for (int u = 0; u < 4; u++) {
          const int c = idx + u * 256;
          const int row = c >> 4, col8 = (c & 15) * 8;
          *reinterpret_cast<short8*>(
              kl + row * 256 + ((col8 * 2) ^ ((row & 7) << 4))) = kreg[u];
        }
      }
      return;
    }
#pragma unroll
    for (int u = 0; u < NCH; u++) {
      const int c = threadIdx.x + u * GW * 64;
      const int row = c >> 4, col8 = (c & 15) * 8;
      *reinterpret_cast<short8*>(
          kl + row * 256 + ((col8 * 2) ^ ((row & 7) << 4))) = kreg[u];
#pragma unroll
      for (int i = 0; i < 8; i++) {
        const int d = col8 + i;
        const int key = XK ? ((d ^ (d >> 3)) & 7) : (d & 7);
        *(short*)(vl + d * VST + ((row * 2) ^ (key << 4))) = vreg[u][i];
      }
    }
  };

  if constexpr (ASTAGE) load_tile(0);
  if constexpr (ASTAGE == 2) {
    // double-buffer prologue: buf0 holds tile 0; next tile's loads in regs
    store_tile(0);
    __syncthreads();
    if (KVBLK < kv_end) load_tile(KVBLK);
  }

  for (int t0 = 0; t0 < kv_end; t0 += KVBLK) {
    if constexpr (ASTAGE == 2) {
      const int cur = (t0 / KVBLK) & 1;
      k_lds = (short*)((char*)lds + cur * (kHalfB));
      v_lds = (short*)((char*)k_lds + KVBLK * 256);
    } else {
      __syncthreads();
      if constexpr (!ASTAGE) load_tile(t0);
      store_tile(0);
      __syncthreads();
      // async-stage: issue next tile's global loads now; they complete
      // under this tile's QK^T + softmax + PV
      if constexpr (ASTAGE == 1) {
        if (t0 + KVBLK < kv_end) load_tile(t0 + KVBLK);
      }
    }

    // ---- QK^T: S^T tiles [32tok x 32q], toks 0-31 and 32-63 -------------
    f32x16 s0 = {}, s1 = {};
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ds = 0; ds < 8; ds++) {
      const int koff = (ds * 32 + hi * 16);
      const bf16x8 a0 = *reinterpret_cast<const bf16x8*>(
          (char*)k_lds + lo * 256 + (koff ^ ((lo & 7) << 4)));
      const bf16x8 a1 = *reinterpret_cast<const bf16x8*>(
          (char*)k_lds + (lo + 32) * 256 + (koff ^ ((lo & 7) << 4)));
      s0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, qreg[ds], s0, 0, 0, 0);
      s1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, qreg[ds], s1, 0, 0, 0);
    }
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);

    // ---- mask + scale + in-register softmax -----------------------------
    const int qpos = q0 + lo;
    float p[32];
    float mt = -1e30f;
#pragma unroll
    for (int r = 0; r < 16; r++) {
      const int trow = (r & 3) + 8 * (r >> 2) + 4 * hi;
      p[r] = (t0 + trow <= qpos) ? s0[r] * scale : -1e30f;
      p[16 + r] = (t0 + 32 + trow <= qpos) ? s1[r] * scale : -1e30f;
      mt = fmaxf(mt, fmaxf(p[r], p[16 + r]));
    }
    mt = fmaxf(mt, xor32_swap(mt, hi));         // combine lane halves
    float m_new = fmaxf(m_run, mt);
    bool skip_rescale = false;
    if constexpr (DEFER) {
      // defer-max: if the tile max is within 8 of the running max on every
      // lane, keep m_old (P bounded by e^8) and skip the O-rescale.
      if (__all(mt - m_run <= 8.0f)) { m_new = m_run; skip_rescale = true; }
    }
    const float alpha = skip_rescale ? 1.f : __expf(m_run - m_new);
    float ls = 0.f;
#pragma unroll
    for (int r = 0; r < 32; r++) {
      p[r] = __expf(p[r] - m_new);
      ls += p[r];
    }
    ls += xor32_swap(ls, hi);
    l_run = l_run * alpha + ls;
    m_run = m_new;
    if (!skip_rescale) {
      // O lives in D-layout: each reg r is q-row (r&3)+8*(r>>2)+4*hi, NOT
      // the lane's softmax row (q=lo) — rescale with THAT row's alpha
      // (alpha is half-replicated, so lane index `row` suffices).
      float arow[16];
#pragma unroll
      for (int r = 0; r < 16; r++)
        arow[r] = __shfl(alpha, (r & 3) + 8 * (r >> 2) + 4 * hi, 64);
#pragma unroll
      for (int dt = 0; dt < 4; dt++)
#pragma unroll
        for (int r = 0; r < 16; r++) o[dt][r] *= arow[r];
    }

    // ---- P -> A-fragments: pa[ks] = P[q=lo][16ks + 8hi + i] -------------
    // cvt_pk + permlane32_swap (guide rung): pack pairs of P values to
    // bf16x2 with v_cvt_pk_bf16_f32, then ONE permlane32_swap delivers
    // word0 to this lane and word2's source from the partner (and the
    // second swap words 1/3) — 16 cvt_pk + 8 permlane replaces 16 shfl +
    // 32 scalar converts. Derivation: slot ks's 8 own values live at regs
    // b..b+7 with b = 8*(ks&1)+16*(ks>>1); swap(pack(p[b],p[b+1]),
    // pack(p[b+4],p[b+5])) returns (word0, word2) on BOTH halves.
    auto cvtpk = [](float a, float b) {
      unsigned int r;
      asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
      return r;
    };
    bf16x8 pa[4];
#pragma unroll
    for (int ks = 0; ks < 4; ks++) {
      const int b0 = 8 * (ks & 1) + 16 * (ks >> 1);
      uint2_t rA = __builtin_amdgcn_permlane32_swap(
          cvtpk(p[b0], p[b0 + 1]), cvtpk(p[b0 + 4], p[b0 + 5]), false, false);
      uint2_t rB = __builtin_amdgcn_permlane32_swap(
          cvtpk(p[b0 + 2], p[b0 + 3]), cvtpk(p[b0 + 6], p[b0 + 7]),
          false, false);
      unsigned int w[4] = {rA.x, rB.x, rA.y, rB.y};
      pa[ks] = *reinterpret_cast<bf16x8*>(w);
    }

    // ---- PV: O[q][d] += P[q][k] V[k][d] ---------------------------------
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int dt = 0; dt < 4; dt++) {
      const int drow = dt * 32 + lo;
#pragma unroll
      for (int ks = 0; ks < 4; ks++) {
        const int toff = (ks * 16 + 8 * hi) * 2;
        const int rkey = XK ? ((drow ^ (drow >> 3)) & 7) : (drow & 7);
        const bf16x8 vb = *reinterpret_cast<const bf16x8*>(
            (char*)v_lds + drow * VST + (toff ^ (rkey << 4)));
        o[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[ks], vb, o[dt], 0, 0, 0);
      }
    }
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);

    if constexpr (ASTAGE == 2) {
      // stage the (already loaded) next tile into the other buffer while
      // other warps are still computing this one, then prefetch tile t+2
      if (t0 + KVBLK < kv_end) {
        store_tile(((t0 / KVBLK) & 1) ^ 1);
        if (t0 + 2 * KVBLK < kv_end) load_tile(t0 + 2 * KVBLK);
      }
      __syncthreads();
    }
  }

  // ---- epilogue: O /= l (per q-ROW l, D-layout), write bf16 -------------
  float lrow[16];
#pragma unroll
  for (int r = 0; r < 16; r++)
    lrow[r] = __shfl(l_run, (r & 3) + 8 * (r >> 2) + 4 * hi, 64);
#pragma unroll
  for (int dt = 0; dt < 4; dt++)
#pragma unroll
    for (int r = 0; r < 16; r++) {
      const int qrow_i = (r & 3) + 8 * (r >> 2) + 4 * hi;
      const int d = dt * 32 + lo;
      out[(((int64_t)bb * S + q0 + qrow_i) * Hq + h) * 128 + d] =
          f2bf(o[dt][r] / lrow[r]);
    }
}

// ---------------------------------------------------------------------------
// T15 "att[2]" double-pipeline variant (guide: +7-11% attn): iteration t
// issues QK^T MFMAs for tile t, then finishes softmax + PV of tile t-1 —
// the softmax VALU chain of the PREVIOUS tile overlaps the current tile's
// MFMA cluster (separate pipes). Two LDS buffers as before; two barriers
// per tile (PV(t-1) reads vl[cur^1] which store(t+1) then overwrites).
// Requires VSTAGE-style role-split staging (GW=8 only).
template <int GW, int DEFER, int PRIO>
__global__ __launch_bounds__(GW * 64) void prefill32_pipe_kernel(
    short* __restrict__ out, const short* __restrict__ q,
    const short* __restrict__ kc, const short* __restrict__ vc,
    int S, int Hq, int Hkv, float scale) {
  static_assert(GW == 8, "pipe variant: role-split staging needs 8 waves");
  const int qt = blockIdx.x;
  const int kvh = blockIdx.y;
  const int bb = blockIdx.z;
  const int w = threadIdx.x / 64;
  const int l = threadIdx.x % 64;
  const int h = kvh * GW + w;
  const int q0 = qt * QBLK;
  const int lo = l & 31, hi = l >> 5;

  extern __shared__ char lds[];
  constexpr int kHalf = KVBLK * 256 + 128 * 128;   // K + V^T per buffer

  const short* qrow = q + (((int64_t)bb * S + q0 + lo) * Hq + h) * 128;
  bf16x8 qreg[8];
#pragma unroll
  for (int ds = 0; ds < 8; ds++)
    qreg[ds] = *reinterpret_cast<const bf16x8*>(qrow + ds * 16 + 8 * hi);

  f32x16 o[4] = {};
  float m_run = -1e30f, l_run = 0.f;
  const int kv_end = q0 + QBLK;
  const short* kbase = kc + (((int64_t)bb * Hkv + kvh) * S) * 128;
  const short* vbase = vc + (((int64_t)bb * Hkv + kvh) * S) * 128;

  const int vrole = threadIdx.x < 256;
  short8 kreg[4];
  auto load_tile = [&](int t0) {
    if (vrole) {
      const int unit = threadIdx.x;
      const int row0 = (unit >> 4) * 4, d0 = (unit & 15) * 8;
#pragma unroll
      for (int j = 0; j < 4; j++)
        kreg[j] = *reinterpret_cast<const short8*>(
            vbase + (int64_t)(t0 + row0 + j) * 128 + d0);
    } else {
      const int idx = threadIdx.x - 256;
#pragma unroll
      for (int u = 0; u < 4; u++) {
        const int c = idx + u * 256;
        const int row = c >> 4, col8 = (c & 15) * 8;
        kreg[u] = *reinterpret_cast<const short8*>(
            kbase + (int64_t)(t0 + row) * 128 + col8);
      }
    }
  };
  auto store_tile = [&](int buf) {
    char* kl = lds + buf * kHalf;
    char* vl = kl + KVBLK * 256;
    if (vrole) {
      typedef __attribute__((ext_vector_type(4))) short short4_t;
      const int unit = threadIdx.x;
      const int row0 = (unit >> 4) * 4, d0 = (unit & 15) * 8;
#pragma unroll
      for (int i = 0; i < 8; i++) {
        const int d = d0 + i;
        short4_t pk = {kreg[0][i], kreg[1][i], kreg[2][i], kreg[3][i]};
        *(short4_t*)(vl + d * 128 + ((row0 * 2) ^ ((d & 7) << 4))) = pk;
      }
    } else {
      const int idx = threadIdx.x - 256;
#pragma unroll
      for (int u = 0; u < 4; u++) {
        const int c = idx + u * 256;
        const int row = c >> 4, col8 = (c & 15) * 8;
        *reinterpret_cast<short8*>(
            kl + row * 256 + ((col8 * 2) ^ ((row & 7) << 4))) = kreg[u];
      }
    }
  };

  auto qk = [&](int t0, f32x16& s0, f32x16& s1) {
    const char* kl = lds + ((t0 / KVBLK) & 1) * kHalf;
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ds = 0; ds < 8; ds++) {
      const int koff = (ds * 32 + hi * 16);
      const bf16x8 a0 = *reinterpret_cast<const bf16x8*>(
          kl + lo * 256 + (koff ^ ((lo & 7) << 4)));
      const bf16x8 a1 = *reinterpret_cast<const bf16x8*>(
          kl + (lo + 32) * 256 + (koff ^ ((lo & 7) << 4)));
      s0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, qreg[ds], s0, 0, 0, 0);
      s1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, qreg[ds], s1, 0, 0, 0);
    }
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
  };

  auto finish = [&](int t0, const f32x16& s0, const f32x16& s1) {
    const char* vl = lds + ((t0 / KVBLK) & 1) * kHalf + KVBLK * 256;
    const int qpos = q0 + lo;
    float p[32];
    float mt = -1e30f;
#pragma unroll
    for (int r = 0; r < 16; r++) {
      const int trow = (r & 3) + 8 * (r >> 2) + 4 * hi;
      p[r] = (t0 + trow <= qpos) ? s0[r] * scale : -1e30f;
      p[16 + r] = (t0 + 32 + trow <= qpos) ? s1[r] * scale : -1e30f;
      mt = fmaxf(mt, fmaxf(p[r], p[16 + r]));
    }
    mt = fmaxf(mt, xor32_swap(mt, hi));
    float m_new = fmaxf(m_run, mt);
    bool skip_rescale = false;
    if constexpr (DEFER) {
      if (__all(mt - m_run <= 8.0f)) { m_new = m_run; skip_rescale = true; }
    }
    const float alpha = skip_rescale ? 1.f : __expf(m_run - m_new);
    float ls = 0.f;
#pragma unroll
    for (int r = 0; r < 32; r++) {
      p[r] = __expf(p[r] - m_new);
      ls += p[r];
    }
    ls += xor32_swap(ls, hi);
    l_run = l_run * alpha + ls;
    m_run = m_new;
    if (!skip_rescale) {
      float arow[16];
#pragma unroll
      for (int r = 0; r < 16; r++)
        arow[r] = __shfl(alpha, (r & 3) + 8 * (r >> 2) + 4 * hi, 64);
#pragma unroll
      for (int dt = 0; dt < 4; dt++)
#pragma unroll
        for (int r = 0; r < 16; r++) o[dt][r] *= arow[r];
    }
    auto cvtpk = [](float a, float b) {
      unsigned int r;
      asm("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(a), "v"(b));
      return r;
    };
    bf16x8 pa[4];
#pragma unroll
    for (int ks = 0; ks < 4; ks++) {
      const int b0 = 8 * (ks & 1) + 16 * (ks >> 1);
      uint2_t rA = __builtin_amdgcn_permlane32_swap(
          cvtpk(p[b0], p[b0 + 1]), cvtpk(p[b0 + 4], p[b0 + 5]), false, false);
      uint2_t rB = __builtin_amdgcn_permlane32_swap(
          cvtpk(p[b0 + 2], p[b0 + 3]), cvtpk(p[b0 + 6], p[b0 + 7]),
          false, false);
      unsigned int wds[4] = {rA.x, rB.x, rA.y, rB.y};
      pa[ks] = *reinterpret_cast<bf16x8*>(wds);
    }
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int dt = 0; dt < 4; dt++) {
      const int drow = dt * 32 + lo;
#pragma unroll
      for (int ks = 0; ks < 4; ks++) {
        const int toff = (ks * 16 + 8 * hi) * 2;
        const bf16x8 vb = *reinterpret_cast<const bf16x8*>(
            vl + drow * 128 + (toff ^ ((drow & 7) << 4)));
        o[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[ks], vb, o[dt],
                                                        0, 0, 0);
      }
    }
    if constexpr (PRIO) __builtin_amdgcn_s_setprio(0);
  };

  // prologue: stage tile 0, prefetch tile 1 into regs
  load_tile(0);
  store_tile(0);
  __syncthreads();
  if (KVBLK < kv_end) load_tile(KVBLK);

  f32x16 sp0 = {}, sp1 = {};
  int prev = -1;
  for (int t0 = 0; t0 < kv_end; t0 += KVBLK) {
    f32x16 s0 = {}, s1 = {};
    qk(t0, s0, s1);                 // MFMA cluster for tile t ...
    if (prev >= 0) finish(prev, sp0, sp1);   // ... overlaps VALU of t-1
    sp0 = s0; sp1 = s1; prev = t0;
    __syncthreads();                // PV(t-1) readers done with buf cur^1
    if (t0 + KVBLK < kv_end) {
      store_tile(((t0 / KVBLK) & 1) ^ 1);    // tile t+1 -> buf cur^1
      if (t0 + 2 * KVBLK < kv_end) load_tile(t0 + 2 * KVBLK);
    }
    __syncthreads();                // store(t+1) visible before QK(t+1)
  }
  if (prev >= 0) finish(prev, sp0, sp1);     // drain the pipeline

  float lrow[16];
#pragma unroll
  for (int r = 0; r < 16; r++)
    lrow[r] = __shfl(l_run, (r & 3) + 8 * (r >> 2) + 4 * hi, 64);
#pragma unroll
  for (int dt = 0; dt < 4; dt++)
#pragma unroll
    for (int r = 0; r < 16; r++) {
      const int qrow_i = (r & 3) + 8 * (r >> 2) + 4 * hi;
      const int d = dt * 32 + lo;
      out[(((int64_t)bb * S + q0 + qrow_i) * Hq + h) * 128 + d] =
          f2bf(o[dt][r] / lrow[r]);
    }
}

// ---------------------------------------------------------------------------
struct Cfg { int B, S, Hq, Hkv; };

static void cpu_ref(std::vector<float>& o, const std::vector<short>& q,
                    const std::vector<short>& k, const std::vector<short>& v,
                    Cfg c, float scale) {
  const int G = c.Hq / c.Hkv, D = 128;
  for (int b = 0; b < c.B; b++)
    for (int h = 0; h < c.Hq; h++) {
      const int kh = h / G;
      for (int qi = 0; qi < c.S; qi++) {
        std::vector<float> s(qi + 1);
        float m = -1e30f;
        for (int t = 0; t <= qi; t++) {
          float acc = 0;
          for (int d = 0; d < D; d++)
            acc += bf2f(q[(((int64_t)b * c.S + qi) * c.Hq + h) * D + d]) *
                   bf2f(k[(((int64_t)b * c.Hkv + kh) * c.S + t) * D + d]);
          s[t] = acc * scale;
          m = fmaxf(m, s[t]);
        }
        float l = 0;
        for (int t = 0; t <= qi; t++) { s[t] = expf(s[t] - m); l += s[t]; }
        for (int d = 0; d < D; d++) {
          float acc = 0;
          for (int t = 0; t <= qi; t++)
            acc += s[t] * bf2f(v[(((int64_t)b * c.Hkv + kh) * c.S + t) * D + d]);
          o[(((int64_t)b * c.S + qi) * c.Hq + h) * D + d] = acc / l;
        }
      }
    }
}

template <int GW, int ASTAGE, int VSTAGE = 0, int DEFER = 0, int PRIO = 0, int VST = 128, int XK = 0>
static void run(Cfg c, bool check, int iters) {
  const int D = 128;
  const float scale = 1.f / sqrtf((float)D);
  size_t qe = (size_t)c.B * c.S * c.Hq * D, ke = (size_t)c.B * c.Hkv * c.S * D;
  std::vector<short> hq(qe), hk(ke), hv(ke);
  srand(42);
  for (auto& x : hq) x = f2bf((rand() % 2001 - 1000) / 1000.f);
  for (auto& x : hk) x = f2bf((rand() % 2001 - 1000) / 1000.f);
  for (auto& x : hv) x = f2bf((rand() % 2001 - 1000) / 1000.f);
  short *dq, *dk, *dv, *dout;
  CK(hipMalloc(&dq, qe * 2)); CK(hipMalloc(&dk, ke * 2));
  CK(hipMalloc(&dv, ke * 2)); CK(hipMalloc(&dout, qe * 2));
  CK(hipMemcpy(dq, hq.data(), qe * 2, hipMemcpyHostToDevice));
  CK(hipMemcpy(dk, hk.data(), ke * 2, hipMemcpyHostToDevice));
  CK(hipMemcpy(dv, hv.data(), ke * 2, hipMemcpyHostToDevice));
  dim3 grid(c.S / QBLK, c.Hkv, c.B);
  int lds = KVBLK * 256 + 128 * VST;
  if (ASTAGE == 2) lds *= 2;                  // double-buffered
  if (lds > 65536)
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(
            &prefill32_kernel<GW, ASTAGE, VSTAGE, DEFER, PRIO, VST, XK>),
        hipFuncAttributeMaxDynamicSharedMemorySize, lds);
  auto launch = [&] {
    prefill32_kernel<GW, ASTAGE, VSTAGE, DEFER, PRIO, VST, XK><<<grid, GW * 64, lds>>>(
        dout, dq, dk, dv, c.S, c.Hq, c.Hkv, scale);
  };
  launch();
  CK(hipDeviceSynchronize());
  if (check) {
    std::vector<short> got(qe);
    CK(hipMemcpy(got.data(), dout, qe * 2, hipMemcpyDeviceToHost));
    std::vector<float> ref(qe);
    cpu_ref(ref, hq, hk, hv, c, scale);
    double maxerr = 0; int bad = 0;
    for (size_t i = 0; i < qe; i++) {
      const double err = fabs(bf2f(got[i]) - ref[i]);
      if (err > maxerr) maxerr = err;
      if (err > 0.05 && bad++ < 8)
        printf("  mismatch [%zu] got %f want %f\n", i, bf2f(got[i]), ref[i]);
    }
    printf("check GW=%d AS=%d VS=%d DF=%d PR=%d B=%d S=%d Hq=%d: maxerr=%.4f %s\n",
           GW, ASTAGE, VSTAGE, DEFER, PRIO, c.B, c.S, c.Hq, maxerr,
           bad ? "FAIL" : "PASS");
  }
  if (iters > 0) {
    for (int i = 0; i < 3; i++) launch();
    CK(hipDeviceSynchronize());
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    hipEventRecord(e0);
    for (int i = 0; i < iters; i++) launch();
    hipEventRecord(e1);
    CK(hipDeviceSynchronize());
    float ms; hipEventElapsedTime(&ms, e0, e1);
    ms /= iters;
    const double fl = 2.0 * c.B * c.Hq * (double)c.S * c.S * D;
    printf("bench GW=%d AS=%d VS=%d DF=%d PR=%d VT=%d XK=%d B=%d S=%d Hq=%d Hkv=%d: %.3f ms  %.1f TF\n",
           GW, ASTAGE, VSTAGE, DEFER, PRIO, VST, XK, c.B, c.S, c.Hq, c.Hkv, ms,
           fl / (ms * 1e-3) / 1e12);
  }
  hipFree(dq); hipFree(dk); hipFree(dv); hipFree(dout);
}

template <int GW, int DEFER, int PRIO>
static void run_pipe(Cfg c, bool check, int iters) {
  const int D = 128;
  const float scale = 1.f / sqrtf((float)D);
  size_t qe = (size_t)c.B * c.S * c.Hq * D, ke = (size_t)c.B * c.Hkv * c.S * D;
  std::vector<short> hq(qe), hk(ke), hv(ke);
  srand(42);
  for (auto& x : hq) x = f2bf((rand() % 2001 - 1000) / 1000.f);
  for (auto& x : hk) x = f2bf((rand() % 2001 - 1000) / 1000.f);
  for (auto& x : hv) x = f2bf((rand() % 2001 - 1000) / 1000.f);
  short *dq, *dk, *dv, *dout;
  CK(hipMalloc(&dq, qe * 2)); CK(hipMalloc(&dk, ke * 2));
  CK(hipMalloc(&dv, ke * 2)); CK(hipMalloc(&dout, qe * 2));
  CK(hipMemcpy(dq, hq.data(), qe * 2, hipMemcpyHostToDevice));
  CK(hipMemcpy(dk, hk.data(), ke * 2, hipMemcpyHostToDevice));
  CK(hipMemcpy(dv, hv.data(), ke * 2, hipMemcpyHostToDevice));
  dim3 grid(c.S / QBLK, c.Hkv, c.B);
  int lds = 2 * (KVBLK * 256 + 128 * 128);
  auto launch = [&] {
    prefill32_pipe_kernel<GW, DEFER, PRIO><<<grid, GW * 64, lds>>>(
        dout, dq, dk, dv, c.S, c.Hq, c.Hkv, scale);
  };
  launch();
  CK(hipDeviceSynchronize());
  if (check) {
    std::vector<short> got(qe);
    CK(hipMemcpy(got.data(), dout, qe * 2, hipMemcpyDeviceToHost));
    std::vector<float> ref(qe);
    cpu_ref(ref, hq, hk, hv, c, scale);
    double maxerr = 0; int bad = 0;
    for (size_t i = 0; i < qe; i++) {
      const double err = fabs(bf2f(got[i]) - ref[i]);
      if (err > maxerr) maxerr = err;
      if (err > 0.05 && bad++ < 8)
        printf("  mismatch [%zu] got %f want %f\n", i, bf2f(got[i]), ref[i]);
    }
    printf("check PIPE GW=%d DF=%d PR=%d B=%d S=%d Hq=%d: maxerr=%.4f %s\n",
           GW, DEFER, PRIO, c.B, c.S, c.Hq, maxerr, bad ? "FAIL" : "PASS");
  }
  if (iters > 0) {
    for (int i = 0; i < 3; i++) launch();
    CK(hipDeviceSynchronize());
    hipEvent_t e0, e1; hipEventCreate(&e0); hipEventCreate(&e1);
    hipEventRecord(e0);
    for (int i = 0; i < iters; i++) launch();
    hipEventRecord(e1);
    CK(hipDeviceSynchronize());
    float ms; hipEventElapsedTime(&ms, e0, e1);
    ms /= iters;
    const double fl = 2.0 * c.B * c.Hq * (double)c.S * c.S * D;
    printf("bench PIPE GW=%d DF=%d PR=%d B=%d S=%d Hq=%d Hkv=%d: %.3f ms  %.1f TF\n",
           GW, DEFER, PRIO, c.B, c.S, c.Hq, c.Hkv, ms,
           fl / (ms * 1e-3) / 1e12);
  }
  (void)hipFree(dq); (void)hipFree(dk); (void)hipFree(dv); (void)hipFree(dout);
}

int main() {
  if (!probe32()) return 1;
  permprobe();
  run<8, 2, 1, 1, 0, 128, 1>({1, 256, 16, 2}, true, 0);
  run<8, 2, 1, 1>({1, 8192, 64, 8}, false, 20);
  run<8, 2, 1, 1, 0, 128, 1>({1, 8192, 64, 8}, false, 20);
  run<8, 2, 1, 1, 1, 128, 1>({1, 8192, 64, 8}, false, 20);
  run<8, 1, 1, 1, 0, 128, 1>({1, 8192, 64, 8}, false, 20);
  run<8, 2, 1, 1, 0, 128, 1>({2, 4096, 64, 8}, false, 20);
  return 0;
}
