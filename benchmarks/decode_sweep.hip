// Standalone sweep over decode-attention template combos (DP, HS, DEPTH)
// on synthetic data. Build:
//   hipcc --offload-arch=gfx950 -O3 -std=c++17 \
//     -I dynamo_amd/csrc benchmarks/decode_sweep.hip -o benchmarks/decode_sweep
// Run on the GPU box: ./benchmarks/decode_sweep
#include "../dynamo_amd/csrc/attention_decode_impl.h"
#include <hip/hip_runtime.h>
#include <cstdio>
#include <vector>

using namespace decode_attn;

#define CK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  printf("HIP error %s @%d\n", hipGetErrorString(e), __LINE__); exit(1); } } while (0)

constexpr int B = 16, Hkv = 8, PS = 64, HD = 128, CTX = 8192;

// probe: permlane16_swap semantics (the swapped-MFMA P-exchange relies on
// x = {l%32<16: v0(l), else: v1(l-16)}, y = {l%32<16: v0(l+16), else: v1(l)})
__global__ void perm16_probe_kernel(unsigned int* out) {
  typedef __attribute__((ext_vector_type(2))) unsigned int uint2_t;
  unsigned int v0 = threadIdx.x, v1 = 1000 + threadIdx.x;
  uint2_t r = __builtin_amdgcn_permlane16_swap(v0, v1, false, false);
  out[threadIdx.x * 2] = r.x;
  out[threadIdx.x * 2 + 1] = r.y;
}

static void perm16_probe() {
  unsigned int* d; CK(hipMalloc(&d, 128 * 4));
  perm16_probe_kernel<<<1, 64>>>(d);
  CK(hipDeviceSynchronize());
  std::vector<unsigned int> h(128);
  CK(hipMemcpy(h.data(), d, 128 * 4, hipMemcpyDeviceToHost));
  printf("permlane16_swap l0:(%u,%u) l16:(%u,%u) l5:(%u,%u) l21:(%u,%u) "
         "l37:(%u,%u) l53:(%u,%u)\n",
         h[0], h[1], h[32], h[33], h[10], h[11], h[42], h[43],
         h[74], h[75], h[106], h[107]);
  (void)hipFree(d);
}

struct Bufs {
  short *kc, *vc, *vct, *q, *out;
  float *partial, *ml;
  int32_t *pt, *ctx;
  int npages, C;
};

static Bufs make(int G) {
  Bufs bf;
  bf.npages = B * (CTX / PS);
  bf.C = (CTX + kChunk - 1) / kChunk;
  size_t cache_e = (size_t)bf.npages * Hkv * PS * HD;
  CK(hipMalloc(&bf.kc, cache_e * 2));
  CK(hipMalloc(&bf.vc, cache_e * 2));
  int Hq = G * Hkv;
  CK(hipMalloc(&bf.q, (size_t)B * Hq * HD * 2));
  CK(hipMalloc(&bf.out, (size_t)B * Hq * HD * 2));
  CK(hipMalloc(&bf.partial, (size_t)B * Hq * bf.C * HD * 4));
  CK(hipMalloc(&bf.ml, (size_t)B * Hq * bf.C * 2 * 4));
  std::vector<int32_t> pt(B * (CTX / PS));
  for (int i = 0; i < (int)pt.size(); i++) pt[i] = i;
  CK(hipMalloc(&bf.pt, pt.size() * 4));
  CK(hipMemcpy(bf.pt, pt.data(), pt.size() * 4, hipMemcpyHostToDevice));
  std::vector<int32_t> cl(B, CTX);
  CK(hipMalloc(&bf.ctx, B * 4));
  CK(hipMemcpy(bf.ctx, cl.data(), B * 4, hipMemcpyHostToDevice));
  // fill kv with RANDOM bf16 in [-1,1] (numerics cross-checks need it;
  // bandwidth timing is data-independent)
  auto fill = [](short* dptr, size_t n) {
    std::vector<short> h(n);
    for (auto& x : h) {
      float f = (rand() % 2001 - 1000) / 1000.f;
      union { float ff; unsigned int i; } c; c.ff = f;
      unsigned int r = c.i + 0x7fff + ((c.i >> 16) & 1);
      h[&x - h.data()] = (short)(r >> 16);
    }
    CK(hipMemcpy(dptr, h.data(), n * 2, hipMemcpyHostToDevice));
  };
  srand(7);
  fill(bf.kc, cache_e);
  fill(bf.vc, cache_e);
  fill(bf.q, (size_t)B * Hq * HD);
  // transposed V cache (VT layout): per (page, head) block,
  // vt[d * PS + t] = v[t * HD + d]
  CK(hipMalloc(&bf.vct, cache_e * 2));
  {
    std::vector<short> hv(cache_e), ht(cache_e);
    CK(hipMemcpy(hv.data(), bf.vc, cache_e * 2, hipMemcpyDeviceToHost));
    const size_t nblk = (size_t)bf.npages * Hkv;
    for (size_t blk = 0; blk < nblk; blk++) {
      const size_t base = blk * PS * HD;
      for (int t = 0; t < PS; t++)
        for (int d = 0; d < HD; d++)
          ht[base + (size_t)d * PS + t] = hv[base + (size_t)t * HD + d];
    }
    CK(hipMemcpy(bf.vct, ht.data(), cache_e * 2, hipMemcpyHostToDevice));
  }
  return bf;
}

// swapped-operand MFMA variant: bench + elementwise compare vs the proven
// paged_decode_mfma output
template <int G, int DEFER = 1, int PRIO = 1, int KPF = 0, int VS = 80, int XK2 = 0,
          int VT = 0, int LG2 = 0, int MINW = 1>
static void run_mfma_swapped(const Bufs& bf, bool check) {
  dim3 grid(B, Hkv, bf.C);
  // VT4/5 stage K (and V) through the per-wave v_lds region; VT1-3 skip
  // all staging LDS
  const int lds = (VT >= 4) ? mfma_swapped_lds_bytes(G, HD, VS)
                : VT        ? mfma_swapped_vt_lds_bytes(G, HD)
                            : mfma_swapped_lds_bytes(G, HD, VS);
  const int iters = 30;
  if (lds > 65536)
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&paged_decode_mfma_swapped<DEFER, PRIO, KPF, 0, VS, XK2, VT, LG2, MINW>),
        hipFuncAttributeMaxDynamicSharedMemorySize, lds);
  auto launch = [&] {
    paged_decode_mfma_swapped<DEFER, PRIO, KPF, 0, VS, XK2, VT, LG2, MINW><<<grid, kBlock, lds>>>(
        bf.partial, bf.ml, bf.out, bf.q, bf.kc, VT ? bf.vct : bf.vc, bf.pt, bf.ctx,
        0.0883883f, kChunk, G, B, Hkv, bf.C, CTX / PS, 6, HD, nullptr);
    paged_decode_phase2<<<dim3(B, G * Hkv), 128>>>(
        bf.out, bf.partial, bf.ml, bf.ctx, kChunk, G * Hkv, bf.C, HD);
  };
  if (check) {
    const size_t n = (size_t)B * G * Hkv * HD;
    launch();
    CK(hipDeviceSynchronize());
    std::vector<short> got(n);
    CK(hipMemcpy(got.data(), bf.out, n * 2, hipMemcpyDeviceToHost));
    // reference: the existing (validated) kernel
    const int lds0 = mfma_lds_bytes(G, HD);
    if (lds0 > 65536)
      (void)hipFuncSetAttribute(
          reinterpret_cast<const void*>(&paged_decode_mfma<0, 0>),
          hipFuncAttributeMaxDynamicSharedMemorySize, lds0);
    paged_decode_mfma<0, 0><<<grid, kBlock, lds0>>>(
        bf.partial, bf.ml, bf.out, bf.q, bf.kc, bf.vc, bf.pt, bf.ctx,
        0.0883883f, kChunk, G, B, Hkv, bf.C, CTX / PS, 6, HD);
    paged_decode_phase2<<<dim3(B, G * Hkv), 128>>>(
        bf.out, bf.partial, bf.ml, bf.ctx, kChunk, G * Hkv, bf.C, HD);
    CK(hipDeviceSynchronize());
    std::vector<short> ref(n);
    CK(hipMemcpy(ref.data(), bf.out, n * 2, hipMemcpyDeviceToHost));
    auto b2f = [](short u) {
      union { float f; unsigned int i; } c;
      c.i = ((unsigned int)(unsigned short)u) << 16; return c.f;
    };
    double maxerr = 0; int bad = 0;
    for (size_t i = 0; i < n; i++) {
      double e = fabs(b2f(got[i]) - b2f(ref[i]));
      if (e > maxerr) maxerr = e;
      if (e > 0.03 && bad++ < 6)
        printf("  sw mismatch [%zu] got %f want %f\n", i,
               b2f(got[i]), b2f(ref[i]));
    }
    printf("check G%d MFMA_SW vs MFMA: maxerr=%.4f %s\n", G, maxerr,
           bad ? "FAIL" : "PASS");
  }
  for (int i = 0; i < 5; i++) launch();
  CK(hipDeviceSynchronize());
  hipEvent_t e0, e1;
  CK(hipEventCreate(&e0)); CK(hipEventCreate(&e1));
  CK(hipEventRecord(e0));
  for (int i = 0; i < iters; i++) launch();
  CK(hipEventRecord(e1));
  CK(hipEventSynchronize(e1));
  float ms;
  CK(hipEventElapsedTime(&ms, e0, e1));
  double t = ms / 1000.0 / iters;
  double gb = 2.0 * B * CTX * Hkv * HD * 2 / 1e9;
  printf("G%d MFMA_SW DF%d PR%d KP%d VS%d X%d VT%d L%d W%d %8.1f us  %7.0f GB/s\n", G,
         DEFER, PRIO, KPF, VS, XK2, VT, LG2, MINW, t * 1e6, gb / t);
  fflush(stdout);
}

template <int G, int DEFER = 0, int PRIO = 0>
static void run_mfma(const Bufs& bf) {
  dim3 grid(B, Hkv, bf.C);
  const int lds = mfma_lds_bytes(G, HD);
  const int iters = 30;
  if (lds > 65536)
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(&paged_decode_mfma<DEFER, PRIO>),
        hipFuncAttributeMaxDynamicSharedMemorySize, lds);
  for (int i = 0; i < 5; i++)
    paged_decode_mfma<DEFER, PRIO><<<grid, kBlock, lds>>>(
        bf.partial, bf.ml, bf.out, bf.q, bf.kc, bf.vc, bf.pt, bf.ctx,
        0.0883883f, kChunk, G, B, Hkv, bf.C, CTX / PS, 6, HD);
  CK(hipDeviceSynchronize());
  hipEvent_t e0, e1;
  CK(hipEventCreate(&e0)); CK(hipEventCreate(&e1));
  CK(hipEventRecord(e0));
  for (int i = 0; i < iters; i++) {
    paged_decode_mfma<DEFER, PRIO><<<grid, kBlock, lds>>>(
        bf.partial, bf.ml, bf.out, bf.q, bf.kc, bf.vc, bf.pt, bf.ctx,
        0.0883883f, kChunk, G, B, Hkv, bf.C, CTX / PS, 6, HD);
    paged_decode_phase2<<<dim3(B, G * Hkv), 128>>>(
        bf.out, bf.partial, bf.ml, bf.ctx, kChunk, G * Hkv, bf.C, HD);
  }
  CK(hipEventRecord(e1));
  CK(hipEventSynchronize(e1));
  float ms;
  CK(hipEventElapsedTime(&ms, e0, e1));
  double t = ms / 1000.0 / iters;
  double gb = 2.0 * B * CTX * Hkv * HD * 2 / 1e9;
  printf("G%d MFMA DF%d PR%d   %8.1f us  %7.0f GB/s\n", G, DEFER, PRIO,
         t * 1e6, gb / t);
  fflush(stdout);
}

template <int G, int DP, int HS, int DEPTH>
static void run(const Bufs& bf, const char* tag) {
  dim3 grid(B, Hkv, bf.C);
  const int lds = phase1_lds_bytes(G, HS, HD);
  const int iters = 30;
  // warmup
  for (int i = 0; i < 5; i++)
    paged_decode_phase1<G, DP, HS, DEPTH><<<grid, kBlock, lds>>>(
        bf.partial, bf.ml, bf.out, bf.q, bf.kc, bf.vc, bf.pt, bf.ctx,
        0.0883883f, kChunk, B, Hkv, bf.C, CTX / PS, 6, HD);
  CK(hipDeviceSynchronize());
  hipEvent_t e0, e1;
  CK(hipEventCreate(&e0)); CK(hipEventCreate(&e1));
  CK(hipEventRecord(e0));
  for (int i = 0; i < iters; i++) {
    paged_decode_phase1<G, DP, HS, DEPTH><<<grid, kBlock, lds>>>(
        bf.partial, bf.ml, bf.out, bf.q, bf.kc, bf.vc, bf.pt, bf.ctx,
        0.0883883f, kChunk, B, Hkv, bf.C, CTX / PS, 6, HD);
    paged_decode_phase2<<<dim3(B, G * Hkv), 128>>>(
        bf.out, bf.partial, bf.ml, bf.ctx, kChunk, G * Hkv, bf.C, HD);
  }
  CK(hipEventRecord(e1));
  CK(hipEventSynchronize(e1));
  float ms;
  CK(hipEventElapsedTime(&ms, e0, e1));
  double t = ms / 1000.0 / iters;
  double gb = 2.0 * B * CTX * Hkv * HD * 2 / 1e9;
  printf("G%d DP%-2d HS%d D%d  %-10s %8.1f us  %7.0f GB/s\n", G, DP, HS, DEPTH,
         tag, t * 1e6, gb / t);
  fflush(stdout);
}

int main() {
  perm16_probe();
  {
    Bufs bf = make(8);
    run_mfma_swapped<8, 0, 0, 0, 72, 1>(bf, true);
    run_mfma_swapped<8, 0, 0, 0, 72, 0, 1>(bf, true);   // VT check
    run_mfma_swapped<8, 1, 1, 0, 72, 1>(bf, false);
    run_mfma_swapped<8, 1, 1, 0, 72, 0, 1>(bf, false);  // VT
    run_mfma_swapped<8, 1, 0, 0, 72, 0, 1>(bf, false);  // VT no-prio
    run_mfma_swapped<8, 0, 0, 0, 72, 0, 2>(bf, true);   // VT2 check
    run_mfma_swapped<8, 0, 0, 0, 72, 0, 2, 1>(bf, true);  // VT2+LG2 check
    run_mfma_swapped<8, 1, 1, 0, 72, 0, 2>(bf, false);  // VT2 pipelined
    run_mfma_swapped<8, 0, 0, 0, 72, 0, 2, 0, 2>(bf, true);   // VT2 minw2 chk
    run_mfma_swapped<8, 1, 1, 0, 72, 0, 2, 0, 2>(bf, false);  // VT2 minw2
    run_mfma_swapped<8, 1, 1, 0, 72, 0, 1, 0, 3>(bf, false);  // VT1 minw3
    run_mfma_swapped<8, 0, 0, 0, 72, 0, 3, 0, 1>(bf, true);   // VT3 64tok chk
    run_mfma_swapped<8, 1, 1, 0, 72, 0, 3, 0, 2>(bf, false);  // VT3 minw2
    run_mfma_swapped<8, 0, 0, 0, 72, 0, 4, 0, 1>(bf, true);   // VT4 chk
    run_mfma_swapped<8, 1, 1, 0, 72, 0, 4, 0, 1>(bf, false);  // VT4
    run_mfma_swapped<8, 0, 0, 0, 72, 0, 5, 0, 1>(bf, true);   // VT5 chk
    run_mfma_swapped<8, 1, 1, 0, 72, 0, 5, 0, 1>(bf, false);  // VT5
    run_mfma_swapped<8, 1, 1, 0, 72, 0, 5, 0, 2>(bf, false);  // VT5 minw2
    run_mfma_swapped<8, 1, 1, 0, 72, 0, 5, 0, 3>(bf, false);  // VT5 minw3
    run<8, 16, 4, 2>(bf, "");
    run<8, 16, 4, 3>(bf, "");
    run<8, 16, 4, 4>(bf, "");
    run<8, 16, 2, 2>(bf, "");
    run<8, 16, 2, 3>(bf, "");
    run<8, 8, 4, 2>(bf, "");
    run<8, 8, 4, 3>(bf, "");
    run<8, 16, 1, 2>(bf, "");
    run<8, 8, 2, 2>(bf, "");
    run_mfma<8, 0, 0>(bf);
    run_mfma<8, 1, 0>(bf);
    run_mfma<8, 0, 1>(bf);
    run_mfma<8, 1, 1>(bf);
  }
  {
    Bufs bf = make(4);
    run<4, 16, 1, 2>(bf, "");
    run<4, 16, 1, 3>(bf, "");
    run<4, 16, 2, 2>(bf, "");
    run<4, 16, 2, 3>(bf, "");
    run<4, 16, 2, 4>(bf, "");
    run<4, 16, 4, 2>(bf, "");
    run<4, 8, 2, 2>(bf, "");
    run<4, 8, 1, 2>(bf, "");
    run_mfma<4, 1, 0>(bf);
    run_mfma_swapped<4, 1, 1, 0, 88>(bf, false);
    run_mfma_swapped<4, 0, 0, 0, 72, 0, 2>(bf, true);   // G4 VT2 check
    run_mfma_swapped<4, 1, 1, 0, 72, 0, 2>(bf, false);  // G4 VT2
    run_mfma_swapped<4, 1, 1, 0, 72, 0, 2, 1>(bf, false);  // G4 VT2+LG2
    run_mfma_swapped<4, 1, 1, 0, 72, 0, 2, 0, 2>(bf, false);  // G4 VT2 minw2
    run_mfma_swapped<4, 1, 1, 0, 72, 0, 1, 0, 3>(bf, false);  // G4 VT1 minw3
    run_mfma_swapped<4, 1, 1, 0, 72, 0, 3, 0, 1>(bf, false);  // G4 VT3
    run_mfma_swapped<4, 1, 1, 0, 72, 0, 3, 0, 2>(bf, false);  // G4 VT3 minw2
    run_mfma_swapped<2, 0, 0, 0, 72, 0, 2>(bf, true);   // G2 VT2 check
    run_mfma_swapped<2, 1, 1, 0, 72, 0, 2>(bf, false);  // G2 VT2
  }
  {
    Bufs bf = make(1);
    run<1, 8, 1, 2>(bf, "");
    run<1, 8, 1, 4>(bf, "");
    run<1, 16, 1, 4>(bf, "");
  }
  return 0;
}
