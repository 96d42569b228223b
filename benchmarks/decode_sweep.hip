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

struct Bufs {
  short *kc, *vc, *q, *out;
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
  // fill kv with a pattern (values don't matter for timing)
  CK(hipMemset(bf.kc, 0x3c, cache_e * 2));
  CK(hipMemset(bf.vc, 0x3c, cache_e * 2));
  CK(hipMemset(bf.q, 0x3c, (size_t)B * Hq * HD * 2));
  return bf;
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
  {
    Bufs bf = make(8);
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
  }
  {
    Bufs bf = make(1);
    run<1, 8, 1, 2>(bf, "");
    run<1, 8, 1, 4>(bf, "");
    run<1, 16, 1, 4>(bf, "");
  }
  return 0;
}
