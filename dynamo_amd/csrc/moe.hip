// MoE kernels: top-k gating + grouped GEMM for expert FFNs.
//
// Decode-regime grouped GEMM (few tokens per expert, e.g. conc 16 x top-2
// over 8 experts) is WEIGHTS-bandwidth bound: every live expert's weights
// stream from HBM once regardless of token count. The kernel assigns one
// N-row of the current expert's weight to each lane and streams D in
// 128-element chunks, dotting against <=16 segment tokens staged in LDS
// (broadcast reads). Arithmetic intensity M*2 flops / 2 B keeps VALU far
// from the ceiling, so the kernel runs at the HBM roofline like a dense
// skinny GEMM but with ONE launch for all experts (vs E x hipBLASLt
// launches). Large (prefill) segments go through hipBLASLt per expert at
// the Python layer instead.
//
// Capability parity: the reference delegates MoE math to its engines
// (SURVEY.md §2.4 "engine kernels the reference outsources").
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr int kBlock = 256;
constexpr int kMaxM = 16;   // tokens per tile (host splits larger segments)
constexpr int kDC = 128;    // D chunk staged per iteration

// Segment offsets live on DEVICE (seg_start [E+1], from a cumsum of the
// routing counts) so the launch is sync-free and hipGraph-capturable: the
// grid covers worst-case m-tiles per expert and blocks with no tokens
// exit. tiles mode (tiles != nullptr) is kept for host-built tile lists.

__global__ __launch_bounds__(kBlock) void moe_gemm_kernel(
    short* __restrict__ y,        // [T, N] bf16 (gathered order)
    const short* __restrict__ x,  // [T, D] bf16 (gathered by expert)
    const short* __restrict__ w,  // [E, N, D] bf16
    const int32_t* __restrict__ tiles,      // [ntiles,3] or null
    const int32_t* __restrict__ seg_start,  // [E+1] or null
    int D, int N, int ntiles) {
  typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2_t;
  typedef __attribute__((ext_vector_type(4))) short short4_t;
  int e, r0, m;
  if (tiles != nullptr) {
    const int tile = blockIdx.x;
    if (tile >= ntiles) return;
    e = tiles[3 * tile];
    r0 = tiles[3 * tile + 1];
    m = tiles[3 * tile + 2];
  } else {
    // blockIdx.x = expert * max_m_tiles + m_tile
    e = blockIdx.x / ntiles;            // ntiles = max m-tiles per expert
    const int mt = blockIdx.x % ntiles;
    const int s = seg_start[e], cnt = seg_start[e + 1] - s;
    if (mt * kMaxM >= cnt) return;
    r0 = s + mt * kMaxM;
    m = min(kMaxM, cnt - mt * kMaxM);
  }
  // lane-per-W-row with the row streamed in 16-deep load bursts: the
  // stride-D access is uncoalesced but L2 soaks it, and the 16 independent
  // short8 loads per chunk keep HBM latency covered. (Two "coalesced"
  // redesigns — wave-per-row with LDS-staged X and with L2-read X —
  // measured 2x and 3x SLOWER: only 1-2 loads in flight per lane.)
  const int n = blockIdx.y * kBlock + threadIdx.x;  // this lane's W row
  const short* wrow = w + ((int64_t)e * N + n) * D;

  __shared__ short x_lds[kMaxM * kDC];

  float acc[kMaxM];
#pragma unroll
  for (int i = 0; i < kMaxM; i++) acc[i] = 0.f;

  for (int dc = 0; dc < D; dc += kDC) {
    __syncthreads();
    // stage X chunk [m][kDC]
    for (int i = threadIdx.x; i < m * (kDC / 8); i += kBlock) {
      const int mi = i / (kDC / 8);
      const int d8 = i % (kDC / 8);
      *reinterpret_cast<short8*>(x_lds + mi * kDC + d8 * 8) =
          *reinterpret_cast<const short8*>(
              x + ((int64_t)(r0 + mi)) * D + dc + d8 * 8);
    }
    __syncthreads();
    if (n < N) {
      // stream this lane's weight chunk once; dot against every token
      short8 wv[kDC / 8];
#pragma unroll
      for (int i = 0; i < kDC / 8; i++)
        wv[i] = *reinterpret_cast<const short8*>(wrow + dc + i * 8);
      for (int mi = 0; mi < m; mi++) {
        float d = acc[mi];
#pragma unroll
        for (int i = 0; i < kDC / 8; i++) {
          const bf16x2_t* w2 = reinterpret_cast<const bf16x2_t*>(&wv[i]);
          const bf16x2_t* x2 = reinterpret_cast<const bf16x2_t*>(
              x_lds + mi * kDC + i * 8);
#pragma unroll
          for (int p = 0; p < 4; p++)
            d = __builtin_amdgcn_fdot2_f32_bf16(w2[p], x2[p], d, false);
        }
        acc[mi] = d;
      }
    }
  }
  if (n < N) {
    for (int mi = 0; mi < m; mi++)
      y[((int64_t)(r0 + mi)) * N + n] = f32_to_bf16(acc[mi]);
  }
}

// MFMA grouped-GEMM variant: the W stream rides MFMA B-fragments loaded
// DIRECTLY from global (the decode-attention K-load structure, which
// sustains 3.7-3.9 TB/s on this chip, vs 2.0 TB/s effective for the
// lane-per-row VALU kernel above). Per wave: one 16-column N-tile;
// A = X tokens (<=16) staged once in LDS (XOR-swizzled rows); per 32-dim
// K chunk one v_mfma_f32_16x16x32_bf16 with B[k=8*lg+j][col=lr] =
// W[n0+lr][kc*32+8lg+j] as a single b128 global load per lane. 4 waves
// per block cover 64 N columns sharing the X tile.
template <int NT>   // 16-column N-tiles per wave (grid fill vs depth)
__global__ __launch_bounds__(kBlock) void moe_gemm_mfma_kernel(
    short* __restrict__ y,        // [T, N]
    const short* __restrict__ x,  // [T, D]
    const short* __restrict__ w,  // [E, N, D]
    const int32_t* __restrict__ tiles,      // [ntiles,3] or null
    const int32_t* __restrict__ seg_start,  // [E+1] or null
    int D, int N, int ntiles) {
  typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
  int e, r0, m;
  if (tiles != nullptr) {
    const int tile = blockIdx.x;
    if (tile >= ntiles) return;
    e = tiles[3 * tile];
    r0 = tiles[3 * tile + 1];
    m = tiles[3 * tile + 2];
  } else {
    e = blockIdx.x / ntiles;
    const int mt = blockIdx.x % ntiles;
    const int s = seg_start[e], cnt = seg_start[e + 1] - s;
    if (mt * kMaxM >= cnt) return;
    r0 = s + mt * kMaxM;
    m = min(kMaxM, cnt - mt * kMaxM);
  }
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int lr = lane & 15;
  const int lg = lane >> 4;
  // NT 16-column N-tiles per wave: NT independent B loads in flight per
  // kc per lane (the VALU kernel's lesson: in-flight depth beats
  // everything on the W stream), all sharing one A fragment. NT=4 for
  // wide N (gate_up), NT=2 for narrow N (down proj) to keep the grid
  // filling 256 CUs.
  const int n0 = (blockIdx.y * 4 + wid) * (NT * 16);
  const short* wb[NT];
  bool ok[NT];
#pragma unroll
  for (int t = 0; t < NT; t++) {
    wb[t] = w + ((int64_t)e * N + n0 + t * 16 + lr) * D;
    ok[t] = (n0 + t * 16 + lr) < N;
  }

  // X tile [16][kXC] staged per 512-dim chunk (16 KB LDS), XOR-swizzled
  // rows; the wide chunk keeps 16 B-loads + MFMAs between barriers so the
  // W stream pipelines like the decode-attention K loop
  constexpr int kXC = 512;
  __shared__ short x_lds[kMaxM * kXC];

  f32x4 accs[NT] = {};
  for (int dc = 0; dc < D; dc += kXC) {
    const int cw = min(kXC, D - dc);
    __syncthreads();
    for (int i = threadIdx.x; i < kMaxM * (cw / 8); i += kBlock) {
      const int mi = i / (cw / 8);
      const int d8 = i % (cw / 8);
      short8 v{};
      if (mi < m)
        v = *reinterpret_cast<const short8*>(
            x + ((int64_t)(r0 + mi)) * D + dc + d8 * 8);
      *reinterpret_cast<short8*>(
          (char*)x_lds + mi * (kXC * 2) + ((d8 * 16) ^ ((mi & 7) << 4))) = v;
    }
    __syncthreads();
#pragma unroll 4
    for (int kc = 0; kc < cw / 32; kc++) {
      // A: X[token=lr][dc + kc*32 + 8lg + j] (zero rows beyond m)
      short8 a = *reinterpret_cast<const short8*>(
          (char*)x_lds + lr * (kXC * 2) +
          ((kc * 64 + lg * 16) ^ ((lr & 7) << 4)));
      // B: W[n][dc + kc*32 + 8lg + j] — direct global b128, four tiles
      short8 bt[NT];
#pragma unroll
      for (int t = 0; t < NT; t++)
        bt[t] = ok[t]
            ? *reinterpret_cast<const short8*>(wb[t] + dc + kc * 32 + lg * 8)
            : short8{};
#pragma unroll
      for (int t = 0; t < NT; t++)
        accs[t] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            *reinterpret_cast<bf16x8_t*>(&a),
            *reinterpret_cast<bf16x8_t*>(&bt[t]), accs[t], 0, 0, 0);
    }
  }
  // D[row=token lg*4+r][col=n lr]
#pragma unroll
  for (int t = 0; t < NT; t++)
#pragma unroll
    for (int r = 0; r < 4; r++) {
      const int mi = lg * 4 + r;
      if (mi < m && ok[t])
        y[((int64_t)(r0 + mi)) * N + n0 + t * 16 + lr] =
            f32_to_bf16(accs[t][r]);
    }
}

// top-k gating: softmax over E experts, renormalized top-k weights.
// one wave per token; E <= 64.
__global__ void topk_gating_kernel(float* __restrict__ topw,   // [T, k]
                                   int32_t* __restrict__ topi, // [T, k]
                                   const float* __restrict__ logits,  // [T,E]
                                   int T, int E, int K) {
  const int t = blockIdx.x * (blockDim.x / 64) + (threadIdx.x / 64);
  if (t >= T) return;
  const int lane = threadIdx.x & 63;
  float x = (lane < E) ? logits[(int64_t)t * E + lane] : -1e30f;
  // softmax denominator over E
  float mx = wave_reduce_max(x);
  float ex = (lane < E) ? __expf(x - mx) : 0.f;
  float denom = wave_reduce_sum(ex);
  float p = ex / denom;
  // iterative top-k (K <= 8; static indices to keep arrays in registers)
  float mine = p;
  float wsum = 0.f;
  float myw[8];
  int myi[8];
#pragma unroll
  for (int k = 0; k < 8; k++) {
    if (k >= K) break;
    float best = wave_reduce_max(mine);
    // first lane holding `best` wins
    unsigned long long mask = __ballot(mine == best && lane < E);
    int win = __ffsll((long long)mask) - 1;
    myw[k] = best;
    myi[k] = win;
    wsum += best;
    if (lane == win) mine = -1.f;  // remove from candidates
  }
#pragma unroll
  for (int k = 0; k < 8; k++) {
    if (k >= K) break;
    if (lane == k) {
      topw[(int64_t)t * K + k] = myw[k] / wsum;
      topi[(int64_t)t * K + k] = myi[k];
    }
  }
}

}  // namespace

void moe_grouped_gemm(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                      torch::Tensor tiles) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16);
  TORCH_CHECK(tiles.dtype() == torch::kInt32 && tiles.size(1) == 3);
  const int D = x.size(1);
  const int N = w.size(1);
  TORCH_CHECK(w.size(2) == D && y.size(1) == N);
  TORCH_CHECK(D % kDC == 0, "in-features must be a multiple of 128");
  const int ntiles = tiles.size(0);
  if (ntiles == 0) return;
  auto stream = at::cuda::getCurrentHIPStream();
  static const bool use_mfma = [] {
    const char* env = getenv("DYNAMO_MOE_MFMA");
    return env == nullptr || env[0] != '0';
  }();
  if (use_mfma && D % 32 == 0) {
    if (N >= 16384) {
      dim3 grid(ntiles, (N + 255) / 256);
      moe_gemm_mfma_kernel<4><<<grid, kBlock, 0, stream>>>(
          (short*)y.data_ptr(), (const short*)x.data_ptr(),
          (const short*)w.data_ptr(), tiles.data_ptr<int32_t>(), nullptr,
          D, N, ntiles);
    } else {
      dim3 grid(ntiles, (N + 127) / 128);
      moe_gemm_mfma_kernel<2><<<grid, kBlock, 0, stream>>>(
          (short*)y.data_ptr(), (const short*)x.data_ptr(),
          (const short*)w.data_ptr(), tiles.data_ptr<int32_t>(), nullptr,
          D, N, ntiles);
    }
  } else {
    dim3 grid(ntiles, (N + kBlock - 1) / kBlock);
    moe_gemm_kernel<<<grid, kBlock, 0, stream>>>(
        (short*)y.data_ptr(), (const short*)x.data_ptr(),
        (const short*)w.data_ptr(), tiles.data_ptr<int32_t>(), nullptr,
        D, N, ntiles);
  }
  HIP_CHECK_KERNEL();
}

// sync-free variant: segment offsets on device (hipGraph-capturable).
void moe_grouped_gemm_seg(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                          torch::Tensor seg_start, int64_t max_tokens) {
  TORCH_CHECK(x.is_cuda() && x.dtype() == torch::kBFloat16);
  TORCH_CHECK(seg_start.dtype() == torch::kInt32);
  const int D = x.size(1);
  const int N = w.size(1);
  const int E = w.size(0);
  TORCH_CHECK(seg_start.numel() == E + 1);
  TORCH_CHECK(D % kDC == 0, "in-features must be a multiple of 128");
  const int max_mt = (int)((max_tokens + kMaxM - 1) / kMaxM);
  if (max_mt == 0) return;
  auto stream = at::cuda::getCurrentHIPStream();
  static const bool use_mfma = [] {
    const char* env = getenv("DYNAMO_MOE_MFMA");
    return env == nullptr || env[0] != '0';
  }();
  if (use_mfma && D % 32 == 0) {
    if (N >= 16384) {
      dim3 grid(E * max_mt, (N + 255) / 256);
      moe_gemm_mfma_kernel<4><<<grid, kBlock, 0, stream>>>(
          (short*)y.data_ptr(), (const short*)x.data_ptr(),
          (const short*)w.data_ptr(), nullptr, seg_start.data_ptr<int32_t>(),
          D, N, max_mt);
    } else {
      dim3 grid(E * max_mt, (N + 127) / 128);
      moe_gemm_mfma_kernel<2><<<grid, kBlock, 0, stream>>>(
          (short*)y.data_ptr(), (const short*)x.data_ptr(),
          (const short*)w.data_ptr(), nullptr, seg_start.data_ptr<int32_t>(),
          D, N, max_mt);
    }
  } else {
    dim3 grid(E * max_mt, (N + kBlock - 1) / kBlock);
    moe_gemm_kernel<<<grid, kBlock, 0, stream>>>(
        (short*)y.data_ptr(), (const short*)x.data_ptr(),
        (const short*)w.data_ptr(), nullptr, seg_start.data_ptr<int32_t>(),
        D, N, max_mt);
  }
  HIP_CHECK_KERNEL();
}

void topk_gating(torch::Tensor topw, torch::Tensor topi, torch::Tensor logits) {
  TORCH_CHECK(logits.is_cuda() && logits.dtype() == torch::kFloat32);
  const int T = logits.size(0);
  const int E = logits.size(1);
  const int K = topw.size(1);
  TORCH_CHECK(E <= 64 && K <= 8);
  if (T == 0) return;
  auto stream = at::cuda::getCurrentHIPStream();
  const int waves_per_block = 4;
  const int grid = (T + waves_per_block - 1) / waves_per_block;
  topk_gating_kernel<<<grid, waves_per_block * 64, 0, stream>>>(
      topw.data_ptr<float>(), topi.data_ptr<int32_t>(),
      logits.data_ptr<float>(), T, E, K);
  HIP_CHECK_KERNEL();
}
