// MFMA layout probe: computes one 16x16x32 bf16 MFMA with the fragment
// mappings the engine kernels assume, so a GPU test can verify them against
// a torch reference (guide §3 G9: asymmetric-input check — a transposed
// mapping is invisible to symmetric tests).
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;

__global__ void mfma_probe_kernel(float* __restrict__ D,      // [16,16]
                                  const short* __restrict__ A,  // [16,32] bf16
                                  const short* __restrict__ B)  // [32,16] bf16
{
  const int l = threadIdx.x;
  bf16x8_t a, b;
#pragma unroll
  for (int i = 0; i < 8; i++) {
    union { short s; __bf16 h; } ua, ub;
    ua.s = A[(l % 16) * 32 + 8 * (l / 16) + i];
    ub.s = B[(8 * (l / 16) + i) * 16 + (l % 16)];
    a[i] = ua.h;
    b[i] = ub.h;
  }
  f32x4 c{0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; r++) D[((l / 16) * 4 + r) * 16 + (l % 16)] = c[r];
}

__global__ void tr16_probe_kernel(int32_t* __restrict__ out) {  // [64, 8]
  __shared__ short lds[512];
  for (int i = threadIdx.x; i < 512; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  const int l = threadIdx.x & 63;
  const unsigned base = (unsigned)(unsigned long long)(void*)&lds[0];
  // production address pattern (attention_decode_impl.h MFMA PV path)
  const unsigned a = base + (l >> 4) * 128 + ((l & 15) >> 2) * 32 + (l & 3) * 2;
  unsigned long long v0, v1;
  asm volatile("ds_read_b64_tr_b16 %0, %2 offset:0\n\t"
               "ds_read_b64_tr_b16 %1, %2 offset:512"
               : "=v"(v0), "=v"(v1) : "v"(a) : "memory");
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
  if (threadIdx.x < 64) {
#pragma unroll
    for (int j = 0; j < 4; j++) {
      out[l * 8 + j] = (int32_t)((v0 >> (16 * j)) & 0xffff);
      out[l * 8 + 4 + j] = (int32_t)((v1 >> (16 * j)) & 0xffff);
    }
  }
}

// Replicates the MFMA decode kernel's V staging + transposed gather in
// isolation: input v[t][d] = t*200+d (int16), output [64 lanes][8 db][8 j]
// of gathered values. Expected: out[l][db][j] = (8*(l>>4)+j)*200 + db*16 + (l&15).
__global__ void vstage_probe_kernel(int32_t* __restrict__ out,
                                    const short* __restrict__ v) {  // [32,128]
  __shared__ short v_lds[4096];
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4;
  const int lr = lane & 15;
#pragma unroll
  for (int it = 0; it < 8; it++) {
    const int slot = lane + it * 64;
    const int d16 = slot & 15;
    const int tl = slot >> 4;
    short8 vv = *reinterpret_cast<const short8*>(v + tl * 128 + d16 * 8);
    *reinterpret_cast<short8*>(
        (char*)v_lds + tl * 256 + ((d16 * 16) ^ ((tl & 7) << 4))) = vv;
  }
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_sched_barrier(0);
#pragma unroll
  for (int db = 0; db < 8; db++) {
#pragma unroll
    for (int j = 0; j < 8; j++) {
      const int tok = 8 * lg + j;
      short val = *(const short*)(
          (const char*)v_lds + tok * 256 +
          (((db * 16 + lr) * 2) ^ ((tok & 7) << 4)));
      out[(lane * 8 + db) * 8 + j] = (int32_t)val;
    }
  }
}

// Full PV-path probe: stage patterned V (exact kernel layout), P = one-hot
// on token `hot`, run the PV mfma; out[row][dim] should equal V[hot][dim].
__global__ void pv_probe_kernel(float* __restrict__ out,   // [16, 128]
                                const short* __restrict__ v,  // [32,128] bf16
                                int hot) {
  typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8_t;
  __shared__ short v_lds[4096];
  __shared__ short p_lds[512];
  const int lane = threadIdx.x & 63;
  const int lg = lane >> 4;
  const int lr = lane & 15;
#pragma unroll
  for (int it = 0; it < 8; it++) {
    const int slot = lane + it * 64;
    const int d16 = slot & 15;
    const int tl = slot >> 4;
    short8 vv = *reinterpret_cast<const short8*>(v + tl * 128 + d16 * 8);
    *reinterpret_cast<short8*>(
        (char*)v_lds + tl * 256 + ((d16 * 16) ^ ((tl & 7) << 4))) = vv;
  }
  // P writes exactly like the kernel: rows lg*4+r, cols lr and 16+lr
#pragma unroll
  for (int r = 0; r < 4; r++) {
    const int row = lg * 4 + r;
    const int x = (row & 3) << 4;
    const float pA = (lr == hot) ? 1.f : 0.f;
    const float pB = (16 + lr == hot) ? 1.f : 0.f;
    *(short*)((char*)p_lds + row * 64 + ((lr * 2) ^ x)) = f32_to_bf16(pA);
    *(short*)((char*)p_lds + row * 64 + (((16 + lr) * 2) ^ x)) = f32_to_bf16(pB);
  }
  __syncthreads();
  short8 pa_s;
  {
    const int row = lr;
    const int x = (row & 3) << 4;
    pa_s = *reinterpret_cast<const short8*>(
        (char*)p_lds + row * 64 + ((lg * 16) ^ x));
  }
  bf16x8_t pa = *reinterpret_cast<bf16x8_t*>(&pa_s);
  f32x4 acc[8];
#pragma unroll
  for (int d = 0; d < 8; d++) acc[d] = f32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int db = 0; db < 8; db++) {
    short8 vb_s;
#pragma unroll
    for (int j = 0; j < 8; j++) {
      const int tok = 8 * lg + j;
      vb_s[j] = *(const short*)(
          (const char*)v_lds + tok * 256 +
          (((db * 16 + lr) * 2) ^ ((tok & 7) << 4)));
    }
    bf16x8_t vbf = *reinterpret_cast<bf16x8_t*>(&vb_s);
    acc[db] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pa, vbf, acc[db], 0, 0, 0);
  }
#pragma unroll
  for (int r = 0; r < 4; r++) {
    const int row = lg * 4 + r;
#pragma unroll
    for (int db = 0; db < 8; db++)
      out[row * 128 + db * 16 + lr] = acc[db][r];
  }
}

}  // namespace

torch::Tensor pv_probe(int64_t hot) {
  auto opts16 = torch::TensorOptions().dtype(torch::kBFloat16)
                    .device(torch::kCUDA, 0);
  auto v = (torch::arange(32 * 128, opts16.dtype(torch::kFloat32)
                                        .device(torch::kCUDA, 0))
                .view({32, 128}) / 1000.0).to(torch::kBFloat16);
  auto out = torch::empty({16, 128}, torch::TensorOptions()
                                         .dtype(torch::kFloat32)
                                         .device(torch::kCUDA, 0));
  auto stream = at::cuda::getCurrentHIPStream();
  pv_probe_kernel<<<1, 64, 0, stream>>>(out.data_ptr<float>(),
                                        (const short*)v.data_ptr(), (int)hot);
  HIP_CHECK_KERNEL();
  return out;
}

torch::Tensor vstage_probe() {
  auto opts = torch::TensorOptions().dtype(torch::kInt16)
                  .device(torch::kCUDA, 0);
  auto v = torch::empty({32, 128}, opts);
  {
    auto hv = torch::empty({32, 128}, torch::kInt16);
    auto acc = hv.accessor<int16_t, 2>();
    for (int t = 0; t < 32; t++)
      for (int d = 0; d < 128; d++) acc[t][d] = (int16_t)(t * 200 + d);
    v.copy_(hv);
  }
  auto out = torch::empty({64, 8, 8}, torch::TensorOptions()
                                          .dtype(torch::kInt32)
                                          .device(torch::kCUDA, 0));
  auto stream = at::cuda::getCurrentHIPStream();
  vstage_probe_kernel<<<1, 64, 0, stream>>>(out.data_ptr<int32_t>(),
                                            (const short*)v.data_ptr());
  HIP_CHECK_KERNEL();
  return out;
}

// Returns [64, 8]: the LDS bf16 element indices each lane's two
// ds_read_b64_tr_b16 reads delivered (lds was filled with iota), so tests
// can verify the transpose mapping the PV path assumes.
torch::Tensor tr16_probe() {
  auto out = torch::empty({64, 8}, torch::TensorOptions()
                                       .dtype(torch::kInt32)
                                       .device(torch::kCUDA, 0));
  auto stream = at::cuda::getCurrentHIPStream();
  tr16_probe_kernel<<<1, 64, 0, stream>>>(out.data_ptr<int32_t>());
  HIP_CHECK_KERNEL();
  return out;
}

torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kBFloat16);
  TORCH_CHECK(A.sizes() == torch::IntArrayRef({16, 32}));
  TORCH_CHECK(B.sizes() == torch::IntArrayRef({32, 16}));
  auto D = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentHIPStream();
  mfma_probe_kernel<<<1, 64, 0, stream>>>(
      D.data_ptr<float>(), (const short*)A.contiguous().data_ptr(),
      (const short*)B.contiguous().data_ptr());
  HIP_CHECK_KERNEL();
  return D;
}
