#include "hip/hip_runtime.h"
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

}  // namespace

// Returns [64, 8]: the LDS bf16 element indices each lane's two
// ds_read_b64_tr_b16 reads delivered (lds was filled with iota), so tests
// can verify the transpose mapping the PV path assumes.
torch::Tensor tr16_probe() {
  auto out = torch::empty({64, 8}, torch::TensorOptions()
                                       .dtype(torch::kInt32)
                                       .device(torch::kCUDA, 0));
  auto stream = at::cuda::getCurrentHIPStream();
 hipLaunchKernelGGL(( tr16_probe_kernel), dim3(1), dim3(64), 0, stream, out.data_ptr<int32_t>());
  HIP_CHECK_KERNEL();
  return out;
}

torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kBFloat16);
  TORCH_CHECK(A.sizes() == torch::IntArrayRef({16, 32}));
  TORCH_CHECK(B.sizes() == torch::IntArrayRef({32, 16}));
  auto D = torch::empty({16, 16}, A.options().dtype(torch::kFloat32));
  auto stream = at::cuda::getCurrentHIPStream();
 hipLaunchKernelGGL(( mfma_probe_kernel), dim3(1), dim3(64), 0, stream, 
      D.data_ptr<float>(), (const short*)A.contiguous().data_ptr(),
      (const short*)B.contiguous().data_ptr());
  HIP_CHECK_KERNEL();
  return D;
}
