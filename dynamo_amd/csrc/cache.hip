// Paged KV-cache maintenance kernels.
//
// Cache layout per layer: k_cache/v_cache = [num_pages, Hkv, page_size, hd]
// bf16 — page rows per head are contiguous, which keeps both the decode
// attention gather and page-granular transfers coalesced.
//
// These are the MI355X-native equivalents of the reference's block-copy
// machinery (ai-dynamo/dynamo lib/llm/src/kernels/block_copy.cu:41-164 and
// lib/kvbm-kernels/cuda/tensor_kernels.cu:494-543): 16 B vector copies,
// 64-wide wavefronts, grid-stride.
#include "common.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

namespace {

constexpr int kBlock = 256;

template <int VT = 0>  // VT: vcache is d-major [P, Hkv, hd, ps]
__global__ void kv_append_kernel(short* __restrict__ kcache,
                                 short* __restrict__ vcache,
                                 const short* __restrict__ k,  // [T, Hkv*hd]
                                 const short* __restrict__ v,
                                 const int64_t* __restrict__ slots,  // [T]
                                 int T, int Hkv, int page_size, int hd) {
  const int t = blockIdx.x;
  if (t >= T) return;
  const int64_t slot = slots[t];
  if (slot < 0) return;  // padding token
  const int64_t page = slot / page_size;
  const int off = (int)(slot % page_size);
  const int nvec = Hkv * hd / 8;
  for (int i = threadIdx.x; i < nvec; i += kBlock) {
    const int h = (i * 8) / hd;
    const int d = (i * 8) % hd;
    const int64_t dst = (((page * Hkv + h) * page_size + off) * hd + d);
    *reinterpret_cast<short8*>(kcache + dst) =
        *reinterpret_cast<const short8*>(k + (int64_t)t * Hkv * hd + i * 8);
    const short8 vv =
        *reinterpret_cast<const short8*>(v + (int64_t)t * Hkv * hd + i * 8);
    if constexpr (VT) {
      const int64_t vb0 = ((page * Hkv + h) * hd + d) * page_size + off;
#pragma unroll
      for (int e = 0; e < 8; e++)
        vcache[vb0 + (int64_t)e * page_size] = vv[e];
    } else {
      *reinterpret_cast<short8*>(vcache + dst) = vv;
    }
  }
}

typedef __attribute__((ext_vector_type(4))) int int4v;

// Gather whole pages into a contiguous staging buffer (for xGMI transfer or
// host offload) or scatter them back. `page_elems` = Hkv*page_size*hd.
template <bool GATHER>
__global__ void page_copy_kernel(short* __restrict__ staging,  // [N, page_elems]
                                 short* __restrict__ cache,    // [P, page_elems]
                                 const int32_t* __restrict__ page_ids,  // [N]
                                 int64_t page_elems, int n_pages) {
  const int64_t nvec = page_elems / 8;
  for (int p = blockIdx.y; p < n_pages; p += gridDim.y) {
    const int64_t page = page_ids[p];
    short* st = staging + (int64_t)p * page_elems;
    short* ca = cache + page * page_elems;
    for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < nvec;
         i += (int64_t)gridDim.x * kBlock) {
      if constexpr (GATHER)
        *reinterpret_cast<int4v*>(st + i * 8) = *reinterpret_cast<const int4v*>(ca + i * 8);
      else
        *reinterpret_cast<int4v*>(ca + i * 8) = *reinterpret_cast<const int4v*>(st + i * 8);
    }
  }
}

// Direct cache-to-cache page copy by (src,dst) id pairs (intra-GPU COW /
// defrag, and peer-mapped transfer when both pools are visible).
__global__ void page_pair_copy_kernel(short* __restrict__ dst_cache,
                                      const short* __restrict__ src_cache,
                                      const int32_t* __restrict__ pairs,  // [N,2]
                                      int64_t page_elems, int n_pairs) {
  const int64_t nvec = page_elems / 8;
  for (int p = blockIdx.y; p < n_pairs; p += gridDim.y) {
    const short* s = src_cache + (int64_t)pairs[2 * p] * page_elems;
    short* d = dst_cache + (int64_t)pairs[2 * p + 1] * page_elems;
    for (int64_t i = (int64_t)blockIdx.x * kBlock + threadIdx.x; i < nvec;
         i += (int64_t)gridDim.x * kBlock) {
      *reinterpret_cast<int4v*>(d + i * 8) = *reinterpret_cast<const int4v*>(s + i * 8);
    }
  }
}

}  // namespace

void kv_cache_append(torch::Tensor kcache, torch::Tensor vcache,
                     torch::Tensor k, torch::Tensor v,
                     torch::Tensor slot_mapping, bool v_transposed) {
  TORCH_CHECK(kcache.is_cuda() && kcache.dtype() == torch::kBFloat16);
  TORCH_CHECK(slot_mapping.dtype() == torch::kInt64);
  const int Hkv = kcache.size(1);
  const int page_size = kcache.size(2);
  const int hd = kcache.size(3);
  const int T = slot_mapping.size(0);
  TORCH_CHECK(hd % 8 == 0);
  if (T == 0) return;
  auto stream = at::cuda::getCurrentHIPStream();
  auto launch = [&](auto kern) {
    kern<<<T, kBlock, 0, stream>>>(
        (short*)kcache.data_ptr(), (short*)vcache.data_ptr(),
        (const short*)k.data_ptr(), (const short*)v.data_ptr(),
        slot_mapping.data_ptr<int64_t>(), T, Hkv, page_size, hd);
  };
  if (v_transposed) launch(kv_append_kernel<1>);
  else launch(kv_append_kernel<0>);
  HIP_CHECK_KERNEL();
}

static void pages_launch_dims(int64_t page_elems, int n, dim3& grid) {
  int gx = (int)std::min<int64_t>((page_elems / 8 + kBlock - 1) / kBlock, 64);
  int gy = std::min(n, 4096);
  grid = dim3(gx, gy);
}

void gather_pages(torch::Tensor staging, torch::Tensor cache, torch::Tensor page_ids) {
  TORCH_CHECK(staging.is_cuda() && cache.is_cuda());
  TORCH_CHECK(page_ids.dtype() == torch::kInt32);
  // elem-size-agnostic: kernels copy in 2-byte units (fp8 caches pass
  // half as many "short elems" per page)
  const int64_t page_elems =
      cache.numel() / cache.size(0) * cache.element_size() / 2;
  const int n = page_ids.size(0);
  TORCH_CHECK(staging.numel() * staging.element_size() >= n * page_elems * 2);
  if (n == 0) return;
  dim3 grid; pages_launch_dims(page_elems, n, grid);
  auto stream = at::cuda::getCurrentHIPStream();
  page_copy_kernel<true><<<grid, kBlock, 0, stream>>>(
      (short*)staging.data_ptr(), (short*)cache.data_ptr(),
      page_ids.data_ptr<int32_t>(), page_elems, n);
  HIP_CHECK_KERNEL();
}

void scatter_pages(torch::Tensor staging, torch::Tensor cache, torch::Tensor page_ids) {
  TORCH_CHECK(staging.is_cuda() && cache.is_cuda());
  TORCH_CHECK(page_ids.dtype() == torch::kInt32);
  // elem-size-agnostic: kernels copy in 2-byte units (fp8 caches pass
  // half as many "short elems" per page)
  const int64_t page_elems =
      cache.numel() / cache.size(0) * cache.element_size() / 2;
  const int n = page_ids.size(0);
  if (n == 0) return;
  dim3 grid; pages_launch_dims(page_elems, n, grid);
  auto stream = at::cuda::getCurrentHIPStream();
  page_copy_kernel<false><<<grid, kBlock, 0, stream>>>(
      (short*)staging.data_ptr(), (short*)cache.data_ptr(),
      page_ids.data_ptr<int32_t>(), page_elems, n);
  HIP_CHECK_KERNEL();
}

void copy_pages(torch::Tensor dst_cache, torch::Tensor src_cache, torch::Tensor pairs) {
  TORCH_CHECK(pairs.dtype() == torch::kInt32 && pairs.dim() == 2 && pairs.size(1) == 2);
  const int64_t page_elems =
      src_cache.numel() / src_cache.size(0) * src_cache.element_size() / 2;
  const int n = pairs.size(0);
  if (n == 0) return;
  dim3 grid; pages_launch_dims(page_elems, n, grid);
  auto stream = at::cuda::getCurrentHIPStream();
  page_pair_copy_kernel<<<grid, kBlock, 0, stream>>>(
      (short*)dst_cache.data_ptr(), (const short*)src_cache.data_ptr(),
      pairs.data_ptr<int32_t>(), page_elems, n);
  HIP_CHECK_KERNEL();
}
