// Torch bindings + production template choices for paged-attention decode.
// Kernel implementation: attention_decode_impl.h (shared with the
// standalone sweep tool benchmarks/decode_sweep.hip, which picked these
// (DP, HS, DEPTH) combos on MI355X hardware).
#include "attention_decode_impl.h"
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

using namespace decode_attn;

int64_t paged_decode_num_chunks(int64_t max_ctx) {
  const int chunk = decode_chunk_tokens((int)max_ctx);
  return std::max<int64_t>(1, (max_ctx + chunk - 1) / chunk);
}

int64_t decode_chunk_tokens_py(int64_t max_ctx) {
  return decode_chunk_tokens((int)max_ctx);
}

void paged_attention_decode(torch::Tensor out, torch::Tensor q,
                            torch::Tensor kcache, torch::Tensor vcache,
                            torch::Tensor page_table, torch::Tensor ctx_lens,
                            torch::Tensor partial, torch::Tensor ml,
                            double scale, int64_t chunk_tokens,
                            bool v_transposed,
                            c10::optional<torch::Tensor> chunk_cnt) {
  TORCH_CHECK(q.is_cuda() && q.dtype() == torch::kBFloat16);
  TORCH_CHECK(page_table.dtype() == torch::kInt32 && ctx_lens.dtype() == torch::kInt32);
  const bool fp8 = kcache.dtype() == torch::kFloat8_e4m3fn;
  TORCH_CHECK(fp8 || kcache.dtype() == torch::kBFloat16,
              "kv cache must be bf16 or float8_e4m3fn");
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
  const int chunk = (int)chunk_tokens;
  TORCH_CHECK(chunk >= 128 && chunk % 128 == 0, "bad chunk_tokens ", chunk);
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid(B, Hkv, C);

  if (v_transposed) {
    // d-major V pages: the 64-token-tile swapped kernel (VT=3) is the
    // production consumer (benchmarks/decode_sweep.hip ladder)
    TORCH_CHECK(vcache.size(2) == hd && vcache.size(3) == ps,
                "v_transposed expects vcache [P, Hkv, hd, ps]");
    TORCH_CHECK(ps % 32 == 0 && hd == 128 && G >= 2 && G <= 16,
                "v_transposed decode needs page_size%32==0, head_dim==128, "
                "2<=G<=16 (got G=", G, ")");
    // VT4 stages K through the per-wave LDS region (reuses the V-staging
    // space of the token-major layout: 8 KB/wave of the VS=72 slab)
    const int lds = mfma_swapped_lds_bytes(G, hd, 72);
    // fused chunk merge (DYNAMO_FUSED_MERGE=1): MEASURED NET NEGATIVE on
    // the flagship (38.75 -> 47.35 ms/step): the two __threadfence()s per
    // block cost ~6 ms/step across 640 blocks x 80 layers, an order of
    // magnitude more than the 5.6 us phase2 launches they replace. Kept
    // behind the env flag as a recorded negative result.
    static const bool use_fused_merge = [] {
      const char* e = getenv("DYNAMO_FUSED_MERGE");
      return e != nullptr && e[0] == '1';
    }();
    int32_t* cnt = nullptr;
    if (use_fused_merge && C > 1 && chunk_cnt.has_value() &&
        chunk_cnt->numel() >= (int64_t)B * Hkv)
      cnt = chunk_cnt->data_ptr<int32_t>();
    auto launch_vt = [&](auto* kern) {
      if (lds > 65536)
        (void)hipFuncSetAttribute(reinterpret_cast<const void*>(kern),
                                  hipFuncAttributeMaxDynamicSharedMemorySize,
                                  lds);
      kern<<<grid, kBlock, lds, stream>>>(
          partial.data_ptr<float>(), ml.data_ptr<float>(),
          (short*)out.data_ptr(), (const short*)q.data_ptr(),
          (const short*)kcache.data_ptr(), (const short*)vcache.data_ptr(),
          page_table.data_ptr<int32_t>(), ctx_lens.data_ptr<int32_t>(),
          (float)scale, chunk, G, B, Hkv, C, max_pages, log2_ps, hd, cnt);
    };
    // VT4 = 64-token tiles + K staged via per-wave LDS in contiguous
    // 1KB bursts (sweep same-box: 5357 GB/s vs 5150 VT3 / 4922 VT2);
    // DYNAMO_VT3_VARIANT selects the bring-up fallbacks
    static const int vt3v = [] {
      const char* e = getenv("DYNAMO_VT3_VARIANT");
      return e ? atoi(e) : 0;
    }();
    if (fp8)
      launch_vt(&paged_decode_mfma_swapped<1, 1, 0, 1, 72, 0, 4>);
    else if (vt3v == 1)
      launch_vt(&paged_decode_mfma_swapped<1, 1, 0, 0, 72, 0, 3, 0, 2>);
    else if (vt3v == 2)
      launch_vt(&paged_decode_mfma_swapped<1, 1, 0, 0, 72, 0, 2>);
    else if (vt3v == 5 && ps == 64)
      // VT5 (V through LDS too): measured NEGATIVE once the sweep's LDS
      // under-allocation was fixed - 4266 vs 5222 GB/s: PV serializes on
      // the V stage. Kept selectable for the record.
      launch_vt(&paged_decode_mfma_swapped<1, 1, 0, 0, 72, 0, 5>);
    else
      launch_vt(&paged_decode_mfma_swapped<1, 1, 0, 0, 72, 0, 4>);
    HIP_CHECK_KERNEL();
    if (C > 1 && cnt == nullptr) {
      dim3 grid2(B, Hq);
      paged_decode_phase2<<<grid2, 128, 0, stream>>>(
          (short*)out.data_ptr(), partial.data_ptr<float>(),
          ml.data_ptr<float>(), ctx_lens.data_ptr<int32_t>(), chunk, Hq, C,
          hd);
      HIP_CHECK_KERNEL();
    }
    return;
  }

#define LAUNCH_G(GG, DP, HS, DEPTH)                                           \
  paged_decode_phase1<GG, DP, HS, DEPTH>                                      \
      <<<grid, kBlock, phase1_lds_bytes(GG, HS, hd), stream>>>(               \
      partial.data_ptr<float>(), ml.data_ptr<float>(), (short*)out.data_ptr(),\
      (const short*)q.data_ptr(), (const short*)kcache.data_ptr(),            \
      (const short*)vcache.data_ptr(), page_table.data_ptr<int32_t>(),        \
      ctx_lens.data_ptr<int32_t>(), (float)scale, chunk, B, Hkv, C,            \
      max_pages, log2_ps, hd)
  static const bool use_mfma = [] {
    const char* e = getenv("DYNAMO_DECODE_MFMA");
    return e == nullptr || e[0] != '0';  // default ON (sweep-verified)
  }();
  // swapped-operand variant (tokens on MFMA M, in-register P exchange via
  // permlane swaps): sweep-verified bit-match + faster at every measured
  // (G, chunk); DYNAMO_DECODE_SWAPPED=0 falls back to the A-variant
  static const bool use_swapped = [] {
    const char* e = getenv("DYNAMO_DECODE_SWAPPED");
    return e == nullptr || e[0] != '0';
  }();
#define LAUNCH_MFMA(GG)                                                       \
  do {                                                                        \
    if (use_swapped) {                                                        \
      if (mfma_swapped_lds_bytes(GG, hd, 72) > 65536)                             \
        (void)hipFuncSetAttribute(                                            \
            reinterpret_cast<const void*>(                                    \
                &paged_decode_mfma_swapped<1, 1, 0, 0, 72, 1>),                         \
            hipFuncAttributeMaxDynamicSharedMemorySize,                       \
            mfma_swapped_lds_bytes(GG, hd, 72));                                  \
      paged_decode_mfma_swapped<1, 1, 0, 0, 72, 1>                                      \
          <<<grid, kBlock, mfma_swapped_lds_bytes(GG, hd, 72), stream>>>(         \
          partial.data_ptr<float>(), ml.data_ptr<float>(),                    \
          (short*)out.data_ptr(), (const short*)q.data_ptr(),                 \
          (const short*)kcache.data_ptr(),                                    \
          (const short*)vcache.data_ptr(), page_table.data_ptr<int32_t>(),    \
          ctx_lens.data_ptr<int32_t>(), (float)scale, chunk, GG, B, Hkv, C,    \
          max_pages, log2_ps, hd, nullptr);                                   \
      break;                                                                  \
    }                                                                         \
    if (mfma_lds_bytes(GG, hd) > 65536)                                       \
      (void)hipFuncSetAttribute(                                              \
          reinterpret_cast<const void*>(&paged_decode_mfma<1, 1>),            \
          hipFuncAttributeMaxDynamicSharedMemorySize,                         \
          mfma_lds_bytes(GG, hd));                                            \
    paged_decode_mfma<1, 1><<<grid, kBlock, mfma_lds_bytes(GG, hd), stream>>>(\
      partial.data_ptr<float>(), ml.data_ptr<float>(), (short*)out.data_ptr(),\
      (const short*)q.data_ptr(), (const short*)kcache.data_ptr(),            \
      (const short*)vcache.data_ptr(), page_table.data_ptr<int32_t>(),        \
      ctx_lens.data_ptr<int32_t>(), (float)scale, chunk, GG, B, Hkv, C,        \
      max_pages, log2_ps, hd);                                                \
  } while (0)
  const bool mfma_ok = use_mfma && ps % 32 == 0 && hd == 128;
  if (fp8) {
    // fp8 KV: the runtime-G swapped kernel is the ONLY fp8 consumer
    TORCH_CHECK(ps % 32 == 0 && hd == 128 && G <= 16,
                "fp8 KV cache needs page_size%32==0, head_dim==128, G<=16");
    if (mfma_swapped_lds_bytes(G, hd, 72) > 65536)
      (void)hipFuncSetAttribute(
          reinterpret_cast<const void*>(&paged_decode_mfma_swapped<1, 1, 0, 1, 72, 1>),
          hipFuncAttributeMaxDynamicSharedMemorySize,
          mfma_swapped_lds_bytes(G, hd, 72));
    paged_decode_mfma_swapped<1, 1, 0, 1, 72, 1>
        <<<grid, kBlock, mfma_swapped_lds_bytes(G, hd, 72), stream>>>(
        partial.data_ptr<float>(), ml.data_ptr<float>(),
        (short*)out.data_ptr(), (const short*)q.data_ptr(),
        (const short*)kcache.data_ptr(), (const short*)vcache.data_ptr(),
        page_table.data_ptr<int32_t>(), ctx_lens.data_ptr<int32_t>(),
        (float)scale, chunk, G, B, Hkv, C, max_pages, log2_ps, hd, nullptr);
    HIP_CHECK_KERNEL();
    if (C > 1) {
      dim3 grid2(B, Hq);
      paged_decode_phase2<<<grid2, 128, 0, stream>>>(
          (short*)out.data_ptr(), partial.data_ptr<float>(),
          ml.data_ptr<float>(), ctx_lens.data_ptr<int32_t>(), chunk, Hq, C,
          hd);
      HIP_CHECK_KERNEL();
    }
    return;
  }
  switch (G) {  // combos picked by benchmarks/decode_sweep on MI355X
    case 1: LAUNCH_G(1, 8, 1, 2); break;
    case 2: LAUNCH_G(2, 8, 2, 2); break;
    case 4:
      // sweep: VALU DP16/HS2/D4 streams 3951 GB/s vs 3348 for the MFMA
      // variant at G=4 (occupancy beats MFMA here) — VALU is production
      LAUNCH_G(4, 16, 2, 4);
      break;
    case 8:
      if (mfma_ok) { LAUNCH_MFMA(8); } else { LAUNCH_G(8, 8, 2, 2); }
      break;
    case 16:
      TORCH_CHECK(mfma_ok, "G=16 requires the MFMA decode path");
      LAUNCH_MFMA(16);
      break;
    default:
      // odd groups (e.g. qwen2 G=7): the runtime-G MFMA kernel is the ONLY
      // handler, so the DYNAMO_DECODE_MFMA preference toggle is ignored
      TORCH_CHECK(ps % 32 == 0 && hd == 128 && G <= 16,
                  "GQA group ", G, " needs the MFMA decode path "
                  "(page_size%32==0, head_dim==128)");
      LAUNCH_MFMA(G);
      break;
  }
#undef LAUNCH_G
#undef LAUNCH_MFMA
  HIP_CHECK_KERNEL();
  if (C > 1) {
    dim3 grid2(B, Hq);
    paged_decode_phase2<<<grid2, 128, 0, stream>>>(
        (short*)out.data_ptr(), partial.data_ptr<float>(), ml.data_ptr<float>(),
        ctx_lens.data_ptr<int32_t>(), chunk, Hq, C, hd);
    HIP_CHECK_KERNEL();
  }
}
