// Python bindings for the dynamo_amd native CDNA4 kernels (dynamo_amd._hip).
#include <torch/extension.h>
#include <pybind11/pybind11.h>

namespace py = pybind11;

// norm.hip
void rmsnorm(torch::Tensor out, torch::Tensor input, torch::Tensor weight, double eps);
void fused_add_rmsnorm(torch::Tensor input, torch::Tensor residual,
                       torch::Tensor weight, double eps);
// rope.hip
void rope_append_qkv(torch::Tensor q_out, torch::Tensor kcache,
                     torch::Tensor vcache, torch::Tensor qkv,
                     c10::optional<torch::Tensor> bias,
                     torch::Tensor positions, torch::Tensor slot_mapping,
                     torch::Tensor cos_sin_cache, bool v_transposed);
void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
                  torch::Tensor cos_sin_cache, int64_t num_q_heads,
                  int64_t num_k_heads, int64_t head_dim);
// activation.hip
void silu_mul(torch::Tensor out, torch::Tensor gate_up);
void gelu(torch::Tensor out, torch::Tensor input);
// cache.hip
void kv_cache_append(torch::Tensor kcache, torch::Tensor vcache, torch::Tensor k,
                     torch::Tensor v, torch::Tensor slot_mapping,
                     bool v_transposed);
void gather_pages(torch::Tensor staging, torch::Tensor cache, torch::Tensor page_ids);
void scatter_pages(torch::Tensor staging, torch::Tensor cache, torch::Tensor page_ids);
void copy_pages(torch::Tensor dst_cache, torch::Tensor src_cache, torch::Tensor pairs);
// attention_decode.hip
int64_t paged_decode_num_chunks(int64_t max_ctx);
int64_t decode_chunk_tokens_py(int64_t max_ctx);
void paged_attention_decode(torch::Tensor out, torch::Tensor q, torch::Tensor kcache,
                            torch::Tensor vcache, torch::Tensor page_table,
                            torch::Tensor ctx_lens, torch::Tensor partial,
                            torch::Tensor ml, double scale,
                            int64_t chunk_tokens, bool v_transposed,
                            c10::optional<torch::Tensor> chunk_cnt);
// attention_prefill.hip
void attention_prefill_paged(torch::Tensor out, torch::Tensor q,
                             torch::Tensor kcache, torch::Tensor vcache,
                             torch::Tensor page_table, torch::Tensor tile_seq,
                             torch::Tensor tile_q0, torch::Tensor seq_q_start,
                             torch::Tensor seq_q_len, torch::Tensor seq_ctx_len,
                             double scale, bool v_transposed);
// sampling.hip
void greedy_sample(torch::Tensor out, torch::Tensor logits);
void topkp_sample(torch::Tensor out, torch::Tensor logits,
                  torch::Tensor inv_temp, torch::Tensor top_k,
                  torch::Tensor top_p, int64_t seed,
                  c10::optional<torch::Tensor> row_seeds);
void gumbel_sample(torch::Tensor out, torch::Tensor logits, torch::Tensor inv_temp,
                   int64_t seed, c10::optional<torch::Tensor> row_seeds);
// moe.hip
void moe_grouped_gemm(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                      torch::Tensor tiles);
void moe_grouped_gemm_seg(torch::Tensor y, torch::Tensor x, torch::Tensor w,
                          torch::Tensor seg_start, int64_t max_tokens);
void topk_gating(torch::Tensor topw, torch::Tensor topi, torch::Tensor logits);
// ipc.hip
torch::Tensor ipc_alloc(int64_t nbytes, int64_t device);
py::bytes ipc_export(torch::Tensor t);
torch::Tensor ipc_open(py::bytes handle_bytes, int64_t nbytes, int64_t device);
void enable_peer_access(int64_t device, int64_t peer);
// probe.hip
torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B);
torch::Tensor tr16_probe();
torch::Tensor vstage_probe();
torch::Tensor pv_probe(int64_t hot);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "dynamo_amd native MI355X (gfx950) kernels";
  m.def("rmsnorm", &rmsnorm);
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm);
  m.def("rope_inplace", &rope_inplace);
  m.def("rope_append_qkv", &rope_append_qkv,
        py::arg("q_out"), py::arg("kcache"), py::arg("vcache"),
        py::arg("qkv"), py::arg("bias"), py::arg("positions"),
        py::arg("slot_mapping"), py::arg("cos_sin"),
        py::arg("v_transposed") = false);
  m.def("silu_mul", &silu_mul);
  m.def("gelu", &gelu);
  m.def("kv_cache_append", &kv_cache_append, py::arg("kcache"),
        py::arg("vcache"), py::arg("k"), py::arg("v"),
        py::arg("slot_mapping"), py::arg("v_transposed") = false);
  m.def("gather_pages", &gather_pages);
  m.def("scatter_pages", &scatter_pages);
  m.def("copy_pages", &copy_pages);
  m.def("paged_decode_num_chunks", &paged_decode_num_chunks);
  m.def("decode_chunk_tokens", &decode_chunk_tokens_py);
  m.def("paged_attention_decode", &paged_attention_decode,
        py::arg("out"), py::arg("q"), py::arg("kcache"), py::arg("vcache"),
        py::arg("page_table"), py::arg("ctx_lens"), py::arg("partial"),
        py::arg("ml"), py::arg("scale"), py::arg("chunk_tokens"),
        py::arg("v_transposed") = false,
        py::arg("chunk_cnt") = py::none());
  m.def("attention_prefill_paged", &attention_prefill_paged,
        py::arg("out"), py::arg("q"), py::arg("kcache"), py::arg("vcache"),
        py::arg("page_table"), py::arg("tile_seq"), py::arg("tile_q0"),
        py::arg("seq_q_start"), py::arg("seq_q_len"), py::arg("seq_ctx_len"),
        py::arg("scale"), py::arg("v_transposed") = false);
  m.def("greedy_sample", &greedy_sample);
  m.def("gumbel_sample", &gumbel_sample, py::arg("out"), py::arg("logits"),
        py::arg("inv_temp"), py::arg("seed"),
        py::arg("row_seeds") = py::none());
  m.def("topkp_sample", &topkp_sample, py::arg("out"), py::arg("logits"),
        py::arg("inv_temp"), py::arg("top_k"), py::arg("top_p"),
        py::arg("seed"), py::arg("row_seeds") = py::none());
  m.def("moe_grouped_gemm", &moe_grouped_gemm);
  m.def("moe_grouped_gemm_seg", &moe_grouped_gemm_seg);
  m.def("topk_gating", &topk_gating);
  m.def("ipc_alloc", &ipc_alloc);
  m.def("ipc_export", &ipc_export);
  m.def("ipc_open", &ipc_open);
  m.def("enable_peer_access", &enable_peer_access);
  m.def("mfma_probe", &mfma_probe);
  m.def("tr16_probe", &tr16_probe);
  m.def("vstage_probe", &vstage_probe);
  m.def("pv_probe", &pv_probe);
}
