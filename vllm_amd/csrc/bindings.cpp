// torch.library bindings for the vllm_amd CDNA4 kernels.
//
// Loaded via torch.ops.load_library("vllm_amd/_C.so"); ops appear as
// torch.ops.vllm_amd.<name>. No Python C API — pure libtorch, same
// pattern as the reference's csrc/libtorch_stable/torch_bindings.cpp.

#include <torch/all.h>
#include <torch/library.h>

namespace vllm_amd {

torch::Tensor rms_norm(torch::Tensor x, torch::Tensor weight, double eps);
void fused_add_rms_norm(torch::Tensor x, torch::Tensor residual,
                        torch::Tensor weight, double eps);
torch::Tensor silu_and_mul(torch::Tensor x);
torch::Tensor gelu_and_mul(torch::Tensor x);
void rotary_embedding(torch::Tensor positions, torch::Tensor q,
                      torch::Tensor k, torch::Tensor cos_sin_cache,
                      int64_t rot);
void reshape_and_cache(torch::Tensor key, torch::Tensor value,
                       torch::Tensor kv_cache, torch::Tensor slot_mapping);
void paged_decode_attention(torch::Tensor out, torch::Tensor q,
                            torch::Tensor kv_cache,
                            torch::Tensor block_table,
                            torch::Tensor seq_lens, double scale,
                            int64_t max_seq_len, int64_t sliding_window,
                            torch::Tensor tmp_out, torch::Tensor tmp_lse);
void prefill_attention(torch::Tensor out, torch::Tensor q,
                       torch::Tensor kv_cache, torch::Tensor block_table,
                       torch::Tensor query_start_loc, torch::Tensor seq_lens,
                       double scale, int64_t num_decodes,
                       int64_t max_query_len, int64_t sliding_window);
torch::Tensor lt_linear(torch::Tensor a, torch::Tensor w,
                        c10::optional<torch::Tensor> bias);
torch::Tensor lt_linear_fp8(torch::Tensor a, torch::Tensor w);
void dynamic_quant_fp8(torch::Tensor out, torch::Tensor scales,
                       torch::Tensor x);
void scale_rows_cols(torch::Tensor y, torch::Tensor row_scales,
                     torch::Tensor col_scales,
                     c10::optional<torch::Tensor> bias);
void moe_align(torch::Tensor topk_ids, int64_t num_experts,
               torch::Tensor sorted, torch::Tensor expert_tiles,
               torch::Tensor inv_perm, torch::Tensor counts,
               torch::Tensor fill, torch::Tensor off);
void moe_gemm(torch::Tensor a, torch::Tensor b, torch::Tensor c,
              torch::Tensor sorted, torch::Tensor expert_tiles,
              int64_t topk_div, int64_t total_flat);
void moe_combine(torch::Tensor out, torch::Tensor y,
                 torch::Tensor topk_weights, torch::Tensor inv_perm);
void moe_gemm_shuf(torch::Tensor a, torch::Tensor b_shuf, torch::Tensor c,
                   torch::Tensor sorted, torch::Tensor expert_tiles,
                   int64_t n, int64_t k, int64_t topk_div,
                   int64_t total_flat);
void mla_decode(torch::Tensor out, torch::Tensor q_nope, torch::Tensor q_pe,
                torch::Tensor kv_cache, torch::Tensor block_table,
                torch::Tensor seq_lens, double scale, int64_t max_seq_len,
                torch::Tensor tmp_out, torch::Tensor tmp_lse);
torch::Tensor car_init(int64_t rank, int64_t world, int64_t max_bytes);
void car_connect(torch::Tensor handles);
bool car_is_ready();
int64_t car_max_bytes();
int64_t car_error();
void car_destroy();
void car_all_reduce(torch::Tensor t);
void car_all_gather(torch::Tensor out, torch::Tensor t);
void car_reduce_scatter(torch::Tensor out, torch::Tensor t);

}  // namespace vllm_amd

TORCH_LIBRARY(vllm_amd, m) {
  m.def("rms_norm(Tensor x, Tensor weight, float eps) -> Tensor");
  m.def("fused_add_rms_norm(Tensor(a!) x, Tensor(b!) residual, "
        "Tensor weight, float eps) -> ()");
  m.def("silu_and_mul(Tensor x) -> Tensor");
  m.def("gelu_and_mul(Tensor x) -> Tensor");
  m.def("rotary_embedding(Tensor positions, Tensor(a!) q, Tensor(b!) k, "
        "Tensor cos_sin_cache, int rot) -> ()");
  m.def("reshape_and_cache(Tensor key, Tensor value, Tensor(a!) kv_cache, "
        "Tensor slot_mapping) -> ()");
  m.def("paged_decode_attention(Tensor(a!) out, Tensor q, Tensor kv_cache, "
        "Tensor block_table, Tensor seq_lens, float scale, int max_seq_len, "
        "int sliding_window, Tensor(b!) tmp_out, Tensor(c!) tmp_lse) -> ()");
  m.def("prefill_attention(Tensor(a!) out, Tensor q, Tensor kv_cache, "
        "Tensor block_table, Tensor query_start_loc, Tensor seq_lens, "
        "float scale, int num_decodes, int max_query_len, "
        "int sliding_window) -> ()");
  m.def("lt_linear(Tensor a, Tensor w, Tensor? bias) -> Tensor");
  m.def("lt_linear_fp8(Tensor a, Tensor w) -> Tensor");
  m.def("dynamic_quant_fp8(Tensor(a!) out, Tensor(b!) scales, Tensor x)"
        " -> ()");
  m.def("scale_rows_cols(Tensor(a!) y, Tensor row_scales, Tensor col_scales,"
        " Tensor? bias) -> ()");
  m.def("moe_align(Tensor topk_ids, int num_experts, Tensor(a!) sorted, "
        "Tensor(b!) expert_tiles, Tensor(c!) inv_perm, Tensor(d!) counts, "
        "Tensor(e!) fill, Tensor(f!) off) -> ()");
  m.def("moe_gemm(Tensor a, Tensor b, Tensor(a!) c, Tensor sorted, "
        "Tensor expert_tiles, int topk_div, int total_flat) -> ()");
  m.def("moe_combine(Tensor(a!) out, Tensor y, Tensor topk_weights, "
        "Tensor inv_perm) -> ()");
  m.def("moe_gemm_shuf(Tensor a, Tensor b_shuf, Tensor(a!) c, "
        "Tensor sorted, Tensor expert_tiles, int n, int k, int topk_div, "
        "int total_flat) -> ()");
  m.def("mla_decode(Tensor(a!) out, Tensor q_nope, Tensor q_pe, "
        "Tensor kv_cache, Tensor block_table, Tensor seq_lens, "
        "float scale, int max_seq_len, Tensor(b!) tmp_out, "
        "Tensor(c!) tmp_lse) -> ()");
  // Custom xGMI collectives (comms.hip). init/connect/destroy are
  // host-side control ops and live on the catch-all dispatch below.
  m.def("car_init(int rank, int world, int max_bytes) -> Tensor");
  m.def("car_connect(Tensor handles) -> ()");
  m.def("car_is_ready() -> bool");
  m.def("car_max_bytes() -> int");
  m.def("car_error() -> int");
  m.def("car_destroy() -> ()");
  m.def("car_all_reduce(Tensor(a!) t) -> ()");
  m.def("car_all_gather(Tensor(a!) out, Tensor t) -> ()");
  m.def("car_reduce_scatter(Tensor(a!) out, Tensor t) -> ()");
  m.impl("car_init", &vllm_amd::car_init);
  m.impl("car_connect", &vllm_amd::car_connect);
  m.impl("car_is_ready", &vllm_amd::car_is_ready);
  m.impl("car_max_bytes", &vllm_amd::car_max_bytes);
  m.impl("car_error", &vllm_amd::car_error);
  m.impl("car_destroy", &vllm_amd::car_destroy);
}

TORCH_LIBRARY_IMPL(vllm_amd, CUDA, m) {
  m.impl("rms_norm", &vllm_amd::rms_norm);
  m.impl("fused_add_rms_norm", &vllm_amd::fused_add_rms_norm);
  m.impl("silu_and_mul", &vllm_amd::silu_and_mul);
  m.impl("gelu_and_mul", &vllm_amd::gelu_and_mul);
  m.impl("rotary_embedding", &vllm_amd::rotary_embedding);
  m.impl("reshape_and_cache", &vllm_amd::reshape_and_cache);
  m.impl("paged_decode_attention", &vllm_amd::paged_decode_attention);
  m.impl("prefill_attention", &vllm_amd::prefill_attention);
  m.impl("lt_linear", &vllm_amd::lt_linear);
  m.impl("lt_linear_fp8", &vllm_amd::lt_linear_fp8);
  m.impl("dynamic_quant_fp8", &vllm_amd::dynamic_quant_fp8);
  m.impl("scale_rows_cols", &vllm_amd::scale_rows_cols);
  m.impl("car_all_reduce", &vllm_amd::car_all_reduce);
  m.impl("car_all_gather", &vllm_amd::car_all_gather);
  m.impl("car_reduce_scatter", &vllm_amd::car_reduce_scatter);
  m.impl("moe_align", &vllm_amd::moe_align);
  m.impl("moe_gemm", &vllm_amd::moe_gemm);
  m.impl("moe_gemm_shuf", &vllm_amd::moe_gemm_shuf);
  m.impl("moe_combine", &vllm_amd::moe_combine);
  m.impl("mla_decode", &vllm_amd::mla_decode);
}
