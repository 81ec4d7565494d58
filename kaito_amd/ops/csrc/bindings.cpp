// torch op registration for kaito_amd's gfx950 kernels.
// Loaded via torch.ops.load_library; no pybind dependency.
#include <torch/library.h>
#include <ATen/ATen.h>

namespace kaito {
void rms_norm(at::Tensor out, at::Tensor input, at::Tensor weight, double eps);
void fused_add_rms_norm(at::Tensor out, at::Tensor input, at::Tensor residual,
                        at::Tensor weight, double eps);
void rotary_embedding(at::Tensor positions, at::Tensor q, at::Tensor k,
                      int64_t head_dim, at::Tensor cos_sin_cache);
void silu_and_mul(at::Tensor out, at::Tensor x);
void gelu_and_mul(at::Tensor out, at::Tensor x);
void gelu(at::Tensor out, at::Tensor x);
void layer_norm(at::Tensor out, at::Tensor input, at::Tensor weight,
                at::Tensor bias, double eps);
void fused_add_layer_norm(at::Tensor out, at::Tensor input,
                          at::Tensor residual, at::Tensor weight,
                          at::Tensor bias, double eps);
void reshape_and_cache(at::Tensor k, at::Tensor v, at::Tensor k_cache,
                       at::Tensor v_cache, at::Tensor slot_mapping);
void paged_attention(at::Tensor out, at::Tensor query, at::Tensor k_cache,
                     at::Tensor v_cache, at::Tensor block_tables,
                     at::Tensor seq_lens, double scale);
void mla_decode(at::Tensor out, at::Tensor q, at::Tensor cache,
                at::Tensor block_tables, at::Tensor seq_lens, double scale);
void mla_cache_write(at::Tensor cache, at::Tensor c_kv, at::Tensor k_pe,
                     at::Tensor slots);
void paged_attention_sp(at::Tensor out, at::Tensor query, at::Tensor k_cache,
                        at::Tensor v_cache, at::Tensor block_tables,
                        at::Tensor seq_lens, double scale, int64_t window,
                        at::Tensor sinks);
void prefill_attention(at::Tensor out, at::Tensor q, at::Tensor k, at::Tensor v,
                       at::Tensor tile_seq, at::Tensor tile_qbase,
                       at::Tensor cu_seqlens, double scale, int64_t window,
                       at::Tensor sinks);
void context_attention(at::Tensor out, at::Tensor q, at::Tensor k_cache,
                       at::Tensor v_cache, at::Tensor tile_seq,
                       at::Tensor tile_qbase, at::Tensor cu_seqlens_q,
                       at::Tensor kv_lens, at::Tensor block_tables,
                       double scale, int64_t window, at::Tensor sinks);
at::Tensor mfma_tile_gemm(at::Tensor a, at::Tensor b);
void moe_gate_silu(at::Tensor act, at::Tensor x, at::Tensor w_gate_up,
                   at::Tensor sorted_ids, at::Tensor offsets, at::Tensor bias,
                   int64_t act_mode, int64_t e_base, int64_t n_local_experts);
void moe_down_scatter(at::Tensor out, at::Tensor act, at::Tensor w_down,
                      at::Tensor sorted_ids, at::Tensor gates,
                      at::Tensor offsets, at::Tensor bias, int64_t e_base,
                      int64_t n_local_experts);
void topk(at::Tensor out_vals, at::Tensor out_idx, at::Tensor scores,
          int64_t k);
void lora_shrink(at::Tensor tmp, at::Tensor x, at::Tensor A, at::Tensor idx,
                 double scale);
void lora_expand(at::Tensor y, at::Tensor tmp, at::Tensor B, at::Tensor idx);
void paged_read_bw(at::Tensor out, at::Tensor k_cache, at::Tensor v_cache,
                   at::Tensor block_tables, at::Tensor seq_lens,
                   int64_t mode);
void w4a16_gemv(at::Tensor out, at::Tensor x, at::Tensor qweight,
                at::Tensor scales, at::Tensor zeros, int64_t group);
void w4a16_dequant(at::Tensor out, at::Tensor qweight, at::Tensor scales,
                   at::Tensor zeros, int64_t group);
void w4a16_gemm(at::Tensor out, at::Tensor x, at::Tensor qweight,
                at::Tensor scales, at::Tensor zeros, int64_t group);
void allreduce_rmsnorm(at::Tensor out, at::Tensor ptrs, at::Tensor weight,
                       double eps);
void one_shot_ar_rmsnorm(at::Tensor out, at::Tensor residual, at::Tensor ptrs,
                         at::Tensor sig_ptrs, at::Tensor counter,
                         at::Tensor weight, double eps, int64_t rank);
at::Tensor ipc_handle(at::Tensor t);
int64_t ipc_open(at::Tensor handle_bytes);
void ipc_close(int64_t ptr);
}  // namespace kaito

TORCH_LIBRARY(kaito, m) {
  m.def("rms_norm(Tensor(a!) out, Tensor input, Tensor weight, float eps) -> ()");
  m.def("fused_add_rms_norm(Tensor(a!) out, Tensor input, Tensor(b!) residual, Tensor weight, float eps) -> ()");
  m.def("rotary_embedding(Tensor positions, Tensor(a!) q, Tensor(b!) k, int head_dim, Tensor cos_sin_cache) -> ()");
  m.def("silu_and_mul(Tensor(a!) out, Tensor x) -> ()");
  m.def("gelu_and_mul(Tensor(a!) out, Tensor x) -> ()");
  m.def("gelu(Tensor(a!) out, Tensor x) -> ()");
  m.def("layer_norm(Tensor(a!) out, Tensor input, Tensor weight, Tensor bias, float eps) -> ()");
  m.def("fused_add_layer_norm(Tensor(a!) out, Tensor input, Tensor(b!) residual, Tensor weight, Tensor bias, float eps) -> ()");
  m.def("reshape_and_cache(Tensor k, Tensor v, Tensor(a!) k_cache, Tensor(b!) v_cache, Tensor slot_mapping) -> ()");
  m.def("paged_attention(Tensor(a!) out, Tensor query, Tensor k_cache, Tensor v_cache, Tensor block_tables, Tensor seq_lens, float scale) -> ()");
  m.def("paged_attention_sp(Tensor(a!) out, Tensor query, Tensor k_cache, Tensor v_cache, Tensor block_tables, Tensor seq_lens, float scale, int window, Tensor sinks) -> ()");
  m.def("mla_decode(Tensor(a!) out, Tensor q, Tensor cache, Tensor block_tables, Tensor seq_lens, float scale) -> ()");
  m.def("mla_cache_write(Tensor(a!) cache, Tensor c_kv, Tensor k_pe, Tensor slots) -> ()");
  m.def("prefill_attention(Tensor(a!) out, Tensor q, Tensor k, Tensor v, Tensor tile_seq, Tensor tile_qbase, Tensor cu_seqlens, float scale, int window, Tensor sinks) -> ()");
  m.def("context_attention(Tensor(a!) out, Tensor q, Tensor k_cache, Tensor v_cache, Tensor tile_seq, Tensor tile_qbase, Tensor cu_seqlens_q, Tensor kv_lens, Tensor block_tables, float scale, int window, Tensor sinks) -> ()");
  m.def("mfma_tile_gemm(Tensor a, Tensor b) -> Tensor");
  m.def("moe_gate_silu(Tensor(a!) act, Tensor x, Tensor w_gate_up, Tensor sorted_ids, Tensor offsets, Tensor bias, int act_mode, int e_base, int n_local_experts) -> ()");
  m.def("moe_down_scatter(Tensor(a!) out, Tensor act, Tensor w_down, Tensor sorted_ids, Tensor gates, Tensor offsets, Tensor bias, int e_base, int n_local_experts) -> ()");
  m.def("topk(Tensor(a!) out_vals, Tensor(b!) out_idx, Tensor scores, int k) -> ()");
  m.def("lora_shrink(Tensor(a!) tmp, Tensor x, Tensor A, Tensor idx, float scale) -> ()");
  m.def("lora_expand(Tensor(a!) y, Tensor tmp, Tensor B, Tensor idx) -> ()");
  m.def("paged_read_bw(Tensor(a!) out, Tensor k_cache, Tensor v_cache, Tensor block_tables, Tensor seq_lens, int mode) -> ()");
  m.def("w4a16_gemv(Tensor(a!) out, Tensor x, Tensor qweight, Tensor scales, Tensor zeros, int group) -> ()");
  m.def("w4a16_dequant(Tensor(a!) out, Tensor qweight, Tensor scales, Tensor zeros, int group) -> ()");
  m.def("w4a16_gemm(Tensor(a!) out, Tensor x, Tensor qweight, Tensor scales, Tensor zeros, int group) -> ()");
  m.def("allreduce_rmsnorm(Tensor(a!) out, Tensor ptrs, Tensor weight, float eps) -> ()");
  m.def("one_shot_ar_rmsnorm(Tensor(a!) out, Tensor(b!) residual, Tensor ptrs, Tensor sig_ptrs, Tensor(c!) counter, Tensor weight, float eps, int rank) -> ()");
  m.def("ipc_handle(Tensor t) -> Tensor");
  m.def("ipc_open(Tensor handle_bytes) -> int");
  m.def("ipc_close(int ptr) -> ()");
}

TORCH_LIBRARY_IMPL(kaito, CUDA, m) {
  m.impl("rms_norm", &kaito::rms_norm);
  m.impl("fused_add_rms_norm", &kaito::fused_add_rms_norm);
  m.impl("rotary_embedding", &kaito::rotary_embedding);
  m.impl("silu_and_mul", &kaito::silu_and_mul);
  m.impl("gelu_and_mul", &kaito::gelu_and_mul);
  m.impl("gelu", &kaito::gelu);
  m.impl("layer_norm", &kaito::layer_norm);
  m.impl("fused_add_layer_norm", &kaito::fused_add_layer_norm);
  m.impl("reshape_and_cache", &kaito::reshape_and_cache);
  m.impl("paged_attention", &kaito::paged_attention);
  m.impl("paged_attention_sp", &kaito::paged_attention_sp);
  m.impl("mla_decode", &kaito::mla_decode);
  m.impl("mla_cache_write", &kaito::mla_cache_write);
  m.impl("prefill_attention", &kaito::prefill_attention);
  m.impl("context_attention", &kaito::context_attention);
  m.impl("mfma_tile_gemm", &kaito::mfma_tile_gemm);
  m.impl("moe_gate_silu", &kaito::moe_gate_silu);
  m.impl("moe_down_scatter", &kaito::moe_down_scatter);
  m.impl("topk", &kaito::topk);
  m.impl("lora_shrink", &kaito::lora_shrink);
  m.impl("lora_expand", &kaito::lora_expand);
  m.impl("paged_read_bw", &kaito::paged_read_bw);
  m.impl("w4a16_gemv", &kaito::w4a16_gemv);
  m.impl("w4a16_dequant", &kaito::w4a16_dequant);
  m.impl("w4a16_gemm", &kaito::w4a16_gemm);
  m.impl("allreduce_rmsnorm", &kaito::allreduce_rmsnorm);
  m.impl("one_shot_ar_rmsnorm", &kaito::one_shot_ar_rmsnorm);
  m.impl("ipc_handle", &kaito::ipc_handle);
  m.impl("ipc_open", &kaito::ipc_open);
  m.impl("ipc_close", &kaito::ipc_close);
}
