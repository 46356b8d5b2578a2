// pybind bindings for colossalai_amd._C (gfx950 HIP kernels).
#include <torch/extension.h>
#include <vector>

namespace cai {

// multi_tensor_adam.hip
void multi_tensor_adam(std::vector<at::Tensor> grads, std::vector<at::Tensor> params,
                       std::vector<at::Tensor> exp_avgs, std::vector<at::Tensor> exp_avg_sqs,
                       std::vector<at::Tensor> param_outs, double lr, double beta1, double beta2,
                       double eps, long step, bool adamw_mode, bool bias_correction, double weight_decay,
                       double div_scale, long chunk_size);
void multi_tensor_scale(std::vector<at::Tensor> inputs, std::vector<at::Tensor> outputs, double scale,
                        long chunk_size);
at::Tensor multi_tensor_l2norm(std::vector<at::Tensor> inputs, long chunk_size);
void multi_tensor_lamb(std::vector<at::Tensor> grads, std::vector<at::Tensor> params,
                       std::vector<at::Tensor> exp_avgs, std::vector<at::Tensor> exp_avg_sqs,
                       std::vector<at::Tensor> param_outs, double lr, double beta1, double beta2,
                       double eps, long step, bool bias_correction, double weight_decay,
                       double div_scale, long chunk_size);
void multi_tensor_sgd(std::vector<at::Tensor> grads, std::vector<at::Tensor> params,
                      std::vector<at::Tensor> momentum_bufs, std::vector<at::Tensor> param_outs,
                      double lr, double momentum, double dampening, double weight_decay,
                      bool nesterov, double div_scale, long chunk_size);

// rmsnorm.hip
std::vector<at::Tensor> rmsnorm_fwd(at::Tensor input, at::Tensor weight, double eps, bool save_inv_rms);
std::vector<at::Tensor> rmsnorm_fused_add_fwd(at::Tensor input, at::Tensor residual, at::Tensor weight,
                                              double eps, bool save_inv_rms);
std::vector<at::Tensor> rmsnorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor weight, at::Tensor inv_rms);

// layernorm.hip
std::vector<at::Tensor> layernorm_fwd(at::Tensor input, at::Tensor gamma, c10::optional<at::Tensor> beta,
                                      double eps, bool save_stats);
std::vector<at::Tensor> layernorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor gamma, at::Tensor mean,
                                      at::Tensor invstd);

// rope.hip
void rope_inplace(at::Tensor q, c10::optional<at::Tensor> k, at::Tensor table,
                  c10::optional<at::Tensor> positions, bool backward);

// swiglu.hip
at::Tensor swiglu_fwd(at::Tensor gate_up);
at::Tensor swiglu_bwd(at::Tensor dout, at::Tensor gate_up);

// scaled_softmax.hip
at::Tensor scaled_masked_softmax_fwd(at::Tensor x, c10::optional<at::Tensor> mask, double scale, bool causal);
at::Tensor scaled_masked_softmax_bwd(at::Tensor dy, at::Tensor y, double scale);

// mfma_selftest.hip
std::vector<at::Tensor> mfma_selftest(at::Tensor A16, at::Tensor B16, at::Tensor A32, at::Tensor B32);

void kv_cache_append(at::Tensor k, at::Tensor v, at::Tensor kpool, at::Tensor vpool,
                     at::Tensor slot_rows);

// decode_attn.hip
at::Tensor decode_attention(at::Tensor q, at::Tensor kcache, at::Tensor vcache, at::Tensor seq_lens,
                            double scale, int64_t n_splits);
at::Tensor decode_attention_paged(at::Tensor q, at::Tensor kpool, at::Tensor vpool,
                                  at::Tensor block_tables, at::Tensor seq_lens, double scale,
                                  int64_t n_splits);

// moe.hip
at::Tensor moe_combine_fwd(at::Tensor y, at::Tensor inv, at::Tensor topw);
std::vector<at::Tensor> moe_combine_bwd(at::Tensor dout, at::Tensor y, at::Tensor inv, at::Tensor topw);

// grouped_gemm.hip
at::Tensor grouped_gemm_fwd(at::Tensor x, at::Tensor w, std::vector<long> offsets);
at::Tensor grouped_gemm_dgrad(at::Tensor dy, at::Tensor w, std::vector<long> offsets);
at::Tensor grouped_gemm_wgrad(at::Tensor dy, at::Tensor x, std::vector<long> offsets);
std::vector<at::Tensor> moe_cumsum(at::Tensor experts, long n_experts);
at::Tensor moe_dispatch_fwd(at::Tensor x, at::Tensor src);
at::Tensor moe_dispatch_bwd(at::Tensor grad, at::Tensor src);

// flash_attn.hip
std::vector<at::Tensor> flash_attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v, bool causal, double scale,
                                       c10::optional<at::Tensor> seqlens,
                                       c10::optional<at::Tensor> seqlens_k);
std::vector<at::Tensor> flash_attn_bwd(at::Tensor dout, at::Tensor q, at::Tensor k, at::Tensor v,
                                       at::Tensor out, at::Tensor lse, bool causal, double scale,
                                       at::Tensor dq, at::Tensor dk, at::Tensor dv,
                                       c10::optional<at::Tensor> seqlens,
                                       c10::optional<at::Tensor> seqlens_k);
std::vector<at::Tensor> flash_attn_varlen_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                              at::Tensor cu_seqlens, long max_seqlen,
                                              bool causal, double scale);
std::vector<at::Tensor> flash_attn_varlen_bwd(at::Tensor dout, at::Tensor q, at::Tensor k, at::Tensor v,
                                              at::Tensor out, at::Tensor lse, at::Tensor cu_seqlens,
                                              long max_seqlen, bool causal, double scale);

}  // namespace cai

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("multi_tensor_adam", &cai::multi_tensor_adam, "fused multi-tensor Adam/AdamW (gfx950)");
  m.def("multi_tensor_scale", &cai::multi_tensor_scale, "fused multi-tensor scale");
  m.def("multi_tensor_l2norm", &cai::multi_tensor_l2norm, "fused multi-tensor L2 norm");
  m.def("multi_tensor_sgd", &cai::multi_tensor_sgd, "fused multi-tensor SGD (momentum/nesterov)");
  m.def("multi_tensor_lamb", &cai::multi_tensor_lamb,
        "fused multi-tensor LAMB (two-stage: Adam update + trust-ratio scale)");
  m.def("rmsnorm_fwd", &cai::rmsnorm_fwd, "RMSNorm forward");
  m.def("rmsnorm_fused_add_fwd", &cai::rmsnorm_fused_add_fwd, "fused residual-add RMSNorm forward");
  m.def("rmsnorm_bwd", &cai::rmsnorm_bwd, "RMSNorm backward");
  m.def("layernorm_fwd", &cai::layernorm_fwd, "LayerNorm forward");
  m.def("layernorm_bwd", &cai::layernorm_bwd, "LayerNorm backward");
  m.def("rope_inplace", &cai::rope_inplace, "in-place rotary embedding (fwd/bwd)");
  m.def("swiglu_fwd", &cai::swiglu_fwd, "fused SwiGLU forward");
  m.def("swiglu_bwd", &cai::swiglu_bwd, "fused SwiGLU backward");
  m.def("scaled_masked_softmax_fwd", &cai::scaled_masked_softmax_fwd, "fused scale+mask+softmax fwd");
  m.def("scaled_masked_softmax_bwd", &cai::scaled_masked_softmax_bwd, "fused scale+mask+softmax bwd");
  m.def("mfma_selftest", &cai::mfma_selftest, "MFMA layout self-test probes");
  m.def("kv_cache_append", &cai::kv_cache_append,
        "decode-step K+V scatter into paged pools (one launch)");
  m.def("decode_attention", &cai::decode_attention, "single-token attention over KV cache",
        py::arg("q"), py::arg("kcache"), py::arg("vcache"), py::arg("seq_lens"), py::arg("scale"),
        py::arg("n_splits") = 0);
  m.def("decode_attention_paged", &cai::decode_attention_paged,
        "single-token attention over a paged (block-table) KV pool",
        py::arg("q"), py::arg("kpool"), py::arg("vpool"), py::arg("block_tables"),
        py::arg("seq_lens"), py::arg("scale"), py::arg("n_splits") = 0);
  m.def("moe_combine_fwd", &cai::moe_combine_fwd, "fused MoE un-permute + weighted top-k sum");
  m.def("moe_combine_bwd", &cai::moe_combine_bwd, "MoE combine backward (dy + routing-weight grads)");
  m.def("grouped_gemm_fwd", &cai::grouped_gemm_fwd, "grouped per-expert GEMM: y = x @ w[g]^T");
  m.def("grouped_gemm_dgrad", &cai::grouped_gemm_dgrad, "grouped GEMM data grad: dx = dy @ w[g]");
  m.def("grouped_gemm_wgrad", &cai::grouped_gemm_wgrad, "grouped GEMM weight grad: dw[g] = dy_g^T @ x_g");
  m.def("moe_cumsum", &cai::moe_cumsum, "deterministic per-expert rank/count of flat routing ids");
  m.def("moe_dispatch_fwd", &cai::moe_dispatch_fwd, "MoE dispatch: row gather by permutation");
  m.def("moe_dispatch_bwd", &cai::moe_dispatch_bwd, "MoE dispatch backward: row scatter (bijection)");
  m.def("flash_attn_fwd", &cai::flash_attn_fwd,
        "flash attention forward (bf16, causal, GQA; optional right-padding seqlens)",
        py::arg("q"), py::arg("k"), py::arg("v"), py::arg("causal"), py::arg("scale"),
        py::arg("seqlens") = py::none(), py::arg("seqlens_k") = py::none());
  m.def("flash_attn_bwd", &cai::flash_attn_bwd, "flash attention backward",
        py::arg("dout"), py::arg("q"), py::arg("k"), py::arg("v"), py::arg("out"), py::arg("lse"),
        py::arg("causal"), py::arg("scale"), py::arg("dq"), py::arg("dk"), py::arg("dv"),
        py::arg("seqlens") = py::none(), py::arg("seqlens_k") = py::none());
  m.def("flash_attn_varlen_fwd", &cai::flash_attn_varlen_fwd,
        "flash attention forward over a packed ragged batch (cu_seqlens)");
  m.def("flash_attn_varlen_bwd", &cai::flash_attn_varlen_bwd, "varlen flash attention backward");
}
