// Python bindings for the d9d_amd CDNA4 kernel extension.
#include <torch/extension.h>

// rms_norm.hip
std::vector<torch::Tensor> rms_norm_fwd(torch::Tensor x, torch::Tensor w, double eps, bool zero_centered);
std::vector<torch::Tensor> rms_norm_bwd(torch::Tensor x, torch::Tensor w, torch::Tensor dy, torch::Tensor inv_rms, bool zero_centered);

// silu_mul.hip
torch::Tensor silu_mul_fwd(torch::Tensor a, torch::Tensor b);
torch::Tensor silu_mul_packed_fwd(torch::Tensor x);
torch::Tensor silu_mul_packed_bwd(torch::Tensor x, torch::Tensor g);
std::vector<torch::Tensor> silu_mul_bwd(torch::Tensor a, torch::Tensor b, torch::Tensor g);

// stochastic.hip
void copy_fp32_to_bf16_stochastic_(torch::Tensor dst, torch::Tensor src, int64_t seed);
void add_bf16_into_f32_(torch::Tensor acc, torch::Tensor x);
void adamw_stochastic_bf16_(torch::Tensor p, torch::Tensor g, torch::Tensor m, torch::Tensor v,
                            double lr, double beta1, double beta2, double eps, double weight_decay,
                            int64_t step, int64_t seed);

// attention.hip
std::vector<torch::Tensor> flash_attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                           c10::optional<torch::Tensor> sinks,
                                           c10::optional<torch::Tensor> cu_seqlens_q,
                                           c10::optional<torch::Tensor> cu_seqlens_k,
                                          bool causal, double softmax_scale, int64_t window_left,
                                          int64_t q_offset);
std::vector<torch::Tensor> flash_attn_bwd(torch::Tensor dout, torch::Tensor q, torch::Tensor k,
                                          torch::Tensor v, torch::Tensor out, torch::Tensor lse,
                                          c10::optional<torch::Tensor> cu_seqlens_q,
                                          c10::optional<torch::Tensor> cu_seqlens_k,
                                          bool causal, double softmax_scale, int64_t window_left,
                                          int64_t q_offset);
torch::Tensor mfma_selfcheck(torch::Tensor a, torch::Tensor b);

// moe_router.hip
std::vector<torch::Tensor> router_topk_fwd(torch::Tensor logits, c10::optional<torch::Tensor> bias,
                                           int64_t K, bool renormalize);
torch::Tensor router_topk_bwd(torch::Tensor logits, torch::Tensor top_idx, torch::Tensor dtop,
                              bool renormalize);

torch::Tensor causal_conv_silu_fwd(torch::Tensor x, torch::Tensor w);
std::vector<torch::Tensor> causal_conv_silu_bwd(torch::Tensor x, torch::Tensor w,
                                                torch::Tensor dy);
std::vector<torch::Tensor> gdn_chunk_fwd(torch::Tensor q, torch::Tensor k,
                                         torch::Tensor v, torch::Tensor beta,
                                         torch::Tensor decay_log,
                                         bool return_state, bool return_aux,
                                         bool skip_out);
std::vector<torch::Tensor> gdn_chunk_bwd_scan(torch::Tensor q, torch::Tensor k,
                                              torch::Tensor dout,
                                              torch::Tensor beta,
                                              torch::Tensor decay_log);

torch::Tensor cce_dlogits_(torch::Tensor logits, torch::Tensor lse, torch::Tensor targets,
                           torch::Tensor dl, c10::optional<torch::Tensor> dlse,
                           int64_t vocab_start, int64_t ignore_index,
                           double filter_eps);

void adamw_stochastic_bf16_multi_(
    std::vector<torch::Tensor> params, std::vector<torch::Tensor> grads,
    std::vector<torch::Tensor> exp_avgs, std::vector<torch::Tensor> exp_avg_sqs,
    double lr, double beta1, double beta2, double eps, double weight_decay,
    std::vector<int64_t> steps, std::vector<int64_t> seeds);

// gmm.hip
torch::Tensor gmm(torch::Tensor a, torch::Tensor b, torch::Tensor batch_sizes);
torch::Tensor gmm_nt(torch::Tensor a, torch::Tensor w, torch::Tensor batch_sizes);
torch::Tensor gmm_db(torch::Tensor a, torch::Tensor g, torch::Tensor batch_sizes, int64_t num_experts);

// rope.hip
std::vector<torch::Tensor> rope_qk(torch::Tensor q, torch::Tensor k,
                                   torch::Tensor cos_t, torch::Tensor sin_t, double sin_sign);

// cce.hip
std::vector<torch::Tensor> cce_fwd(torch::Tensor e, torch::Tensor c, torch::Tensor targets);

// moe_permute.hip
torch::Tensor moe_gather_rows(torch::Tensor src, torch::Tensor idx, c10::optional<torch::Tensor> scale);
torch::Tensor moe_csr_combine(torch::Tensor expert_out, c10::optional<torch::Tensor> probs,
                              torch::Tensor inv, int64_t T, int64_t K);
torch::Tensor moe_row_dot(torch::Tensor grad_out, torch::Tensor expert_out, torch::Tensor row_to_token);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("router_topk_fwd", &router_topk_fwd, "fused MoE router fwd");
  m.def("router_topk_bwd", &router_topk_bwd, "fused MoE router bwd");
  m.def("cce_dlogits_", &cce_dlogits_, "fused CCE dlogits (in-place)");
  m.def("causal_conv_silu_fwd", &causal_conv_silu_fwd, "fused causal depthwise conv + SiLU fwd");
  m.def("causal_conv_silu_bwd", &causal_conv_silu_bwd, "fused causal depthwise conv + SiLU bwd");
  m.def("gdn_chunk_fwd", &gdn_chunk_fwd, "chunked gated delta rule forward (GDN)");
  m.def("gdn_chunk_bwd_scan", &gdn_chunk_bwd_scan,
        "chunked gated delta rule backward reverse scan (GDN)");
  m.def("adamw_stochastic_bf16_multi_", &adamw_stochastic_bf16_multi_, "multi-tensor fused SR-AdamW");
  m.def("gmm", &gmm, "CDNA4 grouped GEMM (MoE experts)");
  m.def("gmm_nt", &gmm_nt, "CDNA4 grouped GEMM, weight (E,N,K)");
  m.def("gmm_db", &gmm_db, "CDNA4 grouped GEMM weight-grad");
  m.def("rope_qk", &rope_qk, "fused q/k rotary embedding");
  m.def("cce_fwd", &cce_fwd, "fused linear cross-entropy forward (lse + target logit)");
  m.def("moe_gather_rows", &moe_gather_rows, "MoE row gather (optional scale)");
  m.def("moe_csr_combine", &moe_csr_combine, "MoE weighted replica combine");
  m.def("moe_row_dot", &moe_row_dot, "per-row dot for prob grads");
  m.def("flash_attn_fwd", &flash_attn_fwd, "CDNA4 flash attention forward");
  m.def("flash_attn_bwd", &flash_attn_bwd, "CDNA4 flash attention backward");
  m.def("mfma_selfcheck", &mfma_selfcheck, "MFMA fragment-map self check");
  m.def("rms_norm_fwd", &rms_norm_fwd, "RMSNorm forward (bf16, CDNA4)");
  m.def("rms_norm_bwd", &rms_norm_bwd, "RMSNorm backward (bf16, CDNA4)");
  m.def("silu_mul_fwd", &silu_mul_fwd, "fused silu(a)*b forward");
  m.def("silu_mul_packed_fwd", &silu_mul_packed_fwd, "packed SwiGLU fwd");
  m.def("silu_mul_packed_bwd", &silu_mul_packed_bwd, "packed SwiGLU bwd");
  m.def("silu_mul_bwd", &silu_mul_bwd, "fused silu(a)*b backward");
  m.def("add_bf16_into_f32_", &add_bf16_into_f32_,
        "acc(fp32) += x(bf16), single pass");
  m.def("copy_fp32_to_bf16_stochastic_", &copy_fp32_to_bf16_stochastic_,
        "stochastic-rounding fp32->bf16 copy");
  m.def("adamw_stochastic_bf16_", &adamw_stochastic_bf16_,
        "fused AdamW step with SR bf16 param writes");
}
