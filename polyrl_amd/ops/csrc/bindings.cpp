// pybind bindings for the polyrl_amd CDNA4 kernel suite.
#include <torch/extension.h>
#include <vector>

void rmsnorm(torch::Tensor out, torch::Tensor x, torch::Tensor weight,
             double eps);
void fused_add_rmsnorm(torch::Tensor out, torch::Tensor residual,
                       torch::Tensor x, torch::Tensor weight, double eps);
void silu_mul(torch::Tensor out, torch::Tensor gate, torch::Tensor up);
void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
                  torch::Tensor cos_tab, torch::Tensor sin_tab);
void gather_logprobs(torch::Tensor out_lp, torch::Tensor out_ent,
                     torch::Tensor logits, torch::Tensor labels,
                     bool want_entropy);
void sample(torch::Tensor out_tokens, torch::Tensor out_logprobs,
            torch::Tensor logits, torch::Tensor temperature,
            torch::Tensor top_k, torch::Tensor top_p, int64_t seed,
            bool no_filter, torch::Tensor seed_dev);
void kv_cache_append(torch::Tensor k_cache, torch::Tensor v_cache,
                     torch::Tensor k, torch::Tensor v,
                     torch::Tensor slot_mapping);
void paged_attention_decode(torch::Tensor out, torch::Tensor q,
                            torch::Tensor k_cache, torch::Tensor v_cache,
                            torch::Tensor page_table,
                            torch::Tensor context_lens, double scale);
void varlen_prefill_attention(torch::Tensor out, torch::Tensor q,
                              torch::Tensor k, torch::Tensor v,
                              torch::Tensor cu_seqlens_q,
                              torch::Tensor cu_seqlens_k,
                              torch::Tensor tile_seq, torch::Tensor tile_q0,
                              double scale, bool causal, torch::Tensor lse);
void varlen_attention_backward(
    torch::Tensor dq, torch::Tensor dk, torch::Tensor dv,
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor out, torch::Tensor dout, torch::Tensor lse,
    torch::Tensor cu_seqlens_q, torch::Tensor cu_seqlens_k,
    torch::Tensor tile_seq, torch::Tensor tile_k0,
    double scale, bool causal, bool use_v2);
std::vector<torch::Tensor> rmsnorm_train_fwd(torch::Tensor x,
                                             torch::Tensor res,
                                             torch::Tensor w, double eps);
std::vector<torch::Tensor> rmsnorm_train_bwd(torch::Tensor dy,
                                             torch::Tensor h,
                                             torch::Tensor w,
                                             torch::Tensor rstd);
void silu_mul_bwd(torch::Tensor dgate, torch::Tensor dup, torch::Tensor dy,
                  torch::Tensor gate, torch::Tensor up);
void gather_logprobs_train_fwd(torch::Tensor out_lp, torch::Tensor out_lse,
                               torch::Tensor logits, torch::Tensor labels);
torch::Tensor gather_logprobs_train_bwd(torch::Tensor logits,
                                        torch::Tensor labels,
                                        torch::Tensor lse,
                                        torch::Tensor dlp);
torch::Tensor rope_train_apply(torch::Tensor x, torch::Tensor cos_t,
                               torch::Tensor sin_t, bool backward);
torch::Tensor tuned_linear_fwd(torch::Tensor x, torch::Tensor w);
torch::Tensor tuned_linear_dgrad(torch::Tensor dy, torch::Tensor w);
torch::Tensor tuned_linear_wgrad(torch::Tensor dy, torch::Tensor x);
void varlen_attention_backward_v3(
    torch::Tensor dq, torch::Tensor dk, torch::Tensor dv,
    torch::Tensor q, torch::Tensor k, torch::Tensor v,
    torch::Tensor out, torch::Tensor dout, torch::Tensor lse,
    torch::Tensor cu_seqlens_q, torch::Tensor cu_seqlens_k,
    torch::Tensor t64_seq, torch::Tensor t64_q0,
    double scale, bool causal);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm, "RMSNorm (bf16, CDNA4)");
  m.def("fused_add_rmsnorm", &fused_add_rmsnorm,
        "residual-add + RMSNorm, residual updated in place");
  m.def("silu_mul", &silu_mul, "SiLU(gate) * up");
  m.def("rope_inplace", &rope_inplace, "NEOX RoPE in place (table-driven)");
  m.def("gather_logprobs", &gather_logprobs,
        "log-softmax gather at labels (+optional entropy)");
  m.def("sample", &sample,
        "fused temperature/top-k/top-p sampling + logprob capture");
  m.def("kv_cache_append", &kv_cache_append, "paged KV cache scatter-append");
  m.def("paged_attention_decode", &paged_attention_decode,
        "paged decode attention (GQA, bf16)");
  m.def("varlen_prefill_attention", &varlen_prefill_attention,
        "varlen causal prefill attention (MFMA, bf16; optional LSE out)");
  m.def("varlen_attention_backward", &varlen_attention_backward,
        "varlen causal flash-attention backward (MFMA, bf16)");
  m.def("rmsnorm_train_fwd", &rmsnorm_train_fwd,
        "trainer RMSNorm fwd (+optional fused residual): y, h, rstd");
  m.def("rmsnorm_train_bwd", &rmsnorm_train_bwd,
        "trainer RMSNorm bwd: dx, dw(fp32)");
  m.def("silu_mul_bwd", &silu_mul_bwd, "silu(gate)*up backward, one pass");
  m.def("gather_logprobs_train_fwd", &gather_logprobs_train_fwd,
        "training logprob gather: lp + logsumexp (bf16 logits)");
  m.def("gather_logprobs_train_bwd", &gather_logprobs_train_bwd,
        "dlogits = dlp * (onehot - softmax), one pass");
  m.def("rope_train_apply", &rope_train_apply,
        "trainer NEOX RoPE (one pass; backward = rotate by -theta)");
  m.def("tuned_linear_fwd", &tuned_linear_fwd,
        "Y = X W^T via hipBLASLt with per-shape in-process algo search");
  m.def("tuned_linear_dgrad", &tuned_linear_dgrad, "dX = dY W (algo-pinned)");
  m.def("tuned_linear_wgrad", &tuned_linear_wgrad,
        "dW = dY^T X (algo-pinned)");
  m.def("varlen_attention_backward_v3", &varlen_attention_backward_v3,
        "varlen flash-attention backward v3 (atomic-free dkv+dq split, "
        "head_dim 64/128)");
}
