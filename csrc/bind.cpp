// Python bindings for the MI355X (gfx950) HIP kernels.
#include <torch/extension.h>

namespace bpa {

std::vector<torch::Tensor> ln_fwd(torch::Tensor x, torch::Tensor gamma,
                                  torch::Tensor beta, double eps);
std::vector<torch::Tensor> ln_bwd(torch::Tensor dy, torch::Tensor x,
                                  torch::Tensor gamma, torch::Tensor mean,
                                  torch::Tensor rstd);
torch::Tensor bias_gelu_fwd(torch::Tensor x, torch::Tensor bias);
std::vector<torch::Tensor> bias_gelu_bwd(torch::Tensor dy, torch::Tensor x,
                                         torch::Tensor bias);
torch::Tensor col_sum(torch::Tensor x);
std::vector<torch::Tensor> gemm_bias_gelu_fwd(torch::Tensor x,
                                              torch::Tensor w1,
                                              torch::Tensor b1);
std::vector<torch::Tensor> gemm_dgelu_bgrad(torch::Tensor dout,
                                            torch::Tensor w2,
                                            torch::Tensor aux);
bool gemm_gelu_aux_supported();
std::vector<torch::Tensor> bias_dropout_residual_ln_fwd(
    torch::Tensor x, c10::optional<torch::Tensor> bias, torch::Tensor residual,
    torch::Tensor gamma, torch::Tensor beta, double p, double eps,
    int64_t seed, int64_t offset);
std::vector<torch::Tensor> bias_dropout_residual_ln_bwd(
    torch::Tensor dy, torch::Tensor z, torch::Tensor mask, torch::Tensor gamma,
    torch::Tensor mean, torch::Tensor rstd, double p, bool has_bias);
std::vector<torch::Tensor> embedding_ln_dropout_fwd(
    torch::Tensor ids, c10::optional<torch::Tensor> tt, torch::Tensor word,
    torch::Tensor pos, c10::optional<torch::Tensor> tok, torch::Tensor gamma,
    torch::Tensor beta, double p, double eps, int64_t seed, int64_t offset,
    torch::ScalarType out_dtype);
std::vector<torch::Tensor> embedding_ln_dropout_bwd(
    torch::Tensor dy, torch::Tensor ids, c10::optional<torch::Tensor> tt,
    torch::Tensor z, torch::Tensor mask, torch::Tensor gamma,
    torch::Tensor mean, torch::Tensor rstd, double p, int64_t vocab,
    int64_t max_pos, int64_t n_types);
std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor labels,
                                  int64_t ignore_index);
torch::Tensor ce_bwd(torch::Tensor dloss, torch::Tensor logits,
                     torch::Tensor labels, torch::Tensor lse,
                     torch::Tensor count, int64_t ignore_index);
std::vector<torch::Tensor> attention_fwd(torch::Tensor qkv,
                                         torch::Tensor seqlens,
                                         int64_t num_heads, double p,
                                         int64_t seed, int64_t offset);
torch::Tensor tr16_probe(int64_t addr_mode);
torch::Tensor wgrad_tn(torch::Tensor dy, torch::Tensor x,
                       int64_t splitk_override);
std::vector<torch::Tensor> mlm_head_fwd(torch::Tensor h, torch::Tensor w,
                                        torch::Tensor bias,
                                        torch::Tensor labels,
                                        int64_t ignore_index);
bool mlm_head_supported(int64_t P, int64_t V, int64_t K);
bool wgrad_tn_supported(int64_t K, int64_t M, int64_t N);
bool wgrad_tn_profitable(int64_t K, int64_t M, int64_t N);
torch::Tensor attention_bwd(torch::Tensor dout, torch::Tensor qkv,
                            torch::Tensor seqlens, torch::Tensor out,
                            torch::Tensor lse, torch::Tensor dmask,
                            int64_t num_heads, double p, int64_t seed,
                            int64_t offset);
torch::Tensor multi_tensor_l2norm_sq(std::vector<torch::Tensor> tensors);
void multi_tensor_clip_scale(std::vector<torch::Tensor> grads,
                             torch::Tensor gnorm_sq, double max_norm);
void fused_lamb(std::vector<torch::Tensor> params,
                std::vector<torch::Tensor> grads,
                std::vector<torch::Tensor> ms, std::vector<torch::Tensor> vs,
                torch::Tensor gnorm_sq, double lr, double beta1, double beta2,
                double eps, double wd, int64_t step, bool bias_correction,
                bool grad_averaging, double max_grad_norm, bool use_ratio);
void fused_adam(std::vector<torch::Tensor> params,
                std::vector<torch::Tensor> grads,
                std::vector<torch::Tensor> ms, std::vector<torch::Tensor> vs,
                double lr, double beta1, double beta2, double eps, double wd,
                int64_t step, bool bias_correction, bool adam_w);

}  // namespace bpa

namespace bpa_tok {
int64_t create_wordpiece(std::vector<std::string> vocab, std::string unk);
std::pair<std::vector<std::string>, std::vector<int64_t>> encode_wordpiece(
    int64_t handle, std::vector<std::string> words);
int64_t create_bpe(std::vector<std::string> vocab,
                   std::vector<std::string> merge_lines);
std::pair<std::vector<std::string>, std::vector<int64_t>> encode_bpe(
    int64_t handle, std::vector<std::string> pretokens);
std::vector<std::string> train_bpe(std::vector<std::string> words,
                                   std::vector<int64_t> counts,
                                   int64_t num_merges);
}  // namespace bpa_tok

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "bert_pytorch_amd gfx950 HIP kernels";
  m.def("ln_fwd", &bpa::ln_fwd, "fused LayerNorm forward");
  m.def("ln_bwd", &bpa::ln_bwd, "fused LayerNorm backward");
  m.def("bias_gelu_fwd", &bpa::bias_gelu_fwd);
  m.def("bias_gelu_bwd", &bpa::bias_gelu_bwd);
  m.def("col_sum", &bpa::col_sum, "column sum [rows,H] -> fp32 [H]");
  m.def("gemm_bias_gelu_fwd", &bpa::gemm_bias_gelu_fwd,
        "hipblaslt GEMM with fused bias+GELU epilogue (+aux)");
  m.def("gemm_dgelu_bgrad", &bpa::gemm_dgelu_bgrad,
        "hipblaslt dgrad GEMM with fused dGELU+bias-grad epilogue");
  m.def("gemm_gelu_aux_supported", &bpa::gemm_gelu_aux_supported);
  m.def("bias_dropout_residual_ln_fwd", &bpa::bias_dropout_residual_ln_fwd);
  m.def("bias_dropout_residual_ln_bwd", &bpa::bias_dropout_residual_ln_bwd);
  m.def("embedding_ln_dropout_fwd", &bpa::embedding_ln_dropout_fwd);
  m.def("embedding_ln_dropout_bwd", &bpa::embedding_ln_dropout_bwd);
  m.def("ce_fwd", &bpa::ce_fwd);
  m.def("ce_bwd", &bpa::ce_bwd);
  m.def("attention_fwd", &bpa::attention_fwd);
  m.def("attention_bwd", &bpa::attention_bwd);
  m.def("tr16_probe", &bpa::tr16_probe);
  m.def("wgrad_tn", &bpa::wgrad_tn, py::arg("dy"), py::arg("x"),
        py::arg("splitk_override") = 0);
  m.def("wgrad_tn_supported", &bpa::wgrad_tn_supported);
  m.def("mlm_head_fwd", &bpa::mlm_head_fwd,
        "MFMA MLM-decoder GEMM with fused bias + CE-forward epilogue");
  m.def("mlm_head_supported", &bpa::mlm_head_supported);
  m.def("wgrad_tn_profitable", &bpa::wgrad_tn_profitable);
  m.def("multi_tensor_l2norm_sq", &bpa::multi_tensor_l2norm_sq);
  m.def("multi_tensor_clip_scale", &bpa::multi_tensor_clip_scale);
  m.def("fused_lamb", &bpa::fused_lamb);
  m.def("fused_adam", &bpa::fused_adam);
  m.def("tok_create_wordpiece", &bpa_tok::create_wordpiece);
  m.def("tok_encode_wordpiece", &bpa_tok::encode_wordpiece);
  m.def("tok_create_bpe", &bpa_tok::create_bpe);
  m.def("tok_encode_bpe", &bpa_tok::encode_bpe);
  m.def("tok_train_bpe", &bpa_tok::train_bpe);
}
