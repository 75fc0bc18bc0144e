// pybind11 bindings for the unicore_amd gfx950 kernel extension.
// One module (`unicore_amd._kernels`) covers the same API surface as the
// reference's 8 separate CUDA extensions (reference setup.py:141-387).
#include <torch/extension.h>

#include <optional>
#include <vector>

std::vector<at::Tensor> softmax_dropout_forward(bool is_training, at::Tensor input,
                                                std::optional<at::Tensor> mask,
                                                int64_t mask_outer_div,
                                                std::optional<at::Tensor> bias,
                                                int64_t bias_outer_div,
                                                double dropout_prob);
at::Tensor softmax_dropout_backward(at::Tensor grad_output,
                                    at::Tensor softmax_results,
                                    at::Tensor dropout_mask, double dropout_prob);
std::vector<at::Tensor> layernorm_forward(at::Tensor input, at::Tensor gamma,
                                          at::Tensor beta, double eps);
std::vector<at::Tensor> layernorm_backward(at::Tensor grad_out, at::Tensor input,
                                           at::Tensor mean, at::Tensor invvar,
                                           at::Tensor gamma);
std::vector<at::Tensor> rmsnorm_forward(at::Tensor input, at::Tensor gamma,
                                        double eps);
std::vector<at::Tensor> rmsnorm_backward(at::Tensor grad_out, at::Tensor input,
                                         at::Tensor invvar, at::Tensor gamma);
void fused_adam(at::Tensor p, at::Tensor m, at::Tensor v, at::Tensor g, double lr,
                double beta1, double beta2, double eps, double grad_scale,
                int64_t step, bool bias_correction, double weight_decay);
at::Tensor multi_tensor_l2norm(int64_t chunk_size, std::vector<at::Tensor> tensors);
void fp32_to_bf16_sr(at::Tensor src, at::Tensor dst);
std::vector<at::Tensor> qkv_split_forward(at::Tensor qkv,
                                          std::optional<at::Tensor> bias,
                                          int64_t num_heads, double scale);
std::vector<at::Tensor> qkv_split_backward(at::Tensor dq, at::Tensor dk,
                                           at::Tensor dv, int64_t B,
                                           int64_t num_heads, double scale,
                                           bool bias_grad);
at::Tensor gated_mul_forward(at::Tensor x, at::Tensor g,
                             std::optional<at::Tensor> bx,
                             std::optional<at::Tensor> bg);
std::vector<at::Tensor> gated_mul_backward(at::Tensor grad, at::Tensor x,
                                           at::Tensor g,
                                           std::optional<at::Tensor> bx,
                                           std::optional<at::Tensor> bg);
at::Tensor attn_merge(at::Tensor x, int64_t B, int64_t num_heads,
                      bool inverse);
at::Tensor msa_arrange(at::Tensor x, int64_t B, int64_t S, int64_t L,
                       int64_t H, bool col, bool inverse);
std::vector<at::Tensor> gelu_dropout_forward(at::Tensor x,
                                             std::optional<at::Tensor> bias,
                                             double p, bool is_training);
std::vector<at::Tensor> gelu_dropout_backward(at::Tensor grad, at::Tensor x,
                                              std::optional<at::Tensor> bias,
                                              at::Tensor dmask, double p);
at::Tensor mfma_gemm_16x16x32(at::Tensor A, at::Tensor B);
std::vector<at::Tensor> dropout_add_ln_forward(
    at::Tensor x, at::Tensor res, std::optional<at::Tensor> bias,
    at::Tensor gamma, at::Tensor beta, double p, bool is_training, double eps);

std::vector<at::Tensor> dropout_add_forward(at::Tensor x, at::Tensor res,
                                            std::optional<at::Tensor> bias,
                                            double p, bool is_training);
std::vector<at::Tensor> dropout_add_backward(at::Tensor grad, at::Tensor dmask,
                                             double p, int64_t bias_dim);
at::Tensor embedding_backward(at::Tensor grad, at::Tensor indices,
                              int64_t num_embeddings, int64_t padding_idx);
std::vector<at::Tensor> cross_entropy_forward(at::Tensor logits, at::Tensor target,
                                              int64_t ignore_index);
at::Tensor cross_entropy_backward(at::Tensor logits, at::Tensor target,
                                  at::Tensor lse, at::Tensor grad_scale,
                                  int64_t ignore_index);
bool softmax_dropout_backward_bias_supported(int64_t n_batch, int64_t q,
                                             int64_t k, int64_t bb, int64_t bq,
                                             int64_t od);
std::vector<at::Tensor> softmax_dropout_backward_bias(
    at::Tensor grad_output, at::Tensor softmax_results, at::Tensor dropout_mask,
    double dropout_prob, int64_t bb, int64_t bq, int64_t od);
at::Tensor gaussian_basis_forward(at::Tensor coords, at::Tensor means,
                                  at::Tensor stds, at::ScalarType out_dtype);
std::vector<at::Tensor> gaussian_basis_backward(at::Tensor dg, at::Tensor coords,
                                                at::Tensor means, at::Tensor stds);
bool gaussian_basis_supported(int64_t K);
at::Tensor gaussian_pair_bias_forward(at::Tensor coords, at::Tensor means,
                                      at::Tensor stds, at::Tensor W,
                                      at::Tensor bvec,
                                      std::optional<at::Tensor> pad,
                                      double fill, at::ScalarType out_dtype);
std::vector<at::Tensor> gaussian_pair_bias_backward(
    at::Tensor dbias, at::Tensor coords, at::Tensor means, at::Tensor stds,
    at::Tensor W, std::optional<at::Tensor> pad);
bool gaussian_pair_bias_supported(int64_t K, int64_t H);
std::vector<at::Tensor> flash_attn_forward(at::Tensor q, at::Tensor k, at::Tensor v,
                                           std::optional<at::Tensor> bias,
                                           int64_t bias_outer_div,
                                           std::optional<at::Tensor> mask,
                                           int64_t mask_outer_div,
                                           double dropout_p, bool is_training);
std::vector<at::Tensor> flash_attn_backward(
    at::Tensor d_out, at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor o,
    at::Tensor lse, std::optional<at::Tensor> bias, int64_t bias_outer_div,
    bool bias_needs_grad, std::optional<at::Tensor> mask, int64_t mask_outer_div,
    double dropout_p, bool dropped, int64_t seed_in);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("softmax_dropout_forward", &softmax_dropout_forward,
        "fused softmax(+mask,+bias)+dropout forward (gfx950)");
  m.def("softmax_dropout_backward_bias", &softmax_dropout_backward_bias,
        "fused softmax backward + broadcast bias gradient");
  m.def("softmax_dropout_backward_bias_supported",
        &softmax_dropout_backward_bias_supported,
        "fused bias-grad backward supports this shape");
  m.def("softmax_dropout_backward", &softmax_dropout_backward,
        "fused softmax+dropout backward, in-place on grad");
  m.def("layernorm_forward", &layernorm_forward, "fused LayerNorm forward");
  m.def("layernorm_backward", &layernorm_backward,
        "fused LayerNorm backward -> (dx, dgamma, dbeta)");
  m.def("rmsnorm_forward", &rmsnorm_forward, "fused RMSNorm forward");
  m.def("rmsnorm_backward", &rmsnorm_backward,
        "fused RMSNorm backward -> (dx, dgamma)");
  m.def("adam", &fused_adam, "fused AdamW step (in-place)");
  m.def("multi_tensor_l2norm", &multi_tensor_l2norm,
        "global L2 norm over a tensor list");
  m.def("fp32_to_bf16_sr", &fp32_to_bf16_sr,
        "stochastic-rounding fp32 -> bf16 copy");
  m.def("qkv_split_forward", &qkv_split_forward,
        "fused QKV head-split + q-scale -> (q, k, v) each (B*H, L, D)");
  m.def("qkv_split_backward", &qkv_split_backward,
        "fused QKV head-split backward -> dqkv (B, L, 3E)");
  m.def("gated_mul_forward", &gated_mul_forward,
        "fused (x+bx)*sigmoid(g+bg)");
  m.def("gated_mul_backward", &gated_mul_backward,
        "gated-mul backward -> (dx, dg[, bias colsums])");
  m.def("msa_arrange", &msa_arrange,
        "(B,S,L,H*D) <-> row/col head-major MSA layouts, 16 B both sides");
  m.def("attn_merge", &attn_merge,
        "(B*H, L, D) <-> (B, L, H*D) vectorized permute-copy");
  m.def("gelu_dropout_forward", &gelu_dropout_forward,
        "fused exact-GELU + bitfield dropout forward");
  m.def("gelu_dropout_backward", &gelu_dropout_backward,
        "fused GELU + dropout backward (recomputes gelu grad)");
  m.def("mfma_gemm_16x16x32", &mfma_gemm_16x16x32,
        "one-wave bf16 MFMA probe (fragment-layout unit test)");
  m.def("flash_attn_forward", &flash_attn_forward,
        "flash attention forward (bf16, D=64) -> (o, lse, seed)");
  m.def("flash_attn_backward", &flash_attn_backward,
        "flash attention backward -> (dq, dk, dv[, dS])");
  m.def("dropout_add_ln_forward", &dropout_add_ln_forward,
        "fused dropout+residual+LayerNorm forward");
  m.def("dropout_add_forward", &dropout_add_forward,
        "fused dropout + residual add forward");
  m.def("dropout_add_backward", &dropout_add_backward,
        "fused dropout + residual add backward (dx only; d_res = grad)");
  m.def("embedding_backward", &embedding_backward,
        "atomic-scatter embedding gradient (fp32 accumulate)");
  m.def("cross_entropy_forward", &cross_entropy_forward,
        "fused online-logsumexp token cross entropy -> (loss, lse)");
  m.def("gaussian_basis_forward", &gaussian_basis_forward,
        "fused gaussian pair-basis from coords");
  m.def("gaussian_basis_backward", &gaussian_basis_backward,
        "gaussian pair-basis backward (d_coords, d_means, d_stds)");
  m.def("gaussian_basis_supported", &gaussian_basis_supported,
        "kernel supports this K");
  m.def("gaussian_pair_bias_forward", &gaussian_pair_bias_forward,
        "fully-fused gaussian pair bias: coords -> (B,H,L,L)");
  m.def("gaussian_pair_bias_backward", &gaussian_pair_bias_backward,
        "fused pair-bias backward (d_coords, d_means, d_stds, dW, db)");
  m.def("gaussian_pair_bias_supported", &gaussian_pair_bias_supported,
        "fused pair-bias kernel supports this K/H");
  m.def("cross_entropy_backward", &cross_entropy_backward,
        "cross entropy backward (softmax - onehot, no materialized fp32)");
}
