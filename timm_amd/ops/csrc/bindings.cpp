// Python bindings for the timm_amd gfx950 HIP kernel extension.
#include <torch/extension.h>

// layernorm.hip
std::vector<at::Tensor> layer_norm_fwd(at::Tensor x, at::Tensor w, at::Tensor b, double eps);
std::vector<at::Tensor> layer_norm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                       at::Tensor mean, at::Tensor rstd);
std::vector<at::Tensor> rms_norm_fwd(at::Tensor x, at::Tensor w, double eps);
std::vector<at::Tensor> rms_norm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w, at::Tensor rstd);

// elementwise.hip
at::Tensor bias_act_fwd(at::Tensor x, at::Tensor b, long act_id);
std::vector<at::Tensor> bias_act_bwd(at::Tensor dy, at::Tensor x, at::Tensor b, long act_id);
at::Tensor residual_scale_add_fwd(at::Tensor x, at::Tensor y,
                                  c10::optional<at::Tensor> gamma,
                                  c10::optional<at::Tensor> keep);
std::vector<at::Tensor> residual_scale_add_bwd(at::Tensor dout,
                                               c10::optional<at::Tensor> y,
                                               c10::optional<at::Tensor> gamma,
                                               c10::optional<at::Tensor> keep);

// attention.hip
std::vector<at::Tensor> attention_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                      c10::optional<at::Tensor> mask, double scale);
at::Tensor attn_bwd_softmax(at::Tensor s, at::Tensor lse, c10::optional<at::Tensor> mask,
                            double scale);
at::Tensor attn_bwd_ds(at::Tensor p, at::Tensor dp, at::Tensor delta, double scale);
std::vector<at::Tensor> attn_bwd_preprocess(at::Tensor dout, at::Tensor o);

// attention_bwd.hip
void attention_bwd(at::Tensor q, at::Tensor k, at::Tensor v,
                   at::Tensor dov, at::Tensor lse, at::Tensor delta,
                   c10::optional<at::Tensor> mask,
                   at::Tensor dq, at::Tensor dk, at::Tensor dv,
                   double scale);

// depthwise_conv.hip
at::Tensor dwconv_fwd(at::Tensor x, at::Tensor w_t, c10::optional<at::Tensor> bias,
                      long stride, long pad, long K, long Ho, long Wo);
at::Tensor dwconv_bwd_data(at::Tensor dy, at::Tensor w_t, long stride, long pad, long K,
                           long H, long W);
std::vector<at::Tensor> dwconv_bwd_weight(at::Tensor dy, at::Tensor x, long stride, long pad,
                                          long K, bool need_bias);

// muon_ns.hip
at::Tensor ns_gemm_nt(at::Tensor l, at::Tensor r, c10::optional<at::Tensor> s,
                      double alpha, double beta);
at::Tensor ns_gemm_nn(at::Tensor l, at::Tensor r, c10::optional<at::Tensor> s,
                      double alpha, double beta);


// ce_loss.hip
std::vector<at::Tensor> ce_loss_fwd(at::Tensor logits, at::Tensor target, double smoothing);
at::Tensor ce_loss_bwd(at::Tensor logits, at::Tensor target, at::Tensor lse,
                       at::Tensor dloss, double smoothing);

// data_ops.hip
at::Tensor u8_normalize(at::Tensor x, at::Tensor mean, at::Tensor inv_std, at::ScalarType out_dtype);
std::vector<at::Tensor> masked_pool_fwd(at::Tensor x, at::Tensor valid, bool is_max);
at::Tensor masked_pool_bwd(at::Tensor dy, at::Tensor valid, at::Tensor aux, long n_tokens, bool is_max);

// multi_tensor.hip
void multi_tensor_adamw(
    std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
    std::vector<at::Tensor> exp_avgs, std::vector<at::Tensor> exp_avg_sqs,
    double lr, double beta1, double beta2, double eps, double wd, double bc1, double bc2);
void multi_tensor_lerp(std::vector<at::Tensor> dsts, std::vector<at::Tensor> srcs, double weight);
void multi_tensor_lamb(
    std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
    std::vector<at::Tensor> exp_avgs, std::vector<at::Tensor> exp_avg_sqs,
    double lr, double beta1, double beta2, double beta3, double eps, double wd,
    double bc1, double bc2, double clip_norm, bool adapt, bool trust_clip);
at::Tensor multi_tensor_l2norm(std::vector<at::Tensor> tensors);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("layer_norm_fwd", &layer_norm_fwd, "fused LayerNorm fwd (gfx950)");
  m.def("layer_norm_bwd", &layer_norm_bwd, "fused LayerNorm bwd (gfx950)");
  m.def("rms_norm_fwd", &rms_norm_fwd, "fused RMSNorm fwd (gfx950)");
  m.def("rms_norm_bwd", &rms_norm_bwd, "fused RMSNorm bwd (gfx950)");
  m.def("bias_act_fwd", &bias_act_fwd, "fused bias+activation fwd (gfx950)");
  m.def("bias_act_bwd", &bias_act_bwd, "fused bias+activation bwd (gfx950)");
  m.def("residual_scale_add_fwd", &residual_scale_add_fwd, "residual+LayerScale+DropPath fwd");
  m.def("residual_scale_add_bwd", &residual_scale_add_bwd, "residual+LayerScale+DropPath bwd");
  m.def("attention_fwd", &attention_fwd, "flash attention fwd (MFMA, gfx950)");
  m.def("attn_bwd_softmax", &attn_bwd_softmax, "fused softmax recompute for attention bwd");
  m.def("attn_bwd_ds", &attn_bwd_ds, "fused dS epilogue for attention bwd");
  m.def("attn_bwd_preprocess", &attn_bwd_preprocess, "fused dO copy + delta for attention bwd");
  m.def("attention_bwd", &attention_bwd, "fused flash attention bwd (MFMA, gfx950)");
  m.def("dwconv_fwd", &dwconv_fwd, "NHWC depthwise conv fwd (gfx950)");
  m.def("dwconv_bwd_data", &dwconv_bwd_data, "NHWC depthwise conv bwd-data");
  m.def("dwconv_bwd_weight", &dwconv_bwd_weight, "NHWC depthwise conv bwd-weight");
  m.def("ns_gemm_nt", &ns_gemm_nt, "batched bf16 MFMA GEMM (L R^T + S) for Muon NS");
  m.def("ns_gemm_nn", &ns_gemm_nn, "batched bf16 MFMA GEMM (L R + S) for Muon NS");
  m.def("ce_loss_fwd", &ce_loss_fwd, "fused cross-entropy fwd (loss + lse)");
  m.def("ce_loss_bwd", &ce_loss_bwd, "fused cross-entropy bwd (dlogits from lse)");
  m.def("u8_normalize", &u8_normalize, "fused uint8 -> normalized tensor (loader prefetch)");
  m.def("masked_pool_fwd", &masked_pool_fwd, "masked global pool fwd (NaFlex)");
  m.def("masked_pool_bwd", &masked_pool_bwd, "masked global pool bwd (NaFlex)");
  m.def("multi_tensor_adamw", &multi_tensor_adamw, "fused multi-tensor AdamW step");
  m.def("multi_tensor_lerp", &multi_tensor_lerp, "fused multi-tensor lerp (EMA)");
  m.def("multi_tensor_lamb", &multi_tensor_lamb, "fused multi-tensor LAMB step (trust-ratio, gfx950)");
  m.def("multi_tensor_l2norm", &multi_tensor_l2norm, "fused multi-tensor global L2 norm");
}
