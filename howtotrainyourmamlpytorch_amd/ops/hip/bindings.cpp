// Python bindings for the MI355X (gfx950) kernel extension.
#include <torch/extension.h>
#include <vector>

// bn_act.hip
std::vector<torch::Tensor> bn_act_fwd(torch::Tensor x, torch::Tensor gamma,
                                      torch::Tensor beta, double eps,
                                      double slope, bool act);
std::vector<torch::Tensor> bn_act_bwd(torch::Tensor dy, torch::Tensor x,
                                      torch::Tensor mean, torch::Tensor rstd,
                                      torch::Tensor gamma, torch::Tensor beta,
                                      double slope, bool act);
std::vector<torch::Tensor> bn_act_pool_fwd(torch::Tensor x, torch::Tensor gamma,
                                           torch::Tensor beta, double eps,
                                           double slope,
                                           c10::optional<torch::Tensor> sums_in);
std::vector<torch::Tensor> bn_act_pool_bwd(torch::Tensor dyp, torch::Tensor mask,
                                           torch::Tensor x, torch::Tensor mean,
                                           torch::Tensor rstd, torch::Tensor gamma,
                                           torch::Tensor beta, double slope);
// pool.hip
std::vector<torch::Tensor> maxpool2x2_fwd(torch::Tensor x);
torch::Tensor maxpool2x2_bwd(torch::Tensor dy, torch::Tensor mask, long H, long W);
// softmax_ce.hip
std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor labels);
torch::Tensor ce_bwd(torch::Tensor probs, torch::Tensor labels, torch::Tensor gtask);
std::vector<torch::Tensor> ce_dbwd(torch::Tensor probs, torch::Tensor labels,
                                   torch::Tensor gdl, torch::Tensor gtask);
// bn_dbwd.hip
std::vector<torch::Tensor> bn_act_dbwd(torch::Tensor x, torch::Tensor u,
                                       torch::Tensor gx, torch::Tensor mean,
                                       torch::Tensor rstd, torch::Tensor gamma,
                                       torch::Tensor beta, torch::Tensor ggam,
                                       torch::Tensor gbet, double slope, bool act);
// lslr.hip
torch::Tensor lslr_fwd(torch::Tensor arena, torch::Tensor grad, torch::Tensor lr_vec);
std::vector<torch::Tensor> lslr_bwd(torch::Tensor gout, torch::Tensor grad,
                                    torch::Tensor lr_vec);
// tconv.hip
torch::Tensor tconv_repack(torch::Tensor w, bool dgrad);
std::vector<torch::Tensor> tconv_mm(torch::Tensor x, torch::Tensor wp,
                                    c10::optional<torch::Tensor> bias, long pad,
                                    long Ho, long Wo, bool with_stats);
std::vector<torch::Tensor> tconv_wgrad(torch::Tensor dy, torch::Tensor x,
                                       long pad, bool with_bias);
torch::Tensor tconv_repack_v2(torch::Tensor w, bool dgrad);
std::vector<torch::Tensor> tconv_wgrad_v2(torch::Tensor dy, torch::Tensor x,
                                          long pad, bool with_bias);
// dconv.hip
torch::Tensor dconv_fwd(torch::Tensor x, torch::Tensor w,
                        c10::optional<torch::Tensor> bias, long pad,
                        long Ho, long Wo);
std::vector<torch::Tensor> dconv_wgrad(torch::Tensor dy, torch::Tensor x,
                                       long pad, bool with_bias);
// linear.hip
torch::Tensor lin_fwd(torch::Tensor x, torch::Tensor w,
                      c10::optional<torch::Tensor> b);
torch::Tensor lin_dx(torch::Tensor dy, torch::Tensor w);
std::vector<torch::Tensor> lin_wgrad(torch::Tensor dy, torch::Tensor x,
                                     bool with_bias);
std::vector<torch::Tensor> tconv_mm_v2(torch::Tensor x, torch::Tensor wimg,
                                       c10::optional<torch::Tensor> bias,
                                       long pad, long Ho, long Wo, long Co,
                                       bool with_stats);
std::vector<torch::Tensor> mfma_probe(torch::Tensor A, torch::Tensor B);
// adam.hip
void adam_step(std::vector<torch::Tensor> params,
               std::vector<torch::Tensor> grads,
               std::vector<torch::Tensor> exp_avgs,
               std::vector<torch::Tensor> exp_avg_sqs,
               long step, double lr, double beta1, double beta2, double eps,
               double weight_decay, double clamp_v);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("bn_act_fwd", &bn_act_fwd, "fused task-batched BN+leakyReLU fwd");
  m.def("bn_act_bwd", &bn_act_bwd, "fused task-batched BN+leakyReLU bwd");
  m.def("bn_act_pool_fwd", &bn_act_pool_fwd, "fused BN+leakyReLU+maxpool fwd");
  m.def("bn_act_pool_bwd", &bn_act_pool_bwd, "fused BN+leakyReLU+maxpool bwd");
  m.def("maxpool2x2_fwd", &maxpool2x2_fwd, "NHWC maxpool 2x2 fwd");
  m.def("maxpool2x2_bwd", &maxpool2x2_bwd, "NHWC maxpool 2x2 bwd");
  m.def("ce_fwd", &ce_fwd, "fused softmax-CE fwd");
  m.def("ce_dbwd", &ce_dbwd, "softmax-CE double-backward");
  m.def("bn_act_dbwd", &bn_act_dbwd, "BN+leakyReLU analytic double-backward");
  m.def("ce_bwd", &ce_bwd, "fused softmax-CE bwd");
  m.def("lslr_fwd", &lslr_fwd, "fused LSLR arena update fwd");
  m.def("lslr_bwd", &lslr_bwd, "fused LSLR arena update bwd");
  m.def("tconv_repack", &tconv_repack, "repack conv weights for MFMA (fwd/dgrad)");
  m.def("tconv_mm", &tconv_mm, "task-batched MFMA 3x3 conv fwd/dgrad");
  m.def("tconv_wgrad", &tconv_wgrad, "task-batched MFMA 3x3 conv wgrad");
  m.def("tconv_repack_v2", &tconv_repack_v2,
        "repack conv weights into the v2 swizzled LDS image");
  m.def("tconv_mm_v2", &tconv_mm_v2,
        "async-pipelined MFMA 3x3 conv fwd/dgrad (v2)");
  m.def("tconv_wgrad_v2", &tconv_wgrad_v2,
        "async-pipelined wgrad via k-contiguous operand transposes (v2)");
  m.def("dconv_fwd", &dconv_fwd, "direct small-C conv fwd (VALU)");
  m.def("dconv_wgrad", &dconv_wgrad, "direct small-C conv wgrad (VALU)");
  m.def("lin_fwd", &lin_fwd, "task-batched linear head fwd");
  m.def("lin_dx", &lin_dx, "linear head input-grad");
  m.def("lin_wgrad", &lin_wgrad, "linear head weight/bias grads");
  m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA layout probe");
  m.def("adam_step", &adam_step, "fused multi-tensor Adam + grad clamp");
}
