// Python bindings for the NoisyNet-MI355X HIP extension.

#include <torch/extension.h>

// elementwise.hip
torch::Tensor fake_quant_fwd(torch::Tensor x, int64_t num_bits, double min_value,
                             double max_value, double stochastic, int64_t seed);
torch::Tensor ste_mask(torch::Tensor grad, torch::Tensor x, double min_value,
                       double max_value);
torch::Tensor mult_uniform_noise(torch::Tensor x, double a, int64_t seed);
torch::Tensor relu_clip_fwd(torch::Tensor x, bool relu, double act_max);
torch::Tensor relu_clip_bwd(torch::Tensor g, torch::Tensor y, bool relu,
                            double act_max);
std::vector<torch::Tensor> dropout_fwd(torch::Tensor x, double p, int64_t seed);
void set_seed_buffer(torch::Tensor t);
void clear_seed_buffer();

// optimizer.hip
void sgd_step(torch::Tensor p, torch::Tensor g, torch::Tensor buf, double lr,
              double momentum, double wd, bool nesterov, double cmin,
              double cmax);
void adamw_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                torch::Tensor v, int64_t step, double lr, double beta1,
                double beta2, double eps, double wd, double cmin, double cmax);

// softmax_xent.hip
std::vector<torch::Tensor> softmax_xent_fwd(torch::Tensor logits,
                                            torch::Tensor target);
torch::Tensor softmax_xent_bwd(torch::Tensor softmax, torch::Tensor target,
                               double gscale);
torch::Tensor softmax_xent_bwd_t(torch::Tensor softmax, torch::Tensor target,
                                 torch::Tensor gscale);

// bn_act.hip
std::vector<torch::Tensor> bn_stats(torch::Tensor x);
std::vector<torch::Tensor> bn_stats_finalize(torch::Tensor x,
                                             torch::Tensor running_mean,
                                             torch::Tensor running_var,
                                             double momentum, double eps);
torch::Tensor bn_act_fwd(torch::Tensor x, torch::Tensor mean,
                         torch::Tensor invstd, torch::Tensor gamma,
                         torch::Tensor beta, bool relu, double act_max);
std::vector<torch::Tensor> bn_act_bwd(torch::Tensor g, torch::Tensor x,
                                      torch::Tensor y, torch::Tensor mean,
                                      torch::Tensor invstd, torch::Tensor gamma,
                                      bool training, bool relu,
                                      double act_max);
std::vector<torch::Tensor> bn_act_bwd_reduce(torch::Tensor g, torch::Tensor x,
                                             torch::Tensor y, torch::Tensor mean,
                                             torch::Tensor invstd, bool relu,
                                             double act_max);
torch::Tensor bn_act_bwd_apply(torch::Tensor g, torch::Tensor x,
                               torch::Tensor y, torch::Tensor mean,
                               torch::Tensor invstd, torch::Tensor gamma,
                               torch::Tensor sum_g, torch::Tensor sum_gx,
                               double count, bool training, bool relu,
                               double act_max);

// pool.hip
std::vector<torch::Tensor> maxpool2x2_fwd(torch::Tensor x);
torch::Tensor maxpool2x2_bwd(torch::Tensor g, torch::Tensor code, int64_t H,
                             int64_t W);
std::vector<torch::Tensor> maxpool_fwd(torch::Tensor x, int64_t k,
                                       int64_t stride, int64_t pad);
torch::Tensor maxpool_bwd(torch::Tensor g, torch::Tensor code, int64_t H,
                          int64_t W, int64_t k, int64_t stride, int64_t pad);
torch::Tensor avgpool_fwd(torch::Tensor x, int64_t k, int64_t stride,
                          int64_t pad);
torch::Tensor avgpool_bwd(torch::Tensor g, int64_t H, int64_t W, int64_t k,
                          int64_t stride, int64_t pad);

// depthwise.hip
torch::Tensor dwconv_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor bias,
                         int64_t stride, int64_t pad);
torch::Tensor dwconv_dgrad(torch::Tensor gy, torch::Tensor w, int64_t stride,
                           int64_t pad, int64_t H, int64_t W);
torch::Tensor dwconv_wgrad(torch::Tensor gy, torch::Tensor x, int64_t stride,
                           int64_t pad, int64_t R, int64_t S);

// activations.hip
torch::Tensor act_fwd(torch::Tensor x, int64_t act);
torch::Tensor se_scale_fwd(torch::Tensor x, torch::Tensor s, int64_t act);
std::vector<torch::Tensor> se_scale_bwd(torch::Tensor g, torch::Tensor x,
                                        torch::Tensor s, int64_t act);
torch::Tensor act_bwd(torch::Tensor g, torch::Tensor x, int64_t act);

// percentile.hip
torch::Tensor kth_percentile(torch::Tensor x, double pctl);

// conv_mfma.hip
torch::Tensor conv_fwd(torch::Tensor x, torch::Tensor w, int64_t stride,
                       int64_t pad);
std::vector<torch::Tensor> conv_fwd_fused(torch::Tensor x, torch::Tensor wq,
                                          torch::Tensor wraw, torch::Tensor bias,
                                          int64_t stride, int64_t pad,
                                          int64_t sigma_mode,
                                          torch::Tensor factor,
                                          int64_t seed, bool telem);
torch::Tensor conv_dgrad(torch::Tensor gy, torch::Tensor w, int64_t stride,
                         int64_t pad, int64_t H, int64_t W);
torch::Tensor conv_wgrad(torch::Tensor gy, torch::Tensor x, int64_t stride,
                         int64_t pad, int64_t R, int64_t S);
torch::Tensor im2col_materialize(torch::Tensor x, int64_t K, int64_t stride,
                                 int64_t pad, int64_t R, int64_t S);
torch::Tensor conv_wgrad_from_col(torch::Tensor gy, torch::Tensor col,
                                  int64_t C, int64_t R, int64_t S);
torch::Tensor conv_wgrad_im2col(torch::Tensor gy, torch::Tensor x,
                                int64_t stride, int64_t pad, int64_t R,
                                int64_t S);
torch::Tensor conv_wgrad_patch(torch::Tensor gy, torch::Tensor x,
                               int64_t stride, int64_t pad, int64_t R,
                               int64_t S);
torch::Tensor linear_fwd(torch::Tensor x, torch::Tensor w);
torch::Tensor linear_dgrad(torch::Tensor gy, torch::Tensor w);
torch::Tensor linear_wgrad(torch::Tensor gy, torch::Tensor x);
std::vector<torch::Tensor> linear_fwd_fused(torch::Tensor x, torch::Tensor wq,
                                            torch::Tensor wraw,
                                            torch::Tensor bias,
                                            int64_t sigma_mode,
                                            torch::Tensor factor,
                                            int64_t seed, bool telem);
std::vector<torch::Tensor> sigma_noise_conv(torch::Tensor x, torch::Tensor wraw,
                                            int64_t stride, int64_t pad,
                                            int64_t sigma_mode,
                                            torch::Tensor factor,
                                            int64_t seed, bool telem);
std::vector<torch::Tensor> sigma_noise_linear_impl(torch::Tensor x,
                                                   torch::Tensor wraw,
                                                   int64_t sigma_mode,
                                                   torch::Tensor factor,
                                                   int64_t seed,
                                                   bool telem);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("fake_quant_fwd", &fake_quant_fwd);
  m.def("ste_mask", &ste_mask);
  m.def("mult_uniform_noise", &mult_uniform_noise);
  m.def("relu_clip_fwd", &relu_clip_fwd);
  m.def("relu_clip_bwd", &relu_clip_bwd);
  m.def("dropout_fwd", &dropout_fwd);
  m.def("set_seed_buffer", &set_seed_buffer);
  m.def("clear_seed_buffer", &clear_seed_buffer);
  m.def("sgd_step", &sgd_step);
  m.def("adamw_step", &adamw_step);
  m.def("softmax_xent_fwd", &softmax_xent_fwd);
  m.def("softmax_xent_bwd", &softmax_xent_bwd);
  m.def("softmax_xent_bwd_t", &softmax_xent_bwd_t);
  m.def("bn_stats", &bn_stats);
  m.def("bn_stats_finalize", &bn_stats_finalize);
  m.def("bn_act_fwd", &bn_act_fwd);
  m.def("bn_act_bwd", &bn_act_bwd);
  m.def("bn_act_bwd_reduce", &bn_act_bwd_reduce);
  m.def("bn_act_bwd_apply", &bn_act_bwd_apply);
  m.def("maxpool2x2_fwd", &maxpool2x2_fwd);
  m.def("maxpool2x2_bwd", &maxpool2x2_bwd);
  m.def("dwconv_fwd", &dwconv_fwd);
  m.def("dwconv_dgrad", &dwconv_dgrad);
  m.def("dwconv_wgrad", &dwconv_wgrad);
  m.def("act_fwd", &act_fwd);
  m.def("act_bwd", &act_bwd);
  m.def("se_scale_fwd", &se_scale_fwd);
  m.def("se_scale_bwd", &se_scale_bwd);
  m.def("maxpool_fwd", &maxpool_fwd);
  m.def("maxpool_bwd", &maxpool_bwd);
  m.def("avgpool_fwd", &avgpool_fwd);
  m.def("avgpool_bwd", &avgpool_bwd);
  m.def("kth_percentile", &kth_percentile);
  m.def("conv_fwd", &conv_fwd);
  m.def("conv_fwd_fused", &conv_fwd_fused);
  m.def("conv_dgrad", &conv_dgrad);
  m.def("conv_wgrad", &conv_wgrad);
  m.def("conv_wgrad_im2col", &conv_wgrad_im2col);
  m.def("conv_wgrad_patch", &conv_wgrad_patch);
  m.def("conv_wgrad_from_col", &conv_wgrad_from_col);
  m.def("im2col_materialize", &im2col_materialize);
  m.def("linear_fwd", &linear_fwd);
  m.def("linear_dgrad", &linear_dgrad);
  m.def("linear_wgrad", &linear_wgrad);
  m.def("linear_fwd_fused", &linear_fwd_fused);
  m.def("sigma_noise_conv", &sigma_noise_conv);
  m.def("sigma_noise_linear", &sigma_noise_linear_impl);
}
