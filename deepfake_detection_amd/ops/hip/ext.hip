// Python bindings for the gfx950 kernel set (module: _hip_ops).

#include <torch/extension.h>

#include <vector>

std::vector<at::Tensor> bn_act_fwd(at::Tensor x, at::Tensor weight, at::Tensor bias,
                                   at::Tensor running_mean, at::Tensor running_var,
                                   bool training, double momentum, double eps,
                                   std::string act, c10::optional<at::Tensor> residual,
                                   c10::optional<at::Tensor> stats,
                                   c10::optional<at::Tensor> drop_path);
std::vector<at::Tensor> bn_act_bwd(at::Tensor dy, at::Tensor x, at::Tensor weight,
                                   at::Tensor bias, at::Tensor save_mean,
                                   at::Tensor save_invstd, bool training, std::string act,
                                   c10::optional<at::Tensor> drop_path);
at::Tensor normalize_uint8_nhwc(at::Tensor x, at::Tensor mean, at::Tensor std,
                                std::string dtype, bool channels_last);
at::Tensor global_avg_pool_fwd(at::Tensor x);
at::Tensor global_avg_pool_bwd(at::Tensor dy, long long N, long long C, long long H,
                               long long W);
std::vector<at::Tensor> se_fwd(at::Tensor x, at::Tensor w1, at::Tensor b1, at::Tensor w2,
                               at::Tensor b2, std::string act);
std::vector<at::Tensor> se_bwd_reduce(at::Tensor dy, at::Tensor x, at::Tensor g);
void se_bwd_add_pool(at::Tensor dx, at::Tensor ds);
void rmsprop_tf_multi_tensor(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                             std::vector<at::Tensor> square_avgs,
                             std::vector<at::Tensor> momentum_buffers, double lr,
                             double alpha, double eps, double momentum, double weight_decay,
                             bool decoupled_decay, bool lr_in_momentum);
void adamw_multi_tensor(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                        std::vector<at::Tensor> exp_avgs, std::vector<at::Tensor> exp_avg_sqs,
                        long long step, double lr, double beta1, double beta2, double eps,
                        double weight_decay);
void ema_multi_tensor(std::vector<at::Tensor> ema_params, std::vector<at::Tensor> model_params,
                      double decay);
at::Tensor dw_conv2d_fwd(at::Tensor x, at::Tensor w_packed, int64_t sh, int64_t sw,
                         int64_t ph, int64_t pw, c10::optional<at::Tensor> stats);
at::Tensor dw_conv2d_bwd_data(at::Tensor dy, at::Tensor w_packed, int64_t H, int64_t W,
                              int64_t sh, int64_t sw, int64_t ph, int64_t pw);
at::Tensor dw_conv2d_bwd_weight(at::Tensor dy, at::Tensor x, int64_t K, int64_t sh,
                                int64_t sw, int64_t ph, int64_t pw);
at::Tensor pw_conv2d_fwd_mfma(at::Tensor x, at::Tensor w,
                              c10::optional<at::Tensor> stats);
at::Tensor pw_conv2d_bwd_weight_mfma(at::Tensor dy, at::Tensor x);
at::Tensor stem_conv2d_fwd(at::Tensor x, at::Tensor w_packed, int64_t n_out,
                           int64_t kh, int64_t kw, int64_t sh, int64_t sw,
                           int64_t ph, int64_t pw, c10::optional<at::Tensor> stats);
at::Tensor stem_conv2d_bwd_weight(at::Tensor dy, at::Tensor x, int64_t kh,
                                  int64_t kw, int64_t sh, int64_t sw, int64_t ph,
                                  int64_t pw);
std::vector<at::Tensor> head_ce_fwd(at::Tensor x, at::Tensor w, at::Tensor bias,
                                    at::Tensor target, double smoothing);
std::vector<at::Tensor> head_ce_bwd(at::Tensor dloss, at::Tensor logits,
                                    at::Tensor x, at::Tensor w, at::Tensor target,
                                    double smoothing);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "deepfake_detection_amd gfx950 (MI355X/CDNA4) kernels";
  m.def("bn_act_fwd", &bn_act_fwd, "fused BatchNorm+act(+residual) forward (NHWC)",
        py::arg("x"), py::arg("weight"), py::arg("bias"), py::arg("running_mean"),
        py::arg("running_var"), py::arg("training"), py::arg("momentum"),
        py::arg("eps"), py::arg("act"), py::arg("residual") = py::none(),
        py::arg("stats") = py::none(), py::arg("drop_path") = py::none());
  m.def("bn_act_bwd", &bn_act_bwd, "fused BatchNorm+act backward (NHWC)",
        py::arg("dy"), py::arg("x"), py::arg("weight"), py::arg("bias"),
        py::arg("save_mean"), py::arg("save_invstd"), py::arg("training"),
        py::arg("act"), py::arg("drop_path") = py::none());
  m.def("normalize_uint8_nhwc", &normalize_uint8_nhwc, "uint8 NCHW -> norm NHWC");
  m.def("global_avg_pool_fwd", &global_avg_pool_fwd, "global avg pool fwd (NHWC)");
  m.def("global_avg_pool_bwd", &global_avg_pool_bwd, "global avg pool bwd (NHWC)");
  m.def("se_fwd", &se_fwd, "fused squeeze-excite forward (NHWC)");
  m.def("se_bwd_reduce", &se_bwd_reduce, "SE backward: dx-direct + dgate");
  m.def("se_bwd_add_pool", &se_bwd_add_pool, "SE backward: add pooled ds");
  m.def("rmsprop_tf_multi_tensor", &rmsprop_tf_multi_tensor, "fused RMSpropTF step");
  m.def("adamw_multi_tensor", &adamw_multi_tensor, "fused AdamW step");
  m.def("ema_multi_tensor", &ema_multi_tensor, "fused EMA update");
  m.def("dw_conv2d_fwd", &dw_conv2d_fwd, "depthwise conv2d forward (NHWC)",
        py::arg("x"), py::arg("w_packed"), py::arg("sh"), py::arg("sw"),
        py::arg("ph"), py::arg("pw"), py::arg("stats") = py::none());
  m.def("dw_conv2d_bwd_data", &dw_conv2d_bwd_data, "depthwise conv2d bwd data (NHWC)");
  m.def("dw_conv2d_bwd_weight", &dw_conv2d_bwd_weight, "depthwise conv2d bwd weight (NHWC)");
  m.def("pw_conv2d_fwd_mfma", &pw_conv2d_fwd_mfma, "1x1 conv as MFMA GEMM (NHWC)",
        py::arg("x"), py::arg("w"), py::arg("stats") = py::none());
  m.def("pw_conv2d_bwd_weight_mfma", &pw_conv2d_bwd_weight_mfma,
        "1x1 conv weight grad: split-M MFMA + fp32 chunk reduce");
  m.def("stem_conv2d_fwd", &stem_conv2d_fwd, "stem conv implicit-GEMM fwd (NHWC)",
        py::arg("x"), py::arg("w_packed"), py::arg("n_out"), py::arg("kh"),
        py::arg("kw"), py::arg("sh"), py::arg("sw"), py::arg("ph"), py::arg("pw"),
        py::arg("stats") = py::none());
  m.def("stem_conv2d_bwd_weight", &stem_conv2d_bwd_weight,
        "stem conv implicit-GEMM weight grad");
  m.def("head_ce_fwd", &head_ce_fwd, "fused classifier GEMM + smoothed-CE fwd");
  m.def("head_ce_bwd", &head_ce_bwd, "fused classifier + CE backward");
}
