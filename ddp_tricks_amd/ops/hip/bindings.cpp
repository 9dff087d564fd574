// bindings.cpp — pybind module for the gfx950 kernel library.
#include <torch/extension.h>

// elementwise.hip
void multi_tensor_unscale(std::vector<at::Tensor> grads, at::Tensor found_inf,
                          double inv_scale);
void fused_sgd(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
               std::vector<at::Tensor> bufs, double lr, double momentum,
               double wd, double nesterov, c10::optional<at::Tensor> found_inf);
void fused_lookahead(std::vector<at::Tensor> fast, std::vector<at::Tensor> slow,
                     double alpha, c10::optional<at::Tensor> found_inf);
at::Tensor cast_to_bf16(at::Tensor x);
at::Tensor pad8_channels(at::Tensor x);
at::Tensor nhwc_flatten(at::Tensor x);
at::Tensor nhwc_unflatten(at::Tensor dy, long C, long H, long W);
std::vector<at::Tensor> pack_conv_weight(at::Tensor w, bool pad8,
                                         bool want_wt2);

// ce.hip
std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor target);
at::Tensor ce_bwd(at::Tensor logits, at::Tensor target, at::Tensor lse,
                  at::Tensor dloss);
at::Tensor argmax_correct(at::Tensor logits, at::Tensor target);

// pool.hip
std::vector<at::Tensor> maxpool2x2_fwd(at::Tensor x);
at::Tensor maxpool2x2_bwd(at::Tensor dy, at::Tensor idx, long H, long W);
std::vector<at::Tensor> maxpool_fwd(at::Tensor x, long ks, long st, long pad);
at::Tensor maxpool_bwd(at::Tensor dy, at::Tensor idx, long H, long W,
                       long ks, long st, long pad);
at::Tensor global_avgpool_fwd(at::Tensor x);
at::Tensor global_avgpool_bwd(at::Tensor dy, long H, long W);

// bn.hip
std::vector<at::Tensor> bn_fwd_train(at::Tensor x, at::Tensor gamma,
                                     at::Tensor beta, at::Tensor running_mean,
                                     at::Tensor running_var, double momentum,
                                     double eps, bool fuse_relu,
                                     c10::optional<at::Tensor> residual,
                                     c10::optional<at::Tensor> pre_stats);
at::Tensor bn_fwd_eval(at::Tensor x, at::Tensor gamma, at::Tensor beta,
                       at::Tensor running_mean, at::Tensor running_var,
                       double eps, bool fuse_relu,
                       c10::optional<at::Tensor> residual);
std::vector<at::Tensor> bn_fwd_eval_mask(at::Tensor x, at::Tensor gamma,
                                         at::Tensor beta,
                                         at::Tensor running_mean,
                                         at::Tensor running_var, double eps,
                                         bool fuse_relu,
                                         c10::optional<at::Tensor> residual);
std::vector<at::Tensor> bn_bwd(at::Tensor x, at::Tensor dy, at::Tensor gamma,
                               at::Tensor save_mean, at::Tensor save_invstd,
                               at::Tensor mask, bool fuse_relu,
                               bool want_dresid,
                               c10::optional<at::Tensor> pre_slab,
                               bool eval_stats);

// gemm.hip
at::Tensor gemm_tn(at::Tensor A, at::Tensor B, c10::optional<at::Tensor> bias,
                   bool out_f32);
at::Tensor transpose_bf16(at::Tensor x);
at::Tensor col_sum(at::Tensor x);
at::Tensor linear_fwd(at::Tensor x, at::Tensor w, c10::optional<at::Tensor> bias);
at::Tensor linear_dgrad(at::Tensor dy, at::Tensor w);
at::Tensor linear_wgrad(at::Tensor dy, at::Tensor x);

// conv.hip
at::Tensor conv2d_fwd(at::Tensor x, at::Tensor w, c10::optional<at::Tensor> bias,
                      long stride, long pad);
std::vector<at::Tensor> conv2d_fwd_stats(at::Tensor x, at::Tensor w,
                                         c10::optional<at::Tensor> bias,
                                         long stride, long pad);
at::Tensor conv2d_dgrad(at::Tensor dy, at::Tensor wt2, long N, long C, long H,
                        long W, long R, long S, long stride, long pad,
                        c10::optional<at::Tensor> addend);
std::vector<at::Tensor> conv2d_dgrad_bn(at::Tensor dy, at::Tensor wt2,
                                        long N, long C, long H, long W,
                                        long R, long S, long pad,
                                        at::Tensor bn_x, at::Tensor bn_mask,
                                        at::Tensor bn_mean,
                                        at::Tensor bn_invstd);
at::Tensor conv2d_wgrad(at::Tensor dy, at::Tensor x, long R, long S,
                        long stride, long pad);

// comm.cpp
void register_comm(py::module_& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    register_comm(m);
    m.def("multi_tensor_unscale", &multi_tensor_unscale,
          "fused unscale + inf check over grad tensors");
    m.def("fused_sgd", &fused_sgd, "fused nesterov-momentum SGD step",
          py::arg("params"), py::arg("grads"), py::arg("bufs"), py::arg("lr"),
          py::arg("momentum"), py::arg("wd"), py::arg("nesterov"),
          py::arg("found_inf") = c10::nullopt);
    m.def("fused_lookahead", &fused_lookahead, "fused Lookahead interpolation",
          py::arg("fast"), py::arg("slow"), py::arg("alpha"),
          py::arg("found_inf") = c10::nullopt);
    m.def("cast_to_bf16", &cast_to_bf16);
    m.def("pad8_channels", &pad8_channels);
    m.def("nhwc_flatten", &nhwc_flatten);
    m.def("nhwc_unflatten", &nhwc_unflatten);
    m.def("pack_conv_weight", &pack_conv_weight, py::arg("w"),
          py::arg("pad8") = false, py::arg("want_wt2") = true);
    m.def("ce_fwd", &ce_fwd);
    m.def("ce_bwd", &ce_bwd);
    m.def("argmax_correct", &argmax_correct);
    m.def("maxpool2x2_fwd", &maxpool2x2_fwd);
    m.def("maxpool2x2_bwd", &maxpool2x2_bwd);
    m.def("maxpool_fwd", &maxpool_fwd);
    m.def("maxpool_bwd", &maxpool_bwd);
    m.def("global_avgpool_fwd", &global_avgpool_fwd);
    m.def("global_avgpool_bwd", &global_avgpool_bwd);
    m.def("bn_fwd_train", &bn_fwd_train,
          py::arg("x"), py::arg("gamma"), py::arg("beta"),
          py::arg("running_mean"), py::arg("running_var"), py::arg("momentum"),
          py::arg("eps"), py::arg("fuse_relu") = false,
          py::arg("residual") = c10::nullopt,
          py::arg("pre_stats") = c10::nullopt);
    m.def("bn_fwd_eval", &bn_fwd_eval,
          py::arg("x"), py::arg("gamma"), py::arg("beta"),
          py::arg("running_mean"), py::arg("running_var"), py::arg("eps"),
          py::arg("fuse_relu") = false, py::arg("residual") = c10::nullopt);
    m.def("bn_bwd", &bn_bwd,
          py::arg("x"), py::arg("dy"), py::arg("gamma"), py::arg("save_mean"),
          py::arg("save_invstd"), py::arg("mask"), py::arg("fuse_relu") = false,
          py::arg("want_dresid") = false, py::arg("pre_slab") = c10::nullopt,
          py::arg("eval_stats") = false);
    m.def("bn_fwd_eval_mask", &bn_fwd_eval_mask,
          py::arg("x"), py::arg("gamma"), py::arg("beta"),
          py::arg("running_mean"), py::arg("running_var"), py::arg("eps"),
          py::arg("fuse_relu") = false, py::arg("residual") = c10::nullopt);
    m.def("gemm_tn", &gemm_tn, py::arg("A"), py::arg("B"),
          py::arg("bias") = c10::nullopt, py::arg("out_f32") = false);
    m.def("transpose_bf16", &transpose_bf16);
    m.def("col_sum", &col_sum);
    m.def("linear_fwd", &linear_fwd, py::arg("x"), py::arg("w"),
          py::arg("bias") = c10::nullopt);
    m.def("linear_dgrad", &linear_dgrad);
    m.def("linear_wgrad", &linear_wgrad);
    m.def("conv2d_fwd", &conv2d_fwd, py::arg("x"), py::arg("w"),
          py::arg("bias") = c10::nullopt, py::arg("stride") = 1,
          py::arg("pad") = 0);
    m.def("conv2d_fwd_stats", &conv2d_fwd_stats, py::arg("x"), py::arg("w"),
          py::arg("bias") = c10::nullopt, py::arg("stride") = 1,
          py::arg("pad") = 0);
    m.def("conv2d_dgrad", &conv2d_dgrad,
          py::arg("dy"), py::arg("wt2"), py::arg("N"), py::arg("C"),
          py::arg("H"), py::arg("W"), py::arg("R"), py::arg("S"),
          py::arg("stride"), py::arg("pad"),
          py::arg("addend") = c10::nullopt);
    m.def("conv2d_dgrad_bn", &conv2d_dgrad_bn);
    m.def("conv2d_wgrad", &conv2d_wgrad);
}
