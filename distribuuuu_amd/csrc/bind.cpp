// Python bindings for the gfx950 kernel set.
#include <torch/extension.h>

// elementwise.hip
at::Tensor relu_fwd(at::Tensor x);
at::Tensor relu_bwd(at::Tensor gy, at::Tensor y);
at::Tensor add_relu_fwd(at::Tensor a, at::Tensor b);
at::Tensor se_scale_fwd(at::Tensor x, at::Tensor s);
at::Tensor dropout_fwd(at::Tensor x, double p, int64_t seed);
std::vector<at::Tensor> se_scale_bwd(at::Tensor gy, at::Tensor x,
                                     at::Tensor s);
// bn.hip
std::vector<at::Tensor> bn_sums(at::Tensor x);
at::Tensor bn_reduce_partials(at::Tensor part);
std::vector<at::Tensor> bn_stats(at::Tensor x, at::Tensor gamma,
                                 at::Tensor beta,
                                 c10::optional<at::Tensor> rm_opt,
                                 c10::optional<at::Tensor> rv_opt,
                                 double momentum, double eps, bool training,
                                 c10::optional<at::Tensor> part_opt);
at::Tensor bn_apply_act(at::Tensor x, at::Tensor scale, at::Tensor shift,
                        int64_t act, c10::optional<at::Tensor> res);
at::Tensor bn_bwd_stats(at::Tensor gy, at::Tensor x,
                        c10::optional<at::Tensor> res, at::Tensor scale,
                        at::Tensor shift, int64_t act);
std::vector<at::Tensor> bn_bwd_apply(at::Tensor gy, at::Tensor x,
                                     c10::optional<at::Tensor> res,
                                     at::Tensor mean, at::Tensor rstd,
                                     at::Tensor gamma, at::Tensor scale,
                                     at::Tensor shift, at::Tensor sums,
                                     double total_count, int64_t act,
                                     bool training, bool need_gres);
std::vector<at::Tensor> bn_bwd(at::Tensor gy, at::Tensor x,
                               c10::optional<at::Tensor> res, at::Tensor mean,
                               at::Tensor rstd, at::Tensor gamma,
                               at::Tensor scale, at::Tensor shift,
                               int64_t act, bool training, bool need_gres,
                               c10::optional<at::Tensor> mask);
std::vector<at::Tensor> bn_apply_act_mask(at::Tensor x, at::Tensor scale,
                                          at::Tensor shift, int64_t act,
                                          at::Tensor res);
at::Tensor bn_apply_act_pad(at::Tensor x, at::Tensor scale, at::Tensor shift,
                            int64_t act, int64_t ph, int64_t pw);
std::vector<at::Tensor> bn_bwd_pad(at::Tensor gy, at::Tensor x,
                                   at::Tensor mean, at::Tensor rstd,
                                   at::Tensor gamma, at::Tensor scale,
                                   at::Tensor shift, int64_t act,
                                   bool training, int64_t ph, int64_t pw);
// pool.hip
std::vector<at::Tensor> maxpool_fwd(at::Tensor x, int64_t K, int64_t S,
                                    int64_t P);
at::Tensor maxpool_bwd(at::Tensor gy, at::Tensor idx, int64_t H, int64_t W,
                       int64_t K, int64_t S, int64_t P);
at::Tensor gap_fwd(at::Tensor x);
at::Tensor gap_bwd(at::Tensor gy, int64_t H, int64_t W);
at::Tensor avgpool_fwd(at::Tensor x, int64_t K, int64_t S);
at::Tensor avgpool_bwd(at::Tensor gy, int64_t K, int64_t S, int64_t H,
                       int64_t W);
// loss.hip
std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor target);
at::Tensor ce_bwd(at::Tensor logits, at::Tensor target, at::Tensor lse,
                  at::Tensor gl);
std::vector<at::Tensor> topk_acc(at::Tensor logits, at::Tensor target,
                                 int64_t topk);
// conv.hip
at::Tensor conv2d_fwd_v1(at::Tensor x, at::Tensor w, at::Tensor y,
                         int64_t Ho, int64_t Wo, int64_t sh, int64_t sw,
                         int64_t ph, int64_t pw, int64_t dh, int64_t dw,
                         int64_t groups);
at::Tensor conv2d_fwd(at::Tensor x, at::Tensor w, int64_t sh, int64_t sw,
                      int64_t ph, int64_t pw, int64_t dh, int64_t dw,
                      int64_t groups);
std::vector<at::Tensor> conv2d_fwd_bn(at::Tensor x, at::Tensor w, int64_t sh,
                                      int64_t sw, int64_t ph, int64_t pw,
                                      int64_t dh, int64_t dw, int64_t groups);
at::Tensor conv2d_dgrad(at::Tensor gy, at::Tensor w, int64_t H, int64_t W,
                        int64_t sh, int64_t sw, int64_t ph, int64_t pw,
                        int64_t dh, int64_t dw, int64_t groups);
std::vector<at::Tensor> conv2d_dgrad_bn(at::Tensor gy, at::Tensor w,
                                        int64_t H, int64_t W, int64_t sh,
                                        int64_t sw, int64_t ph, int64_t pw,
                                        int64_t dh, int64_t dw,
                                        int64_t groups, at::Tensor bnx,
                                        at::Tensor bnscale, at::Tensor bnshift,
                                        int64_t bnact);

std::tuple<at::Tensor, int64_t> conv2d_dgrad_acc(
    at::Tensor gy, at::Tensor w, int64_t H, int64_t W, int64_t sh, int64_t sw,
    int64_t ph, int64_t pw, int64_t dh, int64_t dw, int64_t groups,
    at::Tensor into, c10::optional<at::Tensor> pre_opt, int64_t pre_kind);

std::vector<at::Tensor> conv2d_dgrad_prep(at::Tensor w, int64_t Kt0,
                                          int64_t sh, int64_t sw, int64_t ph,
                                          int64_t pw, int64_t dh, int64_t dw,
                                          int64_t groups);

at::Tensor conv2d_dgrad_pre(at::Tensor gy, at::Tensor w, int64_t H,
                            int64_t W, int64_t sh, int64_t sw, int64_t ph,
                            int64_t pw, int64_t dh, int64_t dw,
                            int64_t groups, at::Tensor pre,
                            int64_t pre_kind);
at::Tensor gemm_nt(at::Tensor a, at::Tensor b);
at::Tensor conv2d_fwd_v2(at::Tensor x, at::Tensor w, int64_t sh, int64_t sw,
                         int64_t ph, int64_t pw, int64_t dh, int64_t dw);
at::Tensor weight_flip_t(at::Tensor w, int64_t groups);
at::Tensor dilate_nhwc(at::Tensor x, int64_t sh, int64_t sw);
at::Tensor pad_channels(at::Tensor x, int64_t Cn);
at::Tensor conv2d_wgrad(at::Tensor gy, at::Tensor x, int64_t R, int64_t S,
                        int64_t sh, int64_t sw, int64_t ph, int64_t pw,
                        int64_t dh, int64_t dw, int64_t groups);
// depthwise.hip
at::Tensor dwconv_fwd(at::Tensor x, at::Tensor w, int64_t sh, int64_t sw,
                      int64_t ph, int64_t pw);
at::Tensor dwconv_dgrad(at::Tensor gy, at::Tensor w, int64_t H, int64_t W,
                        int64_t sh, int64_t sw, int64_t ph, int64_t pw);
at::Tensor dwconv_wgrad(at::Tensor gy, at::Tensor x, int64_t R, int64_t S,
                        int64_t sh, int64_t sw, int64_t ph, int64_t pw);
// attention.hip
at::Tensor mhsa_fwd(at::Tensor q, at::Tensor k, at::Tensor vt, at::Tensor rw,
                    at::Tensor rh, int64_t H, int64_t W,
                    c10::optional<at::Tensor> pout, int64_t heads,
                    int64_t qpix, int64_t kpix, double scale);
std::vector<at::Tensor> mhsa_bwd(at::Tensor dO, at::Tensor P, at::Tensor q,
                                 at::Tensor kt, at::Tensor v,
                                 at::Tensor rel_w, at::Tensor rel_h,
                                 int64_t H, int64_t W, int64_t heads,
                                 int64_t qpix, int64_t kqoff, int64_t vpix,
                                 int64_t opix, double scale,
                                 at::Tensor dqk_out, at::Tensor dv_out);
at::Tensor mhsa_rel_tables(at::Tensor q, at::Tensor rel, int64_t rows,
                           int64_t heads, int64_t L, int64_t qpix,
                           double scale);
// augment.hip
at::Tensor aug_crop_flip_norm(at::Tensor raw, at::Tensor meta, int64_t S,
                              std::vector<double> mean,
                              std::vector<double> std, at::ScalarType dtype);
// sgd.hip
void sgd_step(at::Tensor table, double lr, double momentum,
              double dampening, double weight_decay, bool nesterov);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("relu_fwd", &relu_fwd);
  m.def("relu_bwd", &relu_bwd);
  m.def("add_relu_fwd", &add_relu_fwd);
  m.def("se_scale_fwd", &se_scale_fwd);
  m.def("dropout_fwd", &dropout_fwd);
  m.def("se_scale_bwd", &se_scale_bwd);
  m.def("bn_sums", &bn_sums);
  m.def("bn_reduce_partials", &bn_reduce_partials);
  m.def("bn_stats", &bn_stats, py::arg("x"), py::arg("gamma"), py::arg("beta"),
        py::arg("rm") = py::none(), py::arg("rv") = py::none(),
        py::arg("momentum") = 0.1, py::arg("eps") = 1e-5,
        py::arg("training") = true, py::arg("part") = py::none());
  m.def("bn_apply_act", &bn_apply_act, py::arg("x"), py::arg("scale"),
        py::arg("shift"), py::arg("act"), py::arg("res") = py::none());
  m.def("bn_apply_act_mask", &bn_apply_act_mask);
  m.def("bn_apply_act_pad", &bn_apply_act_pad);
  m.def("bn_bwd_pad", &bn_bwd_pad);
  m.def("bn_bwd_stats", &bn_bwd_stats, py::arg("gy"), py::arg("x"),
        py::arg("res"), py::arg("scale"), py::arg("shift"),
        py::arg("act"));
  m.def("bn_bwd_apply", &bn_bwd_apply, py::arg("gy"), py::arg("x"),
        py::arg("res"), py::arg("mean"), py::arg("rstd"),
        py::arg("gamma"), py::arg("scale"), py::arg("shift"), py::arg("sums"),
        py::arg("total_count"), py::arg("act"), py::arg("training"),
        py::arg("need_gres"));
  m.def("bn_bwd", &bn_bwd, py::arg("gy"), py::arg("x"),
        py::arg("res"), py::arg("mean"), py::arg("rstd"), py::arg("gamma"),
        py::arg("scale"), py::arg("shift"), py::arg("act"),
        py::arg("training"), py::arg("need_gres"),
        py::arg("mask") = py::none());
  m.def("maxpool_fwd", &maxpool_fwd);
  m.def("maxpool_bwd", &maxpool_bwd);
  m.def("gap_fwd", &gap_fwd);
  m.def("gap_bwd", &gap_bwd);
  m.def("avgpool_fwd", &avgpool_fwd);
  m.def("avgpool_bwd", &avgpool_bwd);
  m.def("ce_fwd", &ce_fwd);
  m.def("ce_bwd", &ce_bwd);
  m.def("topk_acc", &topk_acc);
  m.def("sgd_step", &sgd_step);
  m.def("conv2d_fwd", &conv2d_fwd);
  m.def("conv2d_fwd_v1", &conv2d_fwd_v1);
  m.def("conv2d_fwd_bn", &conv2d_fwd_bn);
  m.def("conv2d_dgrad", &conv2d_dgrad);
  m.def("conv2d_dgrad_bn", &conv2d_dgrad_bn);
  m.def("conv2d_dgrad_acc", &conv2d_dgrad_acc);
  m.def("conv2d_dgrad_prep", &conv2d_dgrad_prep);
  m.def("conv2d_dgrad_pre", &conv2d_dgrad_pre);
  m.def("conv2d_wgrad", &conv2d_wgrad);
  m.def("gemm_nt", &gemm_nt);
  m.def("conv2d_fwd_v2", &conv2d_fwd_v2);
  m.def("weight_flip_t", &weight_flip_t);
  m.def("dilate_nhwc", &dilate_nhwc);
  m.def("pad_channels", &pad_channels);
  m.def("dwconv_fwd", &dwconv_fwd);
  m.def("dwconv_dgrad", &dwconv_dgrad);
  m.def("dwconv_wgrad", &dwconv_wgrad);
  m.def("mhsa_fwd", &mhsa_fwd);
  m.def("mhsa_bwd", &mhsa_bwd);
  m.def("mhsa_rel_tables", &mhsa_rel_tables);
  m.def("aug_crop_flip_norm", &aug_crop_flip_norm);
}
