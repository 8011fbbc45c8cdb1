// Python bindings for the p2pvg_amd gfx950 kernel library.
#include <torch/extension.h>

#include <vector>

std::vector<torch::Tensor> lstm_cell_fwd(torch::Tensor x, torch::Tensor h,
                                         torch::Tensor c, torch::Tensor w_ih,
                                         torch::Tensor w_hh, torch::Tensor b_ih,
                                         torch::Tensor b_hh);
std::vector<torch::Tensor> lstm_cell_bwd_pointwise(
    torch::Tensor dh, c10::optional<torch::Tensor> dc_in, torch::Tensor gates,
    torch::Tensor c_prev, torch::Tensor c_new);
void multi_tensor_adam(std::vector<torch::Tensor> params,
                       std::vector<torch::Tensor> grads,
                       std::vector<torch::Tensor> exp_avgs,
                       std::vector<torch::Tensor> exp_avg_sqs, double lr,
                       double beta1, double beta2, double eps,
                       double weight_decay, torch::Tensor step);
torch::Tensor gaussian_kl_fwd(torch::Tensor mu1, torch::Tensor lv1,
                              torch::Tensor mu2, torch::Tensor lv2,
                              double denom);
std::vector<torch::Tensor> gaussian_kl_bwd(torch::Tensor mu1, torch::Tensor lv1,
                                           torch::Tensor mu2, torch::Tensor lv2,
                                           torch::Tensor dout, double denom);

torch::Tensor affine4_fwd(torch::Tensor h, torch::Tensor g, torch::Tensor s1,
                          torch::Tensor s2, torch::Tensor W,
                          c10::optional<torch::Tensor> b);
std::vector<torch::Tensor> affine4_bwd(
    torch::Tensor gout, torch::Tensor h, torch::Tensor g, torch::Tensor s1,
    torch::Tensor s2, torch::Tensor W, c10::optional<torch::Tensor> dW_acc,
    c10::optional<torch::Tensor> db_acc, bool need_dh, bool need_dg);
std::vector<torch::Tensor> gauss_head_fwd(torch::Tensor hin, torch::Tensor Wm,
                                          torch::Tensor bm, torch::Tensor Wl,
                                          torch::Tensor bl, torch::Tensor eps,
                                          c10::optional<torch::Tensor> Ws_c,
                                          c10::optional<torch::Tensor> bs_c);
torch::Tensor gauss_head_bwd(
    c10::optional<torch::Tensor> dz, c10::optional<torch::Tensor> dmu,
    c10::optional<torch::Tensor> dlv, torch::Tensor eps, torch::Tensor lv,
    torch::Tensor hin, torch::Tensor Ws, long N_,
    c10::optional<torch::Tensor> dWm, c10::optional<torch::Tensor> dbm,
    c10::optional<torch::Tensor> dWl, c10::optional<torch::Tensor> dbl);
torch::Tensor tanh_head_fwd(torch::Tensor hin, torch::Tensor W,
                            torch::Tensor b);
torch::Tensor tanh_head_bwd(torch::Tensor dy, torch::Tensor y,
                            torch::Tensor hin, torch::Tensor W,
                            c10::optional<torch::Tensor> dW_acc,
                            c10::optional<torch::Tensor> db_acc);
torch::Tensor mfma_probe(torch::Tensor A, torch::Tensor B);
torch::Tensor tr16_probe(long mode);
std::vector<torch::Tensor> conv2d_nhwc_fwd(torch::Tensor in, torch::Tensor w,
                                            c10::optional<torch::Tensor> bias,
                                            long stride, long pad, long act,
                                            bool want_stats, long oh, long ow,
                                            long oy0, long ox0);
std::vector<torch::Tensor> conv2d_nhwc_fracstride(
    torch::Tensor in, torch::Tensor w, c10::optional<torch::Tensor> bias,
    long up_stride, long up_pad, long OH, long OW, long act, bool want_stats,
    long in_ring, long out_ring);
std::vector<torch::Tensor> conv2d_glds_fwd(
    torch::Tensor in, torch::Tensor w, c10::optional<torch::Tensor> bias,
    long stride, long act, bool want_stats, long oh, long ow, long oy0,
    long ox0, c10::optional<torch::Tensor> in2,
    c10::optional<torch::Tensor> out2_k1);
std::vector<torch::Tensor> bn_act_fwd_train(
    torch::Tensor x, torch::Tensor stats, torch::Tensor gamma,
    torch::Tensor beta, c10::optional<torch::Tensor> running_mean,
    c10::optional<torch::Tensor> running_var, double momentum, double eps,
    long act, long ring);
torch::Tensor bn_act_fwd_eval(torch::Tensor x, torch::Tensor gamma,
                              torch::Tensor beta, torch::Tensor running_mean,
                              torch::Tensor running_var, double eps, long act,
                              long ring);
std::vector<torch::Tensor> bn_act_bwd(torch::Tensor x, torch::Tensor dy,
                                      torch::Tensor mean, torch::Tensor invstd,
                                      torch::Tensor gamma, torch::Tensor beta,
                                      torch::Tensor scale, long act,
                                      c10::optional<torch::Tensor> dgamma_acc,
                                      c10::optional<torch::Tensor> dbeta_acc,
                                      long ring);
torch::Tensor conv2d_nhwc_wgrad(torch::Tensor Y, torch::Tensor X, long R,
                                long S, long stride, long pad, long splitp,
                                c10::optional<torch::Tensor> acc, long yring,
                                c10::optional<torch::Tensor> X2);
std::vector<torch::Tensor> maxpool2x2_fwd(torch::Tensor in, long ri, long ro);
torch::Tensor maxpool2x2_bwd(torch::Tensor gout, torch::Tensor idx, long H,
                             long W, long ri, long ro);
torch::Tensor upsample2x_fwd(torch::Tensor in, long ri, long ro);
torch::Tensor upsample2x_bwd(torch::Tensor gout, long ri, long ro);
torch::Tensor channel_sum_nhwc(torch::Tensor x,
                               c10::optional<torch::Tensor> acc);
torch::Tensor sqdiff_sum(torch::Tensor a, torch::Tensor b);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("mfma_probe", &mfma_probe, "16x16x32 bf16 MFMA layout probe");
  m.def("affine4_fwd", &affine4_fwd, "concat-free 4-source affine (gfx950)");
  m.def("affine4_bwd", &affine4_bwd, "affine4 backward (managed dW/db)");
  m.def("gauss_head_fwd", &gauss_head_fwd,
        "fused mu/logvar/reparameterize head (gfx950)");
  m.def("gauss_head_bwd", &gauss_head_bwd, "gauss head backward");
  m.def("tanh_head_fwd", &tanh_head_fwd, "fused Linear+Tanh head (gfx950)");
  m.def("tanh_head_bwd", &tanh_head_bwd, "tanh head backward");
  m.def("tr16_probe", &tr16_probe, "ds_read_b64_tr_b16 semantics probe");
  m.def("conv2d_nhwc_fwd", &conv2d_nhwc_fwd,
        "NHWC implicit-GEMM bf16 conv fwd (gfx950 MFMA); returns (out, stats)",
        pybind11::arg("in"), pybind11::arg("w"), pybind11::arg("bias"),
        pybind11::arg("stride"), pybind11::arg("pad"), pybind11::arg("act"),
        pybind11::arg("want_stats") = false, pybind11::arg("oh") = 0,
        pybind11::arg("ow") = 0, pybind11::arg("oy0") = 0,
        pybind11::arg("ox0") = 0);
  m.def("conv2d_nhwc_fracstride", &conv2d_nhwc_fracstride,
        "fractionally-strided conv, in-kernel parity loop (ConvT fwd / s2 dgrad)",
        pybind11::arg("in"), pybind11::arg("w"), pybind11::arg("bias"),
        pybind11::arg("up_stride"), pybind11::arg("up_pad"), pybind11::arg("OH"),
        pybind11::arg("OW"), pybind11::arg("act"),
        pybind11::arg("want_stats") = false, pybind11::arg("in_ring") = 0,
        pybind11::arg("out_ring") = 0);
  m.def("conv2d_glds_fwd", &conv2d_glds_fwd,
        "NHWC conv fwd, glds 3-buffer pipeline (padded-input contract)",
        pybind11::arg("in"), pybind11::arg("w"), pybind11::arg("bias"),
        pybind11::arg("stride"), pybind11::arg("act"),
        pybind11::arg("want_stats") = false, pybind11::arg("oh") = 0,
        pybind11::arg("ow") = 0, pybind11::arg("oy0") = 0,
        pybind11::arg("ox0") = 0, pybind11::arg("in2") = c10::nullopt,
        pybind11::arg("out2") = c10::nullopt);
  m.def("bn_act_fwd_train", &bn_act_fwd_train, "fused BN+act train fwd (gfx950)",
        pybind11::arg("x"), pybind11::arg("stats"), pybind11::arg("gamma"),
        pybind11::arg("beta"), pybind11::arg("running_mean"),
        pybind11::arg("running_var"), pybind11::arg("momentum"),
        pybind11::arg("eps"), pybind11::arg("act"),
        pybind11::arg("ring") = 0);
  m.def("bn_act_fwd_eval", &bn_act_fwd_eval, "fused BN+act eval fwd (gfx950)",
        pybind11::arg("x"), pybind11::arg("gamma"), pybind11::arg("beta"),
        pybind11::arg("running_mean"), pybind11::arg("running_var"),
        pybind11::arg("eps"), pybind11::arg("act"), pybind11::arg("ring") = 0);
  m.def("bn_act_bwd", &bn_act_bwd, "fused BN+act bwd (gfx950, deterministic)",
        pybind11::arg("x"), pybind11::arg("dy"), pybind11::arg("mean"),
        pybind11::arg("invstd"), pybind11::arg("gamma"), pybind11::arg("beta"),
        pybind11::arg("scale"), pybind11::arg("act"),
        pybind11::arg("dgamma_acc") = c10::nullopt,
        pybind11::arg("dbeta_acc") = c10::nullopt,
        pybind11::arg("ring") = 0);
  m.def("channel_sum_nhwc", &channel_sum_nhwc,
        "NHWC per-channel sum, fp32 out (gfx950, deterministic)",
        pybind11::arg("x"), pybind11::arg("acc") = c10::nullopt);
  m.def("conv2d_nhwc_wgrad", &conv2d_nhwc_wgrad,
        "NHWC wgrad, split-K over pixel slabs (gfx950 MFMA, deterministic); "
        "acc: fp32 dense (B,R,S,A)-layout tensor to accumulate into",
        pybind11::arg("Y"), pybind11::arg("X"), pybind11::arg("R"),
        pybind11::arg("S"), pybind11::arg("stride"), pybind11::arg("pad"),
        pybind11::arg("splitp") = 0, pybind11::arg("acc") = c10::nullopt,
        pybind11::arg("yring") = 0, pybind11::arg("X2") = c10::nullopt);
  m.def("maxpool2x2_fwd", &maxpool2x2_fwd, "NHWC 2x2/s2 maxpool fwd (gfx950)",
        pybind11::arg("in"), pybind11::arg("ri") = 0, pybind11::arg("ro") = 0);
  m.def("maxpool2x2_bwd", &maxpool2x2_bwd, "NHWC 2x2/s2 maxpool bwd (gfx950)",
        pybind11::arg("gout"), pybind11::arg("idx"), pybind11::arg("H"),
        pybind11::arg("W"), pybind11::arg("ri") = 0, pybind11::arg("ro") = 0);
  m.def("upsample2x_fwd", &upsample2x_fwd, "NHWC nearest x2 fwd (gfx950)",
        pybind11::arg("in"), pybind11::arg("ri") = 0, pybind11::arg("ro") = 0);
  m.def("upsample2x_bwd", &upsample2x_bwd, "NHWC nearest x2 bwd (gfx950)",
        pybind11::arg("gout"), pybind11::arg("ri") = 0, pybind11::arg("ro") = 0);
  m.def("lstm_cell_fwd", &lstm_cell_fwd, "fused LSTM cell forward (gfx950)");
  m.def("lstm_cell_bwd_pointwise", &lstm_cell_bwd_pointwise,
        "LSTM cell backward pointwise (gfx950)");
  m.def("multi_tensor_adam", &multi_tensor_adam, "multi-tensor Adam (gfx950)");
  m.def("gaussian_kl_fwd", &gaussian_kl_fwd, "fused gaussian KL fwd (gfx950)");
  m.def("gaussian_kl_bwd", &gaussian_kl_bwd, "fused gaussian KL bwd (gfx950)");
  m.def("sqdiff_sum", &sqdiff_sum, "fused sum((a-b)^2), bf16/f32 in fp32 out");
}
