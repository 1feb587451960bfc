// pybind bindings for eventgrad_amd._core (gfx950 HIP kernels).
#include <torch/extension.h>

#include <vector>

namespace eg {
// engine.hip
torch::Tensor sgd_step_norm(torch::Tensor, torch::Tensor, torch::Tensor,
                            torch::Tensor, torch::Tensor, double, double,
                            double);
torch::Tensor seg_sqnorms(torch::Tensor, torch::Tensor, torch::Tensor);
void avg3(torch::Tensor, torch::Tensor, torch::Tensor);
torch::Tensor gather_segments(torch::Tensor, torch::Tensor, torch::Tensor,
                              long);
void scatter_segments(torch::Tensor, torch::Tensor, torch::Tensor,
                      torch::Tensor);
torch::Tensor trigger_update(torch::Tensor, torch::Tensor, torch::Tensor,
                             torch::Tensor, torch::Tensor, torch::Tensor,
                             long, bool, double, double, long, bool);
torch::Tensor trigger_decide(torch::Tensor, torch::Tensor, torch::Tensor,
                             long, bool, double, double, long);
// elementwise.hip
torch::Tensor relu_fwd(torch::Tensor);
torch::Tensor relu_bwd(torch::Tensor, torch::Tensor);
torch::Tensor add_relu_fwd(torch::Tensor, torch::Tensor);
std::vector<torch::Tensor> dropout_fwd(torch::Tensor, double, long, bool);
torch::Tensor dropout_bwd(torch::Tensor, torch::Tensor, double, bool);
torch::Tensor channel_sum(torch::Tensor);
void channel_sum_into(torch::Tensor, torch::Tensor);
torch::Tensor oihw_to_krsc(torch::Tensor);
torch::Tensor krsc_to_crsk(torch::Tensor);
torch::Tensor krsc_to_oihw(torch::Tensor);
void refresh_conv_shadows(torch::Tensor, torch::Tensor, torch::Tensor,
                          torch::Tensor, torch::Tensor, torch::Tensor,
                          torch::Tensor, torch::Tensor, torch::Tensor);
// gemm.hip
torch::Tensor gemm_bias(torch::Tensor, torch::Tensor, torch::Tensor, bool);
// conv.hip
torch::Tensor conv2d_fwd(torch::Tensor, torch::Tensor, torch::Tensor, long,
                         long, bool);
torch::Tensor conv2d_dgrad(torch::Tensor, torch::Tensor, long, long, long,
                           long, c10::optional<torch::Tensor>,
                           c10::optional<torch::Tensor>,
                           c10::optional<torch::Tensor>,
                           c10::optional<torch::Tensor>,
                           c10::optional<torch::Tensor>,
                           c10::optional<torch::Tensor>);
torch::Tensor conv2d_wgrad(torch::Tensor, torch::Tensor, long, long, long,
                           long);
void conv2d_wgrad_into(torch::Tensor, torch::Tensor, torch::Tensor, long,
                       long, long, long);
// bn.hip
std::vector<torch::Tensor> bn_fwd(torch::Tensor, torch::Tensor, torch::Tensor,
                                  torch::Tensor, torch::Tensor, double,
                                  double, bool, bool, bool);
std::vector<torch::Tensor> bn_bwd(torch::Tensor, torch::Tensor, torch::Tensor,
                                  torch::Tensor, torch::Tensor, torch::Tensor,
                                  bool, bool, c10::optional<torch::Tensor>,
                                  c10::optional<torch::Tensor>, bool);
torch::Tensor relu_bwd_bnstats(torch::Tensor, torch::Tensor, torch::Tensor,
                               torch::Tensor, torch::Tensor, torch::Tensor,
                               torch::Tensor);
std::vector<torch::Tensor> bn_stats_finalize(torch::Tensor, torch::Tensor,
                                             torch::Tensor, double, double,
                                             bool);
torch::Tensor bn_norm_add_relu(torch::Tensor, torch::Tensor, torch::Tensor,
                               torch::Tensor, torch::Tensor, torch::Tensor);
// pool.hip
std::vector<torch::Tensor> maxpool2x2_fwd(torch::Tensor);
torch::Tensor maxpool2x2_bwd(torch::Tensor, torch::Tensor, long, long);
torch::Tensor avgpool_fwd(torch::Tensor, long);
torch::Tensor avgpool_bwd(torch::Tensor, long, long, long);
// loss.hip
std::vector<torch::Tensor> logsoftmax_nll_fwd(torch::Tensor, torch::Tensor);
torch::Tensor logsoftmax_nll_bwd(torch::Tensor, torch::Tensor, torch::Tensor);
// topk.hip
std::vector<torch::Tensor> topk_absdiff(torch::Tensor, torch::Tensor, long);
void scatter_update(torch::Tensor, torch::Tensor, torch::Tensor);
torch::Tensor spevent_pack(torch::Tensor, torch::Tensor, torch::Tensor,
                           torch::Tensor, torch::Tensor, torch::Tensor, long,
                           long);
void spevent_unpack(torch::Tensor, torch::Tensor, torch::Tensor,
                    torch::Tensor, torch::Tensor, long);
torch::Tensor tr16_probe(long);
}  // namespace eg

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("sgd_step_norm", &eg::sgd_step_norm);
  m.def("seg_sqnorms", &eg::seg_sqnorms);
  m.def("avg3", &eg::avg3);
  m.def("gather_segments", &eg::gather_segments);
  m.def("scatter_segments", &eg::scatter_segments);
  m.def("trigger_update", &eg::trigger_update);
  m.def("trigger_decide", &eg::trigger_decide);
  m.def("relu_fwd", &eg::relu_fwd);
  m.def("relu_bwd", &eg::relu_bwd);
  m.def("add_relu_fwd", &eg::add_relu_fwd);
  m.def("dropout_fwd", &eg::dropout_fwd);
  m.def("dropout_bwd", &eg::dropout_bwd);
  m.def("channel_sum", &eg::channel_sum);
  m.def("channel_sum_into", &eg::channel_sum_into);
  m.def("oihw_to_krsc", &eg::oihw_to_krsc);
  m.def("krsc_to_crsk", &eg::krsc_to_crsk);
  m.def("krsc_to_oihw", &eg::krsc_to_oihw);
  m.def("refresh_conv_shadows", &eg::refresh_conv_shadows);
  m.def("gemm_bias", &eg::gemm_bias);
  m.def("conv2d_fwd", &eg::conv2d_fwd, py::arg("x"), py::arg("w"),
        py::arg("bias"), py::arg("stride"), py::arg("pad"),
        py::arg("collect_bn_stats") = false);
  m.def("conv2d_dgrad", &eg::conv2d_dgrad, py::arg("dy"), py::arg("wt"),
        py::arg("stride"), py::arg("pad"), py::arg("H"), py::arg("W"),
        py::arg("bs_y1") = c10::nullopt, py::arg("bs_x1") = c10::nullopt,
        py::arg("bs_mean") = c10::nullopt,
        py::arg("bs_invstd") = c10::nullopt,
        py::arg("bs_dgamma") = c10::nullopt,
        py::arg("bs_dbeta") = c10::nullopt);
  m.def("conv2d_wgrad", &eg::conv2d_wgrad);
  m.def("conv2d_wgrad_into", &eg::conv2d_wgrad_into);
  m.def("bn_fwd", &eg::bn_fwd, py::arg("x"), py::arg("gamma"), py::arg("beta"),
        py::arg("running_mean"), py::arg("running_var"), py::arg("momentum"),
        py::arg("eps"), py::arg("training"), py::arg("relu"),
        py::arg("have_stats") = false);
  m.def("bn_bwd", &eg::bn_bwd, py::arg("dy"), py::arg("x"), py::arg("mean"),
        py::arg("invstd"), py::arg("gamma"), py::arg("y"), py::arg("relu"),
        py::arg("training"),
        py::arg("dgamma_out") = c10::nullopt,
        py::arg("dbeta_out") = c10::nullopt,
        py::arg("stats_ready") = false);
  m.def("relu_bwd_bnstats", &eg::relu_bwd_bnstats);
  m.def("bn_stats_finalize", &eg::bn_stats_finalize);
  m.def("bn_norm_add_relu", &eg::bn_norm_add_relu);
  m.def("maxpool2x2_fwd", &eg::maxpool2x2_fwd);
  m.def("maxpool2x2_bwd", &eg::maxpool2x2_bwd);
  m.def("avgpool_fwd", &eg::avgpool_fwd);
  m.def("avgpool_bwd", &eg::avgpool_bwd);
  m.def("logsoftmax_nll_fwd", &eg::logsoftmax_nll_fwd);
  m.def("logsoftmax_nll_bwd", &eg::logsoftmax_nll_bwd);
  m.def("topk_absdiff", &eg::topk_absdiff);
  m.def("scatter_update", &eg::scatter_update);
  m.def("spevent_pack", &eg::spevent_pack);
  m.def("spevent_unpack", &eg::spevent_unpack);
  m.def("tr16_probe", &eg::tr16_probe);
}
