// dtmx._C python bindings (torch extension).
#include <torch/extension.h>

namespace dtmx {
// gemm_conv.hip
at::Tensor linear_fwd(const at::Tensor&, const at::Tensor&, const c10::optional<at::Tensor>&);
at::Tensor linear_dgrad(const at::Tensor&, const at::Tensor&);
at::Tensor linear_wgrad(const at::Tensor&, const at::Tensor&);
at::Tensor conv_fwd(const at::Tensor&, const at::Tensor&, long, long);
at::Tensor conv_dgrad(const at::Tensor&, const at::Tensor&, long, long, long, long,
                      const c10::optional<at::Tensor>&);
at::Tensor conv_wgrad(const at::Tensor&, const at::Tensor&, long, long, long, long);
std::vector<at::Tensor> conv_dgrad_bnfuse(const at::Tensor&, const at::Tensor&,
                                          long, long, long, long,
                                          const c10::optional<at::Tensor>&,
                                          const at::Tensor&, const at::Tensor&,
                                          const at::Tensor&, const at::Tensor&);
std::vector<std::tuple<long, long, double>> wgrad_ws_stats();
// batchnorm.hip
std::vector<at::Tensor> bn_fwd_train(const at::Tensor&, const at::Tensor&,
                                     const at::Tensor&, at::Tensor, at::Tensor,
                                     double, double, bool,
                                     const c10::optional<at::Tensor>&,
                                     const c10::optional<at::Tensor>&,
                                     const c10::optional<at::Tensor>&);
std::vector<at::Tensor> conv_fwd_stats(const at::Tensor&, const at::Tensor&,
                                       long, long);
std::vector<at::Tensor> bn_local_sums(const at::Tensor&);
std::vector<at::Tensor> bn_fwd_presummed(const at::Tensor&, const at::Tensor&,
                                         const at::Tensor&, at::Tensor,
                                         at::Tensor, double, double, bool,
                                         const c10::optional<at::Tensor>&,
                                         const at::Tensor&, const at::Tensor&,
                                         long);
std::vector<at::Tensor> bn_bwd_sums(const at::Tensor&, const at::Tensor&,
                                    const at::Tensor&, const at::Tensor&,
                                    const at::Tensor&, bool);
std::vector<at::Tensor> bn_bwd_finalize_slabs(const at::Tensor&, const at::Tensor&,
                                              const at::Tensor&);
std::vector<at::Tensor> bn_bwd_dx_presummed(const at::Tensor&, const at::Tensor&,
                                            const at::Tensor&, const at::Tensor&,
                                            const at::Tensor&, const at::Tensor&,
                                            const at::Tensor&, const at::Tensor&,
                                            long, bool, bool);
at::Tensor bn_fwd_infer(const at::Tensor&, const at::Tensor&, const at::Tensor&,
                        const at::Tensor&, const at::Tensor&, double, bool,
                        const c10::optional<at::Tensor>&);
std::vector<at::Tensor> bn_bwd(const at::Tensor&, const at::Tensor&,
                               const at::Tensor&, const at::Tensor&,
                               const at::Tensor&, bool, const at::Tensor&, bool);
// pool_elem.hip
std::vector<at::Tensor> maxpool_fwd(const at::Tensor&, long, long, long);
at::Tensor maxpool_bwd(const at::Tensor&, const at::Tensor&, long, long, long,
                       long, long);
at::Tensor global_avgpool_fwd(const at::Tensor&);
at::Tensor global_avgpool_bwd(const at::Tensor&, long, long);
at::Tensor relu_fwd(const at::Tensor&);
at::Tensor relu_bwd(const at::Tensor&, const at::Tensor&);
at::Tensor add_relu_fwd(const at::Tensor&, const at::Tensor&);
// softmax_opt.hip
std::vector<at::Tensor> softmax_ce_fwd(const at::Tensor&, const at::Tensor&);
at::Tensor softmax_ce_bwd(const at::Tensor&, const at::Tensor&, const at::Tensor&);
void sgd_mom_mp(at::Tensor, const at::Tensor&, at::Tensor, at::Tensor, double,
                double, double, double, double);
void sgd_mom_mp_dev(at::Tensor, const at::Tensor&, at::Tensor, at::Tensor,
                    const at::Tensor&);
void sgd_mom_f32(at::Tensor, const at::Tensor&, at::Tensor, double, double,
                 double, double, double);
// dropout_ln.hip
std::vector<at::Tensor> dropout_fwd(const at::Tensor&, double, int64_t);
at::Tensor dropout_bwd(const at::Tensor&, const at::Tensor&, double);
std::vector<at::Tensor> layer_norm_fwd(const at::Tensor&, const at::Tensor&,
                                       const at::Tensor&, double);
std::vector<at::Tensor> layer_norm_bwd(const at::Tensor&, const at::Tensor&,
                                       const at::Tensor&, const at::Tensor&,
                                       const at::Tensor&);
// compress.hip
at::Tensor quantize_2bit(const at::Tensor&, at::Tensor, double);
at::Tensor dequantize_2bit(const at::Tensor&, long, double);
// gemm256.hip
at::Tensor gemm256_nt(const at::Tensor&, const at::Tensor&);
at::Tensor gemm256n_nt(const at::Tensor&, const at::Tensor&);
at::Tensor lrn_fwd(const at::Tensor&, double, double, double);
at::Tensor lrn_bwd(const at::Tensor&, const at::Tensor&, double, double, double);
// indexing.hip
at::Tensor take_fwd(const at::Tensor&, const at::Tensor&);
at::Tensor take_bwd(const at::Tensor&, const at::Tensor&, long);
// recordio.cpp
void register_recordio(py::module_& m);
}  // namespace dtmx

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("linear_fwd", &dtmx::linear_fwd);
  m.def("linear_dgrad", &dtmx::linear_dgrad);
  m.def("linear_wgrad", &dtmx::linear_wgrad);
  m.def("conv_fwd", &dtmx::conv_fwd);
  m.def("conv_fwd_stats", &dtmx::conv_fwd_stats);
  m.def("conv_dgrad", &dtmx::conv_dgrad, py::arg("dy"), py::arg("w"),
        py::arg("stride"), py::arg("pad"), py::arg("H"), py::arg("W"),
        py::arg("acc") = c10::nullopt);
  m.def("conv_wgrad", &dtmx::conv_wgrad);
  m.def("conv_dgrad_bnfuse", &dtmx::conv_dgrad_bnfuse, py::arg("dy"),
        py::arg("w"), py::arg("stride"), py::arg("pad"), py::arg("H"),
        py::arg("W"), py::arg("acc"), py::arg("y"), py::arg("xin"),
        py::arg("mean"), py::arg("invstd"));
  m.def("bn_bwd_finalize_slabs", &dtmx::bn_bwd_finalize_slabs);
  m.def("wgrad_ws_stats", &dtmx::wgrad_ws_stats);
  m.def("bn_fwd_train", &dtmx::bn_fwd_train);
  m.def("bn_fwd_infer", &dtmx::bn_fwd_infer);
  m.def("bn_bwd", &dtmx::bn_bwd);
  m.def("bn_local_sums", &dtmx::bn_local_sums);
  m.def("bn_fwd_presummed", &dtmx::bn_fwd_presummed);
  m.def("bn_bwd_sums", &dtmx::bn_bwd_sums);
  m.def("bn_bwd_dx_presummed", &dtmx::bn_bwd_dx_presummed);
  m.def("maxpool_fwd", &dtmx::maxpool_fwd);
  m.def("maxpool_bwd", &dtmx::maxpool_bwd);
  m.def("global_avgpool_fwd", &dtmx::global_avgpool_fwd);
  m.def("global_avgpool_bwd", &dtmx::global_avgpool_bwd);
  m.def("relu_fwd", &dtmx::relu_fwd);
  m.def("relu_bwd", &dtmx::relu_bwd);
  m.def("add_relu_fwd", &dtmx::add_relu_fwd);
  m.def("softmax_ce_fwd", &dtmx::softmax_ce_fwd);
  m.def("softmax_ce_bwd", &dtmx::softmax_ce_bwd);
  m.def("sgd_mom_mp", &dtmx::sgd_mom_mp);
  m.def("sgd_mom_mp_dev", &dtmx::sgd_mom_mp_dev);
  m.def("sgd_mom_f32", &dtmx::sgd_mom_f32);
  m.def("dropout_fwd", &dtmx::dropout_fwd);
  m.def("dropout_bwd", &dtmx::dropout_bwd);
  m.def("layer_norm_fwd", &dtmx::layer_norm_fwd);
  m.def("layer_norm_bwd", &dtmx::layer_norm_bwd);
  m.def("gemm256_nt", &dtmx::gemm256_nt);
  m.def("gemm256n_nt", &dtmx::gemm256n_nt);
  m.def("lrn_fwd", &dtmx::lrn_fwd);
  m.def("lrn_bwd", &dtmx::lrn_bwd);
  m.def("take_fwd", &dtmx::take_fwd);
  m.def("take_bwd", &dtmx::take_bwd);
  m.def("quantize_2bit", &dtmx::quantize_2bit);
  m.def("dequantize_2bit", &dtmx::dequantize_2bit);
  dtmx::register_recordio(m);
}
