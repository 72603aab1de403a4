// Python bindings for the CDNA4 kernel library (_hipops).
// Compiled by hipcc together with the .hip TUs; see setup.py / build_hip.py.

#include <torch/extension.h>

namespace dvgg {

// gemm_ops.hip
at::Tensor linear_fwd(const at::Tensor&, const at::Tensor&,
                      const c10::optional<at::Tensor>&, bool, bool);
at::Tensor linear_dgrad(const at::Tensor&, const at::Tensor&);
at::Tensor linear_wgrad(const at::Tensor&, const at::Tensor&,
                        const c10::optional<at::Tensor>&);
at::Tensor conv2d_fwd(const at::Tensor&, const at::Tensor&,
                      const c10::optional<at::Tensor>&, int64_t, int64_t,
                      bool, bool);
at::Tensor conv2d_dgrad(const at::Tensor&, const at::Tensor&, int64_t,
                        int64_t, int64_t, int64_t, bool);
at::Tensor conv2d_wgrad(const at::Tensor&, const at::Tensor&, int64_t,
                        int64_t, int64_t, int64_t);
std::vector<at::Tensor> conv2d_bwd(const at::Tensor&, const at::Tensor&,
                                   const at::Tensor&, const at::Tensor&,
                                   int64_t, int64_t, bool, bool, bool,
                                   const c10::optional<at::Tensor>&,
                                   const c10::optional<at::Tensor>&, bool);
void bump_weight_gen();

// mem.hip
at::Tensor relu_bwd(const at::Tensor&, const at::Tensor&);
at::Tensor add_relu_fwd(const at::Tensor&, const at::Tensor&);
at::Tensor bias_grad(const at::Tensor&, const c10::optional<at::Tensor>&);
void sgd_step(at::Tensor&, at::Tensor&, const at::Tensor&, at::Tensor&,
              double, double, double, double);
std::vector<at::Tensor> dropout_fwd(const at::Tensor&, double, int64_t,
                                    int64_t);
at::Tensor dropout_bwd(const at::Tensor&, const at::Tensor&, double);

// lrnpool.hip
at::Tensor lrn_fwd(const at::Tensor&, int64_t, double, double, double);
at::Tensor lrn_bwd(const at::Tensor&, const at::Tensor&, int64_t, double,
                   double, double);
std::vector<at::Tensor> maxpool_fwd(const at::Tensor&, int64_t, int64_t,
                                    int64_t, int64_t);
at::Tensor maxpool_bwd(const at::Tensor&, const at::Tensor&, int64_t, int64_t,
                       int64_t, int64_t, int64_t, int64_t);

// softbn.hip
std::vector<at::Tensor> softmax_xent_fwd(const at::Tensor&, const at::Tensor&);
std::vector<at::Tensor> bn_fwd(const at::Tensor&, const at::Tensor&,
                               const at::Tensor&, double, bool,
                               const c10::optional<at::Tensor>&,
                               const c10::optional<at::Tensor>&, double,
                               const c10::optional<at::Tensor>&);
std::vector<at::Tensor> bn_bwd(const at::Tensor&, const at::Tensor&,
                               const at::Tensor&, const at::Tensor&,
                               const at::Tensor&, const at::Tensor&, bool,
                               const c10::optional<at::Tensor>&,
                               const c10::optional<at::Tensor>&);

}  // namespace dvgg

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "MI355X-native CDNA4 kernels for distributed_vgg_f_amd";
  m.def("linear_fwd", &dvgg::linear_fwd, py::arg("x"), py::arg("w"),
        py::arg("bias"), py::arg("relu"), py::arg("cache_w") = false);
  m.def("linear_dgrad", &dvgg::linear_dgrad);
  m.def("linear_wgrad", &dvgg::linear_wgrad, py::arg("x"), py::arg("dy"),
        py::arg("dw_out") = c10::nullopt);
  m.def("conv2d_fwd", &dvgg::conv2d_fwd, py::arg("x"), py::arg("w"),
        py::arg("bias"), py::arg("stride"), py::arg("pad"), py::arg("relu"),
        py::arg("cache_w") = false);
  m.def("conv2d_dgrad", &dvgg::conv2d_dgrad, py::arg("dy"), py::arg("w"),
        py::arg("H"), py::arg("W"), py::arg("stride"), py::arg("pad"),
        py::arg("cache_w") = false);
  m.def("conv2d_wgrad", &dvgg::conv2d_wgrad);
  m.def("conv2d_bwd", &dvgg::conv2d_bwd, py::arg("dy"), py::arg("y"),
        py::arg("x"), py::arg("w"), py::arg("stride"), py::arg("pad"),
        py::arg("relu"), py::arg("need_dx"), py::arg("need_db"),
        py::arg("dw_out") = c10::nullopt, py::arg("db_out") = c10::nullopt,
        py::arg("cache_w") = false);
  m.def("relu_bwd", &dvgg::relu_bwd);
  m.def("add_relu_fwd", &dvgg::add_relu_fwd);
  m.def("bias_grad", &dvgg::bias_grad, py::arg("dy"),
        py::arg("db_out") = c10::nullopt);
  m.def("sgd_step", &dvgg::sgd_step);
  m.def("bump_weight_gen", &dvgg::bump_weight_gen);
  m.def("dropout_fwd", &dvgg::dropout_fwd);
  m.def("dropout_bwd", &dvgg::dropout_bwd);
  m.def("lrn_fwd", &dvgg::lrn_fwd);
  m.def("lrn_bwd", &dvgg::lrn_bwd);
  m.def("maxpool_fwd", &dvgg::maxpool_fwd);
  m.def("maxpool_bwd", &dvgg::maxpool_bwd);
  m.def("softmax_xent_fwd", &dvgg::softmax_xent_fwd);
  m.def("bn_fwd", &dvgg::bn_fwd, py::arg("x"), py::arg("gamma"),
        py::arg("beta"), py::arg("eps"), py::arg("relu"),
        py::arg("run_mean") = c10::nullopt, py::arg("run_var") = c10::nullopt,
        py::arg("momentum") = 0.9, py::arg("res") = c10::nullopt);
  m.def("bn_bwd", &dvgg::bn_bwd, py::arg("dy"), py::arg("x"),
        py::arg("gamma"), py::arg("mean"), py::arg("rstd"), py::arg("yr"),
        py::arg("relu"), py::arg("dg_out") = c10::nullopt,
        py::arg("db_out") = c10::nullopt);
}
