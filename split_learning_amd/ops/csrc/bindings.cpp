// Python bindings for the split_learning_amd CDNA4 kernel set.
#include <torch/extension.h>

namespace slk {
// gemm_f32.hip
at::Tensor matmul_f32(const at::Tensor&, const at::Tensor&, bool, bool,
                      c10::optional<at::Tensor>, bool);
at::Tensor linear_fwd(const at::Tensor&, const at::Tensor&, c10::optional<at::Tensor>);
at::Tensor colsum_f32(const at::Tensor&);
// wino.hip
at::Tensor conv2d_wino(const at::Tensor&, const at::Tensor&,
                       c10::optional<at::Tensor>, int, bool);
at::Tensor conv2d_wino_bwdw(const at::Tensor&, const at::Tensor&, int);
at::Tensor conv2d_wino_fused(const at::Tensor&, const at::Tensor&,
                             c10::optional<at::Tensor>, int, bool);
at::Tensor conv2d_wino_bwdw_fused(const at::Tensor&, const at::Tensor&, int);
// conv2d.hip
at::Tensor pad_nchw(const at::Tensor&, int);
at::Tensor conv2d_fwd(const at::Tensor&, const at::Tensor&, c10::optional<at::Tensor>,
                      int, int, bool);
at::Tensor conv2d_bwd_data(const at::Tensor&, const at::Tensor&, int, int, int, int);
at::Tensor conv2d_bwd_weight(const at::Tensor&, const at::Tensor&, int, int, int, int,
                             bool);
at::Tensor conv2d_bwd_bias(const at::Tensor&);
// norm.hip
std::vector<at::Tensor> bn2d_stats_fused(const at::Tensor&,
                                         c10::optional<at::Tensor>,
                                         c10::optional<at::Tensor>,
                                         c10::optional<at::Tensor>, double, double);
at::Tensor bn2d_fwd(const at::Tensor&, const at::Tensor&, const at::Tensor&,
                    const at::Tensor&, const at::Tensor&, bool);
std::vector<at::Tensor> bn2d_bwd(const at::Tensor&, const at::Tensor&,
                                 const at::Tensor&, const at::Tensor&,
                                 const at::Tensor&, c10::optional<at::Tensor>);
std::vector<at::Tensor> bn2d_bwd_eval(const at::Tensor&, const at::Tensor&,
                                      const at::Tensor&, const at::Tensor&,
                                      const at::Tensor&, c10::optional<at::Tensor>);
std::vector<at::Tensor> drop_res_ln_fwd(const at::Tensor&, const at::Tensor&,
                                        const at::Tensor&, const at::Tensor&,
                                        double, double, int64_t,
                                        c10::optional<at::Tensor>);
std::vector<at::Tensor> layernorm_fwd(const at::Tensor&, const at::Tensor&,
                                      const at::Tensor&, double);
std::vector<at::Tensor> layernorm_bwd(const at::Tensor&, const at::Tensor&,
                                      const at::Tensor&, const at::Tensor&,
                                      const at::Tensor&);
// elementwise.hip
at::Tensor relu_fwd(const at::Tensor&);
at::Tensor relu_bwd(const at::Tensor&, const at::Tensor&);
at::Tensor gelu_fwd(const at::Tensor&);
at::Tensor gelu_bwd(const at::Tensor&, const at::Tensor&);
at::Tensor tanh_fwd(const at::Tensor&);
at::Tensor tanh_bwd(const at::Tensor&, const at::Tensor&);
std::vector<at::Tensor> dropout_fwd(const at::Tensor&, double, int64_t, int64_t);
std::vector<at::Tensor> dropout_fwd_dev(const at::Tensor&, double, int64_t,
                                        const at::Tensor&);
at::Tensor dropout_bwd(const at::Tensor&, const at::Tensor&, double);
std::vector<at::Tensor> maxpool2x2_fwd(const at::Tensor&);
at::Tensor maxpool2x2_bwd(const at::Tensor&, const at::Tensor&, int, int);
at::Tensor embedding_fwd(const at::Tensor&, const at::Tensor&);
at::Tensor embedding_bwd(const at::Tensor&, const at::Tensor&, int64_t, int64_t);
// attention.hip
std::vector<at::Tensor> attn_fwd(const at::Tensor&, const at::Tensor&,
                                 const at::Tensor&, double, double, int64_t,
                                 c10::optional<at::Tensor>);
// loss.hip
at::Tensor softmax_fwd(const at::Tensor&);
at::Tensor softmax_bwd(const at::Tensor&, const at::Tensor&);
std::vector<at::Tensor> ce_fwd(const at::Tensor&, const at::Tensor&);
at::Tensor ce_bwd(const at::Tensor&, const at::Tensor&, const at::Tensor&);
// optim.hip
void sgd_step(std::vector<at::Tensor>, std::vector<at::Tensor>,
              std::vector<at::Tensor>, double, double, double, bool, bool);
void adamw_step(std::vector<at::Tensor>, std::vector<at::Tensor>,
                std::vector<at::Tensor>, std::vector<at::Tensor>, int64_t, double,
                double, double, double, double, bool);
at::Tensor make_opt_desc(std::vector<at::Tensor>, std::vector<at::Tensor>,
                         std::vector<at::Tensor>, std::vector<at::Tensor>);
void sgd_step_fused(const at::Tensor&, int64_t, int64_t, double, double, double,
                    bool, bool);
void adamw_step_fused(const at::Tensor&, int64_t, int64_t, int64_t, double, double,
                      double, double, double, bool);
}  // namespace slk

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "split_learning_amd CDNA4 (gfx950) HIP kernels";
  m.def("matmul_f32", &slk::matmul_f32, py::arg("a"), py::arg("b"),
        py::arg("trans_a") = false, py::arg("trans_b") = false,
        py::arg("out") = py::none(), py::arg("accumulate") = false);
  m.def("linear_fwd", &slk::linear_fwd);
  m.def("colsum_f32", &slk::colsum_f32);
  m.def("conv2d_wino", &slk::conv2d_wino, py::arg("x"), py::arg("w"),
        py::arg("bias") = py::none(), py::arg("pad") = 1,
        py::arg("flip") = false);
  m.def("conv2d_wino_bwdw", &slk::conv2d_wino_bwdw, py::arg("gy"),
        py::arg("x"), py::arg("pad") = 1);
  m.def("conv2d_wino_fused", &slk::conv2d_wino_fused, py::arg("x"),
        py::arg("w"), py::arg("bias") = py::none(), py::arg("pad") = 1,
        py::arg("flip") = false);
  m.def("conv2d_wino_bwdw_fused", &slk::conv2d_wino_bwdw_fused, py::arg("gy"),
        py::arg("x"), py::arg("pad") = 1);
  m.def("pad_nchw", &slk::pad_nchw);
  m.def("conv2d_fwd", &slk::conv2d_fwd, py::arg("x"), py::arg("w"),
        py::arg("bias"), py::arg("stride"), py::arg("pad"),
        py::arg("x_is_padded") = false);
  m.def("conv2d_bwd_data", &slk::conv2d_bwd_data);
  m.def("conv2d_bwd_weight", &slk::conv2d_bwd_weight, py::arg("gy"), py::arg("x"),
        py::arg("kh"), py::arg("kw"), py::arg("stride"), py::arg("pad"),
        py::arg("x_is_padded") = false);
  m.def("conv2d_bwd_bias", &slk::conv2d_bwd_bias);
  m.def("bn2d_stats_fused", &slk::bn2d_stats_fused);
  m.def("bn2d_fwd", &slk::bn2d_fwd);
  m.def("bn2d_bwd", &slk::bn2d_bwd, py::arg("x"), py::arg("gy"),
        py::arg("gamma"), py::arg("mean"), py::arg("invstd"),
        py::arg("relu_y") = py::none());
  m.def("bn2d_bwd_eval", &slk::bn2d_bwd_eval, py::arg("x"), py::arg("gy"),
        py::arg("gamma"), py::arg("mean"), py::arg("invstd"),
        py::arg("relu_y") = py::none());
  m.def("drop_res_ln_fwd", &slk::drop_res_ln_fwd, py::arg("x"), py::arg("res"),
        py::arg("gamma"), py::arg("beta"), py::arg("eps"), py::arg("p") = 0.0,
        py::arg("seed") = 0, py::arg("offset") = py::none());
  m.def("layernorm_fwd", &slk::layernorm_fwd);
  m.def("layernorm_bwd", &slk::layernorm_bwd);
  m.def("relu_fwd", &slk::relu_fwd);
  m.def("relu_bwd", &slk::relu_bwd);
  m.def("gelu_fwd", &slk::gelu_fwd);
  m.def("gelu_bwd", &slk::gelu_bwd);
  m.def("tanh_fwd", &slk::tanh_fwd);
  m.def("tanh_bwd", &slk::tanh_bwd);
  m.def("dropout_fwd", &slk::dropout_fwd);
  m.def("dropout_fwd_dev", &slk::dropout_fwd_dev);
  m.def("dropout_bwd", &slk::dropout_bwd);
  m.def("maxpool2x2_fwd", &slk::maxpool2x2_fwd);
  m.def("maxpool2x2_bwd", &slk::maxpool2x2_bwd);
  m.def("embedding_fwd", &slk::embedding_fwd);
  m.def("embedding_bwd", &slk::embedding_bwd);
  m.def("attn_fwd", &slk::attn_fwd, py::arg("q"), py::arg("k"), py::arg("v"),
        py::arg("scale"), py::arg("p") = 0.0, py::arg("seed") = 0,
        py::arg("offset") = py::none());
  m.def("softmax_fwd", &slk::softmax_fwd);
  m.def("softmax_bwd", &slk::softmax_bwd);
  m.def("ce_fwd", &slk::ce_fwd);
  m.def("ce_bwd", &slk::ce_bwd);
  m.def("sgd_step", &slk::sgd_step);
  m.def("adamw_step", &slk::adamw_step);
  m.def("make_opt_desc", &slk::make_opt_desc);
  m.def("sgd_step_fused", &slk::sgd_step_fused);
  m.def("adamw_step_fused", &slk::adamw_step_fused);
}
