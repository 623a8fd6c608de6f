// Python bindings for the mxnet_amd gfx950 kernel library.
// Every entry point here is backed by a hand-written CDNA4 HIP kernel
// (see *.hip in this directory); there is no fallback dispatch at this
// layer — if it binds, it runs native.
#include <torch/extension.h>

// gemm.hip
at::Tensor gemm(const at::Tensor&, const at::Tensor&);
at::Tensor gemm_nt(const at::Tensor&, const at::Tensor&,
                   c10::optional<at::Tensor>);
at::Tensor gemm_nn(const at::Tensor&, const at::Tensor&);
at::Tensor gemm_tn(const at::Tensor&, const at::Tensor&);
at::Tensor bgemm(const at::Tensor&, const at::Tensor&);
at::Tensor transpose2d(const at::Tensor&);
at::Tensor gemm_nt_8ph(const at::Tensor&, const at::Tensor&);
// conv.hip
at::Tensor conv2d_nhwc_fwd(const at::Tensor&, const at::Tensor&,
                           c10::optional<at::Tensor>, int64_t, int64_t,
                           int64_t, int64_t, int64_t, int64_t, int64_t);
at::Tensor conv2d_nhwc_bwd_data(const at::Tensor&, const at::Tensor&,
                                int64_t, int64_t, int64_t, int64_t, int64_t,
                                int64_t, int64_t, int64_t, int64_t);
at::Tensor conv2d_nhwc_bwd_weight(const at::Tensor&, const at::Tensor&,
                                  int64_t, int64_t, int64_t, int64_t,
                                  int64_t, int64_t, int64_t, int64_t,
                                  int64_t);
// norm.hip
std::vector<at::Tensor> bn_nhwc_fwd_train(const at::Tensor&,
                                          const at::Tensor&,
                                          const at::Tensor&, at::Tensor,
                                          at::Tensor, double, double, bool,
                                          const at::Tensor&,
                                          c10::optional<at::Tensor>);
at::Tensor bn_nhwc_fwd_infer(const at::Tensor&, const at::Tensor&,
                             const at::Tensor&, const at::Tensor&,
                             const at::Tensor&, double, bool,
                             const at::Tensor&);
std::vector<at::Tensor> bn_nhwc_bwd(const at::Tensor&, const at::Tensor&,
                                    const at::Tensor&, const at::Tensor&,
                                    const at::Tensor&, bool,
                                    const at::Tensor&, bool,
                                    const at::Tensor&);
std::vector<at::Tensor> layernorm_fwd(const at::Tensor&, const at::Tensor&,
                                      const at::Tensor&, double);
std::vector<at::Tensor> layernorm_bwd(const at::Tensor&, const at::Tensor&,
                                      const at::Tensor&, const at::Tensor&,
                                      const at::Tensor&);
// softmax.hip
at::Tensor softmax_fwd(const at::Tensor&, bool, double,
                       c10::optional<at::Tensor>);
at::Tensor softmax_bwd(const at::Tensor&, const at::Tensor&, bool, double);
at::Tensor colsum(const at::Tensor&);
// pool.hip
std::vector<at::Tensor> pool_nhwc_fwd(const at::Tensor&, const std::string&,
                                      int64_t, int64_t, int64_t, int64_t,
                                      int64_t, int64_t, bool);
at::Tensor pool_nhwc_bwd(const at::Tensor&, const at::Tensor&,
                         const std::string&, int64_t, int64_t, int64_t,
                         int64_t, int64_t, int64_t, int64_t, int64_t, bool);
// quant.hip
at::Tensor quantize_i8(const at::Tensor&, double);
at::Tensor dequantize_i8(const at::Tensor&, double,
                         c10::optional<at::ScalarType>);
at::Tensor im2col_nhwc_op(const at::Tensor&, int, int, int, int, int, int,
                          int, int);
std::vector<at::Tensor> conv2d_nhwc_fwd_stats(
    const at::Tensor&, const at::Tensor&, c10::optional<at::Tensor>, int64_t,
    int64_t, int64_t, int64_t, int64_t, int64_t, int64_t);
std::vector<at::Tensor> gemm_tn_fused(const at::Tensor&, const at::Tensor&,
                                      bool);
std::vector<at::Tensor> attention_fwd(const at::Tensor&,
                                      c10::optional<at::Tensor>, int64_t,
                                      double);
at::Tensor attention_bwd(const at::Tensor&, const at::Tensor&,
                         const at::Tensor&, int64_t, double);
at::Tensor gemm_nt_i8(const at::Tensor&, const at::Tensor&, double,
                      c10::optional<at::ScalarType>);
// elemwise.hip
at::Tensor act_fwd(const at::Tensor&, const std::string&);
at::Tensor act_bwd(const at::Tensor&, const at::Tensor&, const std::string&);
void sgd_update(at::Tensor, c10::optional<at::Tensor>, at::Tensor,
                c10::optional<at::Tensor>, double, double, double, double,
                double);
void multi_sgd_update(std::vector<at::Tensor>, std::vector<at::Tensor>,
                      std::vector<at::Tensor>, std::vector<at::Tensor>,
                      std::vector<double>, std::vector<double>, double,
                      double, double);
void adam_update(at::Tensor, c10::optional<at::Tensor>, at::Tensor,
                 at::Tensor, at::Tensor, double, double, double, double,
                 double, double, double, bool);
bool multi_all_finite(std::vector<at::Tensor>);
std::vector<at::Tensor> lstm_cell_fwd(const at::Tensor&, const at::Tensor&);
std::vector<at::Tensor> dropout_fwd(const at::Tensor&, double, int64_t);
at::Tensor dropout_bwd(const at::Tensor&, const at::Tensor&, double);
at::Tensor embedding_fwd(const at::Tensor&, const at::Tensor&);
at::Tensor embedding_bwd(const at::Tensor&, const at::Tensor&, int64_t);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "mxnet_amd native gfx950 kernels (MFMA GEMM/conv, fused "
            "norms, wave64 reductions)";
  m.def("gemm", &gemm);
  m.def("gemm_nt", &gemm_nt, py::arg("x"), py::arg("w"),
        py::arg("bias") = py::none());
  m.def("gemm_nn", &gemm_nn);
  m.def("gemm_tn", &gemm_tn);
  m.def("bgemm", &bgemm);
  m.def("transpose2d", &transpose2d);
  m.def("gemm_nt_8ph", &gemm_nt_8ph);
  m.def("conv2d_nhwc_fwd", &conv2d_nhwc_fwd);
  m.def("conv2d_nhwc_bwd_data", &conv2d_nhwc_bwd_data);
  m.def("conv2d_nhwc_bwd_weight", &conv2d_nhwc_bwd_weight);
  m.def("bn_nhwc_fwd_train", &bn_nhwc_fwd_train, py::arg("x"),
        py::arg("gamma"), py::arg("beta"), py::arg("rmean"),
        py::arg("rvar"), py::arg("momentum"), py::arg("eps"),
        py::arg("fuse_relu"), py::arg("residual"),
        py::arg("presums") = c10::nullopt);
  m.def("bn_nhwc_fwd_infer", &bn_nhwc_fwd_infer);
  m.def("bn_nhwc_bwd", &bn_nhwc_bwd);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("softmax_fwd", &softmax_fwd, py::arg("x"), py::arg("log"),
        py::arg("temperature"), py::arg("mask") = py::none());
  m.def("softmax_bwd", &softmax_bwd);
  m.def("colsum", &colsum);
  m.def("pool_nhwc_fwd", &pool_nhwc_fwd);
  m.def("pool_nhwc_bwd", &pool_nhwc_bwd);
  m.def("act_fwd", &act_fwd);
  m.def("act_bwd", &act_bwd);
  m.def("sgd_update", &sgd_update, py::arg("w"), py::arg("master"),
        py::arg("grad"), py::arg("mom"), py::arg("lr"), py::arg("mu"),
        py::arg("wd"), py::arg("rescale"), py::arg("clip"));
  m.def("adam_update", &adam_update);
  m.def("multi_sgd_update", &multi_sgd_update);
  m.def("multi_all_finite", &multi_all_finite);
  m.def("lstm_cell_fwd", &lstm_cell_fwd);
  m.def("dropout_fwd", &dropout_fwd);
  m.def("dropout_bwd", &dropout_bwd);
  m.def("embedding_fwd", &embedding_fwd);
  m.def("embedding_bwd", &embedding_bwd);
  m.def("quantize_i8", &quantize_i8);
  m.def("dequantize_i8", &dequantize_i8, py::arg("x"), py::arg("scale"),
        py::arg("dtype") = py::none());
  m.def("im2col_nhwc", &im2col_nhwc_op);
  m.def("conv2d_nhwc_fwd_stats", &conv2d_nhwc_fwd_stats);
  m.def("gemm_tn_fused", &gemm_tn_fused);
  m.def("attention_fwd", &attention_fwd, py::arg("qkv"),
        py::arg("mask") = c10::nullopt, py::arg("heads"),
        py::arg("temperature") = 1.0);
  m.def("attention_bwd", &attention_bwd);
  m.def("gemm_nt_i8", &gemm_nt_i8, py::arg("a"), py::arg("b"),
        py::arg("scale"), py::arg("out_dtype") = py::none());
}
