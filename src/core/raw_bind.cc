// Raw kernel-launcher bindings: the Python frontend (torch-tensor path)
// passes (data_ptr, shape, dtype_flag) triples plus its CURRENT stream and
// device; outputs are allocated by the caller.  No torch headers — torch
// tensors are just pointers here (kernel glue without
// ATen/hipify).
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "../ops/ops_api.h"
#include "op.h"

namespace py = pybind11;
using namespace mxcore;

namespace {

Arr ToArr(const py::handle& h) {
  if (h.is_none()) return Arr();
  py::tuple t = py::cast<py::tuple>(h);
  Arr a;
  a.ptr = (void*)t[0].cast<uintptr_t>();
  for (auto d : py::cast<py::tuple>(t[1])) a.shape.push_back(d.cast<int64_t>());
  a.dtype = t[2].cast<int>();
  return a;
}

std::vector<Arr> ToArrs(const py::handle& h) {
  std::vector<Arr> out;
  for (auto item : py::cast<py::list>(h)) out.push_back(ToArr(item));
  return out;
}

// shim launch context: the frontend's stream, lane-1 scratch arena
LaunchCtx MakeLC(int dev, uintptr_t stream) {
  LaunchCtx lc;
  lc.stream = (hipStream_t)stream;
  lc.dev = dev;
  Context c = Context::GPU(dev);
  ArenaReset(c, 1);
  lc.workspace = [c](size_t n) { return ArenaAlloc(c, n, 1); };
  return lc;
}

}  // namespace

void init_raw(py::module_& m) {
  py::module_ r = m.def_submodule("raw", "raw kernel launchers");

  r.def("gemm_nt", [](int dev, uintptr_t stream, py::handle A, py::handle B,
                      py::handle bias, py::handle out, bool relu,
                      py::handle stats) {
    gemm_nt_raw(MakeLC(dev, stream), ToArr(A), ToArr(B), ToArr(bias),
                ToArr(out), relu, ToArr(stats));
  });
  r.def("transpose2d", [](int dev, uintptr_t stream, py::handle x,
                          py::handle out) {
    transpose2d_raw(MakeLC(dev, stream), ToArr(x), ToArr(out));
  });
  r.def("gemm", [](int dev, uintptr_t stream, py::handle a, py::handle b,
                   py::handle out) {
    gemm_raw(MakeLC(dev, stream), ToArr(a), ToArr(b), ToArr(out));
  });
  r.def("gemm_nn", [](int dev, uintptr_t stream, py::handle a, py::handle b,
                      py::handle out) {
    gemm_nn_raw(MakeLC(dev, stream), ToArr(a), ToArr(b), ToArr(out));
  });
  r.def("gemm_tn", [](int dev, uintptr_t stream, py::handle a, py::handle b,
                      py::handle out) {
    gemm_tn_raw(MakeLC(dev, stream), ToArr(a), ToArr(b), ToArr(out));
  });
  r.def("bgemm", [](int dev, uintptr_t stream, py::handle a, py::handle b,
                    py::handle out) {
    bgemm_raw(MakeLC(dev, stream), ToArr(a), ToArr(b), ToArr(out));
  });
  r.def("gemm_nt_8ph", [](int dev, uintptr_t stream, py::handle a,
                          py::handle b, py::handle out) {
    gemm_nt_8ph_raw(MakeLC(dev, stream), ToArr(a), ToArr(b), ToArr(out));
  });
  r.def("gemm_tn_fused", [](int dev, uintptr_t stream, py::handle A,
                            py::handle B, py::handle C, py::handle dbias) {
    gemm_tn_fused_raw(MakeLC(dev, stream), ToArr(A), ToArr(B), ToArr(C),
                      ToArr(dbias));
  });
  r.def("attention_fwd", [](int dev, uintptr_t stream, py::handle qkv,
                            py::handle mask, int H, double temp,
                            py::handle out, py::handle att) {
    attention_fwd_raw(MakeLC(dev, stream), ToArr(qkv), ToArr(mask),
                      0.0, 0, Arr(), H, temp, ToArr(out), ToArr(att));
  });
  r.def("attention_bwd", [](int dev, uintptr_t stream, py::handle dout,
                            py::handle qkv, py::handle att, int H,
                            double temp, py::handle dqkv) {
    attention_bwd_raw(MakeLC(dev, stream), ToArr(dout), ToArr(qkv),
                      0.0, Arr(), ToArr(att), H, temp, ToArr(dqkv));
  });

  r.def("softmax_fwd", [](int dev, uintptr_t stream, py::handle x,
                          py::handle mask, bool log_mode, double temp,
                          py::handle out) {
    softmax_fwd_raw(MakeLC(dev, stream), ToArr(x), ToArr(mask), log_mode,
                    temp, ToArr(out));
  });
  r.def("softmax_bwd", [](int dev, uintptr_t stream, py::handle dy,
                          py::handle y, bool log_mode, double temp,
                          py::handle out) {
    softmax_bwd_raw(MakeLC(dev, stream), ToArr(dy), ToArr(y), log_mode,
                    temp, ToArr(out));
  });
  r.def("colsum", [](int dev, uintptr_t stream, py::handle x,
                     py::handle out) {
    colsum_raw(MakeLC(dev, stream), ToArr(x), ToArr(out));
  });

  r.def("bn_fwd_train", [](int dev, uintptr_t stream, py::handle x,
                           py::handle gamma, py::handle beta,
                           py::handle rmean, py::handle rvar,
                           double momentum, double eps, bool fuse_relu,
                           py::handle residual, py::handle presums,
                           py::handle y, py::handle save_mean,
                           py::handle save_inv, py::handle mask) {
    bn_fwd_train_raw(MakeLC(dev, stream), ToArr(x), ToArr(gamma),
                     ToArr(beta), ToArr(rmean), ToArr(rvar), momentum, eps,
                     fuse_relu, ToArr(residual), ToArr(presums), ToArr(y),
                     ToArr(save_mean), ToArr(save_inv), ToArr(mask));
  });
  r.def("bn_fwd_infer", [](int dev, uintptr_t stream, py::handle x,
                           py::handle gamma, py::handle beta,
                           py::handle rmean, py::handle rvar, double eps,
                           bool fuse_relu, py::handle residual,
                           py::handle y) {
    bn_fwd_infer_raw(MakeLC(dev, stream), ToArr(x), ToArr(gamma),
                     ToArr(beta), ToArr(rmean), ToArr(rvar), eps, fuse_relu,
                     ToArr(residual), ToArr(y));
  });
  r.def("bn_bwd", [](int dev, uintptr_t stream, py::handle dy, py::handle x,
                     py::handle gamma, py::handle save_mean,
                     py::handle save_inv, bool fused_relu, py::handle y,
                     bool has_res, py::handle mask, py::handle dx,
                     py::handle dgamma, py::handle dbeta,
                     py::handle dres) {
    bn_bwd_raw(MakeLC(dev, stream), ToArr(dy), ToArr(x), ToArr(gamma),
               ToArr(save_mean), ToArr(save_inv), fused_relu, ToArr(y),
               has_res, ToArr(mask), ToArr(dx), ToArr(dgamma), ToArr(dbeta),
               ToArr(dres));
  });
  r.def("layernorm_fwd", [](int dev, uintptr_t stream, py::handle x,
                            py::handle gamma, py::handle beta, double eps,
                            py::handle y, py::handle mean,
                            py::handle rstd) {
    layernorm_fwd_raw(MakeLC(dev, stream), ToArr(x), ToArr(gamma),
                      ToArr(beta), eps, ToArr(y), ToArr(mean), ToArr(rstd));
  });
  r.def("layernorm_bwd", [](int dev, uintptr_t stream, py::handle dy,
                            py::handle x, py::handle gamma, py::handle mean,
                            py::handle rstd, py::handle dx,
                            py::handle dgamma, py::handle dbeta) {
    layernorm_bwd_raw(MakeLC(dev, stream), ToArr(dy), ToArr(x), ToArr(gamma),
                      ToArr(mean), ToArr(rstd), ToArr(dx), ToArr(dgamma),
                      ToArr(dbeta));
  });

  r.def("conv2d_fwd", [](int dev, uintptr_t stream, py::handle x,
                         py::handle w, py::handle bias, int sh, int sw,
                         int ph, int pw, int dh, int dw, int groups,
                         py::handle y, py::handle stats) {
    conv2d_fwd_raw(MakeLC(dev, stream), ToArr(x), ToArr(w), ToArr(bias), sh,
                   sw, ph, pw, dh, dw, groups, ToArr(y), ToArr(stats));
  });
  r.def("conv2d_bwd_data", [](int dev, uintptr_t stream, py::handle dy,
                              py::handle w, int sh, int sw, int ph, int pw,
                              int dh, int dw, int groups, int H, int W,
                              py::handle dx) {
    conv2d_bwd_data_raw(MakeLC(dev, stream), ToArr(dy), ToArr(w), sh, sw,
                        ph, pw, dh, dw, groups, H, W, ToArr(dx));
  });
  r.def("conv2d_bwd_weight", [](int dev, uintptr_t stream, py::handle dy,
                                py::handle x, int sh, int sw, int ph,
                                int pw, int dh, int dw, int groups, int R,
                                int S, py::handle dw_out) {
    conv2d_bwd_weight_raw(MakeLC(dev, stream), ToArr(dy), ToArr(x), sh, sw,
                          ph, pw, dh, dw, groups, R, S, ToArr(dw_out));
  });
  r.def("im2col", [](int dev, uintptr_t stream, py::handle x, int R, int S,
                     int sh, int sw, int ph, int pw, int dh, int dw,
                     py::handle col) {
    im2col_raw(MakeLC(dev, stream), ToArr(x), R, S, sh, sw, ph, pw, dh, dw,
               ToArr(col));
  });

  r.def("pool_fwd", [](int dev, uintptr_t stream, py::handle x,
                       const std::string& mode, int kh, int kw, int sh,
                       int sw, int ph, int pw, bool cip, py::handle y,
                       py::handle argmax) {
    pool_fwd_raw(MakeLC(dev, stream), ToArr(x), mode, kh, kw, sh, sw, ph,
                 pw, cip, ToArr(y), ToArr(argmax));
  });
  r.def("pool_bwd", [](int dev, uintptr_t stream, py::handle dy,
                       py::handle argmax, const std::string& mode, int kh,
                       int kw, int sh, int sw, int ph, int pw, int H, int W,
                       bool cip, py::handle dx) {
    pool_bwd_raw(MakeLC(dev, stream), ToArr(dy), ToArr(argmax), mode, kh,
                 kw, sh, sw, ph, pw, H, W, cip, ToArr(dx));
  });

  r.def("act_fwd", [](int dev, uintptr_t stream, py::handle x,
                      const std::string& kind, py::handle y) {
    act_fwd_raw(MakeLC(dev, stream), ToArr(x), kind, ToArr(y));
  });
  r.def("act_bwd", [](int dev, uintptr_t stream, py::handle dy,
                      py::handle saved, const std::string& kind,
                      py::handle dx) {
    act_bwd_raw(MakeLC(dev, stream), ToArr(dy), ToArr(saved), kind,
                ToArr(dx));
  });
  r.def("sgd_update", [](int dev, uintptr_t stream, py::handle w,
                         py::handle master, py::handle grad, py::handle mom,
                         double lr, double mu, double wd, double rescale,
                         double clip) {
    sgd_update_raw(MakeLC(dev, stream), ToArr(w), ToArr(master), ToArr(grad),
                   ToArr(mom), lr, mu, wd, rescale, clip);
  });
  r.def("adam_update", [](int dev, uintptr_t stream, py::handle w,
                          py::handle master, py::handle grad, py::handle m,
                          py::handle v, double lr_t, double b1, double b2,
                          double eps, double wd, double rescale, double clip,
                          bool adamw) {
    adam_update_raw(MakeLC(dev, stream), ToArr(w), ToArr(master),
                    ToArr(grad), ToArr(m), ToArr(v), lr_t, b1, b2, eps, wd,
                    rescale, clip, adamw);
  });
  r.def("multi_sgd_update", [](int dev, uintptr_t stream, py::handle ws,
                               py::handle masters, py::handle grads,
                               py::handle moms, std::vector<double> lrs,
                               std::vector<double> wds, double mu,
                               double rescale, double clip) {
    multi_sgd_update_raw(MakeLC(dev, stream), ToArrs(ws), ToArrs(masters),
                         ToArrs(grads), ToArrs(moms), lrs, wds, mu, rescale,
                         clip);
  });
  r.def("multi_all_finite", [](int dev, uintptr_t stream, py::handle ts,
                               py::handle finite_out) {
    multi_all_finite_raw(MakeLC(dev, stream), ToArrs(ts), ToArr(finite_out));
  });
  r.def("lstm_cell_fwd", [](int dev, uintptr_t stream, py::handle gates,
                            py::handle c_prev, py::handle h_out,
                            py::handle c_out) {
    lstm_cell_fwd_raw(MakeLC(dev, stream), ToArr(gates), ToArr(c_prev),
                      ToArr(h_out), ToArr(c_out));
  });
  r.def("dropout_fwd", [](int dev, uintptr_t stream, py::handle x, double p,
                          int64_t seed, py::handle y, py::handle mask) {
    dropout_fwd_raw(MakeLC(dev, stream), ToArr(x), p, seed, ToArr(y),
                    ToArr(mask));
  });
  r.def("dropout_bwd", [](int dev, uintptr_t stream, py::handle dy,
                          py::handle mask, double p, py::handle dx) {
    dropout_bwd_raw(MakeLC(dev, stream), ToArr(dy), ToArr(mask), p,
                    ToArr(dx));
  });
  r.def("embedding_fwd", [](int dev, uintptr_t stream, py::handle weight,
                            py::handle idx, py::handle out) {
    embedding_fwd_raw(MakeLC(dev, stream), ToArr(weight), ToArr(idx),
                      ToArr(out));
  });
  r.def("embedding_bwd", [](int dev, uintptr_t stream, py::handle dy,
                            py::handle idx, py::handle dweight) {
    embedding_bwd_raw(MakeLC(dev, stream), ToArr(dy), ToArr(idx),
                      ToArr(dweight));
  });

  r.def("quantize_i8", [](int dev, uintptr_t stream, py::handle x,
                          double scale, py::handle out) {
    quantize_i8_raw(MakeLC(dev, stream), ToArr(x), scale, ToArr(out));
  });
  r.def("dequantize_i8", [](int dev, uintptr_t stream, py::handle x,
                            double scale, py::handle out) {
    dequantize_i8_raw(MakeLC(dev, stream), ToArr(x), scale, ToArr(out));
  });
  r.def("gemm_nt_i8", [](int dev, uintptr_t stream, py::handle a,
                         py::handle b, double scale, py::handle out) {
    gemm_nt_i8_raw(MakeLC(dev, stream), ToArr(a), ToArr(b), scale,
                   ToArr(out));
  });
}
