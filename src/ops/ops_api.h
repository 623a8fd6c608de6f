// Native kernel-launcher API: raw pointers + explicit stream, outputs
// allocated by the caller (the registry's imperative invoke, or the
// Python frontend passing torch-tensor data_ptrs).  Replaces the round-1
// at::Tensor glue (no ATen, no torch/extension, no
// hipify anywhere in the kernel library).
#pragma once

#include "native_common.h"

namespace mxcore {

// ---- gemm.hip -------------------------------------------------------------
// out[.., M, N] = A[.., M, K] @ B[.., N, K]^T (+bias) (opt relu);
// stats (optional, fp32 [64, 2, N]) = fused per-channel sum/ssq slices
void gemm_nt_raw(const LaunchCtx& lc, const Arr& A, const Arr& B,
                 const Arr& bias, const Arr& out, bool relu,
                 const Arr& stats);
void transpose2d_raw(const LaunchCtx& lc, const Arr& x, const Arr& out);
void gemm_raw(const LaunchCtx& lc, const Arr& a, const Arr& b,
              const Arr& out);                       // a @ b
void gemm_nn_raw(const LaunchCtx& lc, const Arr& dy, const Arr& w,
                 const Arr& out);                    // dy @ w
void gemm_tn_raw(const LaunchCtx& lc, const Arr& dy, const Arr& x,
                 const Arr& out);                    // dy^T @ x
void bgemm_raw(const LaunchCtx& lc, const Arr& a, const Arr& b,
               const Arr& out);                      // batched a @ b
void gemm_nt_8ph_raw(const LaunchCtx& lc, const Arr& A, const Arr& B,
                     const Arr& out);
// C[I,J] = A[M,I]^T B[M,J]; dbias (optional fp32 [I]) = colsum(A)
void gemm_tn_fused_raw(const LaunchCtx& lc, const Arr& A, const Arr& B,
                       const Arr& C, const Arr& dbias);
// qkv [B,S,3U] -> out [B,S,U], att [B*H,S,S] (saved for backward)
void attention_fwd_raw(const LaunchCtx& lc, const Arr& qkv, const Arr& mask,
                       double p, int64_t seed, const Arr& dropmask,
                       int H, double temperature, const Arr& out,
                       const Arr& att);
void attention_bwd_raw(const LaunchCtx& lc, const Arr& dout, const Arr& qkv,
                       double p, const Arr& dropmask,
                       const Arr& att, int H, double temperature,
                       const Arr& dqkv);

// ---- softmax.hip ----------------------------------------------------------
void softmax_fwd_raw(const LaunchCtx& lc, const Arr& x, const Arr& mask,
                     bool log_mode, double temperature, const Arr& out);
void softmax_bwd_raw(const LaunchCtx& lc, const Arr& dy, const Arr& y,
                     bool log_mode, double temperature, const Arr& out,
                     const Arr& dropmask = Arr(), double p = 0.0);
void colsum_raw(const LaunchCtx& lc, const Arr& x, const Arr& out);

// ---- norm.hip -------------------------------------------------------------
// NHWC BatchNorm: x [N,H,W,C] (any leading dims, C last)
void bn_fwd_train_raw(const LaunchCtx& lc, const Arr& x, const Arr& gamma,
                      const Arr& beta, const Arr& rmean, const Arr& rvar,
                      double momentum, double eps, bool fuse_relu,
                      const Arr& residual, const Arr& presums,
                      const Arr& y, const Arr& save_mean,
                      const Arr& save_inv, const Arr& mask);
void bn_fwd_infer_raw(const LaunchCtx& lc, const Arr& x, const Arr& gamma,
                      const Arr& beta, const Arr& rmean, const Arr& rvar,
                      double eps, bool fuse_relu, const Arr& residual,
                      const Arr& y);
void bn_bwd_raw(const LaunchCtx& lc, const Arr& dy, const Arr& x,
                const Arr& gamma, const Arr& save_mean, const Arr& save_inv,
                bool fused_relu, const Arr& y_or_empty, bool has_residual,
                const Arr& mask, const Arr& dx, const Arr& dgamma,
                const Arr& dbeta, const Arr& dresidual);
void layernorm_fwd_raw(const LaunchCtx& lc, const Arr& x, const Arr& gamma,
                       const Arr& beta, double eps, const Arr& y,
                       const Arr& mean, const Arr& rstd);
void layernorm_bwd_raw(const LaunchCtx& lc, const Arr& dy, const Arr& x,
                       const Arr& gamma, const Arr& mean, const Arr& rstd,
                       const Arr& dx, const Arr& dgamma, const Arr& dbeta);

// ---- conv.hip (NHWC) ------------------------------------------------------
void conv2d_fwd_raw(const LaunchCtx& lc, const Arr& x, const Arr& w,
                    const Arr& bias, int sh, int sw, int ph, int pw, int dh,
                    int dw, int groups, const Arr& y, const Arr& stats);
void conv2d_bwd_data_raw(const LaunchCtx& lc, const Arr& dy, const Arr& w,
                         int sh, int sw, int ph, int pw, int dh, int dw,
                         int groups, int H, int W, const Arr& dx);
void conv2d_bwd_weight_raw(const LaunchCtx& lc, const Arr& dy, const Arr& x,
                           int sh, int sw, int ph, int pw, int dh, int dw,
                           int groups, int R, int S, const Arr& dw_out);
void im2col_raw(const LaunchCtx& lc, const Arr& x, int R, int S, int sh,
                int sw, int ph, int pw, int dh, int dw, const Arr& col);

// ---- pool.hip (NHWC) ------------------------------------------------------
void pool_fwd_raw(const LaunchCtx& lc, const Arr& x, const std::string& mode,
                  int kh, int kw, int sh, int sw, int ph, int pw,
                  bool count_include_pad, const Arr& y, const Arr& argmax);
void pool_bwd_raw(const LaunchCtx& lc, const Arr& dy, const Arr& argmax,
                  const std::string& mode, int kh, int kw, int sh, int sw,
                  int ph, int pw, int H, int W, bool count_include_pad,
                  const Arr& dx);

// ---- elemwise.hip ---------------------------------------------------------
void act_fwd_raw(const LaunchCtx& lc, const Arr& x, const std::string& kind,
                 const Arr& y);
void act_bwd_raw(const LaunchCtx& lc, const Arr& dy, const Arr& saved,
                 const std::string& kind, const Arr& dx);
void sgd_update_raw(const LaunchCtx& lc, const Arr& w, const Arr& master,
                    const Arr& grad, const Arr& mom, double lr, double mu,
                    double wd, double rescale, double clip);
void adam_update_raw(const LaunchCtx& lc, const Arr& w, const Arr& master,
                     const Arr& grad, const Arr& m, const Arr& v,
                     double lr_t, double b1, double b2, double eps,
                     double wd, double rescale, double clip, bool adamw);
void multi_sgd_update_raw(const LaunchCtx& lc, const std::vector<Arr>& ws,
                          const std::vector<Arr>& masters,
                          const std::vector<Arr>& grads,
                          const std::vector<Arr>& moms,
                          const std::vector<double>& lrs,
                          const std::vector<double>& wds, double mu,
                          double rescale, double clip);
// result written into finite_out (int32[1], 1 = all finite)
void multi_adam_update_raw(const LaunchCtx& lc, const std::vector<Arr>& ws,
                           const std::vector<Arr>& gs,
                           const std::vector<Arr>& ms,
                           const std::vector<Arr>& vs,
                           const std::vector<Arr>& masters, double lr_t,
                           double b1, double b2, double eps, double wd,
                           double rescale, double clip, bool adamw);
void multi_copy_raw(const LaunchCtx& lc, const std::vector<Arr>& srcs,
                    const std::vector<Arr>& dsts);
int multi_copy_mode(int src_dtype, int dst_dtype);
void multi_all_finite_raw(const LaunchCtx& lc, const std::vector<Arr>& ts,
                          const Arr& finite_out);
void lstm_cell_fwd_raw(const LaunchCtx& lc, const Arr& gates,
                       const Arr& c_prev, const Arr& h_out, const Arr& c_out);
void dropout_fwd_raw(const LaunchCtx& lc, const Arr& x, double p,
                     int64_t seed, const Arr& y, const Arr& mask);
void dropout_bwd_raw(const LaunchCtx& lc, const Arr& dy, const Arr& mask,
                     double p, const Arr& dx);
void embedding_fwd_raw(const LaunchCtx& lc, const Arr& weight,
                       const Arr& idx, const Arr& out);
void embedding_bwd_raw(const LaunchCtx& lc, const Arr& dy, const Arr& idx,
                       const Arr& dweight);

// ---- quant.hip ------------------------------------------------------------
void quantize_i8_raw(const LaunchCtx& lc, const Arr& x, double scale,
                     const Arr& out);
void dequantize_i8_raw(const LaunchCtx& lc, const Arr& x, double scale,
                       const Arr& out);
void gemm_nt_i8_raw(const LaunchCtx& lc, const Arr& a, const Arr& b,
                    double scale, const Arr& out);

}  // namespace mxcore
