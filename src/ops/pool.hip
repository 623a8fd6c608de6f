// NHWC pooling (max / avg) fwd + bwd.
//
// Reference parity: src/operator/nn/pool.cuh (pool_max_2d / pool_sum_2d and
// unpool kernels).  MI355X design: channels-contiguous NHWC so one thread
// per output element reads coalesced C-segments; max-pool saves the argmax
// plane index, backward scatters with fp32 atomics (windows overlap).
#include "native_common.h"

using namespace mxcore;

template <typename T, bool IS_MAX>
__global__ void pool_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                int* __restrict__ arg, long total, int N,
                                int H, int W, int C, int P, int Q, int kh,
                                int kw, int sh, int sw, int ph, int pw,
                                bool cip) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c = i % C;
    long t = i / C;
    int q = t % Q;
    t /= Q;
    int p = t % P;
    int n = t / P;
    int h0 = p * sh - ph, w0 = q * sw - pw;
    int h1 = min(h0 + kh, H), w1 = min(w0 + kw, W);
    int hs = max(h0, 0), ws = max(w0, 0);
    const T* xn = x + (long)n * H * W * C;
    if (IS_MAX) {
      float best = -INFINITY;
      int best_idx = hs * W + ws;
      for (int h = hs; h < h1; ++h)
        for (int w = ws; w < w1; ++w) {
          float v = (float)xn[((long)h * W + w) * C + c];
          if (v > best) {
            best = v;
            best_idx = h * W + w;
          }
        }
      y[i] = (T)best;
      arg[i] = best_idx;
    } else {
      float s = 0.f;
      for (int h = hs; h < h1; ++h)
        for (int w = ws; w < w1; ++w) s += (float)xn[((long)h * W + w) * C + c];
      int cnt = cip ? kh * kw : (h1 - hs) * (w1 - ws);
      y[i] = (T)(s / cnt);
    }
  }
}

template <typename T, bool IS_MAX>
__global__ void pool_bwd_kernel(const T* __restrict__ dy,
                                const int* __restrict__ arg,
                                float* __restrict__ dx32, long total, int N,
                                int H, int W, int C, int P, int Q, int kh,
                                int kw, int sh, int sw, int ph, int pw,
                                bool cip) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c = i % C;
    long t = i / C;
    int q = t % Q;
    t /= Q;
    int p = t % P;
    int n = t / P;
    float g = (float)dy[i];
    float* dxn = dx32 + (long)n * H * W * C;
    if (IS_MAX) {
      atomicAdd(dxn + (long)arg[i] * C + c, g);
    } else {
      int h0 = p * sh - ph, w0 = q * sw - pw;
      int h1 = min(h0 + kh, H), w1 = min(w0 + kw, W);
      int hs = max(h0, 0), ws = max(w0, 0);
      int cnt = cip ? kh * kw : (h1 - hs) * (w1 - ws);
      float share = g / cnt;
      for (int h = hs; h < h1; ++h)
        for (int w = ws; w < w1; ++w)
          atomicAdd(dxn + ((long)h * W + w) * C + c, share);
    }
  }
}



// max-pool backward, gather form: each input element sums dy over the
// (few) windows whose saved argmax selected it (no zero pass, no
// atomics, no fp32 scratch; 3x3 s2 has at most 4 covering windows)
template <typename T>
__global__ void maxpool_bwd_gather_kernel(const T* __restrict__ dy,
                                          const int* __restrict__ arg,
                                          T* __restrict__ dx, long total,
                                          int N, int H, int W, int C, int P,
                                          int Q, int kh, int kw, int sh,
                                          int sw, int ph, int pw) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c = i % C;
    long t = i / C;
    int w = t % W;
    long t2 = t / W;
    int h = t2 % H;
    int n = t2 / H;
    const long planebase = (long)n * P * Q;
    const int me = h * W + w;
    float acc = 0.f;
    int pn = h + ph - kh + 1;
    int pstart = pn <= 0 ? 0 : (pn + sh - 1) / sh;
    int qn = w + pw - kw + 1;
    int qstart = qn <= 0 ? 0 : (qn + sw - 1) / sw;
    for (int p = pstart; p < P; ++p) {
      if (p * sh - ph > h) break;
      for (int q = qstart; q < Q; ++q) {
        if (q * sw - pw > w) break;
        long o = (planebase + (long)p * Q + q) * C + c;
        if (arg[o] == me) acc += (float)dy[o];
      }
    }
    dx[i] = (T)acc;
  }
}

// avg-pool backward, gather form: each input element sums the shares of
// every window covering it (deterministic, no atomics, no fp32 scratch)
template <typename T>
__global__ void avgpool_bwd_gather_kernel(const T* __restrict__ dy,
                                          T* __restrict__ dx, long total,
                                          int N, int H, int W, int C, int P,
                                          int Q, int kh, int kw, int sh,
                                          int sw, int ph, int pw, bool cip) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    int c = i % C;
    long t = i / C;
    int w = t % W;
    long t2 = t / W;
    int h = t2 % H;
    int n = t2 / H;
    const T* dyn = dy + (long)n * P * Q * C;
    float acc = 0.f;
    int pn = h + ph - kh + 1;          // smallest p with h < p*sh-ph+kh
    int pstart = pn <= 0 ? 0 : (pn + sh - 1) / sh;
    int qn = w + pw - kw + 1;
    int qstart = qn <= 0 ? 0 : (qn + sw - 1) / sw;
    for (int p = pstart; p < P; ++p) {
      int h0 = p * sh - ph;
      if (h0 > h) break;
      int h1 = min(h0 + kh, H), hs = max(h0, 0);
      for (int q = qstart; q < Q; ++q) {
        int w0 = q * sw - pw;
        if (w0 > w) break;
        int w1 = min(w0 + kw, W), ws = max(w0, 0);
        int cnt = cip ? kh * kw : (h1 - hs) * (w1 - ws);
        acc += (float)dyn[((long)p * Q + q) * C + c] / cnt;
      }
    }
    dx[i] = (T)acc;
  }
}


// ===========================================================================
// native host launchers
// ===========================================================================
#include "ops_api.h"

namespace mxcore {

void pool_fwd_raw(const LaunchCtx& lc, const Arr& x, const std::string& kind,
                  int kh, int kw, int sh, int sw, int ph, int pw, bool cip,
                  const Arr& y, const Arr& argmax) {
  MX_CHECK(x.dim() == 4, "pool expects NHWC 4-D input");
  int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int P = y.size(1), Q = y.size(2);
  bool is_max = kind == "max";
  long total = y.numel();
  if (total == 0) return;
  DISPATCH_FLOAT_NATIVE(x.dtype, "pool_fwd", [&] {
    if (is_max)
      pool_fwd_kernel<scalar_t, true><<<ew_grid_n(total), 256, 0,
                                        lc.stream>>>(
          x.data<scalar_t>(), (scalar_t*)y.ptr, argmax.data<int>(), total,
          N, H, W, C, P, Q, kh, kw, sh, sw, ph, pw, cip);
    else
      pool_fwd_kernel<scalar_t, false><<<ew_grid_n(total), 256, 0,
                                         lc.stream>>>(
          x.data<scalar_t>(), (scalar_t*)y.ptr, nullptr, total, N, H, W, C,
          P, Q, kh, kw, sh, sw, ph, pw, cip);
  });
  HIP_CHECK_LAST();
}

void pool_bwd_raw(const LaunchCtx& lc, const Arr& dy, const Arr& argmax,
                  const std::string& kind, int kh, int kw, int sh, int sw,
                  int ph, int pw, int H, int W, bool cip, const Arr& dx) {
  int N = dy.size(0), P = dy.size(1), Q = dy.size(2), C = dy.size(3);
  bool is_max = kind == "max";
  long total = dx.numel();
  DISPATCH_FLOAT_NATIVE(dy.dtype, "pool_bwd", [&] {
    if (!is_max)
      avgpool_bwd_gather_kernel<scalar_t><<<ew_grid_n(total), 256, 0,
                                            lc.stream>>>(
          dy.data<scalar_t>(), (scalar_t*)dx.ptr, total, N, H, W, C, P, Q,
          kh, kw, sh, sw, ph, pw, cip);
    else
      maxpool_bwd_gather_kernel<scalar_t><<<ew_grid_n(total), 256, 0,
                                            lc.stream>>>(
          dy.data<scalar_t>(), argmax.data<int>(), (scalar_t*)dx.ptr, total,
          N, H, W, C, P, Q, kh, kw, sh, sw, ph, pw);
  });
  HIP_CHECK_LAST();
}

}  // namespace mxcore
