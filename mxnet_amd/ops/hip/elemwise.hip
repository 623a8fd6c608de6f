// Elementwise kernels: activations, fused optimizer updates, dropout,
// embedding, LSTM cell pointwise, all-finite check.
//
// Reference parity: src/operator/nn/activation.cu, optimizer_op-inl.h
// (sgd_mom_update / mp_sgd_mom_update / adam_update), dropout-inl.h,
// indexing_op.cu (Embedding), rnn-inl.h LSTM cell.
//
// MI355X design: every kernel is memory-bound -> grid-stride loops over
// 8-element vectors (16 B/lane fp16: guide Guideline 13), fp32 math
// internally, fused single-pass updates (the reference launches 4-6
// separate mshadow kernels per optimizer step; here it is one).
#include "torch_common.h"

// ---------------------------------------------------------------------------
// activation
// ---------------------------------------------------------------------------
enum ActKind { ACT_RELU = 0, ACT_SIGMOID = 1, ACT_TANH = 2, ACT_GELU = 3,
               ACT_SILU = 4 };

DEV_INLINE float act_apply(float x, int kind) {
  switch (kind) {
    case ACT_RELU: return x > 0.f ? x : 0.f;
    case ACT_SIGMOID: return 1.f / (1.f + __expf(-x));
    case ACT_TANH: return tanhf(x);
    case ACT_GELU: {  // tanh approximation (reference LeakyReLU gelu)
      float c = 0.7978845608028654f * (x + 0.044715f * x * x * x);
      return 0.5f * x * (1.f + tanhf(c));
    }
    case ACT_SILU: return x / (1.f + __expf(-x));
  }
  return x;
}

// saved = y for relu/sigmoid/tanh, x for gelu/silu
DEV_INLINE float act_grad(float dy, float s, int kind) {
  switch (kind) {
    case ACT_RELU: return s > 0.f ? dy : 0.f;
    case ACT_SIGMOID: return dy * s * (1.f - s);
    case ACT_TANH: return dy * (1.f - s * s);
    case ACT_GELU: {
      float x = s;
      float u = 0.7978845608028654f * (x + 0.044715f * x * x * x);
      float t = tanhf(u);
      float du = 0.7978845608028654f * (1.f + 3.f * 0.044715f * x * x);
      return dy * (0.5f * (1.f + t) + 0.5f * x * (1.f - t * t) * du);
    }
    case ACT_SILU: {
      float sig = 1.f / (1.f + __expf(-s));
      return dy * sig * (1.f + s * (1.f - sig));
    }
  }
  return dy;
}

// vectorized: VEC elements per thread per grid-stride step
template <typename T, int VEC>
__global__ void act_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                               long n, int kind) {
  using VecT = T __attribute__((ext_vector_type(VEC)));
  long nv = n / VEC;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
       i += (long)gridDim.x * blockDim.x) {
    VecT v = reinterpret_cast<const VecT*>(x)[i];
    VecT o;
#pragma unroll
    for (int j = 0; j < VEC; ++j) o[j] = (T)act_apply((float)v[j], kind);
    reinterpret_cast<VecT*>(y)[i] = o;
  }
  // tail
  long base = nv * VEC;
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (tid < n - base) y[base + tid] = (T)act_apply((float)x[base + tid], kind);
}

template <typename T, int VEC>
__global__ void act_bwd_kernel(const T* __restrict__ dy,
                               const T* __restrict__ saved,
                               T* __restrict__ dx, long n, int kind) {
  using VecT = T __attribute__((ext_vector_type(VEC)));
  long nv = n / VEC;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
       i += (long)gridDim.x * blockDim.x) {
    VecT g = reinterpret_cast<const VecT*>(dy)[i];
    VecT s = reinterpret_cast<const VecT*>(saved)[i];
    VecT o;
#pragma unroll
    for (int j = 0; j < VEC; ++j)
      o[j] = (T)act_grad((float)g[j], (float)s[j], kind);
    reinterpret_cast<VecT*>(dx)[i] = o;
  }
  long base = nv * VEC;
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (tid < n - base)
    dx[base + tid] =
        (T)act_grad((float)dy[base + tid], (float)saved[base + tid], kind);
}

static int act_kind_from_string(const std::string& s) {
  if (s == "relu") return ACT_RELU;
  if (s == "sigmoid") return ACT_SIGMOID;
  if (s == "tanh") return ACT_TANH;
  if (s == "gelu") return ACT_GELU;
  if (s == "silu" || s == "swish") return ACT_SILU;
  TORCH_CHECK(false, "unknown activation ", s);
}

at::Tensor act_fwd(const at::Tensor& x, const std::string& kind) {
  CHECK_GPU(x); CHECK_CONTIG(x);
  auto y = at::empty_like(x);
  long n = x.numel();
  if (n == 0) return y;
  int k = act_kind_from_string(kind);
  DISPATCH_FLOAT_TYPES(x.scalar_type(), "act_fwd", [&] {
    constexpr int VEC = sizeof(scalar_t) == 2 ? 8 : 4;
    act_fwd_kernel<scalar_t, VEC><<<ew_grid(n / VEC + 1), kEwBlock, 0,
                                    cur_stream()>>>(
        (const scalar_t*)x.data_ptr(), (scalar_t*)y.data_ptr(), n, k);
  });
  HIP_CHECK_LAST();
  return y;
}

at::Tensor act_bwd(const at::Tensor& dy, const at::Tensor& saved,
                   const std::string& kind) {
  CHECK_GPU(dy); CHECK_CONTIG(dy); CHECK_CONTIG(saved);
  auto dx = at::empty_like(dy);
  long n = dy.numel();
  if (n == 0) return dx;
  int k = act_kind_from_string(kind);
  DISPATCH_FLOAT_TYPES(dy.scalar_type(), "act_bwd", [&] {
    constexpr int VEC = sizeof(scalar_t) == 2 ? 8 : 4;
    act_bwd_kernel<scalar_t, VEC><<<ew_grid(n / VEC + 1), kEwBlock, 0,
                                    cur_stream()>>>(
        (const scalar_t*)dy.data_ptr(), (const scalar_t*)saved.data_ptr(),
        (scalar_t*)dx.data_ptr(), n, k);
  });
  HIP_CHECK_LAST();
  return dx;
}

// ---------------------------------------------------------------------------
// fused SGD (momentum, multi-precision) — reference mp_sgd_mom_update
// one pass: g = clip(grad*rescale) + wd*w32; m = mu*m + g; w32 -= lr*m;
//           w16 = cast(w32)
// ---------------------------------------------------------------------------
template <typename T>
__global__ void sgd_mp_kernel(T* __restrict__ w, float* __restrict__ master,
                              const T* __restrict__ grad,
                              float* __restrict__ mom, long n, float lr,
                              float mu, float wd, float rescale, float clip) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float wm = master ? master[i] : (float)w[i];
    float g = (float)grad[i] * rescale;
    if (clip > 0.f) g = fminf(fmaxf(g, -clip), clip);
    g += wd * wm;
    // reference rule: lr folded into the momentum buffer
    // (mom = mu*mom - lr*g; w += mom)
    if (mom) {
      float m = mom[i] * mu - lr * g;
      mom[i] = m;
      wm += m;
    } else {
      wm -= lr * g;
    }
    if (master) master[i] = wm;
    w[i] = (T)wm;
  }
}

void sgd_update(at::Tensor w, c10::optional<at::Tensor> master, at::Tensor grad,
                c10::optional<at::Tensor> mom, double lr, double mu, double wd,
                double rescale, double clip) {
  CHECK_GPU(w); CHECK_CONTIG(w); CHECK_CONTIG(grad);
  long n = w.numel();
  if (n == 0) return;
  DISPATCH_FLOAT_TYPES(w.scalar_type(), "sgd_update", [&] {
    sgd_mp_kernel<scalar_t><<<ew_grid(n), kEwBlock, 0, cur_stream()>>>(
        (scalar_t*)w.data_ptr(),
        master ? master->data_ptr<float>() : nullptr,
        (const scalar_t*)grad.data_ptr(),
        mom ? mom->data_ptr<float>() : nullptr, n, (float)lr, (float)mu,
        (float)wd, (float)rescale, (float)clip);
  });
  HIP_CHECK_LAST();
}

// fused Adam — reference adam_update / mp_adam_update
template <typename T>
__global__ void adam_mp_kernel(T* __restrict__ w, float* __restrict__ master,
                               const T* __restrict__ grad,
                               float* __restrict__ m, float* __restrict__ v,
                               long n, float lr_t, float b1, float b2,
                               float eps, float wd, float rescale, float clip,
                               bool adamw) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float wm = master ? master[i] : (float)w[i];
    float g = (float)grad[i] * rescale;
    if (clip > 0.f) g = fminf(fmaxf(g, -clip), clip);
    if (!adamw) g += wd * wm;
    float mi = m[i] = b1 * m[i] + (1.f - b1) * g;
    float vi = v[i] = b2 * v[i] + (1.f - b2) * g * g;
    wm -= lr_t * mi / (sqrtf(vi) + eps);
    if (adamw) wm -= lr_t * wd * wm;  // decoupled decay
    if (master) master[i] = wm;
    w[i] = (T)wm;
  }
}

void adam_update(at::Tensor w, c10::optional<at::Tensor> master,
                 at::Tensor grad, at::Tensor m, at::Tensor v, double lr_t,
                 double b1, double b2, double eps, double wd, double rescale,
                 double clip, bool adamw) {
  CHECK_GPU(w); CHECK_CONTIG(w); CHECK_CONTIG(grad);
  long n = w.numel();
  if (n == 0) return;
  DISPATCH_FLOAT_TYPES(w.scalar_type(), "adam_update", [&] {
    adam_mp_kernel<scalar_t><<<ew_grid(n), kEwBlock, 0, cur_stream()>>>(
        (scalar_t*)w.data_ptr(),
        master ? master->data_ptr<float>() : nullptr,
        (const scalar_t*)grad.data_ptr(), m.data_ptr<float>(),
        v.data_ptr<float>(), n, (float)lr_t, (float)b1, (float)b2, (float)eps,
        (float)wd, (float)rescale, (float)clip, adamw);
  });
  HIP_CHECK_LAST();
}

// ---------------------------------------------------------------------------
// all-finite check over a list of tensors (AMP loss scaler,
// reference multi_all_finite op: contrib/all_finite.cu)
// ---------------------------------------------------------------------------
template <typename T>
__global__ void notfinite_kernel(const T* __restrict__ x, long n,
                                 int* __restrict__ flag) {
  int bad = 0;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float v = (float)x[i];
    bad |= !isfinite(v);
  }
  if (__builtin_amdgcn_ballot_w64(bad) != 0 && (threadIdx.x & 63) == 0)
    atomicOr(flag, 1);
}

bool multi_all_finite(std::vector<at::Tensor> tensors) {
  if (tensors.empty()) return true;
  auto flag = at::zeros({1}, tensors[0].options().dtype(at::kInt));
  for (auto& t : tensors) {
    long n = t.numel();
    if (n == 0) continue;
    DISPATCH_FLOAT_TYPES(t.scalar_type(), "all_finite", [&] {
      notfinite_kernel<scalar_t><<<ew_grid(n), kEwBlock, 0, cur_stream()>>>(
          (const scalar_t*)t.data_ptr(), n, flag.data_ptr<int>());
    });
  }
  HIP_CHECK_LAST();
  return flag.item<int>() == 0;
}

// ---------------------------------------------------------------------------
// LSTM cell pointwise: gates [N,4H] (i,f,g,o mxnet order), c [N,H]
// -> h', c'   (reference rnn-inl.h LSTM cell math, fused here)
// ---------------------------------------------------------------------------
template <typename T>
__global__ void lstm_cell_kernel(const T* __restrict__ gates,
                                 const T* __restrict__ c,
                                 T* __restrict__ h_out, T* __restrict__ c_out,
                                 long n, long H) {
  for (long idx = (long)blockIdx.x * blockDim.x + threadIdx.x; idx < n;
       idx += (long)gridDim.x * blockDim.x) {
    long row = idx / H, col = idx % H;
    const T* g4 = gates + row * 4 * H;
    float i = 1.f / (1.f + __expf(-(float)g4[col]));
    float f = 1.f / (1.f + __expf(-(float)g4[H + col]));
    float g = tanhf((float)g4[2 * H + col]);
    float o = 1.f / (1.f + __expf(-(float)g4[3 * H + col]));
    float cn = f * (float)c[idx] + i * g;
    c_out[idx] = (T)cn;
    h_out[idx] = (T)(o * tanhf(cn));
  }
}

std::vector<at::Tensor> lstm_cell_fwd(const at::Tensor& gates,
                                      const at::Tensor& c) {
  CHECK_GPU(gates); CHECK_CONTIG(gates); CHECK_CONTIG(c);
  auto h_out = at::empty_like(c);
  auto c_out = at::empty_like(c);
  long n = c.numel(), H = c.size(-1);
  DISPATCH_FLOAT_TYPES(gates.scalar_type(), "lstm_cell", [&] {
    lstm_cell_kernel<scalar_t><<<ew_grid(n), kEwBlock, 0, cur_stream()>>>(
        (const scalar_t*)gates.data_ptr(), (const scalar_t*)c.data_ptr(),
        (scalar_t*)h_out.data_ptr(), (scalar_t*)c_out.data_ptr(), n, H);
  });
  HIP_CHECK_LAST();
  return {h_out, c_out};
}

// ---------------------------------------------------------------------------
// dropout (reference dropout-inl.h; philox-style counter hash here)
// ---------------------------------------------------------------------------
DEV_INLINE unsigned hash_u32(unsigned long long x) {
  x ^= x >> 33; x *= 0xff51afd7ed558ccdULL;
  x ^= x >> 33; x *= 0xc4ceb9fe1a85ec53ULL;
  x ^= x >> 33;
  return (unsigned)x;
}

template <typename T>
__global__ void dropout_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   unsigned char* __restrict__ mask, long n,
                                   float p, float inv_keep,
                                   unsigned long long seed) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float u = (hash_u32(seed * 0x9E3779B97F4A7C15ULL + (unsigned long long)i)
               >> 8) * (1.f / 16777216.f);
    unsigned char keep = u >= p;
    mask[i] = keep;
    y[i] = keep ? (T)((float)x[i] * inv_keep) : (T)0;
  }
}

template <typename T>
__global__ void dropout_bwd_kernel(const T* __restrict__ dy,
                                   const unsigned char* __restrict__ mask,
                                   T* __restrict__ dx, long n, float inv_keep) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    dx[i] = mask[i] ? (T)((float)dy[i] * inv_keep) : (T)0;
}

std::vector<at::Tensor> dropout_fwd(const at::Tensor& x, double p,
                                    int64_t seed) {
  CHECK_GPU(x); CHECK_CONTIG(x);
  auto y = at::empty_like(x);
  auto mask = at::empty_like(x, x.options().dtype(at::kByte));
  long n = x.numel();
  float inv_keep = 1.f / (1.f - (float)p);
  DISPATCH_FLOAT_TYPES(x.scalar_type(), "dropout_fwd", [&] {
    dropout_fwd_kernel<scalar_t><<<ew_grid(n), kEwBlock, 0, cur_stream()>>>(
        (const scalar_t*)x.data_ptr(), (scalar_t*)y.data_ptr(),
        mask.data_ptr<unsigned char>(), n, (float)p, inv_keep,
        (unsigned long long)seed);
  });
  HIP_CHECK_LAST();
  return {y, mask};
}

at::Tensor dropout_bwd(const at::Tensor& dy, const at::Tensor& mask,
                       double p) {
  CHECK_GPU(dy); CHECK_CONTIG(dy);
  auto dx = at::empty_like(dy);
  long n = dy.numel();
  float inv_keep = 1.f / (1.f - (float)p);
  DISPATCH_FLOAT_TYPES(dy.scalar_type(), "dropout_bwd", [&] {
    dropout_bwd_kernel<scalar_t><<<ew_grid(n), kEwBlock, 0, cur_stream()>>>(
        (const scalar_t*)dy.data_ptr(), mask.data_ptr<unsigned char>(),
        (scalar_t*)dx.data_ptr(), n, inv_keep);
  });
  HIP_CHECK_LAST();
  return dx;
}

// ---------------------------------------------------------------------------
// embedding (reference indexing_op.cu Embedding fwd/bwd)
// fwd: one wave per row, vectorized row copy
// bwd: scatter-add into fp32 workspace (atomics; indices may repeat)
// ---------------------------------------------------------------------------
template <typename T>
__global__ void embedding_fwd_kernel(const long* __restrict__ idx,
                                     const T* __restrict__ weight,
                                     T* __restrict__ out, long nrows, long D,
                                     long V) {
  long wave_id = ((long)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  int lane = threadIdx.x & 63;
  long nwaves = (long)gridDim.x * blockDim.x / 64;
  for (long r = wave_id; r < nrows; r += nwaves) {
    long v = idx[r];
    const T* src = weight + (v < 0 || v >= V ? 0 : v) * D;
    T* dst = out + r * D;
    bool valid = v >= 0 && v < V;
    for (long d = lane; d < D; d += 64) dst[d] = valid ? src[d] : (T)0;
  }
}

template <typename T>
__global__ void embedding_bwd_kernel(const long* __restrict__ idx,
                                     const T* __restrict__ dy,
                                     float* __restrict__ dw, long nrows,
                                     long D, long V) {
  long wave_id = ((long)blockIdx.x * blockDim.x + threadIdx.x) / 64;
  int lane = threadIdx.x & 63;
  long nwaves = (long)gridDim.x * blockDim.x / 64;
  for (long r = wave_id; r < nrows; r += nwaves) {
    long v = idx[r];
    if (v < 0 || v >= V) continue;
    const T* src = dy + r * D;
    float* dst = dw + v * D;
    for (long d = lane; d < D; d += 64) atomicAdd(dst + d, (float)src[d]);
  }
}

at::Tensor embedding_fwd(const at::Tensor& indices, const at::Tensor& weight) {
  CHECK_GPU(weight); CHECK_CONTIG(weight);
  auto idx = indices.to(at::kLong).contiguous();
  long nrows = idx.numel(), D = weight.size(1), V = weight.size(0);
  std::vector<int64_t> oshape(indices.sizes().begin(), indices.sizes().end());
  oshape.push_back(D);
  auto out = at::empty(oshape, weight.options());
  DISPATCH_FLOAT_TYPES(weight.scalar_type(), "embedding_fwd", [&] {
    embedding_fwd_kernel<scalar_t><<<ew_grid(nrows * 64), kEwBlock, 0,
                                     cur_stream()>>>(
        idx.data_ptr<long>(), (const scalar_t*)weight.data_ptr(),
        (scalar_t*)out.data_ptr(), nrows, D, V);
  });
  HIP_CHECK_LAST();
  return out;
}

at::Tensor embedding_bwd(const at::Tensor& indices, const at::Tensor& dy,
                         int64_t vocab) {
  CHECK_GPU(dy);
  auto idx = indices.to(at::kLong).contiguous();
  auto dyc = dy.contiguous();
  long nrows = idx.numel(), D = dy.size(-1);
  auto dw32 = at::zeros({vocab, D}, dyc.options().dtype(at::kFloat));
  DISPATCH_FLOAT_TYPES(dyc.scalar_type(), "embedding_bwd", [&] {
    embedding_bwd_kernel<scalar_t><<<ew_grid(nrows * 64), kEwBlock, 0,
                                     cur_stream()>>>(
        idx.data_ptr<long>(), (const scalar_t*)dyc.data_ptr(),
        dw32.data_ptr<float>(), nrows, D, vocab);
  });
  HIP_CHECK_LAST();
  return dw32.to(dyc.scalar_type());
}

// multi-tensor fused SGD (reference multi_sgd_mom_update /
// preloaded_multi_sgd, optimizer_op.cc): ONE launch updates every
// parameter; chunk table in device memory, binary search per block.
struct MTChunk {
  void* w;
  float* master;
  const void* grad;
  float* mom;
  long start;   // global element offset of this tensor
  long len;
  float lr, wd;
};

template <typename T>
__global__ void multi_sgd_kernel(const MTChunk* __restrict__ chunks,
                                 int nchunks, long total, float mu,
                                 float rescale, float clip) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    // binary search the owning tensor
    int lo = 0, hi = nchunks - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (chunks[mid].start <= i) lo = mid;
      else hi = mid - 1;
    }
    const MTChunk c = chunks[lo];
    long j = i - c.start;
    if (j >= c.len) continue;
    T* w = (T*)c.w;
    const T* g = (const T*)c.grad;
    float wm = c.master ? c.master[j] : (float)w[j];
    float gv = (float)g[j] * rescale;
    if (clip > 0.f) gv = fminf(fmaxf(gv, -clip), clip);
    gv += c.wd * wm;
    // reference rule: lr folded into the momentum buffer
    if (c.mom) {
      float m = c.mom[j] * mu - c.lr * gv;
      c.mom[j] = m;
      wm += m;
    } else {
      wm -= c.lr * gv;
    }
    if (c.master) c.master[j] = wm;
    w[j] = (T)wm;
  }
}

void multi_sgd_update(std::vector<at::Tensor> ws,
                      std::vector<at::Tensor> masters,
                      std::vector<at::Tensor> grads,
                      std::vector<at::Tensor> moms,
                      std::vector<double> lrs, std::vector<double> wds,
                      double mu, double rescale, double clip) {
  int n = (int)ws.size();
  if (n == 0) return;
  std::vector<MTChunk> host(n);
  long total = 0;
  for (int i = 0; i < n; ++i) {
    host[i].w = ws[i].data_ptr();
    host[i].master = masters[i].numel() ? masters[i].data_ptr<float>()
                                        : nullptr;
    host[i].grad = grads[i].data_ptr();
    host[i].mom = moms[i].numel() ? moms[i].data_ptr<float>() : nullptr;
    host[i].start = total;
    host[i].len = ws[i].numel();
    host[i].lr = (float)lrs[i];
    host[i].wd = (float)wds[i];
    total += host[i].len;
  }
  auto table = at::from_blob(host.data(), {(long)(n * sizeof(MTChunk))},
                             at::TensorOptions().dtype(at::kByte))
                   .to(ws[0].device(), /*non_blocking=*/false);
  DISPATCH_FLOAT_TYPES(ws[0].scalar_type(), "multi_sgd", [&] {
    multi_sgd_kernel<scalar_t><<<ew_grid(total), kEwBlock, 0,
                                 cur_stream()>>>(
        (const MTChunk*)table.data_ptr(), n, total, (float)mu,
        (float)rescale, (float)clip);
  });
  HIP_CHECK_LAST();
}
