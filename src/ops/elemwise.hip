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
#include "native_common.h"

using namespace mxcore;

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
  MX_CHECK(false, "unknown activation " << s);
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


// ===========================================================================
// native host launchers
// ===========================================================================
#include <algorithm>

#include "ops_api.h"

namespace mxcore {

void act_fwd_raw(const LaunchCtx& lc, const Arr& x, const std::string& kind,
                 const Arr& y) {
  long n = x.numel();
  if (n == 0) return;
  int k = act_kind_from_string(kind);
  DISPATCH_FLOAT_NATIVE(x.dtype, "act_fwd", [&] {
    constexpr int VEC = sizeof(scalar_t) == 2 ? 8 : 4;
    act_fwd_kernel<scalar_t, VEC><<<ew_grid_n(n / VEC + 1), 256, 0,
                                    lc.stream>>>(
        x.data<scalar_t>(), (scalar_t*)y.ptr, n, k);
  });
  HIP_CHECK_LAST();
}

void act_bwd_raw(const LaunchCtx& lc, const Arr& dy, const Arr& saved,
                 const std::string& kind, const Arr& dx) {
  long n = dy.numel();
  if (n == 0) return;
  int k = act_kind_from_string(kind);
  DISPATCH_FLOAT_NATIVE(dy.dtype, "act_bwd", [&] {
    constexpr int VEC = sizeof(scalar_t) == 2 ? 8 : 4;
    act_bwd_kernel<scalar_t, VEC><<<ew_grid_n(n / VEC + 1), 256, 0,
                                    lc.stream>>>(
        dy.data<scalar_t>(), saved.data<scalar_t>(), (scalar_t*)dx.ptr, n,
        k);
  });
  HIP_CHECK_LAST();
}

void sgd_update_raw(const LaunchCtx& lc, const Arr& w, const Arr& master,
                    const Arr& grad, const Arr& mom, double lr, double mu,
                    double wd, double rescale, double clip) {
  long n = w.numel();
  if (n == 0) return;
  DISPATCH_FLOAT_NATIVE(w.dtype, "sgd_update", [&] {
    sgd_mp_kernel<scalar_t><<<ew_grid_n(n), 256, 0, lc.stream>>>(
        (scalar_t*)w.ptr, master.defined() ? master.data<float>() : nullptr,
        grad.data<scalar_t>(), mom.defined() ? mom.data<float>() : nullptr,
        n, (float)lr, (float)mu, (float)wd, (float)rescale, (float)clip);
  });
  HIP_CHECK_LAST();
}

void adam_update_raw(const LaunchCtx& lc, const Arr& w, const Arr& master,
                     const Arr& grad, const Arr& m, const Arr& v,
                     double lr_t, double b1, double b2, double eps,
                     double wd, double rescale, double clip, bool adamw) {
  long n = w.numel();
  if (n == 0) return;
  DISPATCH_FLOAT_NATIVE(w.dtype, "adam_update", [&] {
    adam_mp_kernel<scalar_t><<<ew_grid_n(n), 256, 0, lc.stream>>>(
        (scalar_t*)w.ptr, master.defined() ? master.data<float>() : nullptr,
        grad.data<scalar_t>(), m.data<float>(), v.data<float>(), n,
        (float)lr_t, (float)b1, (float)b2, (float)eps, (float)wd,
        (float)rescale, (float)clip, adamw);
  });
  HIP_CHECK_LAST();
}

void multi_all_finite_raw(const LaunchCtx& lc, const std::vector<Arr>& ts,
                          const Arr& finite_out) {
  int* flag = finite_out.data<int>();
  // flag semantics: 0 after this pass = all finite (kernel sets 1 on bad);
  // caller pre-reads as "finite = (flag == 0)"
  MX_HIP_CALL(hipMemsetAsync(flag, 0, 4, lc.stream));
  for (auto& t : ts) {
    long n = t.numel();
    if (n == 0) continue;
    DISPATCH_FLOAT_NATIVE(t.dtype, "all_finite", [&] {
      notfinite_kernel<scalar_t><<<ew_grid_n(n), 256, 0, lc.stream>>>(
          t.data<scalar_t>(), n, flag);
    });
  }
  HIP_CHECK_LAST();
}

void lstm_cell_fwd_raw(const LaunchCtx& lc, const Arr& gates,
                       const Arr& c_prev, const Arr& h_out,
                       const Arr& c_out) {
  long n = c_prev.numel(), H = c_prev.size(-1);
  DISPATCH_FLOAT_NATIVE(gates.dtype, "lstm_cell", [&] {
    lstm_cell_kernel<scalar_t><<<ew_grid_n(n), 256, 0, lc.stream>>>(
        gates.data<scalar_t>(), c_prev.data<scalar_t>(),
        (scalar_t*)h_out.ptr, (scalar_t*)c_out.ptr, n, H);
  });
  HIP_CHECK_LAST();
}

void dropout_fwd_raw(const LaunchCtx& lc, const Arr& x, double p,
                     int64_t seed, const Arr& y, const Arr& mask) {
  long n = x.numel();
  float inv_keep = 1.f / (1.f - (float)p);
  DISPATCH_FLOAT_NATIVE(x.dtype, "dropout_fwd", [&] {
    dropout_fwd_kernel<scalar_t><<<ew_grid_n(n), 256, 0, lc.stream>>>(
        x.data<scalar_t>(), (scalar_t*)y.ptr, mask.data<unsigned char>(), n,
        (float)p, inv_keep, (unsigned long long)seed);
  });
  HIP_CHECK_LAST();
}

void dropout_bwd_raw(const LaunchCtx& lc, const Arr& dy, const Arr& mask,
                     double p, const Arr& dx) {
  long n = dy.numel();
  float inv_keep = 1.f / (1.f - (float)p);
  DISPATCH_FLOAT_NATIVE(dy.dtype, "dropout_bwd", [&] {
    dropout_bwd_kernel<scalar_t><<<ew_grid_n(n), 256, 0, lc.stream>>>(
        dy.data<scalar_t>(), mask.data<unsigned char>(), (scalar_t*)dx.ptr,
        n, inv_keep);
  });
  HIP_CHECK_LAST();
}

void embedding_fwd_raw(const LaunchCtx& lc, const Arr& weight,
                       const Arr& idx, const Arr& out) {
  MX_CHECK(idx.dtype == kInt64, "embedding indices must be int64");
  long nrows = idx.numel(), D = weight.size(1), V = weight.size(0);
  DISPATCH_FLOAT_NATIVE(weight.dtype, "embedding_fwd", [&] {
    embedding_fwd_kernel<scalar_t><<<ew_grid_n(nrows * 64), 256, 0,
                                     lc.stream>>>(
        idx.data<long>(), weight.data<scalar_t>(), (scalar_t*)out.ptr,
        nrows, D, V);
  });
  HIP_CHECK_LAST();
}

template <typename T>
__global__ void cast_from_f32_kernel_ew(const float* __restrict__ x,
                                        T* __restrict__ y, long n) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    y[i] = (T)x[i];
}

void embedding_bwd_raw(const LaunchCtx& lc, const Arr& dy, const Arr& idx,
                       const Arr& dweight) {
  MX_CHECK(idx.dtype == kInt64, "embedding indices must be int64");
  long nrows = idx.numel(), D = dy.size(-1), V = dweight.size(0);
  float* dw32;
  bool direct = dweight.dtype == kFloat32;
  if (direct) dw32 = dweight.data<float>();
  else dw32 = (float*)lc.workspace((size_t)V * D * 4);
  MX_HIP_CALL(hipMemsetAsync(dw32, 0, (size_t)V * D * 4, lc.stream));
  DISPATCH_FLOAT_NATIVE(dy.dtype, "embedding_bwd", [&] {
    embedding_bwd_kernel<scalar_t><<<ew_grid_n(nrows * 64), 256, 0,
                                     lc.stream>>>(
        idx.data<long>(), dy.data<scalar_t>(), dw32, nrows, D, V);
  });
  HIP_CHECK_LAST();
  if (!direct) {
    DISPATCH_HALF_NATIVE(dweight.dtype, "embedding_cast", [&] {
      cast_from_f32_kernel_ew<scalar_t><<<ew_grid_n(V * D), 256, 0,
                                          lc.stream>>>(
          dw32, (scalar_t*)dweight.ptr, V * D);
    });
    HIP_CHECK_LAST();
  }
}

// multi-tensor copy: ONE launch streams every (src -> dst) pair — the
// autograd tape batches the per-leaf gradient writes through this
// (~200 tiny hipMemcpy ops per BERT/ResNet step otherwise).  16-byte
// units with per-chunk tails; optional inline f32<->f16/bf16 cast.
struct CopyChunk {
  const char* src;
  char* dst;
  long start;    // global 16B-unit offset
  long units;    // ceil(dst_bytes / 16)
  long n;        // elements
  int mode;      // 0 raw bytes, 1 f32->f16, 2 f32->bf16, 3 f16->f32,
                 // 4 bf16->f32
};

// chunks ride the kernarg segment (value struct, max kMCPerLaunch) —
// no device-side table, no H2D staging, safe under hipGraph capture
constexpr int kMCPerLaunch = 64;
struct CopyArgs {
  CopyChunk c[kMCPerLaunch];
};

// IMPORTANT: chunks are walked with a UNIFORM index (every wave visits
// chunk k together) — dynamic indexing into a by-value kernarg aggregate
// scratch-spills the whole table per thread (measured 3.5x slower).
__global__ void multi_copy_kernel(const CopyArgs args, int nchunks,
                                  long total_units) {
  const long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long nthreads = (long)gridDim.x * blockDim.x;
  for (int k = 0; k < nchunks; ++k) {
    const CopyChunk c = args.c[k];
    for (long cu = tid; cu < c.units; cu += nthreads) {
      if (c.mode == 0) {
        long off = cu * 16;
        long nbytes = c.n;  // mode 0 stores bytes in n
        if (off + 16 <= nbytes) {
          *(float4*)(c.dst + off) = *(const float4*)(c.src + off);
        } else {
          for (long b = off; b < nbytes; ++b) c.dst[b] = c.src[b];
        }
      } else if (c.mode == 1 || c.mode == 2) {
        long e0 = cu * 8;  // 8 dst elems per 16B unit
        const float* s = (const float*)c.src;
        for (long e = e0; e < min(e0 + 8, c.n); ++e) {
          if (c.mode == 1) ((_Float16*)c.dst)[e] = (_Float16)s[e];
          else ((__bf16*)c.dst)[e] = (__bf16)s[e];
        }
      } else {
        long e0 = cu * 4;  // 4 f32 dst elems per 16B unit
        float* d = (float*)c.dst;
        for (long e = e0; e < min(e0 + 4, c.n); ++e) {
          if (c.mode == 3) d[e] = (float)((const _Float16*)c.src)[e];
          else d[e] = (float)((const __bf16*)c.src)[e];
        }
      }
    }
  }
}

int multi_copy_mode(int src_dtype, int dst_dtype) {
  if (src_dtype == dst_dtype) return 0;
  if (src_dtype == kFloat32 && dst_dtype == kFloat16) return 1;
  if (src_dtype == kFloat32 && dst_dtype == kBFloat16) return 2;
  if (src_dtype == kFloat16 && dst_dtype == kFloat32) return 3;
  if (src_dtype == kBFloat16 && dst_dtype == kFloat32) return 4;
  return -1;
}

void multi_copy_raw(const LaunchCtx& lc, const std::vector<Arr>& srcs,
                    const std::vector<Arr>& dsts) {
  int n = (int)srcs.size();
  for (int base = 0; base < n; base += kMCPerLaunch) {
    int cnt = std::min(n - base, kMCPerLaunch);
    CopyArgs args{};
    long total = 0;
    for (int i = 0; i < cnt; ++i) {
      const Arr& src = srcs[base + i];
      const Arr& dst = dsts[base + i];
      int mode = multi_copy_mode(src.dtype, dst.dtype);
      MX_CHECK(mode >= 0, "multi_copy: unsupported cast "
                              << src.dtype << "->" << dst.dtype);
      CopyChunk& c = args.c[i];
      c.src = (const char*)src.ptr;
      c.dst = (char*)dst.ptr;
      c.mode = mode;
      long nel = dst.numel();
      long dbytes = nel * dtype_size(dst.dtype);
      c.n = mode == 0 ? dbytes : nel;
      c.units = (dbytes + 15) / 16;
      c.start = total;
      total += c.units;
    }
    if (total == 0) continue;
    multi_copy_kernel<<<ew_grid_n(total), 256, 0, lc.stream>>>(args, cnt,
                                                               total);
    HIP_CHECK_LAST();
  }
}

// multi-tensor fused Adam (kernarg chunk table like multi_copy): ONE
// launch per ~56 parameters replaces the per-param adam_update storm
// (~160 launches/step on BERT-base).
struct AdamChunk {
  void* w;
  const void* g;
  float* m;
  float* v;
  float* master;
  long start, len;
};
constexpr int kMAPerLaunch = 56;
struct AdamArgs {
  AdamChunk c[kMAPerLaunch];
};

// uniform chunk walk (see multi_copy_kernel note on kernarg spilling)
template <typename T>
__global__ void multi_adam_kernel(const AdamArgs args, int n, long total,
                                  float lr_t, float b1, float b2, float eps,
                                  float wd, float rescale, float clip,
                                  bool adamw) {
  const long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long nthreads = (long)gridDim.x * blockDim.x;
  for (int k = 0; k < n; ++k) {
    const AdamChunk c = args.c[k];
    T* w = (T*)c.w;
    const T* g = (const T*)c.g;
    for (long j = tid; j < c.len; j += nthreads) {
      float wm = c.master ? c.master[j] : (float)w[j];
      float gv = (float)g[j] * rescale;
      if (clip > 0.f) gv = fminf(fmaxf(gv, -clip), clip);
      if (!adamw) gv += wd * wm;
      float mi = c.m[j] = b1 * c.m[j] + (1.f - b1) * gv;
      float vi = c.v[j] = b2 * c.v[j] + (1.f - b2) * gv * gv;
      wm -= lr_t * mi / (sqrtf(vi) + eps);
      if (adamw) wm -= lr_t * wd * wm;
      if (c.master) c.master[j] = wm;
      w[j] = (T)wm;
    }
  }
}

void multi_adam_update_raw(const LaunchCtx& lc, const std::vector<Arr>& ws,
                           const std::vector<Arr>& gs,
                           const std::vector<Arr>& ms,
                           const std::vector<Arr>& vs,
                           const std::vector<Arr>& masters, double lr_t,
                           double b1, double b2, double eps, double wd,
                           double rescale, double clip, bool adamw) {
  int n = (int)ws.size();
  for (int base = 0; base < n; base += kMAPerLaunch) {
    int cnt = std::min(n - base, kMAPerLaunch);
    AdamArgs args{};
    long total = 0;
    for (int i = 0; i < cnt; ++i) {
      AdamChunk& c = args.c[i];
      c.w = ws[base + i].ptr;
      c.g = gs[base + i].ptr;
      c.m = ms[base + i].data<float>();
      c.v = vs[base + i].data<float>();
      c.master = masters[base + i].defined() && masters[base + i].numel()
                     ? masters[base + i].data<float>()
                     : nullptr;
      c.start = total;
      c.len = ws[base + i].numel();
      total += c.len;
    }
    if (total == 0) continue;
    DISPATCH_FLOAT_NATIVE(ws[base].dtype, "multi_adam", [&] {
      multi_adam_kernel<scalar_t><<<ew_grid_n(total), 256, 0, lc.stream>>>(
          args, cnt, total, (float)lr_t, (float)b1, (float)b2, (float)eps,
          (float)wd, (float)rescale, (float)clip, adamw);
    });
    HIP_CHECK_LAST();
  }
}

void multi_sgd_update_raw(const LaunchCtx& lc, const std::vector<Arr>& ws,
                          const std::vector<Arr>& masters,
                          const std::vector<Arr>& grads,
                          const std::vector<Arr>& moms,
                          const std::vector<double>& lrs,
                          const std::vector<double>& wds, double mu,
                          double rescale, double clip) {
  int n = (int)ws.size();
  if (n == 0) return;
  std::vector<MTChunk> host(n);
  long total = 0;
  for (int i = 0; i < n; ++i) {
    host[i].w = ws[i].ptr;
    host[i].master =
        masters[i].defined() && masters[i].numel() ? masters[i].data<float>()
                                                   : nullptr;
    host[i].grad = grads[i].ptr;
    host[i].mom =
        moms[i].defined() && moms[i].numel() ? moms[i].data<float>() : nullptr;
    host[i].start = total;
    host[i].len = ws[i].numel();
    host[i].lr = (float)lrs[i];
    host[i].wd = (float)wds[i];
    total += host[i].len;
  }
  void* table = lc.workspace(n * sizeof(MTChunk));
  // pageable H2D: the runtime stages the copy, host vector may die after
  MX_HIP_CALL(hipMemcpyAsync(table, host.data(), n * sizeof(MTChunk),
                             hipMemcpyHostToDevice, lc.stream));
  DISPATCH_FLOAT_NATIVE(ws[0].dtype, "multi_sgd", [&] {
    multi_sgd_kernel<scalar_t><<<ew_grid_n(total), 256, 0, lc.stream>>>(
        (const MTChunk*)table, n, total, (float)mu, (float)rescale,
        (float)clip);
  });
  HIP_CHECK_LAST();
}

}  // namespace mxcore
