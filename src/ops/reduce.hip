// Native reduce / broadcast_to / transpose family.
//
// Reference parity: src/operator/tensor/broadcast_reduce-inl.cuh +
// reduce_rtc.cc (NVRTC there; AOT-templated CDNA4 here), matrix_op
// transpose.  The generic axis-reduce runs one thread per output element
// with a serial strided loop (hot reductions — BN/LN stats, softmax rows,
// bias grads — have dedicated fused kernels elsewhere).
#include "ew_common.h"

namespace mxcore {
namespace {

#define HD __host__ __device__ __forceinline__
struct RSum { static HD float f(float a, float b) { return a + b; } };
struct RMax { static HD float f(float a, float b) { return a > b ? a : b; } };
struct RMin { static HD float f(float a, float b) { return a < b ? a : b; } };

template <typename T>
HD float ldf2(const T& v) { return (float)v; }

struct RedPlan {
  int out_ndim = 0, red_ndim = 0;
  int64_t out_shape[8] = {}, out_stride_in[8] = {};
  int64_t red_shape[8] = {}, red_stride[8] = {};
  int64_t red_size = 1;
};

template <typename T, typename OP, bool MEAN>
__global__ void reduce_axes_kernel(const T* __restrict__ x, T* __restrict__ y,
                                   long nout, RedPlan p, float init) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nout;
       i += (long)gridDim.x * blockDim.x) {
    long rem = i, base = 0;
    for (int d = p.out_ndim - 1; d >= 0; --d) {
      long idx = rem % p.out_shape[d];
      rem /= p.out_shape[d];
      base += idx * p.out_stride_in[d];
    }
    float acc = init;
    long ridx[8] = {};
    for (long r = 0; r < p.red_size; ++r) {
      long off = base;
      for (int d = 0; d < p.red_ndim; ++d) off += ridx[d] * p.red_stride[d];
      acc = OP::f(acc, ldf2(x[off]));
      for (int d = p.red_ndim - 1; d >= 0; --d) {
        if (++ridx[d] < p.red_shape[d]) break;
        ridx[d] = 0;
      }
    }
    if (MEAN) acc /= (float)p.red_size;
    y[i] = (T)acc;
  }
}

// fast path: reduce over the contiguous inner tail — one wave per row
template <typename T, typename OP, bool MEAN>
__global__ void reduce_inner_kernel(const T* __restrict__ x, T* __restrict__ y,
                                    long rows, long cols, float init) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int waves = blockDim.x >> 6;
  for (long r = (long)blockIdx.x * waves + wid; r < rows;
       r += (long)gridDim.x * waves) {
    const T* row = x + r * cols;
    float acc = init;
    for (long c = lane; c < cols; c += 64) acc = OP::f(acc, ldf2(row[c]));
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      acc = OP::f(acc, __shfl_xor(acc, off, 64));
    if (lane == 0) y[r] = (T)(MEAN ? acc / (float)cols : acc);
  }
}

template <typename T, typename OP>
__global__ void reduce_all_stage1(const T* __restrict__ x, float* partial,
                                  long n, float init) {
  float acc = init;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    acc = OP::f(acc, ldf2(x[i]));
  __shared__ float smem[16];
  struct W {
    __device__ float operator()(float a, float b) const { return OP::f(a, b); }
  };
  float r = block_reduce(acc, smem, W(), init);
  if (threadIdx.x == 0) partial[blockIdx.x] = r;
}

template <typename T, typename OP, bool MEAN>
__global__ void reduce_all_stage2(const float* partial, T* y, int nb, long n,
                                  float init) {
  float acc = init;
  for (int i = threadIdx.x; i < nb; i += blockDim.x)
    acc = OP::f(acc, partial[i]);
  __shared__ float smem[16];
  struct W {
    __device__ float operator()(float a, float b) const { return OP::f(a, b); }
  };
  float r = block_reduce(acc, smem, W(), init);
  if (threadIdx.x == 0) y[0] = (T)(MEAN ? r / (double)n : r);
}

// scatter for the strided-view backward: dx[off + idx·strides] = dy[i]
// (slice positions are unique — plain stores into a pre-zeroed buffer)
template <typename T>
__global__ void scatter_strided_kernel(const T* __restrict__ dy,
                                       T* __restrict__ dx, long n,
                                       Strides8 st, long off0) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    long rem = i, off = off0;
#pragma unroll 4
    for (int d = st.ndim - 1; d >= 0; --d) {
      long idx = rem % st.shape[d];
      rem /= st.shape[d];
      off += idx * st.s0[d];
    }
    dx[off] = dy[i];
  }
}

template <typename T, bool MAX>
__global__ void arg_reduce_kernel(const T* __restrict__ x,
                                  float* __restrict__ y, long nout,
                                  long len, long inner) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nout;
       i += (long)gridDim.x * blockDim.x) {
    long outer = i / inner, off = i % inner;
    const T* base = x + outer * len * inner + off;
    float best = (float)base[0];
    long bi = 0;
    for (long k = 1; k < len; ++k) {
      float v = (float)base[k * inner];
      if (MAX ? (v > best) : (v < best)) {
        best = v;
        bi = k;
      }
    }
    y[i] = (float)bi;
  }
}

template <typename T>
__global__ void gather_strided_kernel(const T* __restrict__ x,
                                      T* __restrict__ y, long n, Strides8 st) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    long rem = i, off = 0;
#pragma unroll 4
    for (int d = st.ndim - 1; d >= 0; --d) {
      long idx = rem % st.shape[d];
      rem /= st.shape[d];
      off += idx * st.s0[d];
    }
    y[i] = x[off];
  }
}

// ---------------------------------------------------------------------------
// host-side plan building
// ---------------------------------------------------------------------------
std::vector<int> parse_axes(const NodeAttrs& a, int ndim) {
  std::vector<int64_t> ax = a.GetTuple("axis", {});
  std::vector<int> out;
  if (!a.has("axis") || ax.empty()) {
    for (int i = 0; i < ndim; ++i) out.push_back(i);
  } else {
    for (int64_t v : ax) out.push_back(v < 0 ? (int)(v + ndim) : (int)v);
  }
  return out;
}

TShape reduced_shape(const TShape& in, const std::vector<int>& axes,
                     bool keepdims) {
  std::vector<bool> red(in.size(), false);
  for (int a : axes) red[a] = true;
  TShape out;
  for (size_t i = 0; i < in.size(); ++i) {
    if (red[i]) {
      if (keepdims) out.push_back(1);
    } else {
      out.push_back(in[i]);
    }
  }
  if (out.empty()) out.push_back(1);
  return out;
}

RedPlan build_plan(const TShape& in, const std::vector<int>& axes) {
  std::vector<bool> red(in.size(), false);
  for (int a : axes) red[a] = true;
  // row-major input strides
  std::vector<int64_t> istr(in.size());
  int64_t s = 1;
  for (int i = (int)in.size() - 1; i >= 0; --i) {
    istr[i] = s;
    s *= in[i];
  }
  RedPlan p;
  for (size_t i = 0; i < in.size(); ++i) {
    if (red[i]) {
      p.red_shape[p.red_ndim] = in[i];
      p.red_stride[p.red_ndim] = istr[i];
      p.red_size *= in[i];
      ++p.red_ndim;
    } else {
      p.out_shape[p.out_ndim] = in[i];
      p.out_stride_in[p.out_ndim] = istr[i];
      ++p.out_ndim;
    }
  }
  if (p.out_ndim == 0) {
    p.out_shape[0] = 1;
    p.out_stride_in[0] = 0;
    p.out_ndim = 1;
  }
  return p;
}

// is the reduction exactly the contiguous inner tail?
bool inner_contig(const TShape& in, const std::vector<int>& axes, long* rows,
                  long* cols) {
  int nd = (int)in.size(), na = (int)axes.size();
  std::vector<bool> red(nd, false);
  for (int a : axes) red[a] = true;
  for (int i = 0; i < na; ++i)
    if (!red[nd - 1 - i]) return false;
  long r = 1, c = 1;
  for (int i = 0; i < nd - na; ++i) r *= in[i];
  for (int i = nd - na; i < nd; ++i) c *= in[i];
  *rows = r;
  *cols = c;
  return true;
}

template <typename OP, bool MEAN>
void ReduceGPU(const NodeAttrs& a, const OpCtx& o,
               const std::vector<TBlob>& in, const std::vector<TBlob>& out,
               float init) {
  auto axes = parse_axes(a, in[0].ndim());
  long nout = out[0].size();
  MXC_DISPATCH_FLOAT(in[0].dtype, "reduce", {
    auto* x = (const scalar_t*)in[0].dptr;
    auto* y = (scalar_t*)out[0].dptr;
    if ((int)axes.size() == in[0].ndim()) {
      // full reduce: two stages through a float workspace
      long n = in[0].size();
      int nb = grid_for(n);
      float* partial = (float*)o.workspace((size_t)nb * 4);
      reduce_all_stage1<scalar_t, OP><<<nb, kBlock, 0, o.rc.stream>>>(
          x, partial, n, init);
      reduce_all_stage2<scalar_t, OP, MEAN><<<1, kBlock, 0, o.rc.stream>>>(
          partial, y, nb, n, init);
    } else {
      long rows, cols;
      if (inner_contig(in[0].shape, axes, &rows, &cols)) {
        // one wave per row, 4 rows per 256-thread block
        reduce_inner_kernel<scalar_t, OP, MEAN>
            <<<grid_for(rows * 64), kBlock, 0, o.rc.stream>>>(x, y, rows,
                                                              cols, init);
      } else {
        RedPlan p = build_plan(in[0].shape, axes);
        reduce_axes_kernel<scalar_t, OP, MEAN>
            <<<grid_for(nout), kBlock, 0, o.rc.stream>>>(x, y, nout, p, init);
      }
    }
  });
  HIP_CHECK_LAST();
}

template <typename OP, bool MEAN>
void ReduceCPU(const NodeAttrs& a, const OpCtx&, const std::vector<TBlob>& in,
               const std::vector<TBlob>& out, float init) {
  auto axes = parse_axes(a, in[0].ndim());
  RedPlan p = build_plan(in[0].shape, axes);
  long nout = out[0].size();
  MXC_DISPATCH_FLOAT(in[0].dtype, "reduce", {
    auto* x = (const scalar_t*)in[0].dptr;
    auto* y = (scalar_t*)out[0].dptr;
    for (long i = 0; i < nout; ++i) {
      long rem = i, base = 0;
      for (int d = p.out_ndim - 1; d >= 0; --d) {
        long idx = rem % p.out_shape[d];
        rem /= p.out_shape[d];
        base += idx * p.out_stride_in[d];
      }
      double acc = init;
      long ridx[8] = {};
      for (long r = 0; r < p.red_size; ++r) {
        long off = base;
        for (int d = 0; d < p.red_ndim; ++d) off += ridx[d] * p.red_stride[d];
        acc = OP::f((float)acc, (float)x[off]);
        for (int d = p.red_ndim - 1; d >= 0; --d) {
          if (++ridx[d] < p.red_shape[d]) break;
          ridx[d] = 0;
        }
      }
      if (MEAN) acc /= (double)p.red_size;
      y[i] = (scalar_t)acc;
    }
  });
}

FInferShape InferReduce() {
  return [](const NodeAttrs& a, const std::vector<TShape>& is,
            const std::vector<int>& it, std::vector<TShape>* os,
            std::vector<int>* ot) {
    auto axes = parse_axes(a, (int)is[0].size());
    os->assign(1, reduced_shape(is[0], axes, a.GetBool("keepdims", false)));
    ot->assign(1, it[0]);
  };
}

NDArray RunOp2(const char* name, const NodeAttrs& attrs,
               const std::vector<NDArray>& ins) {
  OpEntry* e = OpRegistry::Get()->Find(name);
  MX_CHECK(e, "op not registered: " << name);
  return Imperative::Run(e, attrs, ins)[0];
}

// backward of sum/mean: broadcast dy (with kept dims) back to input shape
std::vector<NDArray> SumBackward(const TapeNode& n,
                                 const std::vector<NDArray>& og, bool mean) {
  const TShape& xshape = n.inputs[0].shape();
  auto axes = parse_axes(n.attrs, (int)xshape.size());
  // target shape with 1s at reduced axes
  TShape keep = reduced_shape(xshape, axes, true);
  NDArray dy = og[0];
  if (dy.shape() != keep && (int64_t)dy.size() == shape_size(keep))
    dy = dy.Reshape(keep);
  NodeAttrs b;
  std::string s = "(";
  for (auto d : xshape) s += std::to_string(d) + ",";
  s += ")";
  b.d["shape"] = s;
  NDArray dx = RunOp2("broadcast_to", b, {dy});
  if (mean) {
    int64_t rs = 1;
    for (int a : axes) rs *= xshape[a];
    NodeAttrs m;
    m.d["alpha"] = std::to_string(1.0 / (double)rs);
    dx = RunOp2("_mul_scalar", m, {dx});
  }
  return {dx};
}

struct Reg2 {
  OpEntry* e;
  explicit Reg2(const char* name) {
    e = &OpRegistry::Get()->Register(name);
    e->n_out = 1;
  }
  Reg2& in(int n) { e->n_in = n; return *this; }
  Reg2& infer(FInferShape f) { e->infer = std::move(f); return *this; }
  Reg2& gpu(FCompute f) { e->fcompute_gpu = std::move(f); return *this; }
  Reg2& cpu(FCompute f) { e->fcompute_cpu = std::move(f); return *this; }
  Reg2& bwd(FBackward f) { e->fbackward = std::move(f); return *this; }
};

bool _registered_reduce = [] {
  using V = const std::vector<TBlob>&;
  Reg2("sum").in(1).infer(InferReduce())
      .gpu([](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        ReduceGPU<RSum, false>(a, o, in, out, 0.f);
      })
      .cpu([](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        ReduceCPU<RSum, false>(a, o, in, out, 0.f);
      })
      .bwd([](const TapeNode& n, const std::vector<NDArray>& og) {
        return SumBackward(n, og, false);
      });
  Reg2("mean").in(1).infer(InferReduce())
      .gpu([](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        ReduceGPU<RSum, true>(a, o, in, out, 0.f);
      })
      .cpu([](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        ReduceCPU<RSum, true>(a, o, in, out, 0.f);
      })
      .bwd([](const TapeNode& n, const std::vector<NDArray>& og) {
        return SumBackward(n, og, true);
      });
  Reg2("max").in(1).infer(InferReduce())
      .gpu([](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        ReduceGPU<RMax, false>(a, o, in, out, -3.4e38f);
      })
      .cpu([](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        ReduceCPU<RMax, false>(a, o, in, out, -3.4e38f);
      });
  Reg2("min").in(1).infer(InferReduce())
      .gpu([](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        ReduceGPU<RMin, false>(a, o, in, out, 3.4e38f);
      })
      .cpu([](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        ReduceCPU<RMin, false>(a, o, in, out, 3.4e38f);
      });

  // _reduce_to: sum to a (broadcast-compatible) target shape
  auto infer_target = [](const NodeAttrs& a, const std::vector<TShape>& is,
                         const std::vector<int>& it, std::vector<TShape>* os,
                         std::vector<int>* ot) {
    auto s = a.GetTuple("shape", {});
    os->assign(1, TShape(s.begin(), s.end()));
    ot->assign(1, it[0]);
  };
  auto reduce_to_axes = [](const TShape& from, const TShape& to) {
    // right-aligned: leading extra dims + dims where to==1!=from
    NodeAttrs a;
    std::string s = "(";
    int nf = (int)from.size(), nt = (int)to.size();
    for (int i = 0; i < nf; ++i) {
      int it = i - (nf - nt);
      if (it < 0 || (to[it] == 1 && from[i] != 1))
        s += std::to_string(i) + ",";
    }
    s += ")";
    a.d["axis"] = s;
    return a;
  };
  Reg2("_reduce_to").in(1).infer(infer_target)
      .gpu([reduce_to_axes](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        NodeAttrs ax = reduce_to_axes(
            in[0].shape, [&] {
              auto s = a.GetTuple("shape", {});
              return TShape(s.begin(), s.end());
            }());
        std::vector<TBlob> tmp_out = {out[0]};
        ReduceGPU<RSum, false>(ax, o, in, tmp_out, 0.f);
      })
      .cpu([reduce_to_axes](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        NodeAttrs ax = reduce_to_axes(
            in[0].shape, [&] {
              auto s = a.GetTuple("shape", {});
              return TShape(s.begin(), s.end());
            }());
        std::vector<TBlob> tmp_out = {out[0]};
        ReduceCPU<RSum, false>(ax, o, in, tmp_out, 0.f);
      })
      .bwd([](const TapeNode& n, const std::vector<NDArray>& og)
               -> std::vector<NDArray> {
        NodeAttrs b;
        std::string s = "(";
        for (auto d : n.inputs[0].shape()) s += std::to_string(d) + ",";
        s += ")";
        b.d["shape"] = s;
        return {RunOp2("broadcast_to", b, {og[0]})};
      });

  // broadcast_to
  Reg2("broadcast_to").in(1).infer(infer_target)
      .gpu([](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        long n = out[0].size();
        Strides8 st = make_strides(out[0].shape, in[0].shape, in[0].shape);
        MXC_DISPATCH_ALL(out[0].dtype, "broadcast_to", {
          gather_strided_kernel<scalar_t><<<grid_for(n), kBlock, 0,
                                            o.rc.stream>>>(
              (const scalar_t*)in[0].dptr, (scalar_t*)out[0].dptr, n, st);
        });
        HIP_CHECK_LAST();
      })
      .cpu([](const NodeAttrs& a, const OpCtx&, V in, V out) {
        long n = out[0].size();
        Strides8 st = make_strides(out[0].shape, in[0].shape, in[0].shape);
        MXC_DISPATCH_ALL(out[0].dtype, "broadcast_to", {
          auto* x = (const scalar_t*)in[0].dptr;
          auto* y = (scalar_t*)out[0].dptr;
          for (long i = 0; i < n; ++i) {
            long rem = i, off = 0;
            for (int d = st.ndim - 1; d >= 0; --d) {
              long idx = rem % st.shape[d];
              rem /= st.shape[d];
              off += idx * st.s0[d];
            }
            y[i] = x[off];
          }
        });
      })
      .bwd([](const TapeNode& n, const std::vector<NDArray>& og)
               -> std::vector<NDArray> {
        NodeAttrs b;
        std::string s = "(";
        for (auto d : n.inputs[0].shape()) s += std::to_string(d) + ",";
        s += ")";
        b.d["shape"] = s;
        return {RunOp2("_reduce_to", b, {og[0]})};
      });

  // _strided_copy: out[i] = in[offset + multi_idx(i)·strides] — the
  // generic strided view materializer (basic __getitem__ slicing,
  // positional-embedding slices, pooler token picks, attention head
  // split/merge).  attrs: shape (gather plan dims), strides (input
  // element strides per plan dim), offset, and optional oshape — a
  // same-numel reinterpretation of the contiguous result, folding a
  // trailing reshape into the same kernel (e.g. gather [B,H,S,D] →
  // declare [B*H,S,D]).
  auto strided_pack = [](const NodeAttrs& a) {
    auto shv = a.GetTuple("shape", {});
    auto strv = a.GetTuple("strides", {});
    MX_CHECK(shv.size() == strv.size() && shv.size() <= 8,
             "_strided_copy: shape/strides rank mismatch");
    Strides8 st;
    st.ndim = (int)shv.size();
    for (int i = 0; i < st.ndim; ++i) {
      st.shape[i] = shv[i];
      st.s0[i] = strv[i];
    }
    return st;
  };
  Reg2("_strided_copy").in(1)
      .infer([](const NodeAttrs& a, const std::vector<TShape>&,
                const std::vector<int>& it, std::vector<TShape>* os,
                std::vector<int>* ot) {
        auto s = a.has("oshape") ? a.GetTuple("oshape", {})
                                 : a.GetTuple("shape", {});
        os->assign(1, TShape(s.begin(), s.end()));
        ot->assign(1, it[0]);
      })
      .gpu([strided_pack](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        long n = out[0].size();
        if (n == 0) return;
        Strides8 st = strided_pack(a);
        long off = a.GetInt("offset", 0);
        MXC_DISPATCH_ALL(out[0].dtype, "_strided_copy", {
          gather_strided_kernel<scalar_t><<<grid_for(n), kBlock, 0,
                                            o.rc.stream>>>(
              (const scalar_t*)in[0].dptr + off, (scalar_t*)out[0].dptr, n,
              st);
        });
        HIP_CHECK_LAST();
      })
      .cpu([strided_pack](const NodeAttrs& a, const OpCtx&, V in, V out) {
        long n = out[0].size();
        Strides8 st = strided_pack(a);
        long off0 = a.GetInt("offset", 0);
        MXC_DISPATCH_ALL(out[0].dtype, "_strided_copy", {
          auto* x = (const scalar_t*)in[0].dptr;
          auto* y = (scalar_t*)out[0].dptr;
          for (long i = 0; i < n; ++i) {
            long rem = i, off = off0;
            for (int d = st.ndim - 1; d >= 0; --d) {
              long idx = rem % st.shape[d];
              rem /= st.shape[d];
              off += idx * st.s0[d];
            }
            y[i] = x[off];
          }
        });
      })
      .bwd([](const TapeNode& n, const std::vector<NDArray>& og)
               -> std::vector<NDArray> {
        NodeAttrs a = n.attrs;
        std::string s = "(";
        for (auto d : n.inputs[0].shape()) s += std::to_string(d) + ",";
        s += ")";
        a.d["xshape"] = s;
        return {RunOp2("_strided_copy_bwd", a, {og[0]})};
      });

  Reg2("_strided_copy_bwd").in(1)
      .infer([](const NodeAttrs& a, const std::vector<TShape>&,
                const std::vector<int>& it, std::vector<TShape>* os,
                std::vector<int>* ot) {
        auto s = a.GetTuple("xshape", {});
        os->assign(1, TShape(s.begin(), s.end()));
        ot->assign(1, it[0]);
      })
      .gpu([strided_pack](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        long n = in[0].size();
        Strides8 st = strided_pack(a);
        long off = a.GetInt("offset", 0);
        MX_HIP_CALL(hipMemsetAsync(
            out[0].dptr, 0, (size_t)out[0].size() * dtype_size(out[0].dtype),
            o.rc.stream));
        MXC_DISPATCH_ALL(in[0].dtype, "_strided_copy_bwd", {
          scatter_strided_kernel<scalar_t><<<grid_for(n), kBlock, 0,
                                             o.rc.stream>>>(
              (const scalar_t*)in[0].dptr, (scalar_t*)out[0].dptr, n, st,
              off);
        });
        HIP_CHECK_LAST();
      })
      .cpu([strided_pack](const NodeAttrs& a, const OpCtx&, V in, V out) {
        long n = in[0].size();
        Strides8 st = strided_pack(a);
        long off0 = a.GetInt("offset", 0);
        memset(out[0].dptr, 0,
               (size_t)out[0].size() * dtype_size(out[0].dtype));
        MXC_DISPATCH_ALL(in[0].dtype, "_strided_copy_bwd", {
          auto* dy = (const scalar_t*)in[0].dptr;
          auto* dx = (scalar_t*)out[0].dptr;
          for (long i = 0; i < n; ++i) {
            long rem = i, off = off0;
            for (int d = st.ndim - 1; d >= 0; --d) {
              long idx = rem % st.shape[d];
              rem /= st.shape[d];
              off += idx * st.s0[d];
            }
            dx[off] = dy[i];
          }
        });
      });

  // argmax/argmin over ONE axis (reference ordering ops use topk; the
  // index-only reductions here run one thread per output element with a
  // serial strided scan)
  auto arg_reduce = [](const char* name, bool is_max) {
    Reg2(name).in(1)
        .infer([](const NodeAttrs& a, const std::vector<TShape>& is,
                  const std::vector<int>&, std::vector<TShape>* os,
                  std::vector<int>* ot) {
          int nd = (int)is[0].size();
          int ax = (int)a.GetInt("axis", -1);
          if (ax < 0) ax += nd;
          TShape out;
          for (int d = 0; d < nd; ++d)
            if (d != ax) out.push_back(is[0][d]);
          if (out.empty()) out.push_back(1);
          os->assign(1, out);
          ot->assign(1, kFloat32);  // mxnet argmax returns float indices
        })
        .gpu([is_max](const NodeAttrs& a, const OpCtx& o, V in, V out) {
          int nd = in[0].ndim();
          int ax = (int)a.GetInt("axis", -1);
          if (ax < 0) ax += nd;
          long n = in[0].size(), len = in[0].shape[ax];
          if (n == 0 || len == 0) {
            MX_HIP_CALL(hipMemsetAsync(out[0].dptr, 0,
                                       (size_t)out[0].size() * 4,
                                       o.rc.stream));
            return;
          }
          long inner = 1;
          for (int d = ax + 1; d < nd; ++d) inner *= in[0].shape[d];
          long nout = n / len;
          MXC_DISPATCH_FLOAT(in[0].dtype, "arg_reduce", {
            if (is_max)
              arg_reduce_kernel<scalar_t, true>
                  <<<grid_for(nout), kBlock, 0, o.rc.stream>>>(
                      (const scalar_t*)in[0].dptr, (float*)out[0].dptr,
                      nout, len, inner);
            else
              arg_reduce_kernel<scalar_t, false>
                  <<<grid_for(nout), kBlock, 0, o.rc.stream>>>(
                      (const scalar_t*)in[0].dptr, (float*)out[0].dptr,
                      nout, len, inner);
          });
          HIP_CHECK_LAST();
        })
        .cpu([is_max](const NodeAttrs& a, const OpCtx&, V in, V out) {
          MX_CHECK(in[0].dtype == kFloat32,
                   "arg_reduce: CPU path is fp32");
          int nd = in[0].ndim();
          int ax = (int)a.GetInt("axis", -1);
          if (ax < 0) ax += nd;
          long n = in[0].size(), len = in[0].shape[ax];
          if (n == 0 || len == 0) {
            memset(out[0].dptr, 0, (size_t)out[0].size() * 4);
            return;
          }
          long inner = 1;
          for (int d = ax + 1; d < nd; ++d) inner *= in[0].shape[d];
          long nout = n / len;
          auto* x = (const float*)in[0].dptr;
          auto* y = (float*)out[0].dptr;
          for (long i = 0; i < nout; ++i) {
            long outer = i / inner, off = i % inner;
            const float* base = x + outer * len * inner + off;
            float best = base[0];
            long bi = 0;
            for (long k = 1; k < len; ++k) {
              float v = base[k * inner];
              if (is_max ? (v > best) : (v < best)) { best = v; bi = k; }
            }
            y[i] = (float)bi;
          }
        });
  };
  arg_reduce("argmax", true);
  arg_reduce("argmin", false);

  // concat along an axis (reference concat.cc): forward scatters each
  // input into its strided slab of the output (scatter_strided, one
  // launch per input, full cover so no memset); backward gathers the
  // matching slices of the output gradient (_strided_copy).
  Reg2("concat").in(-1)
      .infer([](const NodeAttrs& a, const std::vector<TShape>& is,
                const std::vector<int>& it, std::vector<TShape>* os,
                std::vector<int>* ot) {
        int nd = (int)is[0].size();
        int axis = (int)a.GetInt("dim", 0);
        if (axis < 0) axis += nd;
        TShape out = is[0];
        for (size_t i = 1; i < is.size(); ++i) {
          MX_CHECK((int)is[i].size() == nd, "concat: rank mismatch");
          out[axis] += is[i][axis];
        }
        os->assign(1, out);
        ot->assign(1, it[0]);
      })
      .gpu([](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        int nd = out[0].ndim();
        int axis = (int)a.GetInt("dim", 0);
        if (axis < 0) axis += nd;
        std::vector<int64_t> ostr(nd, 1);
        for (int d = nd - 2; d >= 0; --d)
          ostr[d] = ostr[d + 1] * out[0].shape[d + 1];
        long off_axis = 0;
        for (auto& x : in) {
          Strides8 st;
          st.ndim = nd;
          for (int d = 0; d < nd; ++d) {
            st.shape[d] = x.shape[d];
            st.s0[d] = ostr[d];
          }
          long n = x.size();
          long off = off_axis * ostr[axis];
          MXC_DISPATCH_ALL(out[0].dtype, "concat", {
            scatter_strided_kernel<scalar_t><<<grid_for(n), kBlock, 0,
                                               o.rc.stream>>>(
                (const scalar_t*)x.dptr, (scalar_t*)out[0].dptr, n, st,
                off);
          });
          off_axis += x.shape[axis];
        }
        HIP_CHECK_LAST();
      })
      .cpu([](const NodeAttrs& a, const OpCtx&, V in, V out) {
        int nd = out[0].ndim();
        int axis = (int)a.GetInt("dim", 0);
        if (axis < 0) axis += nd;
        std::vector<int64_t> ostr(nd, 1);
        for (int d = nd - 2; d >= 0; --d)
          ostr[d] = ostr[d + 1] * out[0].shape[d + 1];
        long off_axis = 0;
        for (auto& x : in) {
          long n = x.size();
          long off0 = off_axis * ostr[axis];
          MXC_DISPATCH_ALL(out[0].dtype, "concat", {
            auto* src = (const scalar_t*)x.dptr;
            auto* dst = (scalar_t*)out[0].dptr;
            for (long i = 0; i < n; ++i) {
              long rem = i, off = off0;
              for (int d = nd - 1; d >= 0; --d) {
                long idx = rem % x.shape[d];
                rem /= x.shape[d];
                off += idx * ostr[d];
              }
              dst[off] = src[i];
            }
          });
          off_axis += x.shape[axis];
        }
      })
      .bwd([](const TapeNode& n, const std::vector<NDArray>& og)
               -> std::vector<NDArray> {
        int nd = (int)og[0].shape().size();
        int axis = (int)n.attrs.GetInt("dim", 0);
        if (axis < 0) axis += nd;
        std::vector<int64_t> ostr(nd, 1);
        for (int d = nd - 2; d >= 0; --d)
          ostr[d] = ostr[d + 1] * og[0].shape()[d + 1];
        std::vector<NDArray> r(n.inputs.size());
        long off_axis = 0;
        for (size_t i = 0; i < n.inputs.size(); ++i) {
          const TShape& xs = n.inputs[i].shape();
          NodeAttrs a;
          std::string sh = "(", strd = "(";
          for (int d = 0; d < nd; ++d) {
            sh += std::to_string(xs[d]) + ",";
            strd += std::to_string(ostr[d]) + ",";
          }
          a.d["shape"] = sh + ")";
          a.d["strides"] = strd + ")";
          a.d["offset"] = std::to_string(off_axis * ostr[axis]);
          r[i] = RunOp2("_strided_copy", a, {og[0]});
          off_axis += xs[axis];
        }
        return r;
      });

  // transpose (general permute)
  auto infer_perm = [](const NodeAttrs& a, const std::vector<TShape>& is,
                       const std::vector<int>& it, std::vector<TShape>* os,
                       std::vector<int>* ot) {
    auto ax = a.GetTuple("axes", {});
    int nd = (int)is[0].size();
    TShape out(nd);
    if (ax.empty()) {
      for (int i = 0; i < nd; ++i) out[i] = is[0][nd - 1 - i];
    } else {
      for (int i = 0; i < nd; ++i) out[i] = is[0][ax[i]];
    }
    os->assign(1, out);
    ot->assign(1, it[0]);
  };
  auto perm_strides = [](const NodeAttrs& a, const TShape& in_shape,
                         const TShape& out_shape) {
    auto ax = a.GetTuple("axes", {});
    int nd = (int)in_shape.size();
    std::vector<int64_t> perm(nd);
    if (ax.empty())
      for (int i = 0; i < nd; ++i) perm[i] = nd - 1 - i;
    else
      for (int i = 0; i < nd; ++i) perm[i] = ax[i];
    std::vector<int64_t> istr(nd);
    int64_t s = 1;
    for (int i = nd - 1; i >= 0; --i) {
      istr[i] = s;
      s *= in_shape[i];
    }
    Strides8 st;
    st.ndim = nd;
    for (int i = 0; i < nd; ++i) {
      st.shape[i] = out_shape[i];
      st.s0[i] = istr[perm[i]];
    }
    return st;
  };
  Reg2("transpose").in(1).infer(infer_perm)
      .gpu([perm_strides](const NodeAttrs& a, const OpCtx& o, V in, V out) {
        long n = out[0].size();
        Strides8 st = perm_strides(a, in[0].shape, out[0].shape);
        MXC_DISPATCH_ALL(out[0].dtype, "transpose", {
          gather_strided_kernel<scalar_t><<<grid_for(n), kBlock, 0,
                                            o.rc.stream>>>(
              (const scalar_t*)in[0].dptr, (scalar_t*)out[0].dptr, n, st);
        });
        HIP_CHECK_LAST();
      })
      .cpu([perm_strides](const NodeAttrs& a, const OpCtx&, V in, V out) {
        long n = out[0].size();
        Strides8 st = perm_strides(a, in[0].shape, out[0].shape);
        MXC_DISPATCH_ALL(out[0].dtype, "transpose", {
          auto* x = (const scalar_t*)in[0].dptr;
          auto* y = (scalar_t*)out[0].dptr;
          for (long i = 0; i < n; ++i) {
            long rem = i, off = 0;
            for (int d = st.ndim - 1; d >= 0; --d) {
              long idx = rem % st.shape[d];
              rem /= st.shape[d];
              off += idx * st.s0[d];
            }
            y[i] = x[off];
          }
        });
      })
      .bwd([](const TapeNode& n, const std::vector<NDArray>& og)
               -> std::vector<NDArray> {
        auto ax = n.attrs.GetTuple("axes", {});
        NodeAttrs inv;
        if (!ax.empty()) {
          std::vector<int64_t> iax(ax.size());
          for (size_t i = 0; i < ax.size(); ++i) iax[ax[i]] = i;
          std::string s = "(";
          for (auto v : iax) s += std::to_string(v) + ",";
          s += ")";
          inv.d["axes"] = s;
        }
        return {RunOp2("transpose", inv, {og[0]})};
      });

  return true;
}();

}  // namespace
}  // namespace mxcore
