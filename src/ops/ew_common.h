// Shared host/device helpers for the native op implementations.
#pragma once

#include <hip/hip_runtime.h>

#include "../core/op.h"
#include "common.h"  // kernel helpers (wave reductions, vec types)

namespace mxcore {

constexpr int kBlock = 256;

inline int grid_for(int64_t n, int per_thread = 1) {
  int64_t g = (n + (int64_t)kBlock * per_thread - 1) / ((int64_t)kBlock * per_thread);
  // memory-bound cap: ~8 blocks/CU on 256 CUs, grid-stride the rest
  const int64_t cap = env_int("MXNET_EW_BLOCKS", 2048);
  if (g < 1) g = 1;
  return (int)(g < cap ? g : cap);
}

// dispatch over float storage dtypes (host side; maps dtype flag -> ctype)
#define MXC_DISPATCH_FLOAT(DTYPE, NAME, ...)                         \
  switch (DTYPE) {                                                   \
    case ::mxcore::kFloat32: {                                       \
      using scalar_t = float;                                        \
      { __VA_ARGS__ } break;                                         \
    }                                                                \
    case ::mxcore::kFloat16: {                                       \
      using scalar_t = _Float16;                                     \
      { __VA_ARGS__ } break;                                         \
    }                                                                \
    case ::mxcore::kBFloat16: {                                      \
      using scalar_t = __bf16;                                       \
      { __VA_ARGS__ } break;                                         \
    }                                                                \
    default:                                                         \
      MX_CHECK(false, NAME << ": unsupported dtype "                 \
                           << ::mxcore::dtype_name(DTYPE));          \
  }

#define MXC_DISPATCH_ALL(DTYPE, NAME, ...)                           \
  switch (DTYPE) {                                                   \
    case ::mxcore::kFloat32: {                                       \
      using scalar_t = float;                                        \
      { __VA_ARGS__ } break;                                         \
    }                                                                \
    case ::mxcore::kFloat16: {                                       \
      using scalar_t = _Float16;                                     \
      { __VA_ARGS__ } break;                                         \
    }                                                                \
    case ::mxcore::kBFloat16: {                                      \
      using scalar_t = __bf16;                                       \
      { __VA_ARGS__ } break;                                         \
    }                                                                \
    case ::mxcore::kInt32: {                                         \
      using scalar_t = int;                                          \
      { __VA_ARGS__ } break;                                         \
    }                                                                \
    case ::mxcore::kInt64: {                                         \
      using scalar_t = long long;                                    \
      { __VA_ARGS__ } break;                                         \
    }                                                                \
    case ::mxcore::kUint8: {                                         \
      using scalar_t = unsigned char;                                \
      { __VA_ARGS__ } break;                                         \
    }                                                                \
    default:                                                         \
      MX_CHECK(false, NAME << ": unsupported dtype "                 \
                           << ::mxcore::dtype_name(DTYPE));          \
  }

// up-to-8-dim shape/stride pack passed by value into kernels
struct Strides8 {
  int ndim = 0;
  int64_t shape[8] = {};
  int64_t s0[8] = {};  // strides of operand 0 (elements; 0 = broadcast)
  int64_t s1[8] = {};  // strides of operand 1
};

inline Strides8 make_strides(const TShape& out, const TShape& a,
                             const TShape& b) {
  Strides8 st;
  st.ndim = (int)out.size();
  MX_CHECK(st.ndim <= 8, "broadcast ndim > 8");
  int na = (int)a.size(), nb = (int)b.size(), no = st.ndim;
  int64_t sa = 1, sb = 1;
  // row-major strides, right-aligned; dim==1 broadcasts with stride 0
  for (int i = no - 1; i >= 0; --i) {
    st.shape[i] = out[i];
    int ia = i - (no - na), ib = i - (no - nb);
    MX_CHECK(ia < 0 || a[ia] == 1 || a[ia] == out[i],
             "broadcast: dim " << (ia >= 0 ? a[ia] : 1)
                               << " incompatible with " << out[i]);
    MX_CHECK(ib < 0 || b[ib] == 1 || b[ib] == out[i],
             "broadcast: dim " << (ib >= 0 ? b[ib] : 1)
                               << " incompatible with " << out[i]);
    if (ia >= 0 && a[ia] != 1) {
      st.s0[i] = sa;
      sa *= a[ia];
    } else {
      st.s0[i] = 0;
      if (ia >= 0) sa *= a[ia];
    }
    if (ib >= 0 && b[ib] != 1) {
      st.s1[i] = sb;
      sb *= b[ib];
    } else {
      st.s1[i] = 0;
      if (ib >= 0) sb *= b[ib];
    }
  }
  return st;
}

inline TShape broadcast_shape(const TShape& a, const TShape& b) {
  int na = (int)a.size(), nb = (int)b.size();
  int n = na > nb ? na : nb;
  TShape out(n);
  for (int i = n - 1; i >= 0; --i) {
    int ia = i - (n - na), ib = i - (n - nb);
    int64_t da = ia >= 0 ? a[ia] : 1, db = ib >= 0 ? b[ib] : 1;
    MX_CHECK(da == db || da == 1 || db == 1,
             "incompatible broadcast dims " << da << " vs " << db);
    out[i] = da > db ? da : db;
  }
  return out;
}

// standard infer helpers
inline FInferShape InferSame(int which = 0) {
  return [which](const NodeAttrs&, const std::vector<TShape>& is,
                 const std::vector<int>& it, std::vector<TShape>* os,
                 std::vector<int>* ot) {
    os->assign(1, is[which]);
    ot->assign(1, it[which]);
  };
}

inline FInferShape InferBroadcast() {
  return [](const NodeAttrs&, const std::vector<TShape>& is,
            const std::vector<int>& it, std::vector<TShape>* os,
            std::vector<int>* ot) {
    MX_CHECK(it[0] == it[1], "binary op dtype mismatch: "
                                 << dtype_name(it[0]) << " vs "
                                 << dtype_name(it[1]));
    os->assign(1, broadcast_shape(is[0], is[1]));
    ot->assign(1, it[0]);
  };
}

}  // namespace mxcore
