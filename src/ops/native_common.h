// Host-side glue for the native kernel launchers: a lightweight tensor
// handle (raw pointer + shape + dtype flag) and the launch context.
// No torch headers anywhere — the Python frontend passes
// (data_ptr, shape, dtype, stream) and allocates outputs itself; the
// native registry path passes engine TBlobs + the device compute stream.
#pragma once

#include <hip/hip_runtime.h>

#include <functional>
#include <string>
#include <vector>

#include "../core/base.h"
#include "common.h"  // device helpers (DTraits, wave reductions, swizzle)

namespace mxcore {

struct Arr {
  void* ptr = nullptr;
  std::vector<int64_t> shape;
  int dtype = kFloat32;

  Arr() = default;
  Arr(void* p, std::vector<int64_t> s, int dt)
      : ptr(p), shape(std::move(s)), dtype(dt) {}
  Arr(const TBlob& b) : ptr(b.dptr), shape(b.shape), dtype(b.dtype) {}

  bool defined() const { return ptr != nullptr; }
  int dim() const { return (int)shape.size(); }
  int64_t size(int i) const {
    if (i < 0) i += dim();
    return shape[i];
  }
  int64_t numel() const {
    int64_t n = 1;
    for (auto d : shape) n *= d;
    return n;
  }
  void* data_ptr() const { return ptr; }
  template <typename T>
  T* data() const { return (T*)ptr; }
};

struct LaunchCtx {
  hipStream_t stream = nullptr;
  // compute-stream-ordered scratch (valid for this launch only)
  std::function<void*(size_t)> workspace;
  // fresh allocation for multi-launch temporaries that must not alias the
  // arena (rare); returned pointers stay alive until the stream drains the
  // current step — backed by the per-device ring below
  int dev = 0;
};

// dtype dispatch over the kernel storage types
#define DISPATCH_FLOAT_NATIVE(DTYPE, NAME, ...)                       \
  [&] {                                                               \
    switch (DTYPE) {                                                  \
      case ::mxcore::kFloat16: {                                      \
        using scalar_t = _Float16;                                    \
        return __VA_ARGS__();                                         \
      }                                                               \
      case ::mxcore::kBFloat16: {                                     \
        using scalar_t = __bf16;                                      \
        return __VA_ARGS__();                                         \
      }                                                               \
      case ::mxcore::kFloat32: {                                      \
        using scalar_t = float;                                       \
        return __VA_ARGS__();                                         \
      }                                                               \
      default:                                                        \
        MX_CHECK(false, NAME << ": unsupported dtype "                \
                             << ::mxcore::dtype_name(DTYPE));         \
        __builtin_unreachable();                                      \
    }                                                                 \
  }()

#define DISPATCH_HALF_NATIVE(DTYPE, NAME, ...)                        \
  [&] {                                                               \
    switch (DTYPE) {                                                  \
      case ::mxcore::kFloat16: {                                      \
        using scalar_t = _Float16;                                    \
        return __VA_ARGS__();                                         \
      }                                                               \
      case ::mxcore::kBFloat16: {                                     \
        using scalar_t = __bf16;                                      \
        return __VA_ARGS__();                                         \
      }                                                               \
      default:                                                        \
        MX_CHECK(false, NAME << ": MFMA path needs fp16/bf16, got "   \
                             << ::mxcore::dtype_name(DTYPE));         \
        __builtin_unreachable();                                      \
    }                                                                 \
  }()

#define CHECK_SAME_DTYPE(a, b) \
  MX_CHECK((a).dtype == (b).dtype, "dtype mismatch")

// per-device zero page (OOB redirect target for global_load_lds staging:
// the HW needs a valid address; branch-free bounds handling)
const void* zero_page(int dev);

// launch-helper shared with ew kernels
inline int ew_grid_n(long work_items) {
  long cap = env_int("MXNET_EW_BLOCKS", 1024);
  long g = (work_items + 255) / 256;
  if (g < 1) g = 1;
  return (int)(g < cap ? g : cap);
}

}  // namespace mxcore
