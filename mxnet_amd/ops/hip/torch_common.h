// Host-side glue shared by the mxnet_amd HIP translation units:
// torch tensor checks, dtype dispatch, stream access, launch helpers.
#pragma once

#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include "common.h"

// current HIP stream for the active device (PyTorch-ROCm stream pool)
inline hipStream_t cur_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

#define CHECK_GPU(t) TORCH_CHECK((t).is_cuda(), #t " must be on the GPU")
#define CHECK_CONTIG(t) TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")

// Dispatch over the three storage dtypes the compute path supports.
// f(ctype) is instantiated per dtype; half maps to _Float16, bf16 to __bf16
// (the clang native types the MFMA builtins take).
#define DISPATCH_FLOAT_TYPES(DTYPE, NAME, ...)                               \
  [&] {                                                                      \
    switch (DTYPE) {                                                         \
      case at::ScalarType::Half: {                                           \
        using scalar_t = _Float16;                                           \
        return __VA_ARGS__();                                                \
      }                                                                      \
      case at::ScalarType::BFloat16: {                                       \
        using scalar_t = __bf16;                                             \
        return __VA_ARGS__();                                                \
      }                                                                      \
      case at::ScalarType::Float: {                                          \
        using scalar_t = float;                                              \
        return __VA_ARGS__();                                                \
      }                                                                      \
      default:                                                               \
        TORCH_CHECK(false, NAME, ": unsupported dtype ", DTYPE);             \
    }                                                                        \
  }()

// 16-bit-only dispatch (MFMA paths)
#define DISPATCH_HALF_TYPES(DTYPE, NAME, ...)                                \
  [&] {                                                                      \
    switch (DTYPE) {                                                         \
      case at::ScalarType::Half: {                                           \
        using scalar_t = _Float16;                                           \
        return __VA_ARGS__();                                                \
      }                                                                      \
      case at::ScalarType::BFloat16: {                                       \
        using scalar_t = __bf16;                                             \
        return __VA_ARGS__();                                                \
      }                                                                      \
      default:                                                               \
        TORCH_CHECK(false, NAME, ": MFMA path needs fp16/bf16, got ", DTYPE); \
    }                                                                        \
  }()

// cross-TU helpers (gemm.hip)
at::Tensor gemm_nt_core(at::Tensor A, at::Tensor B,
                        c10::optional<at::Tensor> bias, bool relu,
                        at::Tensor* stats_out = nullptr);
at::Tensor transpose2d(const at::Tensor& x);
at::Tensor colsum(const at::Tensor& in);
const void* zero_page(const at::Tensor& like);
at::Tensor softmax_fwd(const at::Tensor& x, bool log_mode, double temperature,
                       c10::optional<at::Tensor> mask);
at::Tensor softmax_bwd(const at::Tensor& dy, const at::Tensor& y,
                       bool log_mode, double temperature);

constexpr int kEwBlock = 256;
// memory-bound launch cap: ~8 blocks/CU on 256 CUs (Guideline 11)
static const long kEwMaxGrid = [] {
  const char* e = getenv("MXNET_EW_BLOCKS");
  return e ? atoi(e) : 1024;  // swept: 1024 edges 2048/4096
}();

inline int ew_grid(long work_items) {
  long g = (work_items + kEwBlock - 1) / kEwBlock;
  return (int)(g < kEwMaxGrid ? (g > 0 ? g : 1) : kEwMaxGrid);
}
