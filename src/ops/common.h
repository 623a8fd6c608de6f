// Common device helpers for the mxnet_amd gfx950 kernels.
// CDNA4 ONLY (MI355X): wave64, MFMA, 160 KiB LDS, 8 XCDs.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

// ---------------------------------------------------------------------------
// vector types
// ---------------------------------------------------------------------------
typedef _Float16 half8_t __attribute__((ext_vector_type(8)));
typedef __bf16 bf168_t __attribute__((ext_vector_type(8)));
typedef float float4_t __attribute__((ext_vector_type(4)));
typedef float float16_t __attribute__((ext_vector_type(16)));
typedef short short8_t __attribute__((ext_vector_type(8)));
typedef int int4_t __attribute__((ext_vector_type(4)));

// ---------------------------------------------------------------------------
// dtype traits: unify f16/bf16 paths (storage, fragment, MFMA intrinsic)
// ---------------------------------------------------------------------------
template <typename T> struct DTraits;

template <> struct DTraits<_Float16> {
  using frag8 = half8_t;
  static DEV_INLINE float4_t mfma_16x16x32(frag8 a, frag8 b, float4_t c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
  }
  static DEV_INLINE float to_float(_Float16 v) { return (float)v; }
  static DEV_INLINE _Float16 from_float(float v) { return (_Float16)v; }
};

template <> struct DTraits<__bf16> {
  using frag8 = bf168_t;
  static DEV_INLINE float4_t mfma_16x16x32(frag8 a, frag8 b, float4_t c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  }
  static DEV_INLINE float to_float(__bf16 v) { return (float)v; }
  static DEV_INLINE __bf16 from_float(float v) { return (__bf16)v; }
};

// float "traits" for kernels templated over storage dtype (elementwise etc.)
template <> struct DTraits<float> {
  static DEV_INLINE float to_float(float v) { return v; }
  static DEV_INLINE float from_float(float v) { return v; }
};

// ---------------------------------------------------------------------------
// wave reductions (64-lane; NOT the CUDA 32-lane idiom)
// ---------------------------------------------------------------------------
DEV_INLINE float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

DEV_INLINE float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// block reduce via LDS (block <= 1024 threads); smem must hold >=16 floats
template <typename Op>
DEV_INLINE float block_reduce(float v, float* smem, Op op, float init) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, 64));
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  const int nwaves = (blockDim.x + 63) >> 6;
  v = (threadIdx.x < nwaves) ? smem[threadIdx.x] : init;
  if (wid == 0) {
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) v = op(v, __shfl_xor(v, off, 64));
    if (lane == 0) smem[0] = v;
  }
  __syncthreads();
  return smem[0];
}

struct SumOp { DEV_INLINE float operator()(float a, float b) const { return a + b; } };
struct MaxOp { DEV_INLINE float operator()(float a, float b) const { return fmaxf(a, b); } };

// ---------------------------------------------------------------------------
// XCD-aware blockIdx swizzle (guide T1, bijective variant m204):
// contiguous grid chunks land on one XCD's L2 so neighbor tiles share
// operand panels.  Safe for any nwg (bijective).
// ---------------------------------------------------------------------------
#define N_XCD 8
DEV_INLINE int xcd_swizzle(int bid, int nwg) {
  if (nwg < 2 * N_XCD) return bid;
  int q = nwg / N_XCD, r = nwg % N_XCD;
  int xcd = bid % N_XCD, idx = bid / N_XCD;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

#include <stdexcept>
#include <string>

#define HIP_CHECK_LAST()                                                      \
  do {                                                                        \
    hipError_t e_ = hipGetLastError();                                        \
    if (e_ != hipSuccess) {                                                   \
      throw std::runtime_error(std::string("HIP kernel launch failed: ") +    \
                               hipGetErrorString(e_));                        \
    }                                                                         \
  } while (0)
