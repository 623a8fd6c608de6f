// int8 quantization family (reference src/operator/quantization/:
// quantize_v2, dequantize, quantized FullyConnected).
//
// MI355X design: gfx950's v_mfma_i32_16x16x64_i8 runs at ~2x the bf16
// MFMA rate (measured 3944-4404 TOPS, guide 3), so the quantized GEMM
// is a first-class MFMA kernel, not an emulation: int8 operands, i32
// accumulation, fused scale to fp16/fp32 on the way out.
#include "native_common.h"

using namespace mxcore;

typedef char char16_t_ __attribute__((ext_vector_type(16)));

// symmetric per-tensor quantize: y = clamp(round(x/scale), -127, 127)
template <typename T>
__global__ void quantize_kernel(const T* __restrict__ x,
                                signed char* __restrict__ y, long n,
                                float inv_scale) {
  using V8 = T __attribute__((ext_vector_type(8)));
  typedef signed char c8 __attribute__((ext_vector_type(8)));
  long nv = n / 8;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
       i += (long)gridDim.x * blockDim.x) {
    V8 v = reinterpret_cast<const V8*>(x)[i];
    c8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float q = rintf((float)v[j] * inv_scale);
      o[j] = (signed char)fminf(fmaxf(q, -127.f), 127.f);
    }
    reinterpret_cast<c8*>(y)[i] = o;
  }
  long base = nv * 8;
  long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (tid < n - base) {
    float q = rintf((float)x[base + tid] * inv_scale);
    y[base + tid] = (signed char)fminf(fmaxf(q, -127.f), 127.f);
  }
}

template <typename T>
__global__ void dequantize_kernel(const signed char* __restrict__ x,
                                  T* __restrict__ y, long n, float scale) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    y[i] = (T)((float)x[i] * scale);
}

// int8 NT GEMM: C[M,N] = scale * (A_i8[M,K] x B_i8[N,K]^T), i32 accum.
// Same 128x128 tile anatomy as the fp16 kernel; BK=128 bytes, one
// 16-byte fragment per lane per mfma_i32_16x16x64_i8.
template <typename T>
__global__ __launch_bounds__(256, 2) void gemm_nt_i8_kernel(
    const signed char* __restrict__ A, const signed char* __restrict__ B,
    T* __restrict__ C, long M, long N, long K, float scale,
    const signed char* __restrict__ zpage) {
  constexpr int BM = 128, BN = 128, BK = 128;
  __shared__ signed char As[2][BM * BK];
  __shared__ signed char Bs[2][BN * BK];

  const int nTn = (N + BN - 1) / BN;
  const int nwg = ((M + BM - 1) / BM) * nTn;
  const int bid = xcd_swizzle(blockIdx.x, nwg);
  const long m0 = (long)(bid / nTn) * BM;
  const long n0 = (long)(bid % nTn) * BN;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wid = t >> 6;
  const int wr = wid >> 1, wc = wid & 1;
  const int s_row = t >> 3;          // + r*32
  const int s_col = (t & 7) * 16;    // byte column of the 16 B segment

  int4_t acc[4][4] = {};
  const int nk = (int)((K + BK - 1) / BK);

  auto stage = [&](int buf, int kt) {
    const long k0 = (long)kt * BK;
    const long kcol = k0 + s_col;
    const bool k_ok = kcol + 16 <= K;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long row_a = m0 + r * 32 + s_row;
      const signed char* ga =
          (row_a < M && k_ok) ? A + row_a * K + kcol : zpage;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)ga,
          (__attribute__((address_space(3))) unsigned int*)(uintptr_t)
              &As[buf][(r * 256 + t) * 16],
          16, 0, 0);
      const long row_b = n0 + r * 32 + s_row;
      const signed char* gb =
          (row_b < N && k_ok) ? B + row_b * K + kcol : zpage;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gb,
          (__attribute__((address_space(3))) unsigned int*)(uintptr_t)
              &Bs[buf][(r * 256 + t) * 16],
          16, 0, 0);
    }
  };

  stage(0, 0);
  __syncthreads();

  const int a_row = lane & 15;
  const int k_off = (lane >> 4) * 16;   // byte offset of this lane's frag

  for (int kt = 0; kt < nk; ++kt) {
    const int buf = kt & 1;
    if (kt + 1 < nk) stage(buf ^ 1, kt + 1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {    // two 64-deep mfma steps per BK
      int4_t af[4], bf[4];
#pragma unroll
      for (int m = 0; m < 4; ++m)
        af[m] = *(const int4_t*)&As[buf][(wr * 64 + m * 16 + a_row) * BK +
                                         kk * 64 + k_off];
#pragma unroll
      for (int n = 0; n < 4; ++n)
        bf[n] = *(const int4_t*)&Bs[buf][(wc * 64 + n * 16 + a_row) * BK +
                                         kk * 64 + k_off];
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_i32_16x16x64_i8(
              af[m], bf[n], acc[m][n], 0, 0, 0);
    }
    __syncthreads();
  }

  const int d_col = lane & 15;
  const int d_row = (lane >> 4) * 4;
#pragma unroll
  for (int n = 0; n < 4; ++n) {
    const long col = n0 + wc * 64 + n * 16 + d_col;
    if (col >= N) continue;
#pragma unroll
    for (int m = 0; m < 4; ++m) {
      const long row_base = m0 + wr * 64 + m * 16 + d_row;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const long row = row_base + j;
        if (row < M) C[row * N + col] = (T)((float)acc[m][n][j] * scale);
      }
    }
  }
}


// -- native host ------------------------------------------------------------
#include "ops_api.h"

namespace mxcore {

void quantize_i8_raw(const LaunchCtx& lc, const Arr& x, double scale,
                     const Arr& out) {
  long n = x.numel();
  DISPATCH_FLOAT_NATIVE(x.dtype, "quantize", [&] {
    quantize_kernel<scalar_t><<<ew_grid_n(n / 8 + 1), 256, 0, lc.stream>>>(
        x.data<scalar_t>(), (signed char*)out.ptr, n, (float)(1.0 / scale));
  });
  HIP_CHECK_LAST();
}

void dequantize_i8_raw(const LaunchCtx& lc, const Arr& x, double scale,
                       const Arr& out) {
  long n = x.numel();
  DISPATCH_FLOAT_NATIVE(out.dtype, "dequantize", [&] {
    dequantize_kernel<scalar_t><<<ew_grid_n(n), 256, 0, lc.stream>>>(
        (const signed char*)x.ptr, (scalar_t*)out.ptr, n, (float)scale);
  });
  HIP_CHECK_LAST();
}

void gemm_nt_i8_raw(const LaunchCtx& lc, const Arr& a, const Arr& b,
                    double scale, const Arr& out) {
  MX_CHECK(a.dtype == kInt8 && b.dtype == kInt8,
           "gemm_nt_i8 expects int8 operands");
  long M = a.size(0), K = a.size(1), N = b.size(0);
  MX_CHECK(b.size(1) == K && K % 16 == 0,
           "gemm_nt_i8: K must match and be a multiple of 16");
  long nwg = ((M + 127) / 128) * ((N + 127) / 128);
  DISPATCH_FLOAT_NATIVE(out.dtype, "gemm_i8", [&] {
    gemm_nt_i8_kernel<scalar_t><<<(unsigned)nwg, 256, 0, lc.stream>>>(
        (const signed char*)a.ptr, (const signed char*)b.ptr,
        (scalar_t*)out.ptr, M, N, K, (float)scale,
        (const signed char*)zero_page(lc.dev));
  });
  HIP_CHECK_LAST();
}

}  // namespace mxcore
