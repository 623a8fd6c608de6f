// BatchNorm (NHWC, fused residual-add + ReLU) and LayerNorm.
//
// Reference parity: src/operator/nn/batch_norm.cu:238-660 (fwd/bwd incl.
// the fused BNReLU/BNAddReLU variants), src/operator/nn/layer_norm.cu.
//
// MI355X design: NHWC puts channels contiguous, so per-channel statistics
// are column reductions: blocks own a 64-column chunk x row range,
// accumulate in fp32 registers, cross-wave reduce in LDS, one atomicAdd
// per column into an fp32 workspace.  The apply pass fuses
// normalize + residual-add + ReLU (the reference needs a graph fusion
// pass for this; here it is a single kernel by construction).
#include "native_common.h"

using namespace mxcore;

// ---------------------------------------------------------------------------
// column-chunk reduction: each block covers cols [c0,c0+64) and a row range,
// threads = 64 cols x 4 row-lanes.
// Computes sum and sum-of-squares (fwd) or the two backward sums.
// ---------------------------------------------------------------------------
// vectorized variant (C % 8 == 0): 256 threads = 8 channel-groups x 32
// row-lanes, each lane streams half8 (16 B - Guideline 13) and keeps 8
// per-channel fp32 partials in registers; LDS tree-reduce over row-lanes.
template <typename T>
__global__ void bn_reduce_vec_kernel(const T* __restrict__ x, long M, long C,
                                     long rows_per_block,
                                     float* __restrict__ sum,
                                     float* __restrict__ sumsq) {
  using V8 = T __attribute__((ext_vector_type(8)));
  __shared__ float b0[256][8];
  __shared__ float b1[256][8];
  const int t = threadIdx.x;
  const int cg = t & 7;            // which 8-channel group
  const int rl = t >> 3;           // row lane 0..31
  const long c0 = (long)blockIdx.x * 64 + cg * 8;
  const long r0 = (long)blockIdx.y * rows_per_block;
  const long r1 = min(M, r0 + rows_per_block);
  float a0[8] = {}, a1[8] = {};
  if (c0 + 8 <= C) {
    for (long r = r0 + rl; r < r1; r += 32) {
      V8 v = *(const V8*)(x + r * C + c0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = (float)v[j];
        a0[j] += f;
        a1[j] += f * f;
      }
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    b0[t][j] = a0[j];
    b1[t][j] = a1[j];
  }
  __syncthreads();
  // threads 0..63 finalize channel c = blockIdx.x*64 + t
  if (t < 64) {
    long c = (long)blockIdx.x * 64 + t;
    if (c < C) {
      int g = t >> 3, j = t & 7;
      float s0 = 0.f, s1 = 0.f;
      for (int r = 0; r < 32; ++r) {
        s0 += b0[r * 8 + g][j];
        s1 += b1[r * 8 + g][j];
      }
      if (gridDim.y == 1) {
        sum[c] = s0;
        sumsq[c] = s1;
      } else {
        atomicAdd(sum + c, s0);
        atomicAdd(sumsq + c, s1);
      }
    }
  }
}

template <typename T>
__global__ void bn_reduce_kernel(const T* __restrict__ x, long M, long C,
                                 long rows_per_block,
                                 float* __restrict__ sum,
                                 float* __restrict__ sumsq) {
  __shared__ float s0[4][64], s1[4][64];
  int cc = threadIdx.x & 63;        // column within chunk
  int rl = threadIdx.x >> 6;        // row lane 0..3
  long c = (long)blockIdx.x * 64 + cc;
  long r0 = (long)blockIdx.y * rows_per_block;
  long r1 = min(M, r0 + rows_per_block);
  float a0 = 0.f, a1 = 0.f;
  if (c < C) {
    for (long r = r0 + rl; r < r1; r += 4) {
      float v = (float)x[r * C + c];
      a0 += v;
      a1 += v * v;
    }
  }
  s0[rl][cc] = a0;
  s1[rl][cc] = a1;
  __syncthreads();
  if (rl == 0 && c < C) {
    a0 = s0[0][cc] + s0[1][cc] + s0[2][cc] + s0[3][cc];
    a1 = s1[0][cc] + s1[1][cc] + s1[2][cc] + s1[3][cc];
    if (gridDim.y == 1) {
      sum[c] = a0;
      sumsq[c] = a1;
    } else {
      atomicAdd(sum + c, a0);
      atomicAdd(sumsq + c, a1);
    }
  }
}

// finalize training stats: mean/istd + running-stat update
// (running = momentum*running + (1-momentum)*batch; running var unbiased,
// matching the torch CPU oracle)
// nslices > 0: sum/sumsq point at the conv epilogue's sliced
// [nslices][2][C] workspace; the 64-way fold happens here (saves the
// separate torch reduction + contiguous sum buffers).
__global__ void bn_finalize_kernel(const float* __restrict__ sum,
                                   const float* __restrict__ sumsq, long M,
                                   long C, float momentum, float eps,
                                   float* __restrict__ save_mean,
                                   float* __restrict__ save_istd,
                                   float* __restrict__ rmean,
                                   float* __restrict__ rvar,
                                   int nslices = 0) {
  for (long c = (long)blockIdx.x * blockDim.x + threadIdx.x; c < C;
       c += (long)gridDim.x * blockDim.x) {
    float s0, s1;
    if (nslices > 0) {
      s0 = 0.f; s1 = 0.f;
      for (int k = 0; k < nslices; ++k) {
        s0 += sum[(long)k * 2 * C + c];
        s1 += sum[(long)k * 2 * C + C + c];
      }
    } else {
      s0 = sum[c];
      s1 = sumsq[c];
    }
    float mean = s0 / M;
    float var = fmaxf(s1 / M - mean * mean, 0.f);
    save_mean[c] = mean;
    save_istd[c] = rsqrtf(var + eps);
    if (rmean) {
      float unbias = M > 1 ? var * M / (M - 1) : var;
      rmean[c] = momentum * rmean[c] + (1.f - momentum) * mean;
      rvar[c] = momentum * rvar[c] + (1.f - momentum) * unbias;
    }
  }
}

// inference: scale/shift from running stats
__global__ void bn_scale_shift_kernel(const float* __restrict__ gamma,
                                      const float* __restrict__ beta,
                                      const float* __restrict__ mean,
                                      const float* __restrict__ var_or_istd,
                                      bool is_istd, float eps, long C,
                                      float* __restrict__ scale,
                                      float* __restrict__ shift) {
  for (long c = (long)blockIdx.x * blockDim.x + threadIdx.x; c < C;
       c += (long)gridDim.x * blockDim.x) {
    float istd = is_istd ? var_or_istd[c] : rsqrtf(var_or_istd[c] + eps);
    float sc = gamma[c] * istd;
    scale[c] = sc;
    shift[c] = beta[c] - mean[c] * sc;
  }
}


// channel-block-resident apply variants: each lane owns one 8-channel
// block so scale/shift (and the backward per-channel terms) load ONCE
// into registers — no LDS, no bank conflicts (PMC showed 6e7 conflicts
// on the LDS-cached variant).  Requires 256 % (C/8) == 0.
template <typename T, bool RELU, bool RES>
__global__ void bn_apply_cb_kernel(const T* __restrict__ x,
                                   const T* __restrict__ res,
                                   T* __restrict__ y, long rows, long C,
                                   const float* __restrict__ scale,
                                   const float* __restrict__ shift,
                                   unsigned char* __restrict__ mask) {
  using VecT = T __attribute__((ext_vector_type(8)));
  const int nblk = (int)(C >> 3);
  const int cb = threadIdx.x % nblk;
  const int r_off = threadIdx.x / nblk;
  const int rows_per_block = 256 / nblk;
  const long c0 = (long)cb * 8;
  float sc[8], sh[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    sc[j] = scale[c0 + j];
    sh[j] = shift[c0 + j];
  }
  const long rstride = (long)gridDim.x * rows_per_block;
  for (long r = (long)blockIdx.x * rows_per_block + r_off; r < rows;
       r += rstride) {
    const long vidx = (r * C + c0) / 8;
    VecT v = reinterpret_cast<const VecT*>(x)[vidx];
    VecT o;
    unsigned char mb = 0;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float z = (float)v[j] * sc[j] + sh[j];
      if (RES) z += (float)res[vidx * 8 + j];
      if (RELU) {
        if (z > 0.f) mb |= (1u << j);
        z = fmaxf(z, 0.f);
      }
      o[j] = (T)z;
    }
    reinterpret_cast<VecT*>(y)[vidx] = o;
    if (RELU && mask) mask[vidx] = mb;
  }
}

template <typename T, bool RELU, bool RES>
__global__ void bn_bwd_apply_cb_kernel(
    const T* __restrict__ dy, const T* __restrict__ x,
    const T* __restrict__ y, T* __restrict__ dx, T* __restrict__ dres,
    long rows, long C, float invM, const float* __restrict__ gamma,
    const float* __restrict__ mean, const float* __restrict__ istd,
    const float* __restrict__ s1, const float* __restrict__ s2,
    const unsigned char* __restrict__ mask) {
  using VecT = T __attribute__((ext_vector_type(8)));
  const int nblk = (int)(C >> 3);
  const int cb = threadIdx.x % nblk;
  const int r_off = threadIdx.x / nblk;
  const int rows_per_block = 256 / nblk;
  const long c0 = (long)cb * 8;
  float gis[8], mu[8], is[8], m1[8], m2[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    gis[j] = gamma[c0 + j] * istd[c0 + j];
    mu[j] = mean[c0 + j];
    is[j] = istd[c0 + j];
    m1[j] = s1[c0 + j] * invM;
    m2[j] = s2[c0 + j] * invM;
  }
  const long rstride = (long)gridDim.x * rows_per_block;
  for (long r = (long)blockIdx.x * rows_per_block + r_off; r < rows;
       r += rstride) {
    const long vidx = (r * C + c0) / 8;
    VecT vg = reinterpret_cast<const VecT*>(dy)[vidx];
    VecT vx = reinterpret_cast<const VecT*>(x)[vidx];
    VecT vy;
    unsigned char mb = 0xff;
    if (RELU) {
      if (mask) mb = mask[vidx];
      else vy = reinterpret_cast<const VecT*>(y)[vidx];
    }
    VecT odx, ores;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float g = (float)vg[j];
      if (RELU) {
        bool on = mask ? ((mb >> j) & 1) : ((float)vy[j] > 0.f);
        if (!on) g = 0.f;
      }
      float xhat = ((float)vx[j] - mu[j]) * is[j];
      odx[j] = (T)(gis[j] * (g - m1[j] - xhat * m2[j]));
      if (RES) ores[j] = (T)g;
    }
    reinterpret_cast<VecT*>(dx)[vidx] = odx;
    if (RES) reinterpret_cast<VecT*>(dres)[vidx] = ores;
  }
}

// apply: y = x*scale[c] + shift[c] (+ residual) (relu)
// scale/shift staged in LDS when C fits (<=4096).
template <typename T, bool RELU, bool RES>
__global__ void bn_apply_kernel(const T* __restrict__ x,
                                const T* __restrict__ res,
                                T* __restrict__ y, long total, long C,
                                const float* __restrict__ scale,
                                const float* __restrict__ shift,
                                unsigned char* __restrict__ mask) {
  extern __shared__ float lds[];
  float* s_scale = lds;
  float* s_shift = lds + C;
  const bool use_lds = C <= 4096;
  if (use_lds) {
    for (long c = threadIdx.x; c < C; c += blockDim.x) {
      s_scale[c] = scale[c];
      s_shift[c] = shift[c];
    }
    __syncthreads();
  }
  const float* sc = use_lds ? s_scale : scale;
  const float* sh = use_lds ? s_shift : shift;
  using VecT = T __attribute__((ext_vector_type(8)));
  if (C % 8 == 0) {
    long nv = total / 8;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
         i += (long)gridDim.x * blockDim.x) {
      long c0 = (i * 8) % C;
      VecT v = reinterpret_cast<const VecT*>(x)[i];
      VecT o;
      unsigned char mb = 0;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float z = (float)v[j] * sc[c0 + j] + sh[c0 + j];
        if (RES) z += (float)res[i * 8 + j];
        if (RELU) {
          if (z > 0.f) mb |= (1u << j);
          z = fmaxf(z, 0.f);
        }
        o[j] = (T)z;
      }
      reinterpret_cast<VecT*>(y)[i] = o;
      if (RELU && mask) mask[i] = mb;
    }
  } else {
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
         i += (long)gridDim.x * blockDim.x) {
      long c = i % C;
      float z = (float)x[i] * sc[c] + sh[c];
      if (RES) z += (float)res[i];
      y[i] = (T)(RELU ? fmaxf(z, 0.f) : z);
    }
  }
}

// backward column reduction: s1 = sum(dy_eff), s2 = sum(dy_eff * xhat)
// dy_eff = relu-masked dy (mask from the saved post-activation y)
// vectorized backward reduction (same geometry as bn_reduce_vec_kernel)
template <typename T, bool RELU>
__global__ void bn_bwd_reduce_vec_kernel(const T* __restrict__ dy,
                                         const T* __restrict__ x,
                                         const T* __restrict__ y, long M,
                                         long C, long rows_per_block,
                                         const float* __restrict__ mean,
                                         const float* __restrict__ istd,
                                         float* __restrict__ s1,
                                         float* __restrict__ s2,
                                         const unsigned char* __restrict__
                                             mask) {
  using V8 = T __attribute__((ext_vector_type(8)));
  __shared__ float b1[256][8];
  __shared__ float b2[256][8];
  const int t = threadIdx.x;
  const int cg = t & 7;
  const int rl = t >> 3;
  const long c0 = (long)blockIdx.x * 64 + cg * 8;
  const long r0 = (long)blockIdx.y * rows_per_block;
  const long r1 = min(M, r0 + rows_per_block);
  float a1[8] = {}, a2[8] = {};
  if (c0 + 8 <= C) {
    float mu[8], is[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      mu[j] = mean[c0 + j];
      is[j] = istd[c0 + j];
    }
    for (long r = r0 + rl; r < r1; r += 32) {
      V8 vg = *(const V8*)(dy + r * C + c0);
      V8 vx = *(const V8*)(x + r * C + c0);
      V8 vy;
      unsigned char mb = 0xff;
      if (RELU) {
        if (mask) mb = mask[(r * C + c0) / 8];
        else vy = *(const V8*)(y + r * C + c0);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = (float)vg[j];
        if (RELU) {
          bool on = mask ? ((mb >> j) & 1) : ((float)vy[j] > 0.f);
          if (!on) g = 0.f;
        }
        a1[j] += g;
        a2[j] += g * ((float)vx[j] - mu[j]) * is[j];
      }
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    b1[t][j] = a1[j];
    b2[t][j] = a2[j];
  }
  __syncthreads();
  if (t < 64) {
    long c = (long)blockIdx.x * 64 + t;
    if (c < C) {
      int g = t >> 3, j = t & 7;
      float r1v = 0.f, r2v = 0.f;
      for (int r = 0; r < 32; ++r) {
        r1v += b1[r * 8 + g][j];
        r2v += b2[r * 8 + g][j];
      }
      if (gridDim.y == 1) {
        s1[c] = r1v;
        s2[c] = r2v;
      } else {
        atomicAdd(s1 + c, r1v);
        atomicAdd(s2 + c, r2v);
      }
    }
  }
}

template <typename T, bool RELU>
__global__ void bn_bwd_reduce_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const T* __restrict__ y, long M, long C,
                                     long rows_per_block,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ istd,
                                     float* __restrict__ s1,
                                     float* __restrict__ s2) {
  __shared__ float l1[4][64], l2[4][64];
  int cc = threadIdx.x & 63;
  int rl = threadIdx.x >> 6;
  long c = (long)blockIdx.x * 64 + cc;
  long r0 = (long)blockIdx.y * rows_per_block;
  long r1 = min(M, r0 + rows_per_block);
  float a1 = 0.f, a2 = 0.f;
  if (c < C) {
    float mu = mean[c], is = istd[c];
    for (long r = r0 + rl; r < r1; r += 4) {
      long idx = r * C + c;
      float g = (float)dy[idx];
      if (RELU && (float)y[idx] <= 0.f) g = 0.f;
      a1 += g;
      a2 += g * ((float)x[idx] - mu) * is;
    }
  }
  l1[rl][cc] = a1;
  l2[rl][cc] = a2;
  __syncthreads();
  if (rl == 0 && c < C) {
    a1 = l1[0][cc] + l1[1][cc] + l1[2][cc] + l1[3][cc];
    a2 = l2[0][cc] + l2[1][cc] + l2[2][cc] + l2[3][cc];
    if (gridDim.y == 1) {
      s1[c] = a1;
      s2[c] = a2;
    } else {
      atomicAdd(s1 + c, a1);
      atomicAdd(s2 + c, a2);
    }
  }
}

// backward apply:
// dx = gamma*istd * (dy_eff - (s1 + xhat*s2)/M); dres = dy_eff
template <typename T, bool RELU, bool RES>
__global__ void bn_bwd_apply_kernel(const T* __restrict__ dy,
                                    const T* __restrict__ x,
                                    const T* __restrict__ y,
                                    T* __restrict__ dx, T* __restrict__ dres,
                                    long total, long C, float invM,
                                    const float* __restrict__ gamma,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ istd,
                                    const float* __restrict__ s1,
                                    const float* __restrict__ s2,
                                    const unsigned char* __restrict__ mask) {
  extern __shared__ float lds[];  // [C] x5: ga*is, mean, istd, s1/M, s2/M
  float* c_gis = lds;
  float* c_mu = lds + C;
  float* c_is = lds + 2 * C;
  float* c_s1 = lds + 3 * C;
  float* c_s2 = lds + 4 * C;
  const bool use_lds = C <= 4096;
  if (use_lds) {
    for (long c = threadIdx.x; c < C; c += blockDim.x) {
      c_gis[c] = gamma[c] * istd[c];
      c_mu[c] = mean[c];
      c_is[c] = istd[c];
      c_s1[c] = s1[c] * invM;
      c_s2[c] = s2[c] * invM;
    }
    __syncthreads();
  }
  using VecT = T __attribute__((ext_vector_type(8)));
  if (use_lds && C % 8 == 0) {
    long nv = total / 8;
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nv;
         i += (long)gridDim.x * blockDim.x) {
      long c0 = (i * 8) % C;
      VecT vg = reinterpret_cast<const VecT*>(dy)[i];
      VecT vx = reinterpret_cast<const VecT*>(x)[i];
      VecT vy;
      unsigned char mb = 0xff;
      if (RELU) {
        if (mask) mb = mask[i];
        else vy = reinterpret_cast<const VecT*>(y)[i];
      }
      VecT odx, ores;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        long c = c0 + j;
        float g = (float)vg[j];
        if (RELU) {
          bool on = mask ? ((mb >> j) & 1) : ((float)vy[j] > 0.f);
          if (!on) g = 0.f;
        }
        float xhat = ((float)vx[j] - c_mu[c]) * c_is[c];
        odx[j] = (T)(c_gis[c] * (g - c_s1[c] - xhat * c_s2[c]));
        if (RES) ores[j] = (T)g;
      }
      reinterpret_cast<VecT*>(dx)[i] = odx;
      if (RES) reinterpret_cast<VecT*>(dres)[i] = ores;
    }
    return;
  }
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    long c = i % C;
    float g = (float)dy[i];
    if (RELU && (float)y[i] <= 0.f) g = 0.f;
    float gis, mu, m1, m2, is;
    if (use_lds) {
      gis = c_gis[c]; mu = c_mu[c]; is = c_is[c]; m1 = c_s1[c]; m2 = c_s2[c];
    } else {
      gis = gamma[c] * istd[c]; mu = mean[c]; is = istd[c];
      m1 = s1[c] * invM; m2 = s2[c] * invM;
    }
    float xhat = ((float)x[i] - mu) * is;
    dx[i] = (T)(gis * (g - m1 - xhat * m2));
    if (RES) dres[i] = (T)g;
  }
}

// -- host wrappers ----------------------------------------------------------

static dim3 bn_reduce_grid(long M, long C, long* rows_per_block) {
  static const long want = [] {
    const char* e = getenv("MXNET_BN_REDUCE_BLOCKS");
    return e ? atol(e) : 1024L;  // swept on HW: 1024 beats 2048/4096
                                 // (322.6 vs 335.9/349.1 us on bn_l1)
  }();
  long xb = (C + 63) / 64;
  long yb = std::max<long>(1, std::min<long>((M + 255) / 256,
                                             want / std::max<long>(xb, 1)));
  *rows_per_block = (M + yb - 1) / yb;
  return dim3((unsigned)xb, (unsigned)yb);
}

// ---------------------------------------------------------------------------
// LayerNorm over the last axis (reference layer_norm.cu:172-560)
// block per row; fp32 stats; saved mean/istd for backward
// ---------------------------------------------------------------------------
template <typename T, int VEC>
__global__ void ln_fwd_kernel(const T* __restrict__ x,
                              const float* __restrict__ gamma,
                              const float* __restrict__ beta,
                              T* __restrict__ y, float* __restrict__ omean,
                              float* __restrict__ oistd, long rows, long C,
                              float eps) {
  __shared__ float sred[16];
  using VecT = T __attribute__((ext_vector_type(VEC)));
  for (long r = blockIdx.x; r < rows; r += gridDim.x) {
    const T* xr = x + r * C;
    T* yr = y + r * C;
    long cv = C / VEC;
    float s = 0.f, sq = 0.f;
    for (long i = threadIdx.x; i < cv; i += blockDim.x) {
      VecT v = reinterpret_cast<const VecT*>(xr)[i];
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float f = (float)v[j];
        s += f;
        sq += f * f;
      }
    }
    for (long i = cv * VEC + threadIdx.x; i < C; i += blockDim.x) {
      float f = (float)xr[i];
      s += f;
      sq += f * f;
    }
    s = block_reduce(s, sred, SumOp(), 0.f);
    __syncthreads();
    sq = block_reduce(sq, sred, SumOp(), 0.f);
    float mean = s / C;
    float istd = rsqrtf(fmaxf(sq / C - mean * mean, 0.f) + eps);
    if (threadIdx.x == 0) {
      omean[r] = mean;
      oistd[r] = istd;
    }
    for (long i = threadIdx.x; i < cv; i += blockDim.x) {
      VecT v = reinterpret_cast<const VecT*>(xr)[i];
      VecT o;
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        long c = i * VEC + j;
        o[j] = (T)(((float)v[j] - mean) * istd * gamma[c] + beta[c]);
      }
      reinterpret_cast<VecT*>(yr)[i] = o;
    }
    for (long i = cv * VEC + threadIdx.x; i < C; i += blockDim.x)
      yr[i] = (T)(((float)xr[i] - mean) * istd * gamma[i] + beta[i]);
    __syncthreads();
  }
}

// dx = istd * (dy*g - mean_c(dy*g) - xhat * mean_c(dy*g*xhat))
template <typename T>
__global__ void ln_bwd_dx_kernel(const T* __restrict__ dy,
                                 const T* __restrict__ x,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ istd,
                                 T* __restrict__ dx, long rows, long C) {
  __shared__ float sred[16];
  for (long r = blockIdx.x; r < rows; r += gridDim.x) {
    const T* gr = dy + r * C;
    const T* xr = x + r * C;
    T* dr = dx + r * C;
    float mu = mean[r], is = istd[r];
    float a = 0.f, b = 0.f;
    for (long i = threadIdx.x; i < C; i += blockDim.x) {
      float gg = (float)gr[i] * gamma[i];
      float xh = ((float)xr[i] - mu) * is;
      a += gg * xh;
      b += gg;
    }
    a = block_reduce(a, sred, SumOp(), 0.f);
    __syncthreads();
    b = block_reduce(b, sred, SumOp(), 0.f);
    a /= C;
    b /= C;
    for (long i = threadIdx.x; i < C; i += blockDim.x) {
      float gg = (float)gr[i] * gamma[i];
      float xh = ((float)xr[i] - mu) * is;
      dr[i] = (T)(is * (gg - b - xh * a));
    }
    __syncthreads();
  }
}

// dgamma[c] = sum_r dy*xhat ; dbeta[c] = sum_r dy  (column reduction)
template <typename T>
__global__ void ln_bwd_dgamma_kernel(const T* __restrict__ dy,
                                     const T* __restrict__ x,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ istd,
                                     float* __restrict__ dgamma,
                                     float* __restrict__ dbeta, long rows,
                                     long C, long rows_per_block) {
  // vectorized like bn_bwd_reduce: 8 col-groups x 32 row-lanes, half8
  // loads (the scalar-column first cut measured 10x off HBM roof)
  using V8 = T __attribute__((ext_vector_type(8)));
  __shared__ float l1[256][8];
  __shared__ float l2[256][8];
  const int t = threadIdx.x;
  const int cg = t & 7;
  const int rl = t >> 3;
  const long c0 = (long)blockIdx.x * 64 + cg * 8;
  const long r0 = (long)blockIdx.y * rows_per_block;
  const long r1 = min(rows, r0 + rows_per_block);
  float a1[8] = {}, a2[8] = {};
  if (c0 + 8 <= C) {
    for (long r = r0 + rl; r < r1; r += 32) {
      V8 g8 = *(const V8*)(dy + r * C + c0);
      V8 x8 = *(const V8*)(x + r * C + c0);
      const float mu = mean[r], is = istd[r];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = (float)g8[j];
        a1[j] += g * ((float)x8[j] - mu) * is;
        a2[j] += g;
      }
    }
  } else if (c0 < C) {
    for (long r = r0 + rl; r < r1; r += 32) {
      const float mu = mean[r], is = istd[r];
      for (int j = 0; j < 8 && c0 + j < C; ++j) {
        float g = (float)dy[r * C + c0 + j];
        a1[j] += g * ((float)x[r * C + c0 + j] - mu) * is;
        a2[j] += g;
      }
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    l1[t][j] = a1[j];
    l2[t][j] = a2[j];
  }
  __syncthreads();
  if (t < 64) {
    const long c = (long)blockIdx.x * 64 + t;
    if (c < C) {
      const int g = t >> 3, j = t & 7;
      float s1 = 0.f, s2 = 0.f;
      for (int r = 0; r < 32; ++r) {
        s1 += l1[r * 8 + g][j];
        s2 += l2[r * 8 + g][j];
      }
      if (gridDim.y == 1) {
        dgamma[c] = s1;
        dbeta[c] = s2;
      } else {
        atomicAdd(dgamma + c, s1);
        atomicAdd(dbeta + c, s2);
      }
    }
  }
}


// ===========================================================================
// native host launchers (gamma/beta/running-stats/saved-stats are fp32 by
// contract; outputs caller-allocated; workspaces from the arena)
// ===========================================================================
#include <algorithm>

#include "ops_api.h"

namespace mxcore {

#define CHECK_F32(a) MX_CHECK((a).dtype == kFloat32, #a " must be fp32")

void bn_fwd_train_raw(const LaunchCtx& lc, const Arr& x, const Arr& gamma,
                      const Arr& beta, const Arr& rmean, const Arr& rvar,
                      double momentum, double eps, bool fuse_relu,
                      const Arr& residual, const Arr& presums, const Arr& y,
                      const Arr& save_mean, const Arr& save_istd,
                      const Arr& mask) {
  CHECK_F32(gamma); CHECK_F32(beta); CHECK_F32(rmean); CHECK_F32(rvar);
  CHECK_F32(save_mean); CHECK_F32(save_istd);
  long C = x.size(-1), M = x.numel() / C;
  bool have_pre = presums.defined() && presums.numel() > 0;
  int nslices = have_pre ? (int)presums.size(0) : 0;
  float *sum, *sumsq;
  if (have_pre) {
    sum = presums.data<float>();  // sliced conv/GEMM epilogue stats
    sumsq = sum;                  // unused in the sliced path
  } else {
    sum = (float*)lc.workspace((size_t)2 * C * 4);
    sumsq = sum + C;
    MX_HIP_CALL(hipMemsetAsync(sum, 0, (size_t)2 * C * 4, lc.stream));
  }
  float* scale = (float*)lc.workspace((size_t)2 * C * 4);
  float* shift = scale + C;
  bool has_res = residual.defined() && residual.numel() > 0;
  bool want_mask = mask.defined() && mask.numel() > 0;
  long rpb;
  dim3 grid = bn_reduce_grid(M, C, &rpb);
  DISPATCH_FLOAT_NATIVE(x.dtype, "bn_fwd", [&] {
    if (!have_pre) {
      if (C % 8 == 0 && sizeof(scalar_t) == 2)
        bn_reduce_vec_kernel<scalar_t><<<grid, 256, 0, lc.stream>>>(
            x.data<scalar_t>(), M, C, rpb, sum, sumsq);
      else
        bn_reduce_kernel<scalar_t><<<grid, 256, 0, lc.stream>>>(
            x.data<scalar_t>(), M, C, rpb, sum, sumsq);
    }
    bn_finalize_kernel<<<(int)((C + 255) / 256), 256, 0, lc.stream>>>(
        sum, sumsq, M, C, (float)momentum, (float)eps,
        save_mean.data<float>(), save_istd.data<float>(),
        rmean.data<float>(), rvar.data<float>(), nslices);
    bn_scale_shift_kernel<<<(int)((C + 255) / 256), 256, 0, lc.stream>>>(
        gamma.data<float>(), beta.data<float>(), save_mean.data<float>(),
        save_istd.data<float>(), true, 0.f, C, scale, shift);
    long total = x.numel();
    size_t lds = C <= 4096 ? 2 * C * sizeof(float) : 0;
    bool cb_ok = C % 8 == 0 && sizeof(scalar_t) == 2 && C <= 2048 &&
                 256 % (C / 8) == 0;
    auto launch_apply = [&](auto relu_c, auto res_c) {
      if (cb_ok)
        bn_apply_cb_kernel<scalar_t, decltype(relu_c)::value,
                           decltype(res_c)::value>
            <<<ew_grid_n(total / 8 + 1), 256, 0, lc.stream>>>(
                x.data<scalar_t>(),
                has_res ? residual.data<scalar_t>() : nullptr,
                (scalar_t*)y.ptr, M, C, scale, shift,
                want_mask ? mask.data<unsigned char>() : nullptr);
      else
        bn_apply_kernel<scalar_t, decltype(relu_c)::value,
                        decltype(res_c)::value>
            <<<ew_grid_n(total / 8 + 1), 256, lds, lc.stream>>>(
                x.data<scalar_t>(),
                has_res ? residual.data<scalar_t>() : nullptr,
                (scalar_t*)y.ptr, total, C, scale, shift,
                want_mask ? mask.data<unsigned char>() : nullptr);
    };
    if (fuse_relu && has_res) launch_apply(std::true_type{}, std::true_type{});
    else if (fuse_relu) launch_apply(std::true_type{}, std::false_type{});
    else if (has_res) launch_apply(std::false_type{}, std::true_type{});
    else launch_apply(std::false_type{}, std::false_type{});
  });
  HIP_CHECK_LAST();
}

void bn_fwd_infer_raw(const LaunchCtx& lc, const Arr& x, const Arr& gamma,
                      const Arr& beta, const Arr& rmean, const Arr& rvar,
                      double eps, bool fuse_relu, const Arr& residual,
                      const Arr& y) {
  CHECK_F32(gamma); CHECK_F32(beta); CHECK_F32(rmean); CHECK_F32(rvar);
  long C = x.size(-1);
  float* scale = (float*)lc.workspace((size_t)2 * C * 4);
  float* shift = scale + C;
  bool has_res = residual.defined() && residual.numel() > 0;
  bn_scale_shift_kernel<<<(int)((C + 255) / 256), 256, 0, lc.stream>>>(
      gamma.data<float>(), beta.data<float>(), rmean.data<float>(),
      rvar.data<float>(), false, (float)eps, C, scale, shift);
  DISPATCH_FLOAT_NATIVE(x.dtype, "bn_infer", [&] {
    long total = x.numel();
    size_t lds = C <= 4096 ? 2 * C * sizeof(float) : 0;
    auto launch_apply = [&](auto relu_c, auto res_c) {
      bn_apply_kernel<scalar_t, decltype(relu_c)::value,
                      decltype(res_c)::value>
          <<<ew_grid_n(total / 8 + 1), 256, lds, lc.stream>>>(
              x.data<scalar_t>(),
              has_res ? residual.data<scalar_t>() : nullptr,
              (scalar_t*)y.ptr, total, C, scale, shift, nullptr);
    };
    if (fuse_relu && has_res) launch_apply(std::true_type{}, std::true_type{});
    else if (fuse_relu) launch_apply(std::true_type{}, std::false_type{});
    else if (has_res) launch_apply(std::false_type{}, std::true_type{});
    else launch_apply(std::false_type{}, std::false_type{});
  });
  HIP_CHECK_LAST();
}

void bn_bwd_raw(const LaunchCtx& lc, const Arr& dy, const Arr& x,
                const Arr& gamma, const Arr& save_mean, const Arr& save_inv,
                bool fused_relu, const Arr& y_or_empty, bool has_res,
                const Arr& mask, const Arr& dx, const Arr& dgamma,
                const Arr& dbeta, const Arr& dres) {
  CHECK_F32(gamma); CHECK_F32(save_mean); CHECK_F32(save_inv);
  CHECK_F32(dgamma); CHECK_F32(dbeta);
  long C = x.size(-1), M = x.numel() / C;
  // dbeta = s1 (sum dy), dgamma = s2 (sum dy*xhat): kernels write directly
  float* s1 = dbeta.data<float>();
  float* s2 = dgamma.data<float>();
  MX_HIP_CALL(hipMemsetAsync(s1, 0, (size_t)C * 4, lc.stream));
  MX_HIP_CALL(hipMemsetAsync(s2, 0, (size_t)C * 4, lc.stream));
  const unsigned char* mask_ptr =
      mask.defined() && mask.numel() > 0 ? mask.data<unsigned char>()
                                         : nullptr;
  const void* yp = y_or_empty.defined() ? y_or_empty.ptr : nullptr;
  long rpb;
  dim3 grid = bn_reduce_grid(M, C, &rpb);
  DISPATCH_FLOAT_NATIVE(x.dtype, "bn_bwd", [&] {
    auto launch_red = [&](auto relu_c) {
      if (C % 8 == 0 && sizeof(scalar_t) == 2)
        bn_bwd_reduce_vec_kernel<scalar_t, decltype(relu_c)::value>
            <<<grid, 256, 0, lc.stream>>>(
                dy.data<scalar_t>(), x.data<scalar_t>(),
                (const scalar_t*)yp, M, C, rpb, save_mean.data<float>(),
                save_inv.data<float>(), s1, s2, mask_ptr);
      else
        bn_bwd_reduce_kernel<scalar_t, decltype(relu_c)::value>
            <<<grid, 256, 0, lc.stream>>>(
                dy.data<scalar_t>(), x.data<scalar_t>(),
                (const scalar_t*)yp, M, C, rpb, save_mean.data<float>(),
                save_inv.data<float>(), s1, s2);
    };
    if (fused_relu) launch_red(std::true_type{});
    else launch_red(std::false_type{});
    long total = x.numel();
    size_t lds = C <= 4096 ? 5 * C * sizeof(float) : 0;
    bool cb_ok = C % 8 == 0 && sizeof(scalar_t) == 2 && C <= 2048 &&
                 256 % (C / 8) == 0;
    auto launch_apply = [&](auto relu_c, auto res_c) {
      if (cb_ok)
        bn_bwd_apply_cb_kernel<scalar_t, decltype(relu_c)::value,
                               decltype(res_c)::value>
            <<<ew_grid_n(total / 8 + 1), 256, 0, lc.stream>>>(
                dy.data<scalar_t>(), x.data<scalar_t>(),
                (const scalar_t*)yp, (scalar_t*)dx.ptr,
                has_res ? (scalar_t*)dres.ptr : nullptr, M, C, 1.f / M,
                gamma.data<float>(), save_mean.data<float>(),
                save_inv.data<float>(), s1, s2, mask_ptr);
      else
        bn_bwd_apply_kernel<scalar_t, decltype(relu_c)::value,
                            decltype(res_c)::value>
            <<<ew_grid_n(total / 8 + 1), 256, lds, lc.stream>>>(
                dy.data<scalar_t>(), x.data<scalar_t>(),
                (const scalar_t*)yp, (scalar_t*)dx.ptr,
                has_res ? (scalar_t*)dres.ptr : nullptr, total, C, 1.f / M,
                gamma.data<float>(), save_mean.data<float>(),
                save_inv.data<float>(), s1, s2, mask_ptr);
    };
    if (fused_relu && has_res)
      launch_apply(std::true_type{}, std::true_type{});
    else if (fused_relu) launch_apply(std::true_type{}, std::false_type{});
    else if (has_res) launch_apply(std::false_type{}, std::true_type{});
    else launch_apply(std::false_type{}, std::false_type{});
  });
  HIP_CHECK_LAST();
}

void layernorm_fwd_raw(const LaunchCtx& lc, const Arr& x, const Arr& gamma,
                       const Arr& beta, double eps, const Arr& y,
                       const Arr& mean, const Arr& rstd) {
  CHECK_F32(gamma); CHECK_F32(beta); CHECK_F32(mean); CHECK_F32(rstd);
  long C = x.size(-1), rows = x.numel() / C;
  int grid = (int)std::min<long>(rows, 4096);
  DISPATCH_FLOAT_NATIVE(x.dtype, "ln_fwd", [&] {
    constexpr int VEC = sizeof(scalar_t) == 2 ? 8 : 4;
    ln_fwd_kernel<scalar_t, VEC><<<grid, 256, 0, lc.stream>>>(
        x.data<scalar_t>(), gamma.data<float>(), beta.data<float>(),
        (scalar_t*)y.ptr, mean.data<float>(), rstd.data<float>(), rows, C,
        (float)eps);
  });
  HIP_CHECK_LAST();
}

void layernorm_bwd_raw(const LaunchCtx& lc, const Arr& dy, const Arr& x,
                       const Arr& gamma, const Arr& mean, const Arr& rstd,
                       const Arr& dx, const Arr& dgamma, const Arr& dbeta) {
  CHECK_F32(gamma); CHECK_F32(mean); CHECK_F32(rstd);
  CHECK_F32(dgamma); CHECK_F32(dbeta);
  long C = x.size(-1), rows = x.numel() / C;
  MX_HIP_CALL(hipMemsetAsync(dgamma.ptr, 0, (size_t)C * 4, lc.stream));
  MX_HIP_CALL(hipMemsetAsync(dbeta.ptr, 0, (size_t)C * 4, lc.stream));
  int grid = (int)std::min<long>(rows, 4096);
  long xb = (C + 63) / 64;
  long yb = std::max<long>(1, std::min<long>((rows + 255) / 256,
                                             2048 / std::max<long>(xb, 1)));
  long rpb = (rows + yb - 1) / yb;
  dim3 cgrid((unsigned)xb, (unsigned)yb);
  DISPATCH_FLOAT_NATIVE(x.dtype, "ln_bwd", [&] {
    ln_bwd_dx_kernel<scalar_t><<<grid, 256, 0, lc.stream>>>(
        dy.data<scalar_t>(), x.data<scalar_t>(), gamma.data<float>(),
        mean.data<float>(), rstd.data<float>(), (scalar_t*)dx.ptr, rows, C);
    ln_bwd_dgamma_kernel<scalar_t><<<cgrid, 256, 0, lc.stream>>>(
        dy.data<scalar_t>(), x.data<scalar_t>(), mean.data<float>(),
        rstd.data<float>(), dgamma.data<float>(), dbeta.data<float>(),
        rows, C, rpb);
  });
  HIP_CHECK_LAST();
}

}  // namespace mxcore
