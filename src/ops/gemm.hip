// MFMA GEMM family for gfx950: the FullyConnected / dot / batched-dot
// compute path, plus the tiled transpose used to canonicalize NN/TN
// problems into the NT ("B^T input") form that reads both operands
// contiguously along K.
//
// Reference parity: src/operator/nn/fully_connected.cc:251 (cuBLAS there),
// src/operator/tensor/dot / batch_dot.
//
// MI355X design (guide §5, the verified 128^2 m97 structure):
//   * 128x128 output tile, BK=64, 256 threads = 4 waves in a 2x2 grid,
//     each wave owns a 64x64 sub-tile = 4x4 fragments of
//     v_mfma_f32_16x16x32_{f16,bf16} accumulating fp32 in AGPRs.
//   * global->LDS staging via __builtin_amdgcn_global_load_lds width 16
//     (the single biggest lever: +67% per guide Common-mistake #1).
//   * out-of-bounds rows/K-segments redirect the source address to a
//     per-device zero page (HW requires a valid address; branch-free).
//   * XCD-aware block swizzle (bijective, guide T1) for L2 locality.
//   * fp32 inputs take a classic LDS-tiled VALU kernel (no fp32 MFMA on
//     CDNA4 - guide §3).
#include "native_common.h"

using namespace mxcore;

// ---------------------------------------------------------------------------
// async global->LDS, 16 bytes per lane (dwordx4).  LDS destination must be
// wave-uniform base + lane*16 (guide §5 caveat); generic->AS3 cast via
// uintptr_t is the CK-proven idiom (amd_buffer_addressing.hpp:1055).
// ---------------------------------------------------------------------------
DEV_INLINE void gload_lds16(const void* g, void* lds) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)g,
      (__attribute__((address_space(3))) unsigned int*)(uintptr_t)lds, 16, 0,
      0);
}

// ---------------------------------------------------------------------------
// NT MFMA kernel: C[M,N] = A[M,K] * B[N,K]^T (+bias[N])
// A, B row-major, contiguous along K; K % 8 == 0 (host pads otherwise).
// Batched via blockIdx.y with element strides (0 = broadcast).
// ---------------------------------------------------------------------------
// split-K: when out32 != nullptr, blockIdx.z owns k-tiles
// [z*tiles_per_slice, ...) and accumulates fp32 partials with atomics
// (small-M*N huge-K problems, e.g. the conv-stem weight gradient).
// row strides + 2-level batch (outer x heads) so strided views (e.g.
// attention Q/K/V slices of a fused [B,S,3U] projection) run with no
// contiguous() copy.  Plain GEMM passes {K, K, N, 1, 0, 0, 0}.
struct GemmLd {
  long lda, ldb, ldc;   // row strides (elements); K-contiguous always
  int bh;               // inner batch extent (heads); 1 = plain
  long sAh, sBh, sCh;   // inner-batch element strides
};

template <typename T, int BN = 128, bool WITH_STATS = false>
__global__ __launch_bounds__(256, 2) void gemm_nt_mfma_kernel(
    const T* __restrict__ A, const T* __restrict__ B,
    const float* __restrict__ bias, T* __restrict__ C, long M, long N, long K,
    long strideA, long strideB, long strideC, const T* __restrict__ zpage,
    bool relu, float* __restrict__ out32, int tiles_per_slice, int nbuf,
    const GemmLd ld, float* __restrict__ stats = nullptr) {
  using Frag = typename DTraits<T>::frag8;
  constexpr int BM = 128, BK = 64;
  constexpr int NW = BN / 32;  // n-fragments per wave
  // dynamic LDS: single-buffered when the K loop has one tile (small-K
  // 1x1 convs / FC) so occupancy isn't paying for an unused prefetch buf
  extern __shared__ char smem_raw[];
  T* As = (T*)smem_raw;               // [nbuf][BM*BK]
  T* Bs = As + (long)nbuf * BM * BK;  // [nbuf][BN*BK]

  const long batch = blockIdx.y;
  const long bo = batch / ld.bh, bi = batch % ld.bh;
  A += bo * strideA + bi * ld.sAh;
  B += bo * strideB + bi * ld.sBh;
  C += bo * strideC + bi * ld.sCh;

  const int nTn = (N + BN - 1) / BN;
  const int nwg = ((M + BM - 1) / BM) * nTn;
  const int bid = xcd_swizzle(blockIdx.x, nwg);
  const long m0 = (long)(bid / nTn) * BM;
  const long n0 = (long)(bid % nTn) * BN;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wid = t >> 6;
  const int wr = wid >> 1, wc = wid & 1;  // wave grid 2x2

  // staging map: round r, thread t -> tile row r*32 + t/8, col (t%8)*8
  const int s_row = t >> 3;
  const int s_col = (t & 7) * 8;

  float4_t acc[4][NW] = {};

  const int nk = (int)((K + BK - 1) / BK);

  auto stage = [&](int buf, int kt) {
    const long k0 = (long)kt * BK;
    const long kcol = k0 + s_col;
    const bool ka_ok = kcol + 8 <= K;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long row_a = m0 + r * 32 + s_row;
      const T* ga = (row_a < M && ka_ok) ? A + row_a * ld.lda + kcol : zpage;
      gload_lds16(ga, As + buf * (BM * BK) + (r * 256 + t) * 8);
    }
#pragma unroll
    for (int r = 0; r < BN / 32; ++r) {
      const long row_b = n0 + r * 32 + s_row;
      const T* gb = (row_b < N && ka_ok) ? B + row_b * ld.ldb + kcol : zpage;
      gload_lds16(gb, Bs + buf * (BN * BK) + (r * 256 + t) * 8);
    }
  };

  int kt0 = 0, kt1 = nk;
  if (out32) {
    kt0 = blockIdx.z * tiles_per_slice;
    kt1 = min(nk, kt0 + tiles_per_slice);
    if (kt0 >= kt1) return;
    out32 += batch * strideC;
  }
  stage(0, kt0);
  __syncthreads();

  const int a_row = (lane & 15);
  const int k_off = (lane >> 4) * 8;

  for (int kt = kt0; kt < kt1; ++kt) {
    const int buf = (kt - kt0) & (nbuf - 1);
    if (kt + 1 < kt1) stage(buf ^ 1, kt + 1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      Frag af[4], bf[NW];
#pragma unroll
      for (int m = 0; m < 4; ++m)
        af[m] = *(const Frag*)&As[buf * (BM * BK) +
                                  (wr * 64 + m * 16 + a_row) * BK +
                                  kk * 32 + k_off];
#pragma unroll
      for (int n = 0; n < NW; ++n)
        bf[n] = *(const Frag*)&Bs[buf * (BN * BK) +
                                  (wc * (BN / 2) + n * 16 + a_row) * BK +
                                  kk * 32 + k_off];
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < NW; ++n)
          acc[m][n] = DTraits<T>::mfma_16x16x32(af[m], bf[n], acc[m][n]);
    }
    __syncthreads();
  }

  // epilogue: D frag layout col=lane&15, row=(lane>>4)*4+j (guide §3)
  const int d_col = lane & 15;
  const int d_row = (lane >> 4) * 4;
  // Split-K accumulates fp32 atomics directly; the normal path restages
  // the tile through LDS so global stores are coalesced 16 B (scalar 2 B
  // stores were the bottleneck on write-heavy small-K GEMMs).
  if (out32) {
#pragma unroll
    for (int n = 0; n < NW; ++n) {
      const long col = n0 + wc * (BN / 2) + n * 16 + d_col;
      if (col >= N) continue;
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        const long row_base = m0 + wr * 64 + m * 16 + d_row;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const long row = row_base + j;
          if (row < M) atomicAdd(out32 + row * N + col, acc[m][n][j]);
        }
      }
    }
    return;
  }
  __syncthreads();                       // done with the K-loop buffers
  // optional fused per-channel sum/ssq of the raw fp32 accumulators
  // (BatchNorm's forward reduction -- saves re-reading y; layout
  // [64 slices][2][N], slice = bid&63 to spread the atomic traffic)
  [[maybe_unused]] __shared__ float s_st[WITH_STATS ? 2 : 1]
                                        [WITH_STATS ? BN : 1];
  if (WITH_STATS && stats) {
    for (int i = t; i < 2 * BN; i += 256) s_st[i / BN][i % BN] = 0.f;
    __syncthreads();
#pragma unroll
    for (int n = 0; n < NW; ++n) {
      const int colL = wc * (BN / 2) + n * 16 + d_col;
      if (n0 + colL >= N) continue;
      float ps = 0.f, pq = 0.f;
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        const long row_base = m0 + wr * 64 + m * 16 + d_row;
#pragma unroll
        for (int j = 0; j < 4; ++j)
          if (row_base + j < M) {
            float v = acc[m][n][j];
            ps += v;
            pq += v * v;
          }
      }
      atomicAdd(&s_st[0][colL], ps);
      atomicAdd(&s_st[1][colL], pq);
    }
    __syncthreads();
    float* slice = stats + (long)(bid & 63) * 2 * N;
    for (int i = t; i < BN && n0 + i < N; i += 256) {
      atomicAdd(slice + n0 + i, s_st[0][i]);
      atomicAdd(slice + N + n0 + i, s_st[1][i]);
    }
    __syncthreads();
  }
  T* tile = As;                          // [BM][BN] fp16 staging (fits)
#pragma unroll
  for (int n = 0; n < NW; ++n) {
    const int colL = wc * (BN / 2) + n * 16 + d_col;
    const long col = n0 + colL;
    const float b = (bias && col < N) ? bias[col] : 0.f;
#pragma unroll
    for (int m = 0; m < 4; ++m) {
      const int rowL = wr * 64 + m * 16 + d_row;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float v = acc[m][n][j] + b;
        if (relu) v = fmaxf(v, 0.f);
        tile[(rowL + j) * BN + colL] = (T)v;
      }
    }
  }
  __syncthreads();
  using V8e = T __attribute__((ext_vector_type(8)));
  constexpr int SEGS = BM * BN / 8;      // 8-half segments in the tile
  for (int sidx = t; sidx < SEGS; sidx += 256) {
    const int rowL = sidx / (BN / 8);
    const int colL = (sidx % (BN / 8)) * 8;
    const long row = m0 + rowL;
    const long col = n0 + colL;
    if (row >= M) continue;
    if (col + 8 <= N && (N % 8) == 0 && (ld.ldc % 8) == 0) {  // 16 B fast
      *(V8e*)(C + row * ld.ldc + col) = *(const V8e*)&tile[rowL * BN + colL];
    } else {
      for (int j = 0; j < 8 && col + j < N; ++j)
        C[row * ld.ldc + col + j] = tile[rowL * BN + colL + j];
    }
  }
}


// ---------------------------------------------------------------------------
// 256x256 8-phase NT kernel (guide "The 256^2 8-phase template", m194-m201):
// 512 threads = 8 waves (2M x 4N), BK=64, per-wave output 128x64
// (acc[8][4]), 128 KiB dynamic LDS (double-buffered A/B tiles),
// st_16x32 XOR swizzle (pre-swizzled global source + swizzled ds_read),
// one counted s_waitcnt vmcnt(8) per K-tile (next tile's 8 staging loads
// stay in flight across the boundary), s_setprio(1) around each MFMA
// quadrant.  Routed for large compute-bound shapes; the 128x128 2-phase
// kernel remains the general path.
template <typename T>
__global__ __launch_bounds__(512, 1) void gemm_nt_8ph_kernel(
    const T* __restrict__ A, const T* __restrict__ B, T* __restrict__ C,
    long M, long N, long K, const T* __restrict__ zpage) {
  using Frag = typename DTraits<T>::frag8;
  constexpr int BM = 256, BN = 256, BK = 64;
  extern __shared__ char smem8[];
  T* As = (T*)smem8;                    // [2][256*64]
  T* Bs = As + 2 * BM * BK;

  const int nTn = (N + BN - 1) / BN;
  const int nwg = ((M + BM - 1) / BM) * nTn;
  const int bid = xcd_swizzle(blockIdx.x, nwg);
  const long m0 = (long)(bid / nTn) * BM;
  const long n0 = (long)(bid % nTn) * BN;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wid = t >> 6;
  const int wr = wid >> 2;              // 0..1: row half
  const int wc = wid & 3;               // 0..3: 64-col panel

  // staging map for ONE half-tile (128 rows x 64 halfs = 16 KiB):
  // 2 rounds of 512 threads x 16 B; row = rnd*64 + t/8, col = (t%8)*8,
  // with the st_16x32 source pre-swizzle col ^= ((row>>2)&1)*16
  const int s_row_base = t >> 3;        // 0..63 (+64 second round)
  const int s_col_base = (t & 7) * 8;

  float4_t acc[8][4] = {};
  const int nk = (int)((K + BK - 1) / BK);

  // stage half-tile h (0:A-rows0,1:A-rows1,2:B-rows0,3:B-rows1) of tile kt
  auto stage_half = [&](int buf, int kt, int h) {
    const long k0 = (long)kt * BK;
    const bool is_a = h < 2;
    const int rh = (h & 1) * 128;
    const T* src = is_a ? A : B;
    const long lim = is_a ? M : N;
    const long base0 = is_a ? m0 : n0;
    T* dst = (is_a ? As : Bs) + buf * (BM * BK) + rh * BK;
#pragma unroll
    for (int rnd = 0; rnd < 2; ++rnd) {
      const int row = rnd * 64 + s_row_base;
      const int col = s_col_base ^ (((row >> 2) & 1) << 4);  // pre-swizzle
      const long grow = base0 + rh + row;
      const long kcol = k0 + col;
      const bool ok = grow < lim && kcol + 8 <= K;
      const T* g = ok ? src + grow * K + kcol : zpage;
      // LDS dest is LINEAR (gload_lds constraint); the source column was
      // pre-swizzled above so the swizzled ds_read finds the right data
      gload_lds16(g, dst + (long)(rnd * 512 + t) * 8);
    }
  };

  // prologue: tile 0 fully staged, then drain
  for (int h = 0; h < 4; ++h) stage_half(0, 0, h);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  const int a_row16 = lane & 15;
  const int k_off = (lane >> 4) * 8;

  for (int kt = 0; kt < nk; ++kt) {
    const int buf = kt & 1;
    const T* Asb = As + buf * (BM * BK);
    const T* Bsb = Bs + buf * (BM * BK);
    const bool more = kt + 1 < nk;
    // 4 phases: quadrant q = (qm, qn); stage one next-tile half per phase
#pragma unroll
    for (int q = 0; q < 4; ++q) {
      const int qm = q >> 1, qn = q & 1;
      if (more) stage_half(buf ^ 1, kt + 1, q);
      Frag af[4][2], bf[2][2];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        const int row = wr * 128 + (qm * 4 + m) * 16 + a_row16;
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          const int col = (kk * 32 + k_off) ^ (((row >> 2) & 1) << 4);
          af[m][kk] = *(const Frag*)&Asb[row * BK + col];
        }
      }
#pragma unroll
      for (int n = 0; n < 2; ++n) {
        const int row = wc * 64 + (qn * 2 + n) * 16 + a_row16;
#pragma unroll
        for (int kk = 0; kk < 2; ++kk) {
          const int col = (kk * 32 + k_off) ^ (((row >> 2) & 1) << 4);
          bf[n][kk] = *(const Frag*)&Bsb[row * BK + col];
        }
      }
      __builtin_amdgcn_s_barrier();        // align waves into the MFMA
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 2; ++n)
#pragma unroll
          for (int kk = 0; kk < 2; ++kk)
            acc[qm * 4 + m][qn * 2 + n] = DTraits<T>::mfma_16x16x32(
                af[m][kk], bf[n][kk], acc[qm * 4 + m][qn * 2 + n]);
      __builtin_amdgcn_s_setprio(0);
      __builtin_amdgcn_s_barrier();
    }
    // tile boundary: drain the staging queue before the next tile's
    // ds_reads touch that buffer (gload_lds->ds_read has no automatic
    // waitcnt — the compiler cannot see the dependency).  The counted
    // T4 form needs >1-tile lookahead (a 3-buffer ring); next round.
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
  }

  // epilogue through LDS for coalesced 16 B stores (tile = 128 KiB fits)
  __syncthreads();
  T* tile = As;
  const int d_col = lane & 15;
  const int d_row = (lane >> 4) * 4;
#pragma unroll
  for (int n = 0; n < 4; ++n) {
    const int colL = wc * 64 + n * 16 + d_col;
#pragma unroll
    for (int m = 0; m < 8; ++m) {
      const int rowL = wr * 128 + m * 16 + d_row;
#pragma unroll
      for (int j = 0; j < 4; ++j)
        tile[(rowL + j) * BN + colL] = (T)acc[m][n][j];
    }
  }
  __syncthreads();
  using V8e = T __attribute__((ext_vector_type(8)));
  constexpr int SEGS = BM * BN / 8;
  const bool vec_ok = (N % 8) == 0;
  for (int sidx = t; sidx < SEGS; sidx += 512) {
    const int rowL = sidx / (BN / 8);
    const int colL = (sidx % (BN / 8)) * 8;
    const long row = m0 + rowL;
    const long col = n0 + colL;
    if (row >= M) continue;
    if (vec_ok && col + 8 <= N) {
      *(V8e*)(C + row * N + col) = *(const V8e*)&tile[rowL * BN + colL];
    } else {
      for (int j = 0; j < 8 && col + j < N; ++j)
        C[row * N + col + j] = tile[rowL * BN + colL + j];
    }
  }
}

// ---------------------------------------------------------------------------
// fp32 NT fallback: classic 64x64 LDS tile, 4x4 per thread, VALU FMA
// (CDNA4 has no fp32 MFMA)
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void gemm_nt_f32_kernel(
    const float* __restrict__ A, const float* __restrict__ B,
    const float* __restrict__ bias, float* __restrict__ C, long M, long N,
    long K, long strideA, long strideB, long strideC, bool relu) {
  constexpr int BM = 64, BN = 64, BK = 16;
  __shared__ float As[BM][BK + 1];
  __shared__ float Bs[BN][BK + 1];
  const long batch = blockIdx.y;
  A += batch * strideA;
  B += batch * strideB;
  C += batch * strideC;
  const int nTn = (N + BN - 1) / BN;
  const long m0 = (long)(blockIdx.x / nTn) * BM;
  const long n0 = (long)(blockIdx.x % nTn) * BN;
  const int t = threadIdx.x;
  const int tx = t & 15, ty = t >> 4;  // 16x16 threads, each 4x4 out
  float acc[4][4] = {};
  for (long k0 = 0; k0 < K; k0 += BK) {
    // stage: thread loads 4 elements per operand
    for (int i = t; i < BM * BK; i += 256) {
      int r = i / BK, kc = i % BK;
      long gr = m0 + r, gk = k0 + kc;
      As[r][kc] = (gr < M && gk < K) ? A[gr * K + gk] : 0.f;
      long br = n0 + r;
      Bs[r][kc] = (br < N && gk < K) ? B[br * K + gk] : 0.f;
    }
    __syncthreads();
#pragma unroll
    for (int k = 0; k < BK; ++k) {
      float a[4], b[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) a[i] = As[ty * 4 + i][k];
#pragma unroll
      for (int j = 0; j < 4; ++j) b[j] = Bs[tx * 4 + j][k];
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j) acc[i][j] += a[i] * b[j];
    }
    __syncthreads();
  }
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    long row = m0 + ty * 4 + i;
    if (row >= M) continue;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      long col = n0 + tx * 4 + j;
      if (col < N) {
        float v = acc[i][j] + (bias ? bias[col] : 0.f);
        if (relu) v = fmaxf(v, 0.f);
        C[row * N + col] = v;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// tiled transpose out[C,R] = in[R,C]^T (batched via blockIdx.z)
// 32x32 LDS tile (+1 pad), coalesced both sides
// ---------------------------------------------------------------------------
template <typename T>
__global__ void transpose_kernel(const T* __restrict__ in, T* __restrict__ out,
                                 long R, long C) {
  __shared__ T tile[32][33];
  const long batch = blockIdx.z;
  in += batch * R * C;
  out += batch * R * C;
  long c0 = (long)blockIdx.x * 32;
  long r0 = (long)blockIdx.y * 32;
  int tx = threadIdx.x, ty = threadIdx.y;  // 32 x 8
  for (int i = 0; i < 32; i += 8) {
    long r = r0 + ty + i, c = c0 + tx;
    if (r < R && c < C) tile[ty + i][tx] = in[r * C + c];
  }
  __syncthreads();
  for (int i = 0; i < 32; i += 8) {
    long c = c0 + ty + i, r = r0 + tx;  // transposed coords
    if (c < C && r < R) out[c * R + r] = tile[tx][ty + i];
  }
}

// strided variant: per (outer, inner) batch reads in[r*ldin + c] from
// base + outer*sOut + inner*sIn, writes contiguous out[batch][C][R]
// (attention builds V^T / Q^T / K^T panels straight from the fused
// [B,S,3U] projection with this -- one pass, no permute+contiguous).
template <typename T>
__global__ void transpose_strided_kernel(const T* __restrict__ in,
                                         T* __restrict__ out, long R, long C,
                                         long ldin, int bh, long sOut,
                                         long sIn) {
  __shared__ T tile[32][33];
  const long batch = blockIdx.z;
  in += (batch / bh) * sOut + (batch % bh) * sIn;
  out += batch * R * C;
  long c0 = (long)blockIdx.x * 32;
  long r0 = (long)blockIdx.y * 32;
  int tx = threadIdx.x, ty = threadIdx.y;
  for (int i = 0; i < 32; i += 8) {
    long r = r0 + ty + i, c = c0 + tx;
    if (r < R && c < C) tile[ty + i][tx] = in[r * ldin + c];
  }
  __syncthreads();
  for (int i = 0; i < 32; i += 8) {
    long c = c0 + ty + i, r = r0 + tx;
    if (c < C && r < R) out[c * R + r] = tile[tx][ty + i];
  }
}


// ---------------------------------------------------------------------------
// direct TN GEMM: C[I,J] = A[M,I]^T @ B[M,J]  (+ optional dbias[i] =
// col-sum of A) — the FullyConnected weight-gradient shape.  The NT
// library canonicalized TN via TWO global transposes (reference used
// cublas TN directly); here both operands are consumed straight from
// their k(=m)-major storage with ds_read_b64_tr_b16 hardware transpose
// reads over permuted-row [16][16] subtiles (same trick as
// conv_bwd_w_igemm_tr_kernel), so no transpose passes at all, and the
// bias gradient rides along on the A tiles (saves the colsum pass).
// ---------------------------------------------------------------------------
typedef short trs4g __attribute__((ext_vector_type(4)));
__device__ inline trs4g tr_read16g(const void* p) {
  return __builtin_amdgcn_ds_read_tr16_b64_v4i16(
      (__attribute__((address_space(3))) trs4g*)(uintptr_t)p);
}

template <typename T, int BT>
__global__ __launch_bounds__(256, 2) void gemm_tn_tr_kernel(
    const T* __restrict__ A, const T* __restrict__ B, T* __restrict__ C,
    long M, long I, long J, long m_per_slice, float* __restrict__ out32,
    float* __restrict__ dbias, const T* __restrict__ zpage) {
  using Frag = typename DTraits<T>::frag8;
  using V8 = T __attribute__((ext_vector_type(8)));
  constexpr int BKM = 64;
  constexpr int R = BT / 32;        // 16x16 fragments per wave per dim
  constexpr int SEGS = BT / 8;      // 16 B column segments per row
  constexpr int ROWS_PER_RND = 256 / SEGS;
  __shared__ T AS[2][BKM * BT];
  __shared__ T BS[2][BKM * BT];

  const int nTj = (int)((J + BT - 1) / BT);
  const int bid = blockIdx.x;
  const long i0 = (long)(bid / nTj) * BT;
  const long j0 = (long)(bid % nTj) * BT;

  const long ms0 = (long)blockIdx.y * m_per_slice;
  const long ms1 = min(M, ms0 + m_per_slice);

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wid = t >> 6;
  const int wr = wid >> 1, wc = wid & 1;

  float4_t acc[R][R] = {};
  float bias_acc[R] = {};

  // async staging straight into the tr-read subtile layout: the row
  // permutation is applied at the SOURCE (which global row lane t
  // loads), so the LDS destination stays wave-linear as
  // global_load_lds requires -- no VGPR round-trip, no ds_writes.
  const int prow_c = (t >> 1) & 15;
  const int lsb_c = t & 1;
  const int shi_c = t >> 5;  // subtile within the rnd (8 per rnd)
  auto stage = [&](T (*dst)[BKM * BT], int buf, const T* __restrict__ src,
                   long ld, long W, long col0, long mc) {
#pragma unroll
    for (int rnd = 0; rnd < BKM / ROWS_PER_RND; ++rnd) {
      const int sgl = shi_c + rnd * 8;
      const int seg_hi = sgl % (SEGS / 2);
      const int kk2tt = sgl / (SEGS / 2);
      const int rem = ((prow_c >> 2) << 3) | ((kk2tt & 1) << 2) |
                      (prow_c & 3);
      const int m_l = (kk2tt >> 1) * 32 + rem;
      const int segv = seg_hi * 2 + lsb_c;
      const long m_g = mc + m_l;
      const long c = col0 + segv * 8;
      const T* ga = (m_g < ms1 && c + 8 <= W) ? src + m_g * ld + c : zpage;
      gload_lds16(ga, &dst[buf][(rnd * 256 + t) * 8]);
    }
  };

  stage(AS, 0, A, I, I, i0, ms0);
  stage(BS, 0, B, J, J, j0, ms0);
  __syncthreads();
  int buf = 0;
  for (long mc = ms0; mc < ms1; mc += BKM) {
    if (mc + BKM < ms1) {
      stage(AS, buf ^ 1, A, I, I, i0, mc + BKM);
      stage(BS, buf ^ 1, B, J, J, j0, mc + BKM);
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      Frag af[R], bf[R];
#pragma unroll
      for (int m = 0; m < R; ++m) {
        const int cblk = wr * R + m;
        union { trs4g h[2]; Frag f; } u;
        u.h[0] = tr_read16g(&AS[buf][(((kk * 2 + 0) * (SEGS / 2) + cblk)
                                      << 8) + lane * 4]);
        u.h[1] = tr_read16g(&AS[buf][(((kk * 2 + 1) * (SEGS / 2) + cblk)
                                      << 8) + lane * 4]);
        af[m] = u.f;
        if (dbias != nullptr && wc == 0)
#pragma unroll
          for (int jj = 0; jj < 8; ++jj)
            bias_acc[m] += (float)af[m][jj];
      }
#pragma unroll
      for (int n = 0; n < R; ++n) {
        const int cblk = wc * R + n;
        union { trs4g h[2]; Frag f; } u;
        u.h[0] = tr_read16g(&BS[buf][(((kk * 2 + 0) * (SEGS / 2) + cblk)
                                      << 8) + lane * 4]);
        u.h[1] = tr_read16g(&BS[buf][(((kk * 2 + 1) * (SEGS / 2) + cblk)
                                      << 8) + lane * 4]);
        bf[n] = u.f;
      }
#pragma unroll
      for (int m = 0; m < R; ++m)
#pragma unroll
        for (int n = 0; n < R; ++n)
          acc[m][n] = DTraits<T>::mfma_16x16x32(af[m], bf[n], acc[m][n]);
    }
    __syncthreads();
    buf ^= 1;
  }

  if (dbias != nullptr && j0 == 0) {
    // wc==0 waves each saw columns cblk = wr*R + m, col = lane&15;
    // zero-padded staging keeps OOB contributions at 0
    __shared__ float bsum[BT];
    for (int i = t; i < BT; i += 256) bsum[i] = 0.f;
    __syncthreads();
    if (wc == 0) {
#pragma unroll
      for (int m = 0; m < R; ++m)
        atomicAdd(&bsum[(wr * R + m) * 16 + (lane & 15)], bias_acc[m]);
    }
    __syncthreads();
    for (int i = t; i < BT; i += 256)
      if (i0 + i < I) atomicAdd(dbias + i0 + i, bsum[i]);
  }

  const int d_col = lane & 15;
  const int d_row = (lane >> 4) * 4;
#pragma unroll
  for (int n = 0; n < R; ++n) {
    const long j = j0 + wc * (BT / 2) + n * 16 + d_col;
    if (j >= J) continue;
#pragma unroll
    for (int m = 0; m < R; ++m) {
      const long i_base = i0 + wr * (BT / 2) + m * 16 + d_row;
#pragma unroll
      for (int jj = 0; jj < 4; ++jj) {
        const long i = i_base + jj;
        if (i >= I) continue;
        if (out32) atomicAdd(out32 + i * J + j, acc[m][n][jj]);
        else C[i * J + j] = (T)acc[m][n][jj];
      }
    }
  }
}


// ---------------------------------------------------------------------------
// fused multi-head attention core on the strided NT GEMM
// (reference transformer attention ran separate transpose/reshape +
// batch_dot ops, src/operator/contrib/transformer.cc interleaved path;
// here the Q/K/V panels are consumed as strided views of the fused
// [B, S, 3U] projection -- zero permute/contiguous copies).
// ---------------------------------------------------------------------------

// NT GEMM over a (outerB x H) batch of strided panels.
template <typename T, int BT>
__global__ __launch_bounds__(256, 2) void gemm_tn_tr_batched_kernel(
    const T* __restrict__ A, const T* __restrict__ B, T* __restrict__ C,
    long M, long I, long J, long sAb, long ldb, int bh, long sBo, long sBh,
    long ldc, long sCo, long sCh, const T* __restrict__ zpage) {
  using Frag = typename DTraits<T>::frag8;
  constexpr int BKM = 64;
  constexpr int R = BT / 32;
  constexpr int SEGS = BT / 8;
  __shared__ T AS[2][BKM * BT];
  __shared__ T BS[2][BKM * BT];

  const long z = blockIdx.z;
  A += z * sAb;
  B += (z / bh) * sBo + (z % bh) * sBh;
  C += (z / bh) * sCo + (z % bh) * sCh;

  const int nTj = (int)((J + BT - 1) / BT);
  const long i0 = (long)(blockIdx.x / nTj) * BT;
  const long j0 = (long)(blockIdx.x % nTj) * BT;

  const int t = threadIdx.x;
  const int lane = t & 63;
  const int wid = t >> 6;
  const int wr = wid >> 1, wc = wid & 1;

  float4_t acc[R][R] = {};

  const int prow_c = (t >> 1) & 15;
  const int lsb_c = t & 1;
  const int shi_c = t >> 5;
  auto stage = [&](T (*dst)[BKM * BT], int buf, const T* __restrict__ src,
                   long ld, long W, long col0, long mc) {
#pragma unroll
    for (int rnd = 0; rnd < BKM / (256 / SEGS); ++rnd) {
      const int sgl = shi_c + rnd * 8;
      const int seg_hi = sgl % (SEGS / 2);
      const int kk2tt = sgl / (SEGS / 2);
      const int rem = ((prow_c >> 2) << 3) | ((kk2tt & 1) << 2) |
                      (prow_c & 3);
      const int m_l = (kk2tt >> 1) * 32 + rem;
      const int segv = seg_hi * 2 + lsb_c;
      const long m_g = mc + m_l;
      const long c = col0 + segv * 8;
      const T* ga = (m_g < M && c + 8 <= W) ? src + m_g * ld + c : zpage;
      gload_lds16(ga, &dst[buf][(rnd * 256 + t) * 8]);
    }
  };

  stage(AS, 0, A, I, I, i0, 0);
  stage(BS, 0, B, ldb, J, j0, 0);
  __syncthreads();
  int buf = 0;
  for (long mc = 0; mc < M; mc += BKM) {
    if (mc + BKM < M) {
      stage(AS, buf ^ 1, A, I, I, i0, mc + BKM);
      stage(BS, buf ^ 1, B, ldb, J, j0, mc + BKM);
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      Frag af[R], bf[R];
#pragma unroll
      for (int m = 0; m < R; ++m) {
        const int cblk = wr * R + m;
        union { trs4g h[2]; Frag f; } u;
        u.h[0] = tr_read16g(&AS[buf][(((kk * 2 + 0) * (SEGS / 2) + cblk)
                                      << 8) + lane * 4]);
        u.h[1] = tr_read16g(&AS[buf][(((kk * 2 + 1) * (SEGS / 2) + cblk)
                                      << 8) + lane * 4]);
        af[m] = u.f;
      }
#pragma unroll
      for (int n = 0; n < R; ++n) {
        const int cblk = wc * R + n;
        union { trs4g h[2]; Frag f; } u;
        u.h[0] = tr_read16g(&BS[buf][(((kk * 2 + 0) * (SEGS / 2) + cblk)
                                      << 8) + lane * 4]);
        u.h[1] = tr_read16g(&BS[buf][(((kk * 2 + 1) * (SEGS / 2) + cblk)
                                      << 8) + lane * 4]);
        bf[n] = u.f;
      }
#pragma unroll
      for (int m = 0; m < R; ++m)
#pragma unroll
        for (int n = 0; n < R; ++n)
          acc[m][n] = DTraits<T>::mfma_16x16x32(af[m], bf[n], acc[m][n]);
    }
    __syncthreads();
    buf ^= 1;
  }

  const int d_col = lane & 15;
  const int d_row = (lane >> 4) * 4;
#pragma unroll
  for (int n = 0; n < R; ++n) {
    const long j = j0 + wc * (BT / 2) + n * 16 + d_col;
    if (j >= J) continue;
#pragma unroll
    for (int m = 0; m < R; ++m) {
      const long i_base = i0 + wr * (BT / 2) + m * 16 + d_row;
#pragma unroll
      for (int jj = 0; jj < 4; ++jj) {
        const long i = i_base + jj;
        if (i < I) C[i * ldc + j] = (T)acc[m][n][jj];
      }
    }
  }
}


// ===========================================================================
// host side — native launchers (raw pointers + explicit stream; outputs
// allocated by the caller, scratch from the launch context's arena)
// ===========================================================================
#include <algorithm>
#include <mutex>
#include <unordered_map>

#include "ops_api.h"

namespace mxcore {

const void* zero_page(int dev) {
  static std::mutex mu;
  static std::unordered_map<int, void*> pages;
  std::lock_guard<std::mutex> g(mu);
  auto it = pages.find(dev);
  if (it == pages.end()) {
    MX_HIP_CALL(hipSetDevice(dev));
    void* p = nullptr;
    MX_HIP_CALL(hipMalloc(&p, 1024));
    MX_HIP_CALL(hipMemset(p, 0, 1024));
    it = pages.emplace(dev, p).first;
  }
  return it->second;
}

namespace {

// pad the last (contiguous) dim to a multiple of 8 with zeros
template <typename T>
__global__ void padk_kernel(const T* __restrict__ x, T* __restrict__ y,
                            long rows, long K, long K8) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < rows * K8;
       i += (long)gridDim.x * blockDim.x) {
    long r = i / K8, c = i % K8;
    y[i] = c < K ? x[r * K + c] : (T)0.f;
  }
}

template <typename T>
__global__ void cast_f32_kernel(const T* __restrict__ x, float* __restrict__ y,
                                long n) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    y[i] = (float)x[i];
}

// split-K finish: cast fp32 accumulators (+bias)(relu) into the output
template <typename T>
__global__ void splitk_finish_kernel(const float* __restrict__ acc,
                                     const float* __restrict__ bias,
                                     T* __restrict__ y, long n, long N,
                                     bool relu) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float v = acc[i] + (bias ? bias[i % N] : 0.f);
    if (relu) v = fmaxf(v, 0.f);
    y[i] = (T)v;
  }
}

// K%8 != 0: pad into the arena (returns original when aligned)
Arr pad_k8_ws(const LaunchCtx& lc, const Arr& x) {
  long K = x.size(-1);
  if (K % 8 == 0) return x;
  long K8 = (K + 7) / 8 * 8;
  long rows = x.numel() / K;
  Arr out;
  out.shape = x.shape;
  out.shape.back() = K8;
  out.dtype = x.dtype;
  out.ptr = lc.workspace((size_t)rows * K8 * dtype_size(x.dtype));
  DISPATCH_FLOAT_NATIVE(x.dtype, "pad_k8", [&] {
    padk_kernel<scalar_t><<<ew_grid_n(rows * K8), 256, 0, lc.stream>>>(
        x.data<scalar_t>(), (scalar_t*)out.ptr, rows, K, K8);
  });
  HIP_CHECK_LAST();
  return out;
}

// bias in any float dtype -> fp32 pointer (arena when casting)
const float* bias_f32_ws(const LaunchCtx& lc, const Arr& bias) {
  if (!bias.defined() || bias.numel() == 0) return nullptr;
  if (bias.dtype == kFloat32) return bias.data<float>();
  float* p = (float*)lc.workspace((size_t)bias.numel() * 4);
  DISPATCH_HALF_NATIVE(bias.dtype, "bias_cast", [&] {
    cast_f32_kernel<scalar_t><<<ew_grid_n(bias.numel()), 256, 0,
                                lc.stream>>>(bias.data<scalar_t>(), p,
                                             bias.numel());
  });
  HIP_CHECK_LAST();
  return p;
}

}  // namespace

void transpose2d_raw(const LaunchCtx& lc, const Arr& x, const Arr& out) {
  long B = x.dim() == 3 ? x.size(0) : 1;
  long R = x.size(-2), C = x.size(-1);
  dim3 grid((unsigned)((C + 31) / 32), (unsigned)((R + 31) / 32),
            (unsigned)B);
  DISPATCH_FLOAT_NATIVE(x.dtype, "transpose", [&] {
    transpose_kernel<scalar_t><<<grid, dim3(32, 8), 0, lc.stream>>>(
        x.data<scalar_t>(), (scalar_t*)out.ptr, R, C);
  });
  HIP_CHECK_LAST();
}

void gemm_nt_raw(const LaunchCtx& lc, const Arr& A_, const Arr& B_,
                 const Arr& bias, const Arr& out, bool relu,
                 const Arr& stats) {
  CHECK_SAME_DTYPE(A_, B_);
  bool batched = A_.dim() == 3;
  long nb = batched ? A_.size(0) : 1;
  long M = A_.size(-2), N = B_.size(-2), K = A_.size(-1);
  MX_CHECK(B_.size(-1) == K, "gemm_nt: K mismatch " << K << " vs "
                                                    << B_.size(-1));
  if (out.numel() == 0) return;
  const float* bias_ptr = bias_f32_ws(lc, bias);
  long sA = batched ? M * K : 0, sB = batched ? N * K : 0,
       sC = batched ? M * N : 0;
  if (A_.dtype == kFloat32) {
    long nwg = ((M + 63) / 64) * ((N + 63) / 64);
    dim3 grid((unsigned)nwg, (unsigned)nb);
    gemm_nt_f32_kernel<<<grid, 256, 0, lc.stream>>>(
        A_.data<float>(), B_.data<float>(), bias_ptr, (float*)out.ptr, M, N,
        K, sA, sB, sC, relu);
    HIP_CHECK_LAST();
    return;
  }
  Arr A = pad_k8_ws(lc, A_);
  Arr B = pad_k8_ws(lc, B_);
  K = A.size(-1);
  sA = batched ? M * K : 0;
  sB = batched ? N * K : 0;
  long nwg = ((M + 127) / 128) * ((N + 127) / 128);
  int nk_total = (int)((K + 63) / 64);
  int ksplit = 1, tps = nk_total;
  float* out32 = nullptr;
  // split-K gate: nk>16 only (widening to nk>=8 measured WORSE on the
  // LSTM recurrent GEMM — workspace zero+cast+atomics cost more than
  // the occupancy win)
  static const long kWant = [] {
    const char* e = getenv("MXNET_GEMM_SPLITK_BLOCKS");
    return e ? atol(e) : 1024L;  // swept round-1: 6369 img/s at 1024
  }();
  if (nwg * nb < 512 && nk_total > 16) {
    ksplit = (int)std::min<long>((kWant + nwg * nb - 1) / (nwg * nb),
                                 (nk_total + 15) / 16);
    tps = (nk_total + ksplit - 1) / ksplit;
    ksplit = (nk_total + tps - 1) / tps;
    out32 = (float*)lc.workspace((size_t)out.numel() * 4);
    MX_HIP_CALL(hipMemsetAsync(out32, 0, (size_t)out.numel() * 4,
                               lc.stream));
  }
  int span = ksplit > 1 ? tps : nk_total;
  int nbuf = span > 1 ? 2 : 1;
  // fused BN-forward per-channel sum/ssq slices ([64][2][N] fp32)
  float* stats_ptr = nullptr;
  if (stats.defined() && ksplit == 1 && !batched) {
    stats_ptr = stats.data<float>();
    MX_HIP_CALL(hipMemsetAsync(stats_ptr, 0, (size_t)stats.numel() * 4,
                               lc.stream));
  }
  // 256^2 8-phase path stays opt-in (parity with 128^2, round-1 note)
  static const bool use8ph = [] {
    const char* e = getenv("MXNET_GEMM_8PH");
    return e && e[0] == '1';
  }();
  bool big = use8ph && !batched && ksplit == 1 && M >= 512 && N >= 256 &&
             K >= 256 && !bias_ptr && !relu && !stats_ptr;
  if (big) {
    long nwg8 = ((M + 255) / 256) * ((N + 255) / 256);
    DISPATCH_HALF_NATIVE(A.dtype, "gemm_nt8", [&] {
      static bool attr_set = false;
      if (!attr_set) {
        hipFuncSetAttribute((const void*)&gemm_nt_8ph_kernel<scalar_t>,
                            hipFuncAttributeMaxDynamicSharedMemorySize,
                            131072);
        attr_set = true;
      }
      gemm_nt_8ph_kernel<scalar_t><<<(unsigned)nwg8, 512, 131072,
                                     lc.stream>>>(
          A.data<scalar_t>(), B.data<scalar_t>(), (scalar_t*)out.ptr, M, N,
          K, (const scalar_t*)zero_page(lc.dev));
    });
    HIP_CHECK_LAST();
    return;
  }
  bool narrow = N <= 64;
  if (narrow) nwg = (long)((M + 127) / 128) * ((N + 63) / 64);
  size_t lds_bytes = (size_t)nbuf * (128 + (narrow ? 64 : 128)) * 64 * 2;
  dim3 grid((unsigned)nwg, (unsigned)nb, (unsigned)ksplit);
  DISPATCH_HALF_NATIVE(A.dtype, "gemm_nt", [&] {
    auto launch = [&](auto narrow_c, auto stats_c) {
      constexpr int BNv = decltype(narrow_c)::value ? 64 : 128;
      gemm_nt_mfma_kernel<scalar_t, BNv, decltype(stats_c)::value>
          <<<grid, 256, lds_bytes, lc.stream>>>(
              A.data<scalar_t>(), B.data<scalar_t>(), bias_ptr,
              (scalar_t*)out.ptr, M, N, K, sA, sB, sC,
              (const scalar_t*)zero_page(lc.dev), relu,
              ksplit > 1 ? out32 : nullptr, tps, nbuf,
              GemmLd{K, K, N, 1, 0, 0, 0},
              ksplit > 1 ? nullptr : stats_ptr);
    };
    bool want_stats = stats_ptr != nullptr && ksplit == 1;
    if (narrow && want_stats) launch(std::true_type{}, std::true_type{});
    else if (narrow) launch(std::true_type{}, std::false_type{});
    else if (want_stats) launch(std::false_type{}, std::true_type{});
    else launch(std::false_type{}, std::false_type{});
  });
  HIP_CHECK_LAST();
  if (ksplit > 1) {
    DISPATCH_HALF_NATIVE(out.dtype, "splitk_finish", [&] {
      splitk_finish_kernel<scalar_t><<<ew_grid_n(out.numel()), 256, 0,
                                       lc.stream>>>(
          out32, bias_ptr, (scalar_t*)out.ptr, out.numel(), N, relu);
    });
    HIP_CHECK_LAST();
  }
}

namespace {
// transpose into the arena: returns [.., C, R] scratch
Arr transpose_ws(const LaunchCtx& lc, const Arr& x) {
  Arr out;
  out.dtype = x.dtype;
  out.shape = x.shape;
  std::swap(out.shape[out.dim() - 1], out.shape[out.dim() - 2]);
  out.ptr = lc.workspace((size_t)x.numel() * dtype_size(x.dtype));
  transpose2d_raw(lc, x, out);
  return out;
}
}  // namespace

void gemm_raw(const LaunchCtx& lc, const Arr& a, const Arr& b,
              const Arr& out) {
  gemm_nt_raw(lc, a, transpose_ws(lc, b), Arr(), out, false, Arr());
}

void gemm_nn_raw(const LaunchCtx& lc, const Arr& dy, const Arr& w,
                 const Arr& out) {
  gemm_nt_raw(lc, dy, transpose_ws(lc, w), Arr(), out, false, Arr());
}

void gemm_tn_raw(const LaunchCtx& lc, const Arr& dy, const Arr& x,
                 const Arr& out) {
  gemm_nt_raw(lc, transpose_ws(lc, dy), transpose_ws(lc, x), Arr(), out,
              false, Arr());
}

void bgemm_raw(const LaunchCtx& lc, const Arr& a, const Arr& b,
               const Arr& out) {
  MX_CHECK(a.dim() == 3 && b.dim() == 3, "bgemm expects 3-D");
  gemm_nt_raw(lc, a, transpose_ws(lc, b), Arr(), out, false, Arr());
}

void gemm_nt_8ph_raw(const LaunchCtx& lc, const Arr& A_, const Arr& B_,
                     const Arr& out) {
  Arr A = pad_k8_ws(lc, A_);
  Arr B = pad_k8_ws(lc, B_);
  long M = A.size(-2), N = B.size(-2), K = A.size(-1);
  long nwg8 = ((M + 255) / 256) * ((N + 255) / 256);
  DISPATCH_HALF_NATIVE(A.dtype, "gemm_nt8x", [&] {
    static bool attr_set = false;
    if (!attr_set) {
      hipFuncSetAttribute((const void*)&gemm_nt_8ph_kernel<scalar_t>,
                          hipFuncAttributeMaxDynamicSharedMemorySize,
                          131072);
      attr_set = true;
    }
    gemm_nt_8ph_kernel<scalar_t><<<(unsigned)nwg8, 512, 131072,
                                   lc.stream>>>(
        A.data<scalar_t>(), B.data<scalar_t>(), (scalar_t*)out.ptr, M, N, K,
        (const scalar_t*)zero_page(lc.dev));
  });
  HIP_CHECK_LAST();
}

void gemm_tn_fused_raw(const LaunchCtx& lc, const Arr& A, const Arr& B,
                       const Arr& C, const Arr& dbias) {
  long M = A.size(0), I = A.size(1), J = B.size(1);
  MX_CHECK(B.size(0) == M, "gemm_tn: M mismatch");
  bool want_bias = dbias.defined() && dbias.numel() > 0;
  bool ok = (A.dtype == kFloat16 || A.dtype == kBFloat16) && I % 8 == 0 &&
            J % 8 == 0;
  if (want_bias)
    MX_HIP_CALL(hipMemsetAsync(dbias.ptr, 0, (size_t)dbias.numel() * 4,
                               lc.stream));
  if (!ok) {
    gemm_nt_raw(lc, transpose_ws(lc, A), transpose_ws(lc, B), Arr(), C,
                false, Arr());
    if (want_bias) colsum_raw(lc, A, dbias);
    return;
  }
  int bt = (I >= 128 && J >= 128) ? 128 : 64;
  long nwg = ((I + bt - 1) / bt) * ((J + bt - 1) / bt);
  long yb = std::max<long>(
      1, std::min<long>((M + 63) / 64, 1024 / std::max<long>(nwg, 1)));
  long m_per_slice = ((M + yb - 1) / yb + 63) / 64 * 64;
  yb = (M + m_per_slice - 1) / m_per_slice;
  float* o32 = nullptr;
  if (yb > 1) {
    o32 = (float*)lc.workspace((size_t)I * J * 4);
    MX_HIP_CALL(hipMemsetAsync(o32, 0, (size_t)I * J * 4, lc.stream));
  }
  dim3 grid((unsigned)nwg, (unsigned)yb);
  DISPATCH_HALF_NATIVE(A.dtype, "gemm_tn_tr", [&] {
    auto launch = [&](auto bt_c) {
      gemm_tn_tr_kernel<scalar_t, decltype(bt_c)::value>
          <<<grid, 256, 0, lc.stream>>>(
              A.data<scalar_t>(), B.data<scalar_t>(), (scalar_t*)C.ptr, M, I,
              J, m_per_slice, o32,
              want_bias ? dbias.data<float>() : nullptr,
              (const scalar_t*)zero_page(lc.dev));
    };
    if (bt == 128) launch(std::integral_constant<int, 128>{});
    else launch(std::integral_constant<int, 64>{});
  });
  HIP_CHECK_LAST();
  if (yb > 1) {
    DISPATCH_HALF_NATIVE(C.dtype, "tn_finish", [&] {
      splitk_finish_kernel<scalar_t><<<ew_grid_n(C.numel()), 256, 0,
                                       lc.stream>>>(
          o32, nullptr, (scalar_t*)C.ptr, C.numel(), J, false);
    });
    HIP_CHECK_LAST();
  }
}

// ---------------------------------------------------------------------------
// attention (strided NT/TN over the fused [B,S,3U] projection)
// ---------------------------------------------------------------------------
namespace {

void launch_nt_strided(const LaunchCtx& lc, int dtype, const void* A,
                       const void* B, void* C, long M, long N, long K,
                       long sAb, long sBb, long sCb, const GemmLd& ld,
                       long nb) {
  MX_CHECK(K % 8 == 0 && ld.lda % 8 == 0 && ld.ldb % 8 == 0,
           "attention: K and row strides must be 8-element aligned");
  bool narrow = N <= 64;
  long nwg = narrow ? (long)((M + 127) / 128) * ((N + 63) / 64)
                    : (long)((M + 127) / 128) * ((N + 127) / 128);
  int nk = (int)((K + 63) / 64);
  int nbuf = nk > 1 ? 2 : 1;
  size_t lds_bytes = (size_t)nbuf * (128 + (narrow ? 64 : 128)) * 64 * 2;
  dim3 grid((unsigned)nwg, (unsigned)nb);
  DISPATCH_HALF_NATIVE(dtype, "attn_nt", [&] {
    auto launch = [&](auto narrow_c) {
      constexpr int BNv = decltype(narrow_c)::value ? 64 : 128;
      gemm_nt_mfma_kernel<scalar_t, BNv><<<grid, 256, lds_bytes,
                                           lc.stream>>>(
          (const scalar_t*)A, (const scalar_t*)B, nullptr, (scalar_t*)C, M,
          N, K, sAb, sBb, sCb, (const scalar_t*)zero_page(lc.dev), false,
          nullptr, 0, nbuf, ld);
    };
    if (narrow) launch(std::true_type{});
    else launch(std::false_type{});
  });
  HIP_CHECK_LAST();
}

// strided transpose into the arena: [R, C] panels -> contiguous [nb, C, R]
Arr transpose_strided_ws(const LaunchCtx& lc, int dtype, const void* base,
                         long R, long C, long ldin, int bh, long sOut,
                         long sIn, long nb) {
  Arr out;
  out.dtype = dtype;
  out.shape = {nb, C, R};
  out.ptr = lc.workspace((size_t)nb * C * R * dtype_size(dtype));
  dim3 grid((unsigned)((C + 31) / 32), (unsigned)((R + 31) / 32),
            (unsigned)nb);
  DISPATCH_HALF_NATIVE(dtype, "transpose_strided", [&] {
    transpose_strided_kernel<scalar_t><<<grid, dim3(32, 8), 0, lc.stream>>>(
        (const scalar_t*)base, (scalar_t*)out.ptr, R, C, ldin, bh, sOut,
        sIn);
  });
  HIP_CHECK_LAST();
  return out;
}

void launch_tn_batched(const LaunchCtx& lc, int dtype, const void* A,
                       const void* B, void* C, long M, long I, long J,
                       long sAb, long ldb, int bh, long sBo, long sBh,
                       long ldc, long sCo, long sCh, long nb) {
  int bt = (I >= 128 && J >= 128) ? 128 : 64;
  long nwg = ((I + bt - 1) / bt) * ((J + bt - 1) / bt);
  dim3 grid((unsigned)nwg, 1, (unsigned)nb);
  DISPATCH_HALF_NATIVE(dtype, "tn_batched", [&] {
    auto launch = [&](auto bt_c) {
      gemm_tn_tr_batched_kernel<scalar_t, decltype(bt_c)::value>
          <<<grid, 256, 0, lc.stream>>>(
              (const scalar_t*)A, (const scalar_t*)B, (scalar_t*)C, M, I, J,
              sAb, ldb, bh, sBo, sBh, ldc, sCo, sCh,
              (const scalar_t*)zero_page(lc.dev));
    };
    if (bt == 128) launch(std::integral_constant<int, 128>{});
    else launch(std::integral_constant<int, 64>{});
  });
  HIP_CHECK_LAST();
}

}  // namespace

void attention_fwd_raw(const LaunchCtx& lc, const Arr& qkv, const Arr& mask,
                       double p, int64_t seed, const Arr& dropmask,
                       int H, double temperature, const Arr& out,
                       const Arr& att) {
  long B = qkv.size(0), S = qkv.size(1), U3 = qkv.size(2);
  long U = U3 / 3, D = U / H, BH = B * H;
  MX_CHECK(D % 8 == 0 && S % 8 == 0, "attention: D, S must be %8");
  long es = dtype_size(qkv.dtype);
  const char* qp = (const char*)qkv.ptr;
  // raw scores into the arena, softmax into the saved-att output
  Arr att_raw;
  att_raw.dtype = qkv.dtype;
  att_raw.shape = {BH, S, S};
  att_raw.ptr = lc.workspace((size_t)BH * S * S * es);
  launch_nt_strided(lc, qkv.dtype, qp, qp + U * es, att_raw.ptr, S, S, D,
                    S * U3, S * U3, (long)H * S * S,
                    GemmLd{U3, U3, S, (int)H, D, D, S * S}, BH);
  softmax_fwd_raw(lc, att_raw, mask, false, temperature, att);
  // attention dropout: att stays PRE-dropout (softmax backward needs
  // it); the dropped probs go to the arena and feed the second GEMM
  const void* att2 = att.ptr;
  if (p > 0) {
    MX_CHECK(dropmask.defined(), "attention: dropout needs a mask output");
    Arr dropped;
    dropped.dtype = att.dtype;
    dropped.shape = att.shape;
    dropped.ptr = lc.workspace((size_t)BH * S * S * es);
    dropout_fwd_raw(lc, att, p, seed, dropped, dropmask);
    att2 = dropped.ptr;
  }
  // V^T panels [BH, D, S]
  Arr vt = transpose_strided_ws(lc, qkv.dtype, qp + 2 * U * es, S, D, U3,
                                (int)H, S * U3, D, BH);
  launch_nt_strided(lc, qkv.dtype, att2, vt.ptr, out.ptr, S, D, S,
                    (long)H * S * S, (long)H * D * S, S * U,
                    GemmLd{S, S, U, (int)H, S * S, D * S, D}, BH);
}

void attention_bwd_raw(const LaunchCtx& lc, const Arr& dout, const Arr& qkv,
                       double p, const Arr& dropmask,
                       const Arr& att, int H, double temperature,
                       const Arr& dqkv) {
  long B = qkv.size(0), S = qkv.size(1), U3 = qkv.size(2);
  long U = U3 / 3, D = U / H, BH = B * H;
  long es = dtype_size(qkv.dtype);
  const char* qp = (const char*)qkv.ptr;
  const char* dp = (const char*)dout.ptr;
  Arr datt;
  datt.dtype = qkv.dtype;
  datt.shape = {BH, S, S};
  datt.ptr = lc.workspace((size_t)BH * S * S * es);
  launch_nt_strided(lc, qkv.dtype, dp, qp + 2 * U * es, datt.ptr, S, S, D,
                    S * U, S * U3, (long)H * S * S,
                    GemmLd{U, U3, S, (int)H, D, D, S * S}, BH);
  // datt above is d(att_dropped); the dropout undo (dy*mask/keep) rides
  // the softmax-backward kernel inline — no separate full-tensor pass
  Arr att2 = att;  // operand of the dV GEMM (dropped probs if p>0)
  if (p > 0) {
    // recompute att_dropped = att * mask / (1-p) (same elementwise op
    // as dropout backward) for the dV GEMM
    Arr dropped;
    dropped.dtype = qkv.dtype;
    dropped.shape = {BH, S, S};
    dropped.ptr = lc.workspace((size_t)BH * S * S * es);
    dropout_bwd_raw(lc, att, dropmask, p, dropped);
    att2 = dropped;
  }
  bool inline_undo = p > 0 && S <= 256;  // rowreg softmax-bwd constraint
  if (p > 0 && !inline_undo) {
    Arr tmp;
    tmp.dtype = qkv.dtype;
    tmp.shape = {BH, S, S};
    tmp.ptr = lc.workspace((size_t)BH * S * S * es);
    dropout_bwd_raw(lc, datt, dropmask, p, tmp);
    datt = tmp;
  }
  Arr ds;
  ds.dtype = qkv.dtype;
  ds.shape = {BH, S, S};
  ds.ptr = lc.workspace((size_t)BH * S * S * es);
  softmax_bwd_raw(lc, datt, att, false, temperature, ds,
                  inline_undo ? dropmask : Arr(), inline_undo ? p : 0.0);
  char* dq = (char*)dqkv.ptr;
  Arr kt = transpose_strided_ws(lc, qkv.dtype, qp + U * es, S, D, U3,
                                (int)H, S * U3, D, BH);
  launch_nt_strided(lc, qkv.dtype, ds.ptr, kt.ptr, dq, S, D, S,
                    (long)H * S * S, (long)H * D * S, S * U3,
                    GemmLd{S, S, U3, (int)H, S * S, D * S, D}, BH);
  launch_tn_batched(lc, qkv.dtype, ds.ptr, qp, dq + U * es, S, S, D, S * S,
                    U3, (int)H, S * U3, D, U3, S * U3, D, BH);
  launch_tn_batched(lc, qkv.dtype, att2.ptr, dp, dq + 2 * U * es, S, S, D,
                    S * S, U, (int)H, S * U, D, U3, S * U3, D, BH);
}

}  // namespace mxcore
