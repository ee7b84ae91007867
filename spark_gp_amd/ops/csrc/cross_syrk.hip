// PPA accumulation kernels for MI355X (gfx950, CDNA4):
//
//  * cross_kernel_tile  — K_nm[c, m] = amp * exp(-sum_d s2_d (x - a)^2),
//    the rectangular kernel block (K6 in SURVEY.md §2.4), bf16 or fp32 out.
//  * syrk_bf16          — KK[m, m] += K_nm^T K_nm over a chunk (K12, the
//    rows/sec-dominating GEMM): MFMA bf16 16x16x32 tiles, LDS-staged with
//    both operand tiles stored k-major so every fragment load is one
//    ds_read_b128; split-K over the chunk for occupancy; fp32 atomic
//    accumulation into the output.
//  * colsum_gemv        — Ky[m] += K_nm^T y (fp64 accumulation).
//
// Replaces ProjectedGaussianProcessHelper.scala:20-36's per-expert
// crossKernel + breeze gemm accumulation.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <math.h>

typedef __bf16 bf16;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

// ---------------------------------------------------------------------------
// cross_kernel_tile: 128x128 output tile per 256-thread block, d-chunked LDS
// ---------------------------------------------------------------------------
// out is written bf16 when out_bf16 != 0, else fp32.

#define CK_TILE 128
#define CK_DBLK 32

extern "C" __global__ void __launch_bounds__(256)
cross_kernel_tile_kernel(const float* __restrict__ X,    // [c, d]
                         const float* __restrict__ A,    // [m, d]
                         const float* __restrict__ s2v,  // [d] scale^2
                         const float amp,
                         const int c, const int m, const int d,
                         bf16* __restrict__ out_bf,      // [c, m] or null
                         bf16* __restrict__ out_lo,      // [c, m] or null:
                                                         // residual v - hi
                         float* __restrict__ out_f32) {  // [c, m] or null
  __shared__ float xs[CK_TILE][CK_DBLK + 1];
  __shared__ float as[CK_TILE][CK_DBLK + 1];
  __shared__ float s2s[CK_DBLK];

  const int row0 = blockIdx.x * CK_TILE;
  const int col0 = blockIdx.y * CK_TILE;
  const int tid = threadIdx.x;
  // 16 x 16 thread grid, each thread owns an 8x8 micro-tile
  const int tr = (tid >> 4) * 8;      // row offset within tile
  const int tc = (tid & 15) * 8;      // col offset within tile

  float acc[8][8];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[i][j] = 0.f;

  for (int d0 = 0; d0 < d; d0 += CK_DBLK) {
    const int dl = min(CK_DBLK, d - d0);
    // stage: 256 threads load 128 rows x dl cols of X and A
    for (int f = tid; f < CK_TILE * dl; f += 256) {
      int r = f / dl, q = f - r * dl;
      int gr = row0 + r;
      xs[r][q] = (gr < c) ? X[(size_t)gr * d + d0 + q] : 0.f;
      int gc = col0 + r;
      as[r][q] = (gc < m) ? A[(size_t)gc * d + d0 + q] : 0.f;
    }
    for (int q = tid; q < dl; q += 256) s2s[q] = s2v[d0 + q];
    __syncthreads();

    for (int q = 0; q < dl; ++q) {
      const float w = s2s[q];
      float xf[8], af[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) xf[i] = xs[tr + i][q];
#pragma unroll
      for (int j = 0; j < 8; ++j) af[j] = as[tc + j][q];
#pragma unroll
      for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float t = xf[i] - af[j];
          acc[i][j] += w * t * t;
        }
    }
    __syncthreads();
  }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
    const int gr = row0 + tr + i;
    if (gr >= c) continue;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int gc = col0 + tc + j;
      if (gc >= m) continue;
      const float v = amp * __expf(-acc[i][j]);
      if (out_bf) {
        const bf16 hi = (bf16)v;
        out_bf[(size_t)gr * m + gc] = hi;
        // two-term bf16 split: hi + lo carries ~16 mantissa bits, so the
        // SYRK's input-quantization error drops to fp32 class
        if (out_lo) out_lo[(size_t)gr * m + gc] = (bf16)(v - (float)hi);
      } else {
        out_f32[(size_t)gr * m + gc] = v;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// syrk_bf16: KK[m, m] += Kc^T Kc, Kc [c, m] bf16 row-major
// ---------------------------------------------------------------------------
// 256-thread block = 4 waves in a 2x2 grid; 128x128 output tile per block;
// each wave computes a 64x64 sub-tile as 4x4 mfma_f32_16x16x32_bf16 tiles.
// K-loop: BK=32 rows of Kc staged k-major in LDS: lt[i][kk] / rt[j][kk]
// with row stride 40 bf16 (80 B, 16-B aligned, conflict-free for the
// 16-lane groups of ds_read_b128 — banks hit 20*i mod 64, a full
// permutation).  Fragment loads are single ds_read_b128 each.

#define SY_BK 32
#define SY_STR 136    // row-major [SY_BK][SY_STR] bf16; 272-B rows (16-B
                      // aligned so staging is plain b128 writes)

template <bool HILO>
__global__ void __launch_bounds__(256)
syrk_bf16_kernel(const bf16* __restrict__ Kc,   // [c, m] hi part
                 const bf16* __restrict__ Kl,   // [c, m] lo part (HILO only)
                 const int c, const int m, const int ntile,
                 const int split_k,
                 float* __restrict__ KK) {      // [m, m] accumulated
  // with HILO: KK += hi^T hi + hi^T lo + lo^T hi  (lo^T lo ~ 2^-32, dropped)
  // Tiles stage ROW-major ([k][col]): global loads are coalesced 16-B
  // vectors and LDS writes are conflict-free b128; the k-strided MFMA
  // fragment gathers are scalar b16 reads (2-way worst case).
  __shared__ __align__(16) bf16 lt[SY_BK * SY_STR];
  __shared__ __align__(16) bf16 rt[SY_BK * SY_STR];
  __shared__ __align__(16) bf16 ltl[HILO ? SY_BK * SY_STR : 1];
  __shared__ __align__(16) bf16 rtl[HILO ? SY_BK * SY_STR : 1];

  const int tile = blockIdx.x;
  const int ti = tile / ntile, tj = tile % ntile;
  if (tj < ti) return;                     // upper-triangle tiles only
  const int i0 = ti * 128, j0 = tj * 128;
  const int slice = blockIdx.y;
  const int kblocks = (c + SY_BK - 1) / SY_BK;
  const int per = (kblocks + split_k - 1) / split_k;
  const int kb0 = slice * per;
  const int kb1 = min(kblocks, kb0 + per);
  if (kb0 >= kb1) return;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;              // 0..3
  const int lane = tid & 63;
  const int wr = (wave >> 1) * 64;        // wave row offset in tile
  const int wc = (wave & 1) * 64;         // wave col offset in tile

  f32x4 acc[4][4];
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int b = 0; b < 4; ++b) acc[a][b] = {0.f, 0.f, 0.f, 0.f};

  const int l16 = lane & 15;              // fragment row/col
  const int kgrp = lane >> 4;             // 0..3 -> k-subblock of 8
  const bool mvec = (m % 8 == 0);         // 16-B global loads legal

  // staging assignment: thread -> (k-row, 8-col segment); 512 slots per
  // operand pair = 2 iterations of 256 threads
  for (int kb = kb0; kb < kb1; ++kb) {
    const int krow0 = kb * SY_BK;
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      const int f = tid + half * 256;
      const int kk = f >> 4;              // 0..31
      const int seg = f & 15;
      const int col = seg * 8;
      const int gk = krow0 + kk;
      auto stage = [&](const bf16* mat, int c0, bf16* dst) {
        bf16 vals[8];
        if (gk < c && mvec && c0 + col + 8 <= m) {
          *(uint4*)vals = *(const uint4*)(mat + (size_t)gk * m + c0 + col);
        } else if (gk < c) {
          const bf16* src = mat + (size_t)gk * m;
#pragma unroll
          for (int u = 0; u < 8; ++u) {
            const int gc = c0 + col + u;
            vals[u] = (gc < m) ? src[gc] : (bf16)0.f;
          }
        } else {
#pragma unroll
          for (int u = 0; u < 8; ++u) vals[u] = (bf16)0.f;
        }
        *(uint4*)&dst[kk * SY_STR + col] = *(uint4*)vals;
      };
      stage(Kc, i0, lt);
      stage(Kc, j0, rt);
      if (HILO) {
        stage(Kl, i0, ltl);
        stage(Kl, j0, rtl);
      }
    }
    __syncthreads();

    {
#pragma unroll
      for (int a = 0; a < 4; ++a) {
        const int arow = wr + a * 16 + l16;
        bf16x8 afrag, afl;
#pragma unroll
        for (int t = 0; t < 8; ++t)
          afrag[t] = lt[(kgrp * 8 + t) * SY_STR + arow];
        if (HILO) {
#pragma unroll
          for (int t = 0; t < 8; ++t)
            afl[t] = ltl[(kgrp * 8 + t) * SY_STR + arow];
        }
#pragma unroll
        for (int b = 0; b < 4; ++b) {
          const int bcol = wc + b * 16 + l16;
          bf16x8 bfrag;
#pragma unroll
          for (int t = 0; t < 8; ++t)
            bfrag[t] = rt[(kgrp * 8 + t) * SY_STR + bcol];
          acc[a][b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag, bfrag, acc[a][b], 0, 0, 0);
          if (HILO) {
            bf16x8 bfl;
#pragma unroll
            for (int t = 0; t < 8; ++t)
              bfl[t] = rtl[(kgrp * 8 + t) * SY_STR + bcol];
            acc[a][b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag, bfl, acc[a][b], 0, 0, 0);
            acc[a][b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afl, bfrag, acc[a][b], 0, 0, 0);
          }
        }
      }
    }
    __syncthreads();
  }

  // epilogue: C/D layout for 16x16x32: col = lane&15, row = (lane>>4)*4 + r
  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int b = 0; b < 4; ++b)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int gi = i0 + wr + a * 16 + crow_base + r;
        const int gj = j0 + wc + b * 16 + ccol;
        if (gi < m && gj < m) {
          atomicAdd(&KK[(size_t)gi * m + gj], acc[a][b][r]);
          if (ti != tj)          // mirror off-diagonal tiles
            atomicAdd(&KK[(size_t)gj * m + gi], acc[a][b][r]);
        }
      }
}

// Wait: for diagonal tiles (ti == tj) the full 128x128 tile is computed and
// written once (it contains both halves), so no mirror is needed there; the
// mirror above fills the strict lower triangle from the strict upper tiles.
// Elements of diagonal tiles below the diagonal are produced by the same
// block (a/b loops cover the whole tile), so KK ends up fully populated.

// ---------------------------------------------------------------------------
// colsum_gemv: Ky[m] += Kc^T y (fp64 accumulate per block, one atomic per
// column)
// ---------------------------------------------------------------------------
extern "C" __global__ void __launch_bounds__(256)
colsum_gemv_kernel(const bf16* __restrict__ Kc,   // [c, m]
                   const float* __restrict__ y,   // [c]
                   const int c, const int m,
                   const int rows_per_block,
                   double* __restrict__ Ky) {     // [m] fp64
  const int col = blockIdx.x * 64 + (threadIdx.x & 63);
  const int rseg = blockIdx.y;
  const int warp = threadIdx.x >> 6;        // 4 waves split the row range
  if (col >= m) return;
  const int r0 = rseg * rows_per_block + warp;
  const int r1 = min(c, rseg * rows_per_block + rows_per_block);
  double acc = 0.0;
  for (int r = r0; r < r1; r += 4) {
    acc += (double)(float)Kc[(size_t)r * m + col] * (double)y[r];
  }
  // one fp64 atomic per (column, wave, rseg)
  atomicAdd(&Ky[col], acc);
}

// ---------------------------------------------------------------------------
// launchers
// ---------------------------------------------------------------------------
extern "C" hipError_t launch_cross_kernel_tile(
    const float* X, const float* A, const float* s2v, float amp,
    int c, int m, int d, void* out, void* out_lo, int out_is_bf16,
    hipStream_t stream) {
  dim3 grid((c + CK_TILE - 1) / CK_TILE, (m + CK_TILE - 1) / CK_TILE);
  hipLaunchKernelGGL(cross_kernel_tile_kernel, grid, dim3(256), 0, stream,
                     X, A, s2v, amp, c, m, d,
                     out_is_bf16 ? (bf16*)out : nullptr,
                     out_is_bf16 ? (bf16*)out_lo : nullptr,
                     out_is_bf16 ? nullptr : (float*)out);
  return hipGetLastError();
}

extern "C" hipError_t launch_syrk_bf16(const void* Kc, const void* Kl,
                                       int c, int m, int split_k, float* KK,
                                       hipStream_t stream) {
  const int ntile = (m + 127) / 128;
  dim3 grid(ntile * ntile, split_k);
  if (Kl) {
    hipLaunchKernelGGL((syrk_bf16_kernel<true>), grid, dim3(256), 0, stream,
                       (const bf16*)Kc, (const bf16*)Kl, c, m, ntile,
                       split_k, KK);
  } else {
    hipLaunchKernelGGL((syrk_bf16_kernel<false>), grid, dim3(256), 0, stream,
                       (const bf16*)Kc, nullptr, c, m, ntile, split_k, KK);
  }
  return hipGetLastError();
}

extern "C" hipError_t launch_colsum_gemv(const void* Kc, const float* y,
                                         int c, int m, double* Ky,
                                         hipStream_t stream) {
  const int rows_per_block = 4096;
  dim3 grid((m + 63) / 64, (c + rows_per_block - 1) / rows_per_block);
  hipLaunchKernelGGL(colsum_gemv_kernel, grid, dim3(256), 0, stream,
                     (const bf16*)Kc, y, c, m, rows_per_block, Ky);
  return hipGetLastError();
}
