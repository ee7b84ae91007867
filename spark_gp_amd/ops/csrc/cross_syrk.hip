// PPA accumulation kernels for MI355X (gfx950, CDNA4):
//
//  * cross_kernel_tile  — K_nm[c, m] = amp * exp(-sum_d s2_d (x - a)^2),
//    the rectangular kernel block (K6 in SURVEY.md §2.4), bf16 or fp32 out.
//  * syrk_bf16          — KK[m, m] += K_nm^T K_nm over a chunk (K12, the
//    rows/sec-dominating GEMM): MFMA bf16 16x16x32 tiles, LDS-staged with
//    both operand tiles stored k-major so every fragment load is one
//    ds_read_b128; split-K over the chunk for occupancy; fp32 atomic
//    accumulation into the output.
//  * cross_mfma         — the round-2 PPA fast path: sqdist via
//    ||x'||^2 + ||a'||^2 - 2 x'.a' on f32 matrix cores, transposed hi/lo
//    outputs only, Ky += K^T y fused into the epilogue.
//  * syrk_bf16_sync     — k-synchronized persistent SYRK for large m
//    (one block owns one tile; bounded best-effort pacing).
//  * colsum_gemv        — Ky[m] += K_nm^T y (fp64 accumulation;
//    superseded on the PPA path by the fused epilogues, kept as the
//    standalone op).
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
                         bf16* __restrict__ out_bfT,     // [m, c] or null
                         bf16* __restrict__ out_loT,     // [m, c] or null
                         float* __restrict__ out_f32,    // [c, m] or null
                         const float* __restrict__ yv,   // [c] or null
                         double* __restrict__ Ky) {      // [m] (with yv)
  __shared__ float xs[CK_TILE][CK_DBLK + 1];
  __shared__ float as[CK_TILE][CK_DBLK + 1];
  __shared__ float s2s[CK_DBLK];
  __shared__ double kyred[16][16 + 1];    // fused colsum partials

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

  // finalize the 8x8 micro-tile once; hi/lo split: hi + lo carries ~16
  // mantissa bits, so the SYRK's input-quantization error drops to fp32 class
  bf16 hv[8][8], lv[8][8];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float v = amp * __expf(-acc[i][j]);
      acc[i][j] = v;
      hv[i][j] = (bf16)v;
      lv[i][j] = (bf16)(v - (float)hv[i][j]);
    }

#pragma unroll
  for (int i = 0; i < 8; ++i) {
    const int gr = row0 + tr + i;
    if (gr >= c) continue;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int gc = col0 + tc + j;
      if (gc >= m) continue;
      if (out_bf) {
        out_bf[(size_t)gr * m + gc] = hv[i][j];
        if (out_lo) out_lo[(size_t)gr * m + gc] = lv[i][j];
      } else if (out_f32) {
        out_f32[(size_t)gr * m + gc] = acc[i][j];
      }
    }
  }

  if (out_bfT) {
    // transposed copies [m, c] so the SYRK can stage k-contiguous: per
    // output column j, the 8 row values are contiguous -> one b128 store
    const int gr0 = row0 + tr;
    const bool vec = (c % 8 == 0) && (gr0 + 8 <= c);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int gc = col0 + tc + j;
      if (gc >= m) continue;
      bf16 th[8], tl[8];
#pragma unroll
      for (int i = 0; i < 8; ++i) { th[i] = hv[i][j]; tl[i] = lv[i][j]; }
      if (vec) {
        *(uint4*)&out_bfT[(size_t)gc * c + gr0] = *(uint4*)th;
        if (out_loT) *(uint4*)&out_loT[(size_t)gc * c + gr0] = *(uint4*)tl;
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          if (gr0 + i >= c) break;
          out_bfT[(size_t)gc * c + gr0 + i] = th[i];
          if (out_loT) out_loT[(size_t)gc * c + gr0 + i] = tl[i];
        }
      }
    }
  }

  if (yv) {
    // fused colsum (K12's K_mn^T y): accumulate this tile's column sums
    // of K * y straight from the fp32 register values — the separate
    // colsum pass used to RE-READ the whole [c, m] block from HBM (and
    // was the only consumer of the non-transposed copies)
    float s[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float sj = 0.f;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int gr = row0 + tr + i;
        const float yw = (gr < c) ? yv[gr] : 0.f;
        sj += acc[i][j] * yw;
      }
      s[j] = sj;
    }
    const int thr = threadIdx.x >> 4;       // tr / 8 in thread units
    const int thc = threadIdx.x & 15;       // tc / 8 in thread units
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __syncthreads();
      kyred[thr][thc] = (double)s[j];
      __syncthreads();
      if (thr == 0) {
        double t = 0.0;
#pragma unroll
        for (int r = 0; r < 16; ++r) t += kyred[r][thc];
        const int gc = col0 + thc * 8 + j;
        if (gc < m) atomicAdd(&Ky[gc], t);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// cross_mfma: the K1 MFMA plan — sqdist via ||x||^2 + ||a||^2 - 2 x.a on
// f32 matrix cores (v_mfma_f32_16x16x4_f32: exact fp32 at 157 TF, vs the
// elementwise kernel's ~3 VALU issues per (element, dim)).  PPA mode only:
// inputs arrive PRE-SCALED (x' = x * s per dim) with per-row squared norms;
// outputs are the transposed hi/lo bf16 tiles + the fused Ky += K^T y.
// ---------------------------------------------------------------------------

#define CM_TILE 128
#define CM_DBLK 32
#define CM_STR 36           // f32 LDS row stride ([i][k] layout)

typedef __attribute__((ext_vector_type(4))) float f32x4_;

extern "C" __global__ void __launch_bounds__(256)
cross_mfma_kernel(const float* __restrict__ Xs,   // [c, d] pre-scaled
                  const float* __restrict__ As,   // [m, d] pre-scaled
                  const float* __restrict__ nx,   // [c] ||x'||^2
                  const float* __restrict__ na,   // [m] ||a'||^2
                  const float amp,
                  const int c, const int m, const int d,
                  bf16* __restrict__ out_bfT,     // [m, c]
                  bf16* __restrict__ out_loT,     // [m, c]
                  const float* __restrict__ yv,   // [c]
                  double* __restrict__ Ky) {      // [m]
  __shared__ float xt[CM_TILE * CM_STR];
  __shared__ float at[CM_TILE * CM_STR];
  __shared__ float kyd[CM_TILE];

  const int row0 = blockIdx.x * CM_TILE;          // x rows
  const int col0 = blockIdx.y * CM_TILE;          // a rows (K columns)
  const int tid = threadIdx.x;
  const int wave = tid >> 6;                      // 4 waves, 2x2 grid
  const int lane = tid & 63;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 64;
  const int l16 = lane & 15, kg = lane >> 4;

  for (int f = tid; f < CM_TILE; f += 256) kyd[f] = 0.f;

  // 4x4 tiles of 16x16 per wave = a 64x64 sub-tile
  f32x4_ acc[4][4];
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int b = 0; b < 4; ++b) acc[a][b] = {0.f, 0.f, 0.f, 0.f};

  for (int d0 = 0; d0 < d; d0 += CM_DBLK) {
    const int dl = min(CM_DBLK, d - d0);
    for (int f = tid; f < CM_TILE * CM_DBLK; f += 256) {
      const int r = f >> 5, q = f & 31;           // [i][k], coalesced on k
      const int gr = row0 + r, gc = col0 + r;
      xt[r * CM_STR + q] = (gr < c && q < dl)
                               ? Xs[(size_t)gr * d + d0 + q] : 0.f;
      at[r * CM_STR + q] = (gc < m && q < dl)
                               ? As[(size_t)gc * d + d0 + q] : 0.f;
    }
    __syncthreads();
#pragma unroll
    for (int kc = 0; kc < CM_DBLK / 4; ++kc) {
      const int fk = kg + kc * 4;
      float xa[4], ab[4];
#pragma unroll
      for (int a = 0; a < 4; ++a)
        xa[a] = xt[(wr + a * 16 + l16) * CM_STR + fk];
#pragma unroll
      for (int b = 0; b < 4; ++b)
        ab[b] = at[(wc + b * 16 + l16) * CM_STR + fk];
#pragma unroll
      for (int a = 0; a < 4; ++a)
#pragma unroll
        for (int b = 0; b < 4; ++b)
          acc[a][b] = __builtin_amdgcn_mfma_f32_16x16x4f32(
              xa[a], ab[b], acc[a][b], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: q = nx + na - 2 dot; v = amp exp(-q); hi/lo bf16; stores
  // transposed (4 consecutive x-rows per fragment reg -> one 8-B store);
  // fused Ky partials through LDS.
  // f32 16x16x4 D map: col = lane&15, row = (lane>>4)*4 + reg (the
  // dtype-independent standard map — unlike the f64 form, which was
  // measured transposed in reg; verified by the parity unit test).
  const int crow = (lane >> 4) * 4;
  const int ccol = l16;
#pragma unroll
  for (int a = 0; a < 4; ++a) {
    const int gr0 = row0 + wr + a * 16 + crow;
#pragma unroll
    for (int b = 0; b < 4; ++b) {
      const int gc = col0 + wc + b * 16 + ccol;
      if (gc >= m) continue;
      const float nav = na[gc];
      bf16 th[4], tl[4];
      float kysum = 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int gr = gr0 + r;
        float v = 0.f;
        if (gr < c) {
          const float q = nx[gr] + nav - 2.0f * acc[a][b][r];
          v = amp * __expf(-(q > 0.f ? q : 0.f));
          kysum += v * yv[gr];
        }
        const bf16 h = (bf16)v;
        th[r] = h;
        tl[r] = (bf16)(v - (float)h);
      }
      if ((c % 4 == 0) && gr0 + 4 <= c) {   // 8-B alignment needs 4 | c
        *(uint2*)&out_bfT[(size_t)gc * c + gr0] = *(uint2*)th;
        *(uint2*)&out_loT[(size_t)gc * c + gr0] = *(uint2*)tl;
      } else {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          if (gr0 + r < c) {
            out_bfT[(size_t)gc * c + gr0 + r] = th[r];
            out_loT[(size_t)gc * c + gr0 + r] = tl[r];
          }
        }
      }
      atomicAdd(&kyd[wc + b * 16 + ccol], kysum);
    }
  }
  __syncthreads();
  for (int f = tid; f < CM_TILE; f += 256) {
    const int gc = col0 + f;
    if (gc < m && kyd[f] != 0.f) atomicAdd(&Ky[gc], (double)kyd[f]);
  }
}

// ---------------------------------------------------------------------------
// syrk_bf16: KK[m, m] += K^T K from the TRANSPOSED chunk KcT [m, c]
// ---------------------------------------------------------------------------
// 512-thread block = 8 waves in a 4x2 grid; 256x256 output tile per block
// (the largest tile the LDS+register budget allows: per-CU VMEM return
// throughput (~10 B/cyc/CU HBM-bound) is the wall, and bytes/flop scales
// as (M+N)/(M*N), so 256^2 halves traffic vs 128^2); each wave computes a
// 64x128 sub-tile as 4x8 mfma_f32_16x16x32_bf16 tiles.  K-loop: BK=32
// rows staged k-major in LDS; split-K over the chunk rows for occupancy.

#define SY_BK 32
#define SY_CT 256     // output tile edge (per block)
#define SY_WG 512
#define SY_STR 40     // k-major [SY_CT][SY_STR] bf16: 32 k + 8 pad = 80-B
                      // rows.  80 B x 16 consecutive lanes covers all 64
                      // LDS banks exactly (20c mod 64 is a permutation of
                      // 4-bank blocks), so staging b128 writes AND fragment
                      // b128 reads are both conflict-free.  4 operands x
                      // 20 KB = 80 KB -> one 8-wave block/CU (2 waves/SIMD).

template <bool HILO>
__global__ void __launch_bounds__(SY_WG, 2)  // force 2 waves/SIMD occupancy
syrk_bf16_kernel(const bf16* __restrict__ KcT,  // [m, cpitch] hi^T
                 const bf16* __restrict__ KlT,  // [m, cpitch] lo^T (HILO)
                 const int c, const int m, const int cpitch, const int ntile,
                 const int split_k,
                 float* __restrict__ KK) {      // [m, m] accumulated
  // with HILO: KK += hi^T hi + hi^T lo + lo^T hi  (lo^T lo ~ 2^-32, dropped)
  // Operands arrive TRANSPOSED ([m, c], written for free by the cross
  // kernel's register tile), so k runs along the contiguous axis: staging
  // is one uint4 global load (8 lanes = one full 128-B line) + one
  // conflict-free b128 LDS write per thread, and every MFMA fragment is
  // one b128 LDS read.  The old [c, m]-operand versions were LDS- or
  // VMEM-instruction bound (8 scalar ops per fragment either way).
  __shared__ __align__(16) bf16 lt[SY_CT * SY_STR];
  __shared__ __align__(16) bf16 rt[SY_CT * SY_STR];
  __shared__ __align__(16) bf16 ltl[HILO ? SY_CT * SY_STR : 1];
  __shared__ __align__(16) bf16 rtl[HILO ? SY_CT * SY_STR : 1];

  const int tile = blockIdx.x;
  const int ti = tile / ntile, tj = tile % ntile;
  if (tj < ti) return;                     // upper-triangle tiles only
  const int i0 = ti * SY_CT, j0 = tj * SY_CT;
  const int slice = blockIdx.y;
  const int kblocks = (c + SY_BK - 1) / SY_BK;
  const int per = (kblocks + split_k - 1) / split_k;
  const int kb0 = slice * per;
  const int kb1 = min(kblocks, kb0 + per);
  if (kb0 >= kb1) return;

  const int tid = threadIdx.x;
  const int wave = tid >> 6;              // 0..7
  const int lane = tid & 63;
  const int wr = (wave >> 1) * 64;        // wave row offset in tile
  const int wc = (wave & 1) * 128;        // wave col offset in tile

  f32x4 acc[4][8];
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int b = 0; b < 8; ++b) acc[a][b] = {0.f, 0.f, 0.f, 0.f};

  const int l16 = lane & 15;              // fragment row/col
  const int kgrp = lane >> 4;             // 0..3 -> k-subblock of 8
  const bool vec_ok = (cpitch % 8 == 0);  // 16-B aligned row starts

  // staging: thread -> (column, 8-deep k segment); 256 cols x 4 segs =
  // 1024 slots per operand = 2 iterations of 512 threads.  Adjacent 4
  // lanes take the 4 k-segments of one column (64-B half-lines; the other
  // half of each 128-B line is the next k-block's data and hits L2).
  //
  // Single-buffer T14 register pipeline: the loads for block kb+1 are
  // issued before the MFMA phase of block kb, which covers their latency;
  // the per-CU VMEM return rate (~10 B/cyc HBM-bound) is the wall, which
  // is why the tile is as large as LDS allows (bytes/flop ~ (M+N)/(M*N)).
  auto load4 = [&](const bf16* mat, int c0, int col, int gk0) -> uint4 {
    uint4 v;
    bf16* vals = (bf16*)&v;
    const int gc = c0 + col;
    if (gc < m && vec_ok && gk0 + 8 <= c) {
      v = *(const uint4*)(mat + (size_t)gc * cpitch + gk0);
    } else if (gc < m) {
#pragma unroll
      for (int u = 0; u < 8; ++u)
        vals[u] = (gk0 + u < c) ? mat[(size_t)gc * cpitch + gk0 + u]
                                : (bf16)0.f;
    } else {
#pragma unroll
      for (int u = 0; u < 8; ++u) vals[u] = (bf16)0.f;
    }
    return v;
  };

  uint4 sv[2][4];                         // [iteration][operand] prefetch
  auto load_all = [&](int kb) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int f = tid + it * SY_WG;     // 0..1023
      const int col = f >> 2;
      const int gk0 = kb * SY_BK + (f & 3) * 8;
      sv[it][0] = load4(KcT, i0, col, gk0);
      sv[it][1] = load4(KcT, j0, col, gk0);
      if (HILO) {
        sv[it][2] = load4(KlT, i0, col, gk0);
        sv[it][3] = load4(KlT, j0, col, gk0);
      }
    }
  };
  auto write_all = [&]() {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int f = tid + it * SY_WG;
      const int o = (f >> 2) * SY_STR + (f & 3) * 8;
      *(uint4*)&lt[o] = sv[it][0];
      *(uint4*)&rt[o] = sv[it][1];
      if (HILO) {
        *(uint4*)&ltl[o] = sv[it][2];
        *(uint4*)&rtl[o] = sv[it][3];
      }
    }
  };
  auto mfma_pass = [&](const bf16* at, const bf16* bt) {
    // one product pass in two b-halves of 16 INDEPENDENT MFMAs (no
    // accumulator chaining between consecutive issues); fb covers 4 of
    // the 8 column tiles at a time to stay inside the register budget
    bf16x8 fa[4], fb[4];
    const int ko = kgrp * 8;
#pragma unroll
    for (int a = 0; a < 4; ++a)
      fa[a] = *(const bf16x8*)&at[(wr + a * 16 + l16) * SY_STR + ko];
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int b = 0; b < 4; ++b)
        fb[b] = *(const bf16x8*)&bt[(wc + (h * 4 + b) * 16 + l16) * SY_STR
                                    + ko];
#pragma unroll
      for (int a = 0; a < 4; ++a)
#pragma unroll
        for (int b = 0; b < 4; ++b)
          acc[a][h * 4 + b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              fa[a], fb[b], acc[a][h * 4 + b], 0, 0, 0);
    }
  };

  load_all(kb0);
  for (int kb = kb0; kb < kb1; ++kb) {
    write_all();                          // LDS free: readers passed barrier
    __syncthreads();
    if (kb + 1 < kb1) load_all(kb + 1);   // covered by the MFMA phase
    mfma_pass(lt, rt);                    // hi * hi
    if (HILO) {
      mfma_pass(lt, rtl);                 // hi * lo
      mfma_pass(ltl, rt);                 // lo * hi
    }
    __syncthreads();
  }

  // epilogue: C/D layout for 16x16x32: col = lane&15, row = (lane>>4)*4 + r
  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int b = 0; b < 8; ++b)
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

// Diagonal tiles (ti == tj) compute the full SY_CT x SY_CT tile and write it
// (the a/b loops cover both halves), so no mirror is needed there; the
// mirror above fills the strict lower triangle from the strict upper tiles,
// leaving KK fully populated.

// ---------------------------------------------------------------------------
// syrk_bf16_sync: k-SYNCHRONIZED variant for large m (round-2, TODO item 1)
// ---------------------------------------------------------------------------
// At m >= 4096 the plain kernel's co-scheduled tiles drift apart in k, so
// no XCD's L2 ever holds the column windows its tiles are reading (round-1
// TCC analysis: 38% hit, ~6x compulsory HBM traffic).  This variant keeps
// every live tile on the SAME k-phase: one block owns ONE output tile for
// the whole launch (accumulators never leave AGPRs), the host passes a
// tile list clustered so each XCD's blocks share row/column windows, and
// blocks pace each other between k-phases through a per-phase arrival
// counter.  The pacing spin is BOUNDED and best-effort: there is no data
// dependency between blocks, so a timeout only loses locality, never
// correctness — no grid-residency assumption, no deadlock risk.

template <bool HILO>
__global__ void __launch_bounds__(SY_WG, 2)
syrk_bf16_sync_kernel(const bf16* __restrict__ KcT,   // [m, cpitch] hi^T
                      const bf16* __restrict__ KlT,   // [m, cpitch] lo^T
                      const int c, const int m, const int cpitch,
                      const int* __restrict__ tiles,  // [nb, 2]; -1 = idle
                      const int kpb,                  // k-blocks per phase
                      int* __restrict__ phase_ctr,    // [nphases], zeroed
                      const int nactive,
                      float* __restrict__ KK) {
  const int ti = tiles[2 * blockIdx.x];
  const int tj = tiles[2 * blockIdx.x + 1];
  if (ti < 0) return;
  const int i0 = ti * SY_CT, j0 = tj * SY_CT;
  const int kblocks = (c + SY_BK - 1) / SY_BK;

  __shared__ __align__(16) bf16 lt[SY_CT * SY_STR];
  __shared__ __align__(16) bf16 rt[SY_CT * SY_STR];
  __shared__ __align__(16) bf16 ltl[HILO ? SY_CT * SY_STR : 1];
  __shared__ __align__(16) bf16 rtl[HILO ? SY_CT * SY_STR : 1];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int wr = (wave >> 1) * 64;
  const int wc = (wave & 1) * 128;

  f32x4 acc[4][8];
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int b = 0; b < 8; ++b) acc[a][b] = {0.f, 0.f, 0.f, 0.f};

  const int l16 = lane & 15;
  const int kgrp = lane >> 4;
  const bool vec_ok = (cpitch % 8 == 0);

  auto load4 = [&](const bf16* mat, int c0, int col, int gk0) -> uint4 {
    uint4 v;
    bf16* vals = (bf16*)&v;
    const int gc = c0 + col;
    if (gc < m && vec_ok && gk0 + 8 <= c) {
      v = *(const uint4*)(mat + (size_t)gc * cpitch + gk0);
    } else if (gc < m) {
#pragma unroll
      for (int u = 0; u < 8; ++u)
        vals[u] = (gk0 + u < c) ? mat[(size_t)gc * cpitch + gk0 + u]
                                : (bf16)0.f;
    } else {
#pragma unroll
      for (int u = 0; u < 8; ++u) vals[u] = (bf16)0.f;
    }
    return v;
  };

  uint4 sv[2][4];
  auto load_all = [&](int kb) {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int f = tid + it * SY_WG;
      const int col = f >> 2;
      const int gk0 = kb * SY_BK + (f & 3) * 8;
      sv[it][0] = load4(KcT, i0, col, gk0);
      sv[it][1] = load4(KcT, j0, col, gk0);
      if (HILO) {
        sv[it][2] = load4(KlT, i0, col, gk0);
        sv[it][3] = load4(KlT, j0, col, gk0);
      }
    }
  };
  auto write_all = [&]() {
#pragma unroll
    for (int it = 0; it < 2; ++it) {
      const int f = tid + it * SY_WG;
      const int o = (f >> 2) * SY_STR + (f & 3) * 8;
      *(uint4*)&lt[o] = sv[it][0];
      *(uint4*)&rt[o] = sv[it][1];
      if (HILO) {
        *(uint4*)&ltl[o] = sv[it][2];
        *(uint4*)&rtl[o] = sv[it][3];
      }
    }
  };
  auto mfma_pass = [&](const bf16* at, const bf16* bt) {
    bf16x8 fa[4], fb[4];
    const int ko = kgrp * 8;
#pragma unroll
    for (int a = 0; a < 4; ++a)
      fa[a] = *(const bf16x8*)&at[(wr + a * 16 + l16) * SY_STR + ko];
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int b = 0; b < 4; ++b)
        fb[b] = *(const bf16x8*)&bt[(wc + (h * 4 + b) * 16 + l16) * SY_STR
                                    + ko];
#pragma unroll
      for (int a = 0; a < 4; ++a)
#pragma unroll
        for (int b = 0; b < 4; ++b)
          acc[a][h * 4 + b] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              fa[a], fb[b], acc[a][h * 4 + b], 0, 0, 0);
    }
  };

  load_all(0);
  const int nphases = (kblocks + kpb - 1) / kpb;
  for (int p = 0; p < nphases; ++p) {
    const int kb1 = min(kblocks, (p + 1) * kpb);
    for (int kb = p * kpb; kb < kb1; ++kb) {
      write_all();
      __syncthreads();
      if (kb + 1 < kblocks) load_all(kb + 1);
      mfma_pass(lt, rt);
      if (HILO) {
        mfma_pass(lt, rtl);
        mfma_pass(ltl, rt);
      }
      __syncthreads();
    }
    // best-effort pacing: arrive, then wait (bounded) for the cohort.
    // Relaxed atomics only — nothing is communicated, so no fences.
    if (p + 1 < nphases) {
      if (tid == 0) {
        __hip_atomic_fetch_add(&phase_ctr[p], 1, __ATOMIC_RELAXED,
                               __HIP_MEMORY_SCOPE_AGENT);
        int spins = 0;
        while (__hip_atomic_load(&phase_ctr[p], __ATOMIC_RELAXED,
                                 __HIP_MEMORY_SCOPE_AGENT) < nactive &&
               ++spins < 1500)
          __builtin_amdgcn_s_sleep(8);
      }
      __syncthreads();
    }
  }

  // epilogue: single owner per element per launch -> plain accumulate
  const int crow_base = (lane >> 4) * 4;
  const int ccol = lane & 15;
#pragma unroll
  for (int a = 0; a < 4; ++a)
#pragma unroll
    for (int b = 0; b < 8; ++b)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int gi = i0 + wr + a * 16 + crow_base + r;
        const int gj = j0 + wc + b * 16 + ccol;
        if (gi < m && gj < m) {
          KK[(size_t)gi * m + gj] += acc[a][b][r];
          if (ti != tj) KK[(size_t)gj * m + gi] += acc[a][b][r];
        }
      }
}

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
    int c, int m, int d, void* out, void* out_lo, void* out_t,
    void* out_lo_t, int out_is_bf16, const float* yv, double* Ky,
    hipStream_t stream) {
  dim3 grid((c + CK_TILE - 1) / CK_TILE, (m + CK_TILE - 1) / CK_TILE);
  hipLaunchKernelGGL(cross_kernel_tile_kernel, grid, dim3(256), 0, stream,
                     X, A, s2v, amp, c, m, d,
                     out_is_bf16 ? (bf16*)out : nullptr,
                     out_is_bf16 ? (bf16*)out_lo : nullptr,
                     (bf16*)out_t, (bf16*)out_lo_t,
                     out_is_bf16 ? nullptr : (float*)out, yv, Ky);
  return hipGetLastError();
}

extern "C" hipError_t launch_cross_mfma(const float* Xs, const float* As,
                                         const float* nx, const float* na,
                                         float amp, int c, int m, int d,
                                         void* out_bfT, void* out_loT,
                                         const float* yv, double* Ky,
                                         hipStream_t stream) {
  dim3 grid((c + CM_TILE - 1) / CM_TILE, (m + CM_TILE - 1) / CM_TILE);
  hipLaunchKernelGGL(cross_mfma_kernel, grid, dim3(256), 0, stream,
                     Xs, As, nx, na, amp, c, m, d,
                     (bf16*)out_bfT, (bf16*)out_loT, yv, Ky);
  return hipGetLastError();
}

extern "C" hipError_t launch_syrk_bf16_sync(const void* KcT, const void* KlT,
                                            int c, int m, int cpitch,
                                            const int* tiles, int nb,
                                            int kpb, int* phase_ctr,
                                            int nactive, float* KK,
                                            hipStream_t stream) {
  if (KlT) {
    hipLaunchKernelGGL((syrk_bf16_sync_kernel<true>), dim3(nb), dim3(SY_WG),
                       0, stream, (const bf16*)KcT, (const bf16*)KlT, c, m,
                       cpitch, tiles, kpb, phase_ctr, nactive, KK);
  } else {
    hipLaunchKernelGGL((syrk_bf16_sync_kernel<false>), dim3(nb), dim3(SY_WG),
                       0, stream, (const bf16*)KcT, nullptr, c, m,
                       cpitch, tiles, kpb, phase_ctr, nactive, KK);
  }
  return hipGetLastError();
}

extern "C" hipError_t launch_syrk_bf16(const void* KcT, const void* KlT,
                                       int c, int m, int cpitch, int split_k,
                                       float* KK, hipStream_t stream) {
  const int ntile = (m + SY_CT - 1) / SY_CT;
  dim3 grid(ntile * ntile, split_k);
  if (KlT) {
    hipLaunchKernelGGL((syrk_bf16_kernel<true>), grid, dim3(SY_WG), 0,
                       stream, (const bf16*)KcT, (const bf16*)KlT, c, m,
                       cpitch, ntile, split_k, KK);
  } else {
    hipLaunchKernelGGL((syrk_bf16_kernel<false>), grid, dim3(SY_WG), 0,
                       stream, (const bf16*)KcT, nullptr, c, m, cpitch,
                       ntile, split_k, KK);
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
