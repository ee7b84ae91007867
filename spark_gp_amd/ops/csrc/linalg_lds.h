// Shared LDS dense-linear-algebra machinery for the CDNA4 GP kernels:
// blocked in-place Cholesky with look-ahead (8x8 register sub-diagonals)
// and the in-place triangular inverse.  Included by expert_nll.hip and
// laplace.hip.  See expert_nll.hip's header comment for the algorithm.
#pragma once
#include <hip/hip_runtime.h>
#include <math.h>

#ifndef WG
#define WG 512
#endif
#define NB 32

__device__ inline double wave_sum(double v) {
#pragma unroll
  for (int off = 32; off; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

__device__ inline double block_sum(double v, double* red, int tid) {
  v = wave_sum(v);
  if ((tid & 63) == 0) red[tid >> 6] = v;
  __syncthreads();
  if (tid == 0) {
    double s = 0;
#pragma unroll
    for (int w = 0; w < WG / 64; ++w) s += red[w];
    red[0] = s;
  }
  __syncthreads();
  double out = red[0];
  __syncthreads();
  return out;
}

// flat f -> (a, b) with a >= b in a lower triangle (incl. diagonal)
__device__ inline void tri_decode(int f, int& a, int& b) {
  a = (int)((sqrtf(8.f * (float)f + 1.f) - 1.f) * 0.5f);
  while ((a + 1) * (a + 2) / 2 <= f) ++a;
  while (a * (a + 1) / 2 > f) --a;
  b = f - a * (a + 1) / 2;
}

// 4-accumulator strided dot over LDS: sum_{c=c0}^{c1-1} p[c*sp] * q[c*sq]
__device__ inline float dot4(const float* p, int sp, const float* q, int sq,
                             int c0, int c1) {
  float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
  int c = c0;
  for (; c + 3 < c1; c += 4) {
    s0 += p[c * sp] * q[c * sq];
    s1 += p[(c + 1) * sp] * q[(c + 1) * sq];
    s2 += p[(c + 2) * sp] * q[(c + 2) * sq];
    s3 += p[(c + 3) * sp] * q[(c + 3) * sq];
  }
  for (; c < c1; ++c) s0 += p[c * sp] * q[c * sq];
  return (s0 + s1) + (s2 + s3);
}


// In place on the lower triangle of Abuf (k x k, row stride SA):
// K -> L -> V = L^{-1} (lower).  Strict upper of Abuf is never touched.
// log|K| accumulates into misc[0]; *bad set to 1 (indefinite) or 2
// (non-finite pivot) on breakdown.  Tbuf: >= max(k*33, 448) floats scratch.
// All WG threads must call (contains __syncthreads).
__device__ inline void chol_invert_lower(float* Abuf, float* Tbuf,
                                         const int k, const int SA,
                                         const int tid, const int lane,
                                         int* bad, double* misc) {
  const int nblk = (k + NB - 1) / NB;
    // ---- C: blocked in-place Cholesky with look-ahead ----------------
  // Right-looking, restructured so the serial diagonal factorization
  // overlaps the data-parallel trailing update: per column block J,
  //   phase1: apply panel J-1's rank-NB update to COLUMN-BLOCK J only
  //   phase2: wave 0 factors+inverts diag J  ||  waves 1-7 apply panel
  //           J-1's update to the remaining trailing columns
  //   phase3: panel solve for block J (copy + GEMM vs inverted diagonal)
  for (int J = 0; J < nblk; ++J) {
    const int jb = J * NB;
    const int bs = min(NB, k - jb);
    float* D = Abuf + (size_t)jb * SA + jb;   // diag block, stride SA
    const int pj = jb - NB;                  // previous panel column offset

    if (J > 0) {
      // phase1: update ONLY the diagonal block (rows/cols jb..jb+bs, c<=i)
      // so wave 0 can start factoring immediately; the rest of column-block
      // J and the trailing matrix are updated by waves 1-7 during phase2.
      for (int f = tid; f < bs * bs; f += WG) {
        const int r = f / bs, c = f - r * bs;
        if (c > r) continue;                 // keep the Kb upper cache
        const int i = jb + r, cc = jb + c;
        Abuf[(size_t)i * SA + cc] -=
            dot4(Abuf + (size_t)i * SA + pj, 1,
                 Abuf + (size_t)cc * SA + pj, 1, 0, NB);
      }
      __syncthreads();
    }

    if (tid < 64) {
      // wave 0: factor + invert the bs x bs diagonal block via 8x8
      // sub-blocks.  Each 8x8 sub-diagonal is Cholesky-factored AND
      // inverted entirely inside lane 0's registers (fully unrolled, the
      // serial dependency chain never touches LDS); panels and trailing
      // updates are lane-parallel; the 32-level inverse assembles from the
      // 8x8 inverses by block back-substitution.  8x8 inverses live in
      // T[0..256); T[256..448) is shell scratch.
      const int row = lane & 31;
      const int nq = (bs + 7) / 8;
      float* Vq = Tbuf;               // [nq][8][8]
      float* TS = Tbuf + 256;         // shell scratch [3][8][8]
      for (int q = 0; q < nq; ++q) {
        const int qb = q * 8;
        const int sbs = min(8, bs - qb);
        {
          // 8x8 factor + trtri on lanes 0..7, column j per lane, cross-lane
          // traffic via __shfl (ds_bpermute).  Replaces a lane-0 register
          // version whose ~1200 single-lane instructions dominated the
          // kernel (a lane-parallel in-LDS variant measured 20% worse
          // still — LDS round-trip latency; see git history).  Working set
          // is ~32 registers, so no VGPR pressure.  Lanes 8..63 compute
          // duplicates and are masked off every store.
          const int j = lane & 7;
          float col[8];                 // this lane's column: col[i]=m[i][j]
#pragma unroll
          for (int i = 0; i < 8; ++i)
            col[i] = (j < sbs && i < sbs && i >= j)
                         ? D[(size_t)(qb + i) * SA + qb + j]
                         : (i == j ? 1.f : 0.f);
          bool ok = true, nonfin = false;
          double ldet = 0.0;
#pragma unroll
          for (int ss = 0; ss < 8; ++ss) {
            float l[8];                 // owner lane ss broadcasts column ss
#pragma unroll
            for (int i = 0; i < 8; ++i) l[i] = __shfl(col[i], ss, 64);
            const float piv = l[ss];
            if (ss < sbs && ok) {       // classify only the FIRST failure
              if (!isfinite(piv)) { ok = false; nonfin = true; }
              else if (!(piv > 0.f)) ok = false;
              else ldet += (double)__logf(piv);
            }
            const float rs = rsqrtf(piv);
#pragma unroll
            for (int i = 0; i < 8; ++i) l[i] *= rs;      // l[ss] = piv*rs
            if (j == ss) {
#pragma unroll
              for (int i = 0; i < 8; ++i)
                if (i >= ss) col[i] = l[i];              // keep scaled column
            } else if (j > ss) {
              const float ljs = l[j];
#pragma unroll
              for (int i = 0; i < 8; ++i)
                if (i >= j) col[i] -= l[i] * ljs;        // rank-1 update
            }
          }
          if (lane < 8 && j < sbs) {    // write L back (column j)
#pragma unroll
            for (int i = 0; i < 8; ++i)
              if (i >= j && i < sbs)
                D[(size_t)(qb + i) * SA + qb + j] = col[i];
          }
          // trtri8: V column j by forward substitution; row i of L is
          // gathered as rowi[c] = shfl(col[i], c)
          float v[8];
#pragma unroll
          for (int i = 0; i < 8; ++i) v[i] = 0.f;
          v[j] = __builtin_amdgcn_rcpf(col[j]);
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            float rowi[8];
#pragma unroll
            for (int c = 0; c < 8; ++c) rowi[c] = __shfl(col[i], c, 64);
            if (j < i) {
              float sacc = 0.f;
#pragma unroll
              for (int c = 0; c < 8; ++c)
                if (c >= j && c < i) sacc += rowi[c] * v[c];
              v[i] = -sacc * __builtin_amdgcn_rcpf(rowi[i]);
            }
          }
          if (lane < 8) {
#pragma unroll
            for (int i = 0; i < 8; ++i) Vq[q * 64 + i * 8 + j] = v[i];
          }
          if (lane == 0) {
            if (!ok) {
              if (*bad == 0) *bad = nonfin ? 2 : 1;  // sticky: first cause
            } else {
              misc[0] += ldet;
            }
          }
        }
        __syncwarp();
        if (*bad) break;               // wave-uniform: skip garbage blocks
        // panel rows within the 32-block: P = A * Vq^T; then trailing
        const int p0 = qb + sbs;       // first panel row (local)
        const int pr = bs - p0;        // panel rows
        if (pr > 0) {
          // panel: (pr x sbs) elements, lanes parallel
          for (int f = lane; f < pr * 8; f += 64) {
            const int r = f >> 3, c = f & 7;
            if (c >= sbs) continue;
            float sacc = 0.f;
            const float* ar = D + (size_t)(p0 + r) * SA + qb;
            const float* vr = Vq + q * 64 + c * 8;
            for (int t = 0; t <= c; ++t) sacc += ar[t] * vr[t];
            TS[f] = sacc;              // hold row piece until all reads done
          }
          __syncwarp();
          for (int f = lane; f < pr * 8; f += 64) {
            const int r = f >> 3, c = f & 7;
            if (c >= sbs) continue;
            D[(size_t)(p0 + r) * SA + qb + c] = TS[f];
          }
          __syncwarp();
          // trailing: lower incl diag of remaining rows
          const int ntri = pr * (pr + 1) / 2;
          for (int f = lane; f < ntri; f += 64) {
            int a, b;
            tri_decode(f, a, b);
            const int i = p0 + a, c = p0 + b;
            Abuf[(size_t)(jb + i) * SA + jb + c] -=
                dot4(D + (size_t)i * SA + qb, 1, D + (size_t)c * SA + qb, 1,
                     0, sbs);
          }
          __syncwarp();
        }
      }
      // ---- assemble V_JJ (32x32 inverse) from the 8x8 inverses --------
      // column-blocks Jq descending; all target blocks of a column in
      // parallel through TS
      for (int Jq = nq - 1; Jq >= 0; --Jq) {
        const int jb8 = Jq * 8;
        const int nblks = nq - 1 - Jq;       // target blocks below
        for (int f = lane; f < nblks * 64; f += 64) {
          const int blk = f >> 6;            // 0..nblks-1
          const int ib = (Jq + 1 + blk) * 8;
          const int i = (f >> 3) & 7, j = f & 7;
          const int gi = ib + i;
          float u = 0.f;
          if (gi < bs && jb8 + j < bs) {
            // u = sum_{c=jb8+8}^{gi} V[gi][c] * L[c][jb8+j]
            u = dot4(D + (size_t)gi * SA, 1, D + jb8 + j, SA,
                     jb8 + 8, min(gi + 1, bs));
          }
          TS[f] = u;
        }
        __syncwarp();
        for (int f = lane; f < nblks * 64; f += 64) {
          const int blk = f >> 6;
          const int ib = (Jq + 1 + blk) * 8;
          const int i = (f >> 3) & 7, j = f & 7;
          const int gi = ib + i;
          if (gi >= bs || jb8 + j >= bs) continue;
          float sacc = 0.f;
          const float* ts = TS + blk * 64 + i * 8;
          const float* vv = Vq + Jq * 64;
#pragma unroll
          for (int t = 0; t < 8; ++t) sacc += ts[t] * vv[t * 8 + j];
          D[(size_t)gi * SA + jb8 + j] = -sacc;
        }
        // this column's diagonal block <- its inverse (consumed as V by
        // the shells of columns further left)
        for (int f = lane; f < 64; f += 64) {
          const int i = (f >> 3) & 7, j = f & 7;
          const int gi = jb8 + i, gj = jb8 + j;
          if (gi < bs && gj <= gi) D[(size_t)gi * SA + gj] = Vq[Jq * 64 + f];
        }
        __syncwarp();
      }
      __syncwarp();
    } else if (J > 0) {
      // waves 1-7: previous panel's update to (a) the panel rows of
      // column-block J and (b) the remaining trailing triangle
      const int t0r = jb + bs;
      const int nrp = k - t0r;
      for (int f = tid - 64; f < nrp * bs; f += WG - 64) {
        const int r = f / bs, c = f - r * bs;
        const int i = t0r + r, cc = jb + c;
        Abuf[(size_t)i * SA + cc] -=
            dot4(Abuf + (size_t)i * SA + pj, 1,
                 Abuf + (size_t)cc * SA + pj, 1, 0, NB);
      }
      const int ntri = nrp * (nrp + 1) / 2;
      for (int f = tid - 64; f < ntri; f += WG - 64) {
        int a, b;
        tri_decode(f, a, b);
        const int i = t0r + a, c = t0r + b;
        Abuf[(size_t)i * SA + c] -=
            dot4(Abuf + (size_t)i * SA + pj, 1,
                 Abuf + (size_t)c * SA + pj, 1, 0, NB);
      }
    }
    __syncthreads();
    if (*bad) break;

    const int t0 = jb + bs;        // first trailing row
    const int nr = k - t0;         // panel rows
    if (nr > 0) {
      // C2: copy panel below the diag block into T (row r-t0, stride 33)
      for (int f = tid; f < nr * bs; f += WG) {
        int r = f / bs, c = f - r * bs;
        Tbuf[r * 33 + c] = Abuf[(size_t)(t0 + r) * SA + jb + c];
      }
      __syncthreads();
      // C2b: panel <- T * V_JJ^T : A[r][jb+c] = sum_{t<=c} T[r][t] V[c][t]
      for (int f = tid; f < nr * bs; f += WG) {
        int r = f / bs, c = f - r * bs;
        Abuf[(size_t)(t0 + r) * SA + jb + c] =
            dot4(Tbuf + r * 33, 1, D + c * SA, 1, 0, c + 1);
      }
      __syncthreads();
    }
  }

  if (*bad) return;
  // ---- D: off-diagonal triangular inverse, in place, J right-to-left
  // V_IJ = -(sum_{K=J+1..I} V_IK L_KJ) L_JJ^-1 ; rows fully parallel.
  for (int J = nblk - 2; J >= 0; --J) {
    const int jb = J * NB;
    const int bs = NB;                       // J < nblk-1 => full block
    const int t0 = jb + bs;
    const int nr = k - t0;
    // U into T: U[r][t] = sum_{c=t0..t0+r} V[t0+r][c] * L[c][jb+t]
    for (int f = tid; f < nr * bs; f += WG) {
      int r = f / bs, t = f - r * bs;
      const int row = t0 + r;
      Tbuf[r * 33 + t] = dot4(Abuf + (size_t)row * SA, 1,
                             Abuf + jb + t, SA, t0, row + 1);
    }
    __syncthreads();
    // V[row][jb+j] = - sum_{t>=j} U[r][t] * V_JJ[t][j]
    for (int f = tid; f < nr * bs; f += WG) {
      int r = f / bs, j = f - r * bs;
      Abuf[(size_t)(t0 + r) * SA + jb + j] =
          -dot4(Tbuf + r * 33, 1, Abuf + (size_t)jb * SA + jb + j, SA, j, bs);
    }
    __syncthreads();
  }

}
