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

typedef __attribute__((ext_vector_type(4))) float mfma_f32x4_t;

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

// Vectorized contiguous-contiguous dot: both pointers 16-B aligned at
// index 0 (callers pad their LDS row strides to multiples of 4 floats);
// scalar head to a 4-aligned c, float4 (ds_read_b128) body, scalar tail.
__device__ inline float dotv(const float* p, const float* q, int c0,
                             int c1) {
  float s = 0.f;
  int c = c0;
  for (; c < c1 && (c & 3); ++c) s += p[c] * q[c];
  float4 a4 = {0.f, 0.f, 0.f, 0.f};
  for (; c + 3 < c1; c += 4) {
    const float4 a = *(const float4*)(p + c);
    const float4 b = *(const float4*)(q + c);
    a4.x += a.x * b.x;
    a4.y += a.y * b.y;
    a4.z += a.z * b.z;
    a4.w += a.w * b.w;
  }
  s += (a4.x + a4.y) + (a4.z + a4.w);
  for (; c < c1; ++c) s += p[c] * q[c];
  return s;
}

// Vectorized-p x strided-q dot: p 16-B aligned at index 0, q any stride.
__device__ inline float dotm(const float* p, const float* q, int sq,
                             int c0, int c1) {
  float s = 0.f;
  int c = c0;
  for (; c < c1 && (c & 3); ++c) s += p[c] * q[c * sq];
  float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
  for (; c + 3 < c1; c += 4) {
    const float4 a = *(const float4*)(p + c);
    s0 += a.x * q[c * sq];
    s1 += a.y * q[(c + 1) * sq];
    s2 += a.z * q[(c + 2) * sq];
    s3 += a.w * q[(c + 3) * sq];
  }
  s += (s0 + s1) + (s2 + s3);
  for (; c < c1; ++c) s += p[c] * q[c * sq];
  return s;
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


// ---------------------------------------------------------------------------
// Pipelined variant: wave-specialized q loop with LDS flag counters
// ---------------------------------------------------------------------------
// Round-2 restructure of the C phase (TODO item 1, option b/c) — kept as
// a documented NEGATIVE RESULT: measured 133 us for the C phase at k=100
// vs 88 us for the barrier version (E=20000, launch 7.32 vs 5.58 ms).
// The group-B panel substitution is a serial 8-step LDS-latency chain
// (~1.5-2 us/step regardless of thread count) that replaces the trtri on
// the critical cycle, and the per-step join cadence adds flag latency;
// the barrier version's all-thread panel GEMM against the inverted 8x8
// is faster than anything that avoids the trtri.  Not referenced by any
// kernel (inline => not emitted); kept for the design record.
// Per sub-block q:
//   wave 0:    spin(head[q-1]) -> 8x8 shfl factor(q) -> bump fact[q]
//   wave 1:    spin(fact[q]) -> trtri8(q) from LDS (feeds ONLY the block
//              inverse assembly, off the critical path) -> Vq
//   waves 2-7: spin(fact[q]) -> panel rows [qb+8, bs) by forward
//              substitution directly against L(q) (no vq needed!) ->
//              group join -> trailing on the 16-row head slice -> bump
//              head[q] (releases wave 0's next factor) -> trailing rest
//              -> group join
// Waves 1-7 also stream the previous panel's cross-block updates
// (consumed only after the end-of-block s_barrier).  All spins are
// bounded; on a bound hit or factor breakdown every wave falls through
// to the barrier with *bad set.
//
// LDS flag block: the caller's 16-byte bad slot doubles as 4 ints —
// [0] bad, [1] factor count, [2] head count, [3] group-B join count;
// monotonic counters, workgroup-scope acquire/release.

#define PIPE_SPIN_CAP 100000000

__device__ inline int pipe_ld(int* p) {
  return __hip_atomic_load(p, __ATOMIC_ACQUIRE, __HIP_MEMORY_SCOPE_WORKGROUP);
}

__device__ inline void pipe_st(int* p, int v) {
  __hip_atomic_store(p, v, __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_WORKGROUP);
}

__device__ inline void pipe_add(int* p, int v) {
  __hip_atomic_fetch_add(p, v, __ATOMIC_RELEASE,
                         __HIP_MEMORY_SCOPE_WORKGROUP);
}

// spin until *p >= tgt (or bad); returns false on abort
__device__ inline bool pipe_wait(int* p, int tgt, int* bad) {
  int n = 0;
  while (pipe_ld(p) < tgt) {
    if (pipe_ld(bad)) return false;
    if (++n > PIPE_SPIN_CAP) {
      if (pipe_ld(bad) == 0) pipe_st(bad, 3);
      return false;
    }
    __builtin_amdgcn_s_sleep(1);
  }
  return true;
}

// In place on the lower triangle of Abuf (k x k, row stride SA):
// K -> L -> V = L^{-1} (lower).  Strict upper of Abuf is never touched.
// log|K| accumulates into misc[0]; *bad set to 1 (indefinite) or 2
// (non-finite pivot) on breakdown.  Tbuf: >= max(k*36, 448) floats
// scratch, 16-B aligned; SA must be a multiple of 4 (vectorized dots).
// All WG threads must call (contains __syncthreads).
__device__ inline void chol_invert_lower(float* Abuf, float* Tbuf,
                                         const int k, const int SA,
                                         const int tid, const int lane,
                                         int* bad, double* misc,
                                         unsigned long long* jclk = nullptr) {
  // jclk (profiling only): accumulated cycles into
  //   [0] phase1  [1] phase2 wall  [2] wave0 span  [3] waves1-7 span
  //   [4] C2 panel solve  [5] D off-diag trtri
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
    unsigned long long jt0 = 0, jt1 = 0;
    if (jclk && tid == 0) jt0 = wall_clock64();

    if (J > 0) {
      // phase1: update ONLY the diagonal block (rows/cols jb..jb+bs, c<=i)
      // so wave 0 can start factoring immediately; the rest of column-block
      // J and the trailing matrix are updated by waves 1-7 during phase2.
      for (int f = tid; f < bs * bs; f += WG) {
        const int r = f / bs, c = f - r * bs;
        if (c > r) continue;                 // keep the Kb upper cache
        const int i = jb + r, cc = jb + c;
        Abuf[(size_t)i * SA + cc] -=
            dotv(Abuf + (size_t)i * SA + pj,
                 Abuf + (size_t)cc * SA + pj, 0, NB);
      }
      __syncthreads();
    }
    if (jclk && (tid == 0 || tid == 64)) jt1 = wall_clock64();

    // phase2, restructured (round-1 profiling: the old wave-0-only
    // version spent 86 us/expert on one wave while waves 1-7 idled 5 us):
    // per 8-column sub-block q, wave 0 runs the shfl factor+trtri while
    // waves 1-7 process chunk q of panel J-1's update to the rows below
    // this column block; then ALL 512 threads apply the intra-block
    // panel/trailing for q.  The 32x32 inverse assembly afterwards also
    // runs on all threads.  8x8 inverses live in T[0..256); T[256..448)
    // is shell scratch.
    const int nq = (bs + 7) / 8;
    float* Vq = Tbuf;               // [nq][8][8]
    float* TS = Tbuf + 256;         // shell scratch [3][8][8]
    const int t0r = jb + bs;        // first row below this column block
    const int nrp = k - t0r;
    const int ntri_rest = nrp * (nrp + 1) / 2;
    for (int q = 0; q < nq; ++q) {
      const int qb = q * 8;
      const int sbs = min(8, bs - qb);
      if (tid < 64) {
        // static priority for the serial factor wave: its dependent
        // shfl/VALU chain is the block's critical path while waves 1-7
        // stream trailing updates — give it the issue-arbitration edge
        // (MI355X_MICROARCH.md "Two waves per SIMD", item 4)
        __builtin_amdgcn_s_setprio(1);
        {
          // 8x8 factor + trtri on lanes 0..7, ROW j per lane, cross-lane
          // traffic via __shfl (ds_bpermute).  Row-per-lane keeps every
          // runtime-varying access at a STATIC register index: the rank-1
          // scalar is the lane's own row[ss], and the update bounds
          // c in (ss, 7] are compile-time — the earlier column-per-lane
          // version needed l[j] with a runtime lane index, which compiles
          // to an 8-deep v_cmp/cndmask select chain per pivot step.
          // Lanes 8..63 compute duplicates and are masked off every store.
          const int j = lane & 7;
          float row[8];                 // this lane's row: row[c] = m[j][c]
#pragma unroll
          for (int c = 0; c < 8; ++c)
            row[c] = (j < sbs && c < sbs && c <= j)
                         ? D[(size_t)(qb + j) * SA + qb + c]
                         : (c == j ? 1.f : 0.f);
          bool ok = true, nonfin = false;
          // log|sub-block| via one logf at the end: accumulate the pivot
          // mantissa product and exponent sum (branchless — a per-step
          // `ldet += (double)__logf(piv)` plus isfinite/positivity
          // branches cost ~35 instructions per pivot)
          float mprod = 1.f;
          int expsum = 0;
          float myrs = 1.f;             // this row's pivot 1/sqrt
          float rsv = 1.f;
#pragma unroll
          for (int ss = 0; ss < 8; ++ss) {
            float l[8];                 // column ss: l[c] = lane c's row[ss]
#pragma unroll
            for (int c = 0; c < 8; ++c) l[c] = __shfl(row[ss], c, 64);
            const float piv = l[ss];
            if (ss < sbs) {
              const bool fin = isfinite(piv);
              // classify only the FIRST failure (nonfin wins only if it
              // happens while still ok)
              nonfin = nonfin || (ok && !fin);
              ok = ok && fin && (piv > 0.f);
              mprod *= __builtin_amdgcn_frexp_mantf(piv);
              expsum += __builtin_amdgcn_frexp_expf(piv);
            }
            const float rs = rsqrtf(piv);
            if (j == ss) myrs = rs;
            // subtract l_j*l_c = (rs*row[ss]) * (rs*l[c]) for c > ss; rows
            // j <= ss get w = 0 (their updates ended at step j)
            const float w = (j > ss) ? rs * row[ss] : 0.f;
#pragma unroll
            for (int c = ss + 1; c < 8; ++c) row[c] -= (rs * l[c]) * w;
          }
          // deferred column scaling: L[j][c] = row[c] * rs_c; gather the
          // rs vector from the diagonal lanes (rs_c = lane c's myrs)
#pragma unroll
          for (int c = 0; c < 8; ++c) {
            rsv = __shfl(myrs, c, 64);
            row[c] *= rsv;
          }
          if (lane < 8 && j < sbs) {    // write L back (row j)
#pragma unroll
            for (int c = 0; c < 8; ++c)
              if (c <= j) D[(size_t)(qb + j) * SA + qb + c] = row[c];
          }
          // trtri8: V column j by forward substitution; row i of L is
          // gathered as rowi[c] = lane i's row[c]
          float v[8];
#pragma unroll
          for (int i = 0; i < 8; ++i) v[i] = (i == j) ? myrs : 0.f;
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            float rowi[8];
#pragma unroll
            for (int c = 0; c < 8; ++c) rowi[c] = __shfl(row[c], i, 64);
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
              misc[0] += (double)__logf(mprod)
                         + (double)expsum * 0.6931471805599453;
            }
          }
        }
        __builtin_amdgcn_s_setprio(0);
      } else if (J > 0) {
        // waves 1-7, chunk q (strided by nq, so any nq covers everything):
        // previous panel's update to (a) the panel rows of column-block J
        // and (b) the remaining trailing triangle
        for (int f = (tid - 64) + q * (WG - 64); f < nrp * bs;
             f += (WG - 64) * nq) {
          const int r = f / bs, c = f - r * bs;
          const int i = t0r + r, cc = jb + c;
          Abuf[(size_t)i * SA + cc] -=
              dotv(Abuf + (size_t)i * SA + pj,
                   Abuf + (size_t)cc * SA + pj, 0, NB);
        }
        for (int f = (tid - 64) + q * (WG - 64); f < ntri_rest;
             f += (WG - 64) * nq) {
          int a, b;
          tri_decode(f, a, b);
          const int i = t0r + a, c = t0r + b;
          Abuf[(size_t)i * SA + c] -=
              dotv(Abuf + (size_t)i * SA + pj,
                   Abuf + (size_t)c * SA + pj, 0, NB);
        }
      }
      __syncthreads();
      if (*bad) break;               // uniform: read after the barrier
      // intra-block panel P = A * Vq^T and trailing for q, ALL threads
      const int p0 = qb + sbs;       // first panel row (local)
      const int pr = bs - p0;        // panel rows
      if (pr > 0) {
        for (int f = tid; f < pr * 8; f += WG) {
          const int r = f >> 3, c = f & 7;
          if (c >= sbs) continue;
          float sacc = 0.f;
          const float* ar = D + (size_t)(p0 + r) * SA + qb;
          const float* vr = Vq + q * 64 + c * 8;
          for (int t = 0; t <= c; ++t) sacc += ar[t] * vr[t];
          TS[f] = sacc;              // hold row piece until all reads done
        }
        __syncthreads();
        for (int f = tid; f < pr * 8; f += WG) {
          const int r = f >> 3, c = f & 7;
          if (c >= sbs) continue;
          D[(size_t)(p0 + r) * SA + qb + c] = TS[f];
        }
        __syncthreads();
        // trailing: lower incl diag of remaining rows of this block
        const int ntri = pr * (pr + 1) / 2;
        for (int f = tid; f < ntri; f += WG) {
          int a, b;
          tri_decode(f, a, b);
          const int i = p0 + a, c = p0 + b;
          Abuf[(size_t)(jb + i) * SA + jb + c] -=
              dotv(D + (size_t)i * SA + qb, D + (size_t)c * SA + qb,
                   0, sbs);
        }
        __syncthreads();
      }
    }
    unsigned long long at0 = 0;
    if (jclk && tid == 0) {
      at0 = wall_clock64();
      jclk[2] += at0 - jt1;          // q-loop wall (factor+chunks+intra)
    }
    if (*bad == 0) {
      // ---- assemble V_JJ (32x32 inverse) from the 8x8 inverses --------
      // column-blocks Jq descending; all target blocks of a column in
      // parallel through TS; ALL threads
      for (int Jq = nq - 1; Jq >= 0; --Jq) {
        const int jb8 = Jq * 8;
        const int nblks = nq - 1 - Jq;       // target blocks below
        for (int f = tid; f < nblks * 64; f += WG) {
          const int blk = f >> 6;            // 0..nblks-1
          const int ib = (Jq + 1 + blk) * 8;
          const int i = (f >> 3) & 7, j = f & 7;
          const int gi = ib + i;
          float u = 0.f;
          if (gi < bs && jb8 + j < bs) {
            // u = sum_{c=jb8+8}^{gi} V[gi][c] * L[c][jb8+j]
            u = dotm(D + (size_t)gi * SA, D + jb8 + j, SA,
                     jb8 + 8, min(gi + 1, bs));
          }
          TS[f] = u;
        }
        __syncthreads();
        for (int f = tid; f < nblks * 64; f += WG) {
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
        for (int f = tid; f < 64; f += WG) {
          const int i = (f >> 3) & 7, j = f & 7;
          const int gi = jb8 + i, gj = jb8 + j;
          if (gi < bs && gj <= gi) D[(size_t)gi * SA + gj] = Vq[Jq * 64 + f];
        }
        __syncthreads();
      }
    }
    if (jclk && tid == 0) jclk[3] += wall_clock64() - at0;  // assembly wall
    __syncthreads();
    if (jclk && tid == 0) {
      const unsigned long long t2 = wall_clock64();
      jclk[0] += jt1 - jt0;
      jclk[1] += t2 - jt1;
      jt0 = t2;
    }
    if (*bad) break;

    const int t0 = jb + bs;        // first trailing row
    const int nr = k - t0;         // panel rows
    if (nr > 0) {
      // C2: copy panel below the diag block into T (row r-t0, stride 36)
      for (int f = tid; f < nr * bs; f += WG) {
        int r = f / bs, c = f - r * bs;
        Tbuf[r * 36 + c] = Abuf[(size_t)(t0 + r) * SA + jb + c];
      }
      __syncthreads();
      // C2b (MFMA): panel <- T @ V_JJ^T:
      //   A[t0+r][jb+c] = sum_{t<=c} T[r][t] V_JJ[c][t]
      // B-fragment = V_JJ[c][t] masked to its lower triangle; per-wave
      // register tiles, written straight back to the panel.
      {
        const int w8 = tid >> 6;
        const int fl16 = lane & 15, fkg = lane >> 4;
        const int nti = (nr + 15) / 16, ntj = (bs + 15) / 16;
        for (int t = w8; t < nti * ntj; t += WG / 64) {
          const int ti = t / ntj, tj = t - ti * ntj;
          mfma_f32x4_t a = {0.f, 0.f, 0.f, 0.f};
          const int fr = ti * 16 + fl16;       // T row
          const int fc = tj * 16 + fl16;       // V_JJ row (output col)
          for (int c0 = 0; c0 < bs; c0 += 4) {
            const int tt = c0 + fkg;
            const float av = (fr < nr && tt < bs)
                                 ? Tbuf[fr * 36 + tt] : 0.f;
            const float bv = (fc < bs && tt <= fc)
                                 ? D[(size_t)fc * SA + tt] : 0.f;
            a = __builtin_amdgcn_mfma_f32_16x16x4f32(av, bv, a, 0, 0, 0);
          }
#pragma unroll
          for (int r = 0; r < 4; ++r) {
            const int gr = ti * 16 + fkg * 4 + r;
            const int gc = tj * 16 + fl16;
            if (gr < nr && gc < bs)
              Abuf[(size_t)(t0 + gr) * SA + jb + gc] = a[r];
          }
        }
      }
      __syncthreads();
    }
    if (jclk && tid == 0) jclk[4] += wall_clock64() - jt0;
  }

  if (*bad) return;
  unsigned long long dt0 = 0;
  if (jclk && tid == 0) dt0 = wall_clock64();
  // ---- D: off-diagonal triangular inverse, in place, J right-to-left
  // V_IJ = -(sum_{K=J+1..I} V_IK L_KJ) L_JJ^-1, both halves as MFMA
  // tiles: the U GEMM masks A-fragments to the lower triangle of the
  // already-inverted trailing V (c <= row) and the second GEMM masks
  // B-fragments to V_JJ's lower triangle (t >= j).
  for (int J = nblk - 2; J >= 0; --J) {
    const int jb = J * NB;
    const int bs = NB;                       // J < nblk-1 => full block
    const int t0 = jb + bs;
    const int nr = k - t0;
    const int w8 = tid >> 6;
    const int fl16 = lane & 15, fkg = lane >> 4;
    const int nti = (nr + 15) / 16, ntj = (bs + 15) / 16;
    // U into T: U[r][t] = sum_{c=t0..t0+r} V[t0+r][c] * L[c][jb+t]
    for (int t = w8; t < nti * ntj; t += WG / 64) {
      const int ti = t / ntj, tj = t - ti * ntj;
      mfma_f32x4_t a = {0.f, 0.f, 0.f, 0.f};
      const int fr = t0 + ti * 16 + fl16;        // V row (A fragment)
      const int fj = tj * 16 + fl16;             // L column (B fragment)
      for (int c0 = t0; c0 < k; c0 += 4) {
        const int cc = c0 + fkg;
        const float av = (fr < k && cc < k && cc <= fr)
                             ? Abuf[(size_t)fr * SA + cc] : 0.f;
        const float bv = (cc < k && fj < bs)
                             ? Abuf[(size_t)cc * SA + jb + fj] : 0.f;
        a = __builtin_amdgcn_mfma_f32_16x16x4f32(av, bv, a, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int gr = ti * 16 + fkg * 4 + r;
        const int gc = tj * 16 + fl16;
        if (gr < nr && gc < bs) Tbuf[gr * 36 + gc] = a[r];
      }
    }
    __syncthreads();
    // V[t0+r][jb+j] = - sum_{t>=j} U[r][t] * V_JJ[t][j]
    for (int t = w8; t < nti * ntj; t += WG / 64) {
      const int ti = t / ntj, tj = t - ti * ntj;
      mfma_f32x4_t a = {0.f, 0.f, 0.f, 0.f};
      const int fr = ti * 16 + fl16;             // U row (A fragment)
      const int fj = tj * 16 + fl16;             // V_JJ column (B frag)
      for (int c0 = 0; c0 < bs; c0 += 4) {
        const int tt = c0 + fkg;
        const float av = (fr < nr && tt < bs)
                             ? Tbuf[fr * 36 + tt] : 0.f;
        const float bv = (tt < bs && fj < bs && tt >= fj)
                             ? Abuf[(size_t)(jb + tt) * SA + jb + fj]
                             : 0.f;
        a = __builtin_amdgcn_mfma_f32_16x16x4f32(av, bv, a, 0, 0, 0);
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int gr = ti * 16 + fkg * 4 + r;
        const int gc = tj * 16 + fl16;
        if (gr < nr && gc < bs)
          Abuf[(size_t)(t0 + gr) * SA + jb + gc] = -a[r];
      }
    }
    __syncthreads();
  }
  if (jclk && tid == 0) jclk[5] += wall_clock64() - dt0;
}

// Pipelined C/D: same contract as chol_invert_lower (K -> L -> V in place,
// logdet into misc[0], bad flag), restructured per the header note above.
// flags = 4-int LDS block: [0] bad (shared with caller), [1] factor count,
// [2] head count, [3] group-B join count — [1..3] reset per block here.
// Used by expert_nll; laplace keeps the barrier version (smaller k budget,
// different wave economics).
__device__ inline void chol_invert_lower_pipe(
    float* Abuf, float* Tbuf, const int k, const int SA, const int tid,
    const int lane, int* flags, double* misc,
    unsigned long long* jclk = nullptr) {
  const int nblk = (k + NB - 1) / NB;
  int* bad = flags;
  int* FC = flags + 1;
  int* FH = flags + 2;
  int* FJ = flags + 3;
  const int wave = tid >> 6;

  for (int J = 0; J < nblk; ++J) {
    const int jb = J * NB;
    const int bs = min(NB, k - jb);
    const int nq = (bs + 7) / 8;
    float* D = Abuf + (size_t)jb * SA + jb;
    const int pj = jb - NB;
    if (tid == 0) { *FC = 0; *FH = 0; *FJ = 0; }
    if (J > 0) {
      // phase1: previous panel's rank-NB update to the FULL 32x32 block
      // (lower incl. diag — every row this block's q loop touches)
      for (int f = tid; f < bs * bs; f += WG) {
        const int r = f / bs, c = f - r * bs;
        if (c > r) continue;
        const int i = jb + r, cc = jb + c;
        Abuf[(size_t)i * SA + cc] -=
            dotv(Abuf + (size_t)i * SA + pj,
                 Abuf + (size_t)cc * SA + pj, 0, NB);
      }
    }
    __syncthreads();
    unsigned long long w0t0 = 0;
    if (jclk && tid == 0) w0t0 = wall_clock64();

    // ---------------- pipelined q loop (no barriers) -----------------
    if (wave == 0) {
      // wave 0: the factor chain only
      for (int q = 0; q < nq; ++q) {
        if (q > 0 && !pipe_wait(FH, q, bad)) break;
        const int qb = q * 8;
        const int sbs = min(8, bs - qb);
        const int j = lane & 7;
        float row[8];
#pragma unroll
        for (int c = 0; c < 8; ++c)
          row[c] = (j < sbs && c < sbs && c <= j)
                       ? D[(size_t)(qb + j) * SA + qb + c]
                       : (c == j ? 1.f : 0.f);
        bool ok = true, nonfin = false;
        float mprod = 1.f;
        int expsum = 0;
        float myrs = 1.f;
#pragma unroll
        for (int ss = 0; ss < 8; ++ss) {
          float l[8];
#pragma unroll
          for (int c = 0; c < 8; ++c) l[c] = __shfl(row[ss], c, 64);
          const float piv = l[ss];
          if (ss < sbs) {
            const bool fin = isfinite(piv);
            nonfin = nonfin || (ok && !fin);
            ok = ok && fin && (piv > 0.f);
            mprod *= __builtin_amdgcn_frexp_mantf(piv);
            expsum += __builtin_amdgcn_frexp_expf(piv);
          }
          const float rs = rsqrtf(piv);
          if (j == ss) myrs = rs;
          const float w = (j > ss) ? rs * row[ss] : 0.f;
#pragma unroll
          for (int c = ss + 1; c < 8; ++c) row[c] -= (rs * l[c]) * w;
        }
#pragma unroll
        for (int c = 0; c < 8; ++c) row[c] *= __shfl(myrs, c, 64);
        if (lane < 8 && j < sbs) {
#pragma unroll
          for (int c = 0; c < 8; ++c)
            if (c <= j) D[(size_t)(qb + j) * SA + qb + c] = row[c];
        }
        if (lane == 0) {
          if (!ok) {
            if (pipe_ld(bad) == 0) pipe_st(bad, nonfin ? 2 : 1);
          } else {
            misc[0] += (double)__logf(mprod)
                       + (double)expsum * 0.6931471805599453;
          }
          pipe_add(FC, 1);
        }
        if (!ok) break;
      }
      if (jclk && tid == 0) jclk[2] += wall_clock64() - w0t0;
    } else if (wave == 1) {
      // wave 1: trtri8 into Vq (assembly input; off the critical path)
      float* Vq = Tbuf;
      for (int q = 0; q < nq; ++q) {
        if (!pipe_wait(FC, q + 1, bad)) break;
        const int qb = q * 8;
        const int sbs = min(8, bs - qb);
        if (lane < 8) {
          const int j = lane;
          float v[8];
#pragma unroll
          for (int i = 0; i < 8; ++i) v[i] = (i == j) ? 1.f : 0.f;
          if (j < sbs) {
            v[j] = __builtin_amdgcn_rcpf(
                D[(size_t)(qb + j) * SA + qb + j]);
            for (int i = j + 1; i < sbs; ++i) {
              float sacc = 0.f;
              for (int c = j; c < i; ++c)
                sacc += D[(size_t)(qb + i) * SA + qb + c] * v[c];
              v[i] = -sacc * __builtin_amdgcn_rcpf(
                  D[(size_t)(qb + i) * SA + qb + i]);
            }
          }
#pragma unroll
          for (int i = 0; i < 8; ++i) Vq[q * 64 + i * 8 + j] = v[i];
        }
        // share of the previous panel's cross-block update (see below)
        if (J > 0) {
          const int t0r = jb + bs;
          const int nrp = k - t0r;
          const int ntri_rest = nrp * (nrp + 1) / 2;
          for (int f = (tid - 64) + q * (WG - 64); f < nrp * bs;
               f += (WG - 64) * nq) {
            const int r = f / bs, c = f - r * bs;
            const int i = t0r + r, cc = jb + c;
            Abuf[(size_t)i * SA + cc] -=
                dotv(Abuf + (size_t)i * SA + pj,
                     Abuf + (size_t)cc * SA + pj, 0, NB);
          }
          for (int f = (tid - 64) + q * (WG - 64); f < ntri_rest;
               f += (WG - 64) * nq) {
            int a, b;
            tri_decode(f, a, b);
            const int i = t0r + a, c = t0r + b;
            Abuf[(size_t)i * SA + c] -=
                dotv(Abuf + (size_t)i * SA + pj,
                     Abuf + (size_t)c * SA + pj, 0, NB);
          }
        }
      }
    } else {
      // waves 2-7 (384 threads): panel substitution + trailing
      const int gt = tid - 128;
      for (int q = 0; q < nq; ++q) {
        if (!pipe_wait(FC, q + 1, bad)) break;
        const int qb = q * 8;
        const int sbs = min(8, bs - qb);
        const int p0 = qb + sbs;
        const int pr = bs - p0;
        // panel rows [p0, bs): solve p L(q)^T = a by forward
        // substitution directly against L (no trtri dependency)
        for (int r = p0 + gt; r < bs; r += 384) {
          float p[8];
          for (int c = 0; c < sbs; ++c)
            p[c] = D[(size_t)r * SA + qb + c];
          for (int c = 0; c < sbs; ++c) {
            float s = p[c];
            for (int t = 0; t < c; ++t)
              s -= p[t] * D[(size_t)(qb + c) * SA + qb + t];
            p[c] = s * __builtin_amdgcn_rcpf(
                D[(size_t)(qb + c) * SA + qb + c]);
          }
          for (int c = 0; c < sbs; ++c)
            D[(size_t)r * SA + qb + c] = p[c];
        }
        if (lane == 0) pipe_add(FJ, 1);                 // join A
        if (!pipe_wait(FJ, 18 * q + 6, bad)) break;
        // trailing, head slice first (rows [p0, p0+16)): releases the
        // next factor as soon as its rows are complete
        const int hend = min(p0 + 16, bs);
        const int hr = hend - p0;
        const int nth = hr * (hr + 1) / 2;
        for (int f = gt; f < nth; f += 384) {
          int a, b;
          tri_decode(f, a, b);
          const int r = p0 + a, c = p0 + b;
          float s = 0.f;
          for (int t = 0; t < sbs; ++t)
            s += D[(size_t)r * SA + qb + t] * D[(size_t)c * SA + qb + t];
          D[(size_t)r * SA + c] -= s;
        }
        if (lane == 0) pipe_add(FJ, 1);                 // join B
        if (!pipe_wait(FJ, 18 * q + 12, bad)) break;
        if (gt == 0) pipe_add(FH, 1);
        // trailing rest: the remaining pairs of the [p0, bs) triangle
        // (flat indices continue past the head block: tri_decode
        // enumerates rows ascending, so head pairs are exactly the
        // first nth indices)
        const int ntri_all = pr * (pr + 1) / 2;
        for (int f = nth + gt; f < ntri_all; f += 384) {
          int a, b;
          tri_decode(f, a, b);
          const int r = p0 + a, c = p0 + b;
          float s = 0.f;
          for (int t = 0; t < sbs; ++t)
            s += D[(size_t)r * SA + qb + t] * D[(size_t)c * SA + qb + t];
          D[(size_t)r * SA + c] -= s;
        }
        // share of the previous panel's cross-block update
        if (J > 0) {
          const int t0r = jb + bs;
          const int nrp = k - t0r;
          const int ntri_rest = nrp * (nrp + 1) / 2;
          for (int f = (tid - 64) + q * (WG - 64); f < nrp * bs;
               f += (WG - 64) * nq) {
            const int r = f / bs, c = f - r * bs;
            const int i = t0r + r, cc = jb + c;
            Abuf[(size_t)i * SA + cc] -=
                dotv(Abuf + (size_t)i * SA + pj,
                     Abuf + (size_t)cc * SA + pj, 0, NB);
          }
          for (int f = (tid - 64) + q * (WG - 64); f < ntri_rest;
               f += (WG - 64) * nq) {
            int a, b;
            tri_decode(f, a, b);
            const int i = t0r + a, c = t0r + b;
            Abuf[(size_t)i * SA + c] -=
                dotv(Abuf + (size_t)i * SA + pj,
                     Abuf + (size_t)c * SA + pj, 0, NB);
          }
        }
        if (lane == 0) pipe_add(FJ, 1);                 // join C
        if (!pipe_wait(FJ, 18 * q + 18, bad)) break;
      }
    }
    __syncthreads();
    if (*bad) break;

    // ---- assemble V_JJ (32x32 inverse) from the 8x8 inverses ---------
    {
      const int nq2 = (bs + 7) / 8;
      float* Vq = Tbuf;
      float* TS = Tbuf + 256;
      for (int Jq = nq2 - 1; Jq >= 0; --Jq) {
        const int jb8 = Jq * 8;
        const int nblks = nq2 - 1 - Jq;
        for (int f = tid; f < nblks * 64; f += WG) {
          const int blk = f >> 6;
          const int ib = (Jq + 1 + blk) * 8;
          const int i = (f >> 3) & 7, j = f & 7;
          const int gi = ib + i;
          float u = 0.f;
          if (gi < bs && jb8 + j < bs) {
            u = dotm(D + (size_t)gi * SA, D + jb8 + j, SA,
                     jb8 + 8, min(gi + 1, bs));
          }
          TS[f] = u;
        }
        __syncthreads();
        for (int f = tid; f < nblks * 64; f += WG) {
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
        for (int f = tid; f < 64; f += WG) {
          const int i = (f >> 3) & 7, j = f & 7;
          const int gi = jb8 + i, gj = jb8 + j;
          if (gi < bs && gj <= gi)
            D[(size_t)gi * SA + gj] = Vq[Jq * 64 + f];
        }
        __syncthreads();
      }
    }

    const int t0 = jb + bs;
    const int nr = k - t0;
    if (nr > 0) {
      // C2: cross-block panel solve against the inverted diagonal
      for (int f = tid; f < nr * bs; f += WG) {
        int r = f / bs, c = f - r * bs;
        Tbuf[r * 36 + c] = Abuf[(size_t)(t0 + r) * SA + jb + c];
      }
      __syncthreads();
      for (int f = tid; f < nr * bs; f += WG) {
        int r = f / bs, c = f - r * bs;
        Abuf[(size_t)(t0 + r) * SA + jb + c] =
            dotv(Tbuf + r * 36, D + c * SA, 0, c + 1);
      }
      __syncthreads();
    }
  }

  if (*bad) return;
  // ---- D: off-diagonal triangular inverse (same as the barrier path)
  for (int J = nblk - 2; J >= 0; --J) {
    const int jb = J * NB;
    const int bs = NB;
    const int t0 = jb + bs;
    const int nr = k - t0;
    for (int f = tid; f < nr * bs; f += WG) {
      int r = f / bs, t = f - r * bs;
      const int row = t0 + r;
      Tbuf[r * 36 + t] = dotm(Abuf + (size_t)row * SA,
                              Abuf + jb + t, SA, t0, row + 1);
    }
    __syncthreads();
    for (int f = tid; f < nr * bs; f += WG) {
      int r = f / bs, j = f - r * bs;
      Abuf[(size_t)(t0 + r) * SA + jb + j] =
          -dotm(Tbuf + r * 36, Abuf + (size_t)jb * SA + jb + j, SA, j, bs);
    }
    __syncthreads();
  }
}
