// Fused per-expert Laplace-Newton kernel for binary GP classification on
// MI355X (gfx950, CDNA4) — the K10 hot loop of SURVEY.md §2.4.
//
// One workgroup (512 threads) per expert runs R&W Algorithm 3.1 (Newton
// with step halving, ``classification/GaussianProcessClassifier.scala:90-111``)
// to ITS OWN convergence — experts converge at different iteration counts
// with no cross-block synchronization (the per-expert convergence-mask
// problem disappears: the mask IS the block).
//
// Per Newton iteration, entirely in LDS:
//   pi = sigmoid(f); w = pi(1-pi); b = w f + (y - pi)
//   B = I + sqrt(w) K sqrt(w)              (K rebuilt from the cached Kb)
//   V = L^-1 via the shared blocked machinery (linalg_lds.h)
//   a = b - sqrt(w) V^T V (sqrt(w) (K b))  (matvecs, no serial solves)
//   f' = (1-s) f + s K a;  psi = -1/2 a.f' + sum log sigmoid((2y-1) f')
//   accept / halve s, stop on |delta psi| <= tol or s <= tol
//
// Outputs: updated latent f (in place), psi, sum log diag L, per-expert
// Newton iteration count, bad flag (fp32 breakdown -> host falls back to
// the batched torch path for that expert).  In EV mode the Algorithm 5.1
// evidence + FULL gradient (K11) run in the same launch at the converged
// latent (see the evidence tail below); outside EV mode — or on any bad
// expert, or for tolerances under LAPLACE_MIN_TOL — the torch
// contraction-form path finishes with reference-exact fp64 semantics.
//
// Single-buffer design (round 2): ONE k x SA working matrix — the base
// kernel kb lives in its STRICT UPPER triangle (kb_aa = 1 implicit) and
// survives every phase that works on the lower (B -> L -> V -> Binv);
// K = amp*kb + noise*I is reconstructed on the fly (lap_kdot/lap_kcdot
// split each row dot into a strided upper half and a contiguous tail).
// With __launch_bounds__(512, 4) the kernel runs 2 workgroups per CU —
// the old separate-K version was 1/CU twice over (2 x k x SA LDS AND
// 232 VGPRs) and ran the 1M x 16 GPC fit ~1.6x slower.
//
// Constraints: k <= 128, d <= k; LDS budget checked host-side.

#include <hip/hip_runtime.h>
#include <math.h>

#define WG 512
#include "linalg_lds.h"

struct LapLds {
  float* X;     // k * dp4: raw features (16-B-aligned rows)
  float* A;     // k * SA: strict upper = Kb cache (kb, no amp/noise;
                // kb_aa = 1 implicit); lower = B -> L -> V per iteration
                // and Binv -> W0 in the evidence tail
  float* T;     // max(k*36, 32*SA, 448) scratch for the chol machinery
  float* yb;    // k
  float* fb;    // k   latent
  float* pi;    // k
  float* sqw;   // k
  float* bv;    // k   b = w f + (y - pi)
  float* av;    // k   a
  float* t1;    // k
  float* t2;    // k
  float* s2;    // d
  // evidence-mode (EV) extras
  float* vv;    // k   v = y - pi
  float* s2v;   // k   Algorithm 5.1 s2 vector
  float* betav; // d   raw base scale (beta), signed
  double* gout; // d+2 gradient accumulators (beta..., amp, noise)
  double* red;  // 8
  double* misc; // 2
  int* bad;     // 1
};

// Regions padded to 16 B (vectorized dots assume aligned bases); keep in
// EXACT sync with the mirror in bindings.cpp.
static __host__ __device__ inline size_t lap_sa(int k) {
  return (size_t)((k + 4) & ~3);                       // k+1 up to mult 4
}

static __host__ __device__ inline size_t lap_tsz(int k) {
  size_t t = (size_t)k * 36;
  if (t < 32 * lap_sa(k)) t = 32 * lap_sa(k);
  if (t < 448) t = 448;
  return t;
}

static __host__ __device__ inline size_t lap_a16(size_t n) {
  return (n + 15) & ~(size_t)15;
}

static __host__ __device__ inline size_t lap_dp4(int d) {
  return (size_t)((d + 4) & ~3);
}

static __host__ __device__ inline size_t lap_lds_bytes(int k, int d,
                                                       int ev = 0) {
  size_t off = lap_a16(sizeof(double) * 10);
  off += lap_a16(sizeof(float) * (size_t)k * lap_sa(k));      // A
  off += lap_a16(sizeof(float) * (size_t)k * lap_dp4(d));     // X
  off += lap_a16(sizeof(float) * lap_tsz(k));
  off += 8 * lap_a16(sizeof(float) * k);
  off += lap_a16(sizeof(float) * d);
  if (ev) {
    off += 2 * lap_a16(sizeof(float) * k);       // vv + s2v
    off += lap_a16(sizeof(float) * d);           // betav
    off += lap_a16(sizeof(double) * (d + 2));    // gout
  }
  off += 16;
  return off;
}

template <bool EV>
__device__ inline LapLds lap_carve(char* base, int k, int d) {
  LapLds L;
  char* p = base;
  L.red = (double*)p;  L.misc = L.red + 8;
  p += lap_a16(sizeof(double) * 10);
  L.A = (float*)p;     p += lap_a16(sizeof(float) * (size_t)k * lap_sa(k));
  L.X = (float*)p;     p += lap_a16(sizeof(float) * (size_t)k * lap_dp4(d));
  L.T = (float*)p;     p += lap_a16(sizeof(float) * lap_tsz(k));
  L.yb = (float*)p;    p += lap_a16(sizeof(float) * k);
  L.fb = (float*)p;    p += lap_a16(sizeof(float) * k);
  L.pi = (float*)p;    p += lap_a16(sizeof(float) * k);
  L.sqw = (float*)p;   p += lap_a16(sizeof(float) * k);
  L.bv = (float*)p;    p += lap_a16(sizeof(float) * k);
  L.av = (float*)p;    p += lap_a16(sizeof(float) * k);
  L.t1 = (float*)p;    p += lap_a16(sizeof(float) * k);
  L.t2 = (float*)p;    p += lap_a16(sizeof(float) * k);
  L.s2 = (float*)p;    p += lap_a16(sizeof(float) * d);
  if (EV) {
    L.vv = (float*)p;    p += lap_a16(sizeof(float) * k);
    L.s2v = (float*)p;   p += lap_a16(sizeof(float) * k);
    L.betav = (float*)p; p += lap_a16(sizeof(float) * d);
    L.gout = (double*)p; p += lap_a16(sizeof(double) * (d + 2));
  }
  L.bad = (int*)p;
  return L;
}

// K = amp * kb + noise * I with kb cached in A's STRICT UPPER triangle
// (kb_aa = 1 implicit): row-i dot against a contiguous vector.  The b < i
// half reads the upper entries [b][i] (stride SA); the b > i half is the
// contiguous row tail.  Never touches A's lower (B/L/V live there).
__device__ inline float lap_kdot(const float* A, int SA, int k, int i,
                                 const float* vec, float amp, float noise) {
  const float s = dot4(A + i, SA, vec, 1, 0, i)
                + dotv(A + (size_t)i * SA, vec, i + 1, k);
  return amp * (s + vec[i]) + noise * vec[i];
}

// same against a strided column (the 8-wide chunk buffers); returns the
// NOISELESS product Kc q = amp * (kb q)
__device__ inline float lap_kcdot(const float* A, int SA, int k, int a,
                                  const float* q, int sq, float amp) {
  const float s = dot4(A + a, SA, q, sq, 0, a)
                + dotm(A + (size_t)a * SA, q, sq, a + 1, k);
  return amp * (s + q[a * sq]);
}

__device__ inline double log_sigmoid(double v) {
  // log(1/(1+exp(-v))) = -log1p(exp(-v)), stable for both signs
  if (v < -30.0) return v;
  return -log1p(exp(-v));
}

template <bool EV>
__global__ void __launch_bounds__(WG, 4)   // 4 waves/SIMD = 2 WGs/CU
fused_laplace_kernel(const float* __restrict__ Xg,   // [E, k, d]
                     const float* __restrict__ yg,   // [E, k]
                     float* __restrict__ fg,         // [E, k] inout
                     const float* __restrict__ scale,
                     const float amp, const float noise,
                     const int k, const int d,
                     const double tol, const int max_newton,
                     double* __restrict__ out_psi,      // [E]
                     double* __restrict__ out_sumlogl,  // [E]
                     int* __restrict__ out_iters,       // [E]
                     int* __restrict__ out_bad,         // [E]
                     double* __restrict__ out_logz,     // [E]     (EV)
                     double* __restrict__ out_grad) {   // [E,d+2] (EV)
  extern __shared__ char lds_raw[];
  const int SA = (int)lap_sa(k);
  LapLds S = lap_carve<EV>(lds_raw, k, d);
  const int e = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const float* Xe = Xg + (size_t)e * k * d;
  const int dp = (int)lap_dp4(d);

  // ---- stage X, y, f, s2 -------------------------------------------
  for (int i = tid; i < k * d; i += WG) {
    int a = i / d, j = i - a * d;
    S.X[a * dp + j] = Xe[i];
  }
  for (int i = tid; i < k; i += WG) {
    S.yb[i] = yg[(size_t)e * k + i];
    S.fb[i] = fg[(size_t)e * k + i];
  }
  for (int j = tid; j < d; j += WG) {
    float s = scale[j];
    S.s2[j] = s * s;
    if (EV) S.betav[j] = s;
  }
  if (tid == 0) *S.bad = 0;
  __syncthreads();

  // ---- Kb cache into A's STRICT UPPER (kb = exp(-q); kb_aa = 1) ----
  // Single-buffer design (round 2): the old separate full-K buffer cost
  // a second k x SA region and held the whole kernel at 1 WG/CU; with
  // the expert-NLL upper-cache idiom the kernel fits 2 WGs/CU.
  {
    const int nlow = k * (k + 1) / 2;
    for (int f = tid; f < nlow; f += WG) {
      int a, b;
      tri_decode(f, a, b);
      if (a == b) continue;
      const float* xa = S.X + (size_t)a * dp;
      const float* xb = S.X + (size_t)b * dp;
      float4 q4 = {0.f, 0.f, 0.f, 0.f};
      int j = 0;
      for (; j + 3 < d; j += 4) {
        const float4 va = *(const float4*)(xa + j);
        const float4 vb = *(const float4*)(xb + j);
        const float4 sv = *(const float4*)(S.s2 + j);
        const float u0 = va.x - vb.x, u1 = va.y - vb.y;
        const float u2 = va.z - vb.z, u3 = va.w - vb.w;
        q4.x += sv.x * u0 * u0;
        q4.y += sv.y * u1 * u1;
        q4.z += sv.z * u2 * u2;
        q4.w += sv.w * u3 * u3;
      }
      float q0 = (q4.x + q4.y) + (q4.z + q4.w);
      for (; j < d; ++j) {
        const float u = xa[j] - xb[j];
        q0 += S.s2[j] * u * u;
      }
      S.A[(size_t)b * SA + a] = __expf(-q0);      // upper: b < a
    }
  }
  __syncthreads();

  // ---- Newton loop (Algorithm 3.1 with step halving) ----------------
  double old_obj = -INFINITY;
  double new_obj = -1.7976931348623157e308;   // -DBL_MAX, as the reference
  double step = 1.0;
  int it = 0;
  for (; it < max_newton; ++it) {
    // pi, w, sqw, b
    for (int i = tid; i < k; i += WG) {
      const float fi = S.fb[i];
      const float p = 1.f / (1.f + __expf(-fi));
      const float w = p * (1.f - p);
      S.pi[i] = p;
      S.sqw[i] = sqrtf(w);
      S.bv[i] = w * fi + (S.yb[i] - p);
    }
    if (tid == 0) S.misc[0] = 0.0;
    __syncthreads();
    // t1 = K b
    for (int i = tid; i < k; i += WG)
      S.t1[i] = lap_kdot(S.A, SA, k, i, S.bv, amp, noise);
    __syncthreads();
    // u = sqw * (K b) into t1; build B (lower) into A from the upper cache
    for (int i = tid; i < k; i += WG) S.t1[i] *= S.sqw[i];
    {
      const int nlow = k * (k + 1) / 2;
      for (int f = tid; f < nlow; f += WG) {
        int a, b;
        tri_decode(f, a, b);
        const float kv = (a == b) ? (amp + noise)
                                  : amp * S.A[(size_t)b * SA + a];
        S.A[(size_t)a * SA + b] =
            (a == b ? 1.f : 0.f) + S.sqw[a] * kv * S.sqw[b];
      }
    }
    __syncthreads();
    chol_invert_lower(S.A, S.T, k, SA, tid, lane, S.bad, S.misc);
    if (*S.bad) break;
    // z = V u (lower); t2 = V^T z
    for (int i = tid; i < k; i += WG)
      S.t2[i] = dotv(S.A + (size_t)i * SA, S.t1, 0, i + 1);
    __syncthreads();
    for (int a = tid; a < k; a += WG)
      S.t1[a] = dotm(S.t2, S.A + a, SA, a, k);
    __syncthreads();
    // a = b - sqw * t1
    for (int i = tid; i < k; i += WG)
      S.av[i] = S.bv[i] - S.sqw[i] * S.t1[i];
    __syncthreads();
    // t1 = K a;  f_cand (t2) = (1-s) f + s K a
    for (int i = tid; i < k; i += WG)
      S.t1[i] = lap_kdot(S.A, SA, k, i, S.av, amp, noise);
    __syncthreads();
    const float sf = (float)step;
    double part = 0.0;
    for (int i = tid; i < k; i += WG) {
      const float fc = (1.f - sf) * S.fb[i] + sf * S.t1[i];
      S.t2[i] = fc;
      part += -0.5 * (double)S.av[i] * (double)fc
              + log_sigmoid((double)((2.f * S.yb[i] - 1.f) * fc));
    }
    const double obj_cand = block_sum(part, S.red, tid);
    if (obj_cand > old_obj) {
      for (int i = tid; i < k; i += WG) S.fb[i] = S.t2[i];
      old_obj = new_obj;
      new_obj = obj_cand;
    } else {
      step *= 0.5;
    }
    __syncthreads();
    if (!(fabs(old_obj - new_obj) > tol && step > tol)) { ++it; break; }
  }

  // ---- evidence tail (Algorithm 5.1, fused — K11) --------------------
  // classification/GaussianProcessClassifier.scala:113-128, computed at
  // the converged latent with the same exit-state semantics as
  // torch_backend.laplace_evidence_compiled (f_eval == f_final).  The
  // dK_i contractions never materialize [p, k, k]: the beta-gradient per
  // column j chains  s3_j = Bm_j - K sqw (V^T V (sqw Bm_j))  through the
  // triangular V from the final factor, 8 columns per pass through T.
  double logZ = 0.0;
  if (EV && !*S.bad) {
    // exit-state recompute at the converged f
    for (int i = tid; i < k; i += WG) {
      const float fi = S.fb[i];
      const float p = 1.f / (1.f + __expf(-fi));
      const float w = p * (1.f - p);
      S.pi[i] = p;
      S.sqw[i] = sqrtf(w);
      S.vv[i] = S.yb[i] - p;
      S.bv[i] = w * fi + S.vv[i];
    }
    if (tid == 0) S.misc[0] = 0.0;
    __syncthreads();
    {
      const int nlow = k * (k + 1) / 2;
      for (int f = tid; f < nlow; f += WG) {
        int a, b;
        tri_decode(f, a, b);
        const float kv = (a == b) ? (amp + noise)
                                  : amp * S.A[(size_t)b * SA + a];
        S.A[(size_t)a * SA + b] =
            (a == b ? 1.f : 0.f) + S.sqw[a] * kv * S.sqw[b];
      }
    }
    __syncthreads();
    chol_invert_lower(S.A, S.T, k, SA, tid, lane, S.bad, S.misc);
  }
  if (EV && !*S.bad) {
    // a = b - sqw V^T V (sqw (K b));  fc = K a;  psi;  logZ
    for (int i = tid; i < k; i += WG)
      S.t1[i] = lap_kdot(S.A, SA, k, i, S.bv, amp, noise) * S.sqw[i];
    __syncthreads();
    for (int i = tid; i < k; i += WG)
      S.t2[i] = dotv(S.A + (size_t)i * SA, S.t1, 0, i + 1);
    __syncthreads();
    for (int a = tid; a < k; a += WG)
      S.t1[a] = dotm(S.t2, S.A + a, SA, a, k);
    __syncthreads();
    for (int i = tid; i < k; i += WG)
      S.av[i] = S.bv[i] - S.sqw[i] * S.t1[i];
    __syncthreads();
    double part = 0.0;
    for (int i = tid; i < k; i += WG) {
      const float fc = lap_kdot(S.A, SA, k, i, S.av, amp, noise);
      part += -0.5 * (double)S.av[i] * (double)fc
              + log_sigmoid((double)((2.f * S.yb[i] - 1.f) * fc));
    }
    const double psi = block_sum(part, S.red, tid);
    logZ = psi - 0.5 * S.misc[0];

    // d3 into pi; zero diagKRK accumulators (t2) and gout
    for (int i = tid; i < k; i += WG) {
      const float p = S.pi[i];
      S.pi[i] = -(2.f * p - 1.f) * p * p * __expf(-S.fb[i]);
      S.t2[i] = 0.f;
    }
    for (int j = tid; j < d + 2; j += WG) S.gout[j] = 0.0;
    __syncthreads();
    // diagKRK_i = || V (sqw o K_:,i) ||^2  (V lower in A; K symmetric).
    // K rows come from the upper cache through a staged 8-row block in T
    // so the inner dots stay contiguous-contiguous.
    for (int i0 = 0; i0 < k; i0 += 8) {
      const int ib = min(8, k - i0);
      for (int f = tid; f < ib * k; f += WG) {
        const int il = f / k, c = f - il * k;
        const int i = i0 + il;
        const float kv = (c == i) ? (amp + noise)
                         : amp * ((c < i) ? S.A[(size_t)c * SA + i]
                                          : S.A[(size_t)i * SA + c]);
        S.T[il * SA + c] = S.sqw[c] * kv;
      }
      __syncthreads();
      for (int f = tid; f < ib * k; f += WG) {
        const int il = f / k, j = f - il * k;
        const float z = dotv(S.A + (size_t)j * SA, S.T + il * SA, 0, j + 1);
        atomicAdd(&S.t2[i0 + il], z * z);
      }
      __syncthreads();
    }
    // s2 vector (K_ii = amp + noise); then u0 = K v - nu v into t2
    for (int i = tid; i < k; i += WG)
      S.s2v[i] = -0.5f * (amp + noise - S.t2[i]) * S.pi[i];
    __syncthreads();
    for (int i = tid; i < k; i += WG)
      S.t1[i] = lap_kdot(S.A, SA, k, i, S.vv, amp, noise)
                - noise * S.vv[i];
    __syncthreads();
    for (int i = tid; i < k; i += WG) S.t2[i] = S.t1[i];   // u0
    __syncthreads();

    // ---- column chunks: d beta-columns, then Ba (amp) and v (noise) --
    float* wk1 = S.T;
    float* wk2 = S.T + 8 * k;
    float* bmb = S.T + 16 * k;
    float* qb = S.T + 24 * k;
    for (int j0 = 0; j0 < d + 2; j0 += 8) {
      const int cl = min(8, d + 2 - j0);
      for (int f = tid; f < k * 8; f += WG) {
        const int a = f >> 3, cc = f & 7;
        const int j = j0 + cc;
        float x = 0.f, w1 = 0.f, w2 = 0.f;
        if (cc < cl && j < d) {
          x = S.X[(size_t)a * dp + j];
          w1 = x * S.vv[a];
          w2 = x * x * S.vv[a];
        }
        qb[f] = x;
        wk1[f] = w1;
        wk2[f] = w2;
      }
      __syncthreads();
      // U1 = Kc (X_j o v) into bmb; then U2 = Kc (X_j^2 o v) into wk1
      for (int f = tid; f < k * 8; f += WG) {
        const int a = f >> 3, cc = f & 7;
        bmb[f] = lap_kcdot(S.A, SA, k, a, wk1 + cc, 8, amp);
      }
      __syncthreads();
      for (int f = tid; f < k * 8; f += WG) {
        const int a = f >> 3, cc = f & 7;
        const float u2 = lap_kcdot(S.A, SA, k, a, wk2 + cc, 8, amp);
        const int j = j0 + cc;
        float bm;
        if (j < d) {
          const float x = qb[f];
          bm = -2.f * S.betav[j]
               * (x * x * S.t2[a] - 2.f * x * bmb[f] + u2);
        } else if (j == d) {
          bm = S.t2[a] / amp;            // Ba = Kb v = u0 / C
        } else if (j == d + 1) {
          bm = S.vv[a];                  // noise column: dK = I
        } else {
          bm = 0.f;
        }
        // in-place into bmb: this thread is the only reader of bmb[f]
        // in this loop (the dotm reads wk2, not bmb)
        bmb[f] = bm;
      }
      __syncthreads();
      // chain: s3 = bm - K sqw (V^T V (sqw bm))
      for (int f = tid; f < k * 8; f += WG) {
        const int a = f >> 3;
        qb[f] = S.sqw[a] * bmb[f];
      }
      __syncthreads();
      for (int f = tid; f < k * 8; f += WG) {
        const int j = f >> 3, cc = f & 7;
        wk1[f] = dotm(S.A + (size_t)j * SA, qb + cc, 8, 0, j + 1);
      }
      __syncthreads();
      for (int f = tid; f < k * 8; f += WG) {
        const int a = f >> 3, cc = f & 7;
        wk2[f] = dot4(S.A + a, SA, wk1 + cc, 8, a, k);
      }
      __syncthreads();
      for (int f = tid; f < k * 8; f += WG) {
        const int a = f >> 3;
        qb[f] = S.sqw[a] * wk2[f];
      }
      __syncthreads();
      for (int f = tid; f < k * 8; f += WG) {
        const int a = f >> 3, cc = f & 7;
        wk2[f] = bmb[f] - (lap_kcdot(S.A, SA, k, a, qb + cc, 8, amp)
                           + noise * qb[a * 8 + cc]);
      }
      __syncthreads();
      // gout[j] += sum_a s2v[a] * s3[a][cc]   (wave cc owns column cc)
      {
        const int ww = tid >> 6;
        if (ww < cl) {
          double g = 0.0;
          for (int a = lane; a < k; a += 64)
            g += (double)S.s2v[a] * (double)wk2[a * 8 + ww];
          g = wave_sum(g);
          if (lane == 0) S.gout[j0 + ww] += g;
        }
      }
      __syncthreads();
    }

    // ---- lauum: Binv = V^T V in place (lower), then mirror -----------
    {
      const int nblk = (k + NB - 1) / NB;
      for (int I = 0; I < nblk; ++I) {
        const int ib = I * NB;
        const int bs = min(NB, k - ib);
        const int ncol = ib + bs;
        for (int f = tid; f < bs * ncol; f += WG) {
          const int r = f / ncol, j = f - r * ncol;
          const int i = ib + r;
          if (j > i) continue;
          S.T[r * SA + j] = dot4(S.A + i, SA, S.A + j, SA, i, k);
        }
        __syncthreads();
        for (int f = tid; f < bs * ncol; f += WG) {
          const int r = f / ncol, j = f - r * ncol;
          if (j > ib + r) continue;
          S.A[(size_t)(ib + r) * SA + j] = S.T[r * SA + j];
        }
        __syncthreads();
      }
    }
    // (no mirror: the W phase reads Binv from the lower triangle only,
    // and the strict upper must keep the Kb cache until W consumes it)

    // ---- W phase: scalars + W0 = (a a^T - R) o Kb in place -----------
    double trR_p = 0.0, sRKb_p = 0.0, aKba_p = 0.0, a2_p = 0.0;
    {
      const int nlow = k * (k + 1) / 2;
      for (int f = tid; f < nlow; f += WG) {
        int a, b;
        tri_decode(f, a, b);
        const float binv = S.A[(size_t)a * SA + b];
        const float R = S.sqw[a] * S.sqw[b] * binv;
        const float kb = (a == b) ? 1.f : S.A[(size_t)b * SA + a];
        const float g = S.av[a] * S.av[b] - R;
        const float w = g * kb;
        if (a == b) {
          trR_p += (double)R;
          sRKb_p += (double)R * kb;
          aKba_p += (double)S.av[a] * S.av[b] * kb;
          a2_p += (double)S.av[a] * S.av[a];
        } else {
          sRKb_p += 2.0 * (double)R * kb;
          aKba_p += 2.0 * (double)S.av[a] * S.av[b] * kb;
        }
        S.A[(size_t)a * SA + b] = w;
        if (a != b) S.A[(size_t)b * SA + a] = w;
      }
    }
    const double trR = block_sum(trR_p, S.red, tid);
    const double sRKb = block_sum(sRKb_p, S.red, tid);
    const double aKba = block_sum(aKba_p, S.red, tid);
    const double a2 = block_sum(a2_p, S.red, tid);
    if (tid == 0) {
      S.gout[d] += 0.5 * aKba - 0.5 * sRKb;        // s1 for amp
      S.gout[d + 1] += 0.5 * a2 - 0.5 * trR;       // s1 for noise
    }
    // r_a = row sums of W0 into t1
    for (int a = tid; a < k; a += WG) {
      const float* wr = S.A + (size_t)a * SA;
      float s0 = 0.f, s1 = 0.f, s2 = 0.f, s3 = 0.f;
      int b = 0;
      for (; b + 3 < k; b += 4) {
        s0 += wr[b]; s1 += wr[b + 1]; s2 += wr[b + 2]; s3 += wr[b + 3];
      }
      for (; b < k; ++b) s0 += wr[b];
      S.t1[a] = (s0 + s1) + (s2 + s3);
    }
    __syncthreads();
    // s1 for beta: per 8-col X chunks: WX = W0 X; s1_j = -b_j C (2t1-2t2)
    for (int d0 = 0; d0 < d; d0 += 8) {
      const int cl = min(8, d - d0);
      for (int f = tid; f < k * 8; f += WG) {
        const int a = f >> 3, cc = f & 7;
        qb[f] = (cc < cl) ? S.X[(size_t)a * dp + d0 + cc] : 0.f;
      }
      __syncthreads();
      for (int f = tid; f < k * 8; f += WG) {
        const int a = f >> 3, cc = f & 7;
        wk1[f] = dotm(S.A + (size_t)a * SA, qb + cc, 8, 0, k);
      }
      __syncthreads();
      {
        const int ww = tid >> 6;
        if (ww < cl) {
          double g = 0.0;
          for (int a = lane; a < k; a += 64) {
            const float x = qb[a * 8 + ww];
            g += 2.0 * (double)x
                 * ((double)x * (double)S.t1[a] - (double)wk1[a * 8 + ww]);
          }
          g = wave_sum(g);
          if (lane == 0)
            S.gout[d0 + ww] += -(double)S.betav[d0 + ww] * (double)amp * g;
        }
      }
      __syncthreads();
    }
  }

  // ---- outputs -------------------------------------------------------
  for (int i = tid; i < k; i += WG) fg[(size_t)e * k + i] = S.fb[i];
  if (EV) {
    if (tid == 0) out_logz[e] = logZ;
    for (int j = tid; j < d + 2; j += WG)
      out_grad[(size_t)e * (d + 2) + j] = (*S.bad) ? 0.0 : S.gout[j];
  }
  if (tid == 0) {
    out_bad[e] = *S.bad;
    out_psi[e] = new_obj;
    // misc[0] = sum log(raw pivots) = 2 sum log diag L  (last iteration)
    out_sumlogl[e] = 0.5 * S.misc[0];
    out_iters[e] = it;
  }
}

extern "C" hipError_t launch_fused_laplace_newton(
    const float* X, const float* y, float* f, const float* scale, float amp,
    float noise, int E, int k, int d, double tol, int max_newton,
    double* out_psi, double* out_sumlogl, int* out_iters, int* out_bad,
    hipStream_t stream, size_t* lds_used) {
  size_t lds = lap_lds_bytes(k, d);
  if (lds_used) *lds_used = lds;
  if (lds > 160 * 1024 || k > 128 || d > k)
    return hipErrorInvalidConfiguration;
  hipLaunchKernelGGL(fused_laplace_kernel<false>, dim3(E), dim3(WG), lds,
                     stream, X, y, f, scale, amp, noise, k, d, tol,
                     max_newton, out_psi, out_sumlogl, out_iters, out_bad,
                     nullptr, nullptr);
  return hipGetLastError();
}

extern "C" hipError_t launch_fused_laplace_evidence(
    const float* X, const float* y, float* f, const float* scale, float amp,
    float noise, int E, int k, int d, double tol, int max_newton,
    double* out_psi, double* out_sumlogl, int* out_iters, int* out_bad,
    double* out_logz, double* out_grad, hipStream_t stream,
    size_t* lds_used) {
  size_t lds = lap_lds_bytes(k, d, 1);
  if (lds_used) *lds_used = lds;
  if (lds > 160 * 1024 || k > 128 || d > k)
    return hipErrorInvalidConfiguration;
  hipLaunchKernelGGL(fused_laplace_kernel<true>, dim3(E), dim3(WG), lds,
                     stream, X, y, f, scale, amp, noise, k, d, tol,
                     max_newton, out_psi, out_sumlogl, out_iters, out_bad,
                     out_logz, out_grad);
  return hipGetLastError();
}
