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
// the batched torch path for that expert).  The evidence/gradient pass
// (Algorithm 5.1) runs on the torch path AT the converged f in contraction
// form (torch_backend.laplace_evidence_compiled — one batched fp64
// Cholesky, no [E, p, k, k] tensor, reference-exact semantics); if any
// expert went bad, the whole batch falls back to the torch Newton loop
// warm-started from f.
//
// Constraints: k <= 128, d <= k (X stages through the A buffer).

#include <hip/hip_runtime.h>
#include <math.h>

#define WG 512
#include "linalg_lds.h"

struct LapLds {
  float* KB;    // k * SA: full K = amp*Kb + noise*I (both triangles)
  float* A;     // k * SA: X staging, then B -> L -> V per iteration
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

static __host__ __device__ inline size_t lap_lds_bytes(int k, int d) {
  size_t off = lap_a16(sizeof(double) * 10);
  off += 2 * lap_a16(sizeof(float) * (size_t)k * lap_sa(k));  // KB + A
  off += lap_a16(sizeof(float) * lap_tsz(k));
  off += 8 * lap_a16(sizeof(float) * k);
  off += lap_a16(sizeof(float) * d);
  off += 16;
  return off;
}

__device__ inline LapLds lap_carve(char* base, int k, int d) {
  LapLds L;
  char* p = base;
  L.red = (double*)p;  L.misc = L.red + 8;
  p += lap_a16(sizeof(double) * 10);
  L.KB = (float*)p;    p += lap_a16(sizeof(float) * (size_t)k * lap_sa(k));
  L.A = (float*)p;     p += lap_a16(sizeof(float) * (size_t)k * lap_sa(k));
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
  L.bad = (int*)p;
  return L;
}

__device__ inline double log_sigmoid(double v) {
  // log(1/(1+exp(-v))) = -log1p(exp(-v)), stable for both signs
  if (v < -30.0) return v;
  return -log1p(exp(-v));
}

extern "C" __global__ void __launch_bounds__(WG)
fused_laplace_newton_kernel(const float* __restrict__ Xg,   // [E, k, d]
                            const float* __restrict__ yg,   // [E, k]
                            float* __restrict__ fg,         // [E, k] inout
                            const float* __restrict__ scale,
                            const float amp, const float noise,
                            const int k, const int d,
                            const double tol, const int max_newton,
                            double* __restrict__ out_psi,      // [E]
                            double* __restrict__ out_sumlogl,  // [E]
                            int* __restrict__ out_iters,       // [E]
                            int* __restrict__ out_bad) {       // [E]
  extern __shared__ char lds_raw[];
  const int SA = (int)lap_sa(k);
  LapLds S = lap_carve(lds_raw, k, d);
  const int e = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const float* Xe = Xg + (size_t)e * k * d;

  // ---- stage X (into A), y, f, s2 ----------------------------------
  for (int i = tid; i < k * d; i += WG) {
    int a = i / d, j = i - a * d;
    S.A[a * SA + j] = Xe[i];
  }
  for (int i = tid; i < k; i += WG) {
    S.yb[i] = yg[(size_t)e * k + i];
    S.fb[i] = fg[(size_t)e * k + i];
  }
  for (int j = tid; j < d; j += WG) {
    float s = scale[j];
    S.s2[j] = s * s;
  }
  if (tid == 0) *S.bad = 0;
  __syncthreads();

  // ---- KB = amp * exp(-q) + noise I (full symmetric) ----------------
  {
    const int nlow = k * (k + 1) / 2;
    for (int f = tid; f < nlow; f += WG) {
      int a, b;
      tri_decode(f, a, b);
      const float* xa = S.A + a * SA;
      const float* xb = S.A + b * SA;
      float q0 = 0.f, q1 = 0.f;
      int j = 0;
      for (; j + 1 < d; j += 2) {
        float u0 = xa[j] - xb[j];
        float u1 = xa[j + 1] - xb[j + 1];
        q0 += S.s2[j] * u0 * u0;
        q1 += S.s2[j + 1] * u1 * u1;
      }
      if (j < d) { float u = xa[j] - xb[j]; q0 += S.s2[j] * u * u; }
      const float kv = amp * __expf(-(q0 + q1)) + (a == b ? noise : 0.f);
      S.KB[a * SA + b] = kv;
      if (a != b) S.KB[b * SA + a] = kv;
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
      S.t1[i] = dotv(S.KB + (size_t)i * SA, S.bv, 0, k);
    __syncthreads();
    // u = sqw * (K b) into t1; build B (lower) into A
    for (int i = tid; i < k; i += WG) S.t1[i] *= S.sqw[i];
    {
      const int nlow = k * (k + 1) / 2;
      for (int f = tid; f < nlow; f += WG) {
        int a, b;
        tri_decode(f, a, b);
        S.A[(size_t)a * SA + b] =
            (a == b ? 1.f : 0.f) + S.sqw[a] * S.KB[(size_t)a * SA + b] * S.sqw[b];
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
      S.t1[i] = dotv(S.KB + (size_t)i * SA, S.av, 0, k);
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

  // ---- outputs -------------------------------------------------------
  for (int i = tid; i < k; i += WG) fg[(size_t)e * k + i] = S.fb[i];
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
  hipLaunchKernelGGL(fused_laplace_newton_kernel, dim3(E), dim3(WG), lds,
                     stream, X, y, f, scale, amp, noise, k, d, tol,
                     max_newton, out_psi, out_sumlogl, out_iters, out_bad);
  return hipGetLastError();
}
