// K19 — device-side synthetic data generation (Philox4x32-10 counter RNG)
// for the benchmark harness (PerformanceBenchmark.scala:24-36 generates on
// the driver; at 100M x 128 rows host generation + H2D is minutes and
// >100 GB of host RAM, so the MI355X path generates in place in HBM).
//
// X[i, j] ~ U[0, 1) fp32, y[i] = sin(2 * sum_j X[i, j]) + noise_sd * N(0,1)
// — the same distribution as data/synthetic.py's benchmark_regression_data
// (counter-based: any (seed, row) pair is reproducible independent of the
// launch geometry).

#include <hip/hip_runtime.h>
#include <math.h>

__device__ inline void philox_round(unsigned int& c0, unsigned int& c1,
                                    unsigned int& c2, unsigned int& c3,
                                    unsigned int k0, unsigned int k1) {
  const unsigned long long m0 =
      (unsigned long long)0xD2511F53u * c0;
  const unsigned long long m1 =
      (unsigned long long)0xCD9E8D57u * c2;
  const unsigned int h0 = (unsigned int)(m0 >> 32), l0 = (unsigned int)m0;
  const unsigned int h1 = (unsigned int)(m1 >> 32), l1 = (unsigned int)m1;
  c0 = h1 ^ c1 ^ k0;
  c1 = l1;
  c2 = h0 ^ c3 ^ k1;
  c3 = l0;
}

// Philox4x32-10: 4 uint32 lanes from a 128-bit counter + 64-bit key
__device__ inline void philox4(unsigned long long ctr, unsigned long long seed,
                               unsigned int out[4]) {
  unsigned int c0 = (unsigned int)ctr, c1 = (unsigned int)(ctr >> 32);
  unsigned int c2 = 0, c3 = 0;
  unsigned int k0 = (unsigned int)seed, k1 = (unsigned int)(seed >> 32);
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    philox_round(c0, c1, c2, c3, k0, k1);
    k0 += 0x9E3779B9u;       // golden-ratio Weyl
    k1 += 0xBB67AE85u;
  }
  out[0] = c0; out[1] = c1; out[2] = c2; out[3] = c3;
}

__device__ inline float u01(unsigned int v) {
  return (float)(v >> 8) * (1.0f / 16777216.0f);      // [0, 1), 24-bit
}

extern "C" __global__ void __launch_bounds__(256)
synth_regression_kernel(float* __restrict__ X,   // [n, d]
                        float* __restrict__ y,   // [n]
                        const long n, const int d,
                        const unsigned long long seed,
                        const float noise_sd) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  float s = 0.f;
  // row i consumes counters [i*(d+3)/4 chunks]: 4 values per philox call
  const long base = i * (long)((d + 3) / 4 + 1);
  unsigned int r[4];
  for (int j0 = 0; j0 < d; j0 += 4) {
    philox4(base + j0 / 4, seed, r);
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      const int j = j0 + u;
      if (j < d) {
        const float v = u01(r[u]);
        X[i * d + j] = v;
        s += v;
      }
    }
  }
  // noise: Box-Muller from the row's last counter
  philox4(base + (d + 3) / 4, seed, r);
  const float u1 = u01(r[0]) + 5.96046448e-08f;       // avoid log(0)
  const float u2 = u01(r[1]);
  const float z = sqrtf(-2.0f * __logf(u1)) * __cosf(6.2831853f * u2);
  y[i] = __sinf(2.0f * s) + noise_sd * z;
}

extern "C" hipError_t launch_synth_regression(float* X, float* y, long n,
                                              int d, unsigned long long seed,
                                              float noise_sd,
                                              hipStream_t stream) {
  const long blocks = (n + 255) / 256;
  hipLaunchKernelGGL(synth_regression_kernel, dim3((unsigned)blocks),
                     dim3(256), 0, stream, X, y, n, d, seed, noise_sd);
  return hipGetLastError();
}
