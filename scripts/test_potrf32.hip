// Standalone check of the register potrf + in-place trtri micro-kernels.
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cmath>
#include <vector>
#define NB 32

__global__ void __launch_bounds__(64)
diag_kernel(const float* in, float* out, int bs, int SA) {
  __shared__ float D[NB * (NB + 1)];
  const int lane = threadIdx.x & 63;
  const int row = lane & 31;
  for (int i = threadIdx.x; i < NB * (NB + 1); i += 64) D[i] = in[i];
  __syncthreads();
  {
    float r[NB];
#pragma unroll
    for (int c = 0; c < NB; ++c) {
      const bool inb = (row < bs) && (c <= row);
      r[c] = inb ? D[row * SA + c] : (c == row ? 1.f : 0.f);
    }
#pragma unroll
    for (int s = 0; s < NB; ++s) {
#ifdef USE_SHFL
      const float ajj = __shfl(r[s], s, 64);
#else
      const float ajj = __builtin_amdgcn_readlane(r[s], s);
#endif
      const float rinv = rsqrtf(ajj);
      const float lis = (row == s) ? ajj * rinv : r[s] * rinv;
      r[s] = lis;
#ifdef NOP_FIX
      asm volatile("s_nop 3");   // VALU-write -> v_readlane hazard
#endif
      if (lane < 32 && row >= s && s < bs && row < bs)
        D[row * SA + s] = lis;
#pragma unroll
      for (int c = s + 1; c < NB; ++c) {
#ifdef USE_SHFL
        const float lcs = __shfl(lis, c, 64);
#else
        const float lcs = __builtin_amdgcn_readlane(lis, c);
#endif
#ifdef GUARD_IF
        if (c <= row) r[c] -= lis * lcs;
#else
        r[c] -= (c <= row ? lis : 0.f) * lcs;
#endif
      }
#ifndef NO_SCHEDBAR
      __builtin_amdgcn_sched_barrier(0);
#endif
    }
  }
  __syncwarp();
#ifndef SKIP_TRTRI
  for (int i = 0; i < bs; ++i) {
    const float* Li = D + i * SA;
    float s0 = 0.f, s1 = 0.f;
    int c = row;
    for (; c + 1 < i; c += 2) {
      s0 += Li[c] * D[c * SA + row];
      s1 += Li[c + 1] * D[(c + 1) * SA + row];
    }
    if (c < i) s0 += Li[c] * D[c * SA + row];
    const float Lii = Li[i];
    const float vi = (row < i) ? (-(s0 + s1) / Lii)
                               : (row == i ? 1.f / Lii : 0.f);
    __syncwarp();
    if (lane < 32 && row <= i) D[i * SA + row] = vi;
    __syncwarp();
  }
#endif
  __syncthreads();
  for (int i = threadIdx.x; i < NB * (NB + 1); i += 64) out[i] = D[i];
}

int main() {
  const int bs = NB, SA = NB + 1;
  std::vector<float> h(NB * SA, 0.f);
  std::vector<double> K(NB * NB);
  srand(7);
  std::vector<double> Xr(NB * 8);
  for (auto& v : Xr) v = (double)rand() / RAND_MAX;
  for (int a = 0; a < NB; ++a)
    for (int b = 0; b < NB; ++b) {
      double q = 0;
      for (int j = 0; j < 8; ++j) {
        double t = Xr[a * 8 + j] - Xr[b * 8 + j];
        q += t * t;
      }
      K[a * NB + b] = exp(-q) + (a == b ? 1e-3 : 0.0);
    }
  for (int a = 0; a < NB; ++a)
    for (int b = 0; b <= a; ++b) h[a * SA + b] = (float)K[a * NB + b];

  // reference: double cholesky then inverse of L
  std::vector<double> L(NB * NB, 0.0);
  for (int j = 0; j < NB; ++j) {
    double s = K[j * NB + j];
    for (int t = 0; t < j; ++t) s -= L[j * NB + t] * L[j * NB + t];
    L[j * NB + j] = sqrt(s);
    for (int i = j + 1; i < NB; ++i) {
      double v = K[i * NB + j];
      for (int t = 0; t < j; ++t) v -= L[i * NB + t] * L[j * NB + t];
      L[i * NB + j] = v / L[j * NB + j];
    }
  }
  std::vector<double> V(NB * NB, 0.0);
  for (int j = 0; j < NB; ++j) {
    V[j * NB + j] = 1.0 / L[j * NB + j];
    for (int i = j + 1; i < NB; ++i) {
      double s = 0;
      for (int c = j; c < i; ++c) s += L[i * NB + c] * V[c * NB + j];
      V[i * NB + j] = -s / L[i * NB + i];
    }
  }

  float *din, *dout;
  hipMalloc(&din, h.size() * 4);
  hipMalloc(&dout, h.size() * 4);
  hipMemcpy(din, h.data(), h.size() * 4, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(diag_kernel, dim3(1), dim3(64), 0, 0, din, dout, bs, SA);
  hipDeviceSynchronize();
  std::vector<float> out(h.size());
  hipMemcpy(out.data(), dout, h.size() * 4, hipMemcpyDeviceToHost);

#ifdef SKIP_TRTRI
  const std::vector<double>& ref = L;
  const char* what = "potrf";
#else
  const std::vector<double>& ref = V;
  const char* what = "trtri";
#endif
  double maxe = 0;
  int bi = -1, bj = -1;
  for (int i = 0; i < NB; ++i)
    for (int j = 0; j <= i; ++j) {
      double e = fabs(out[i * SA + j] - ref[i * NB + j]);
      if (e > maxe) { maxe = e; bi = i; bj = j; }
    }
  printf("%s max err %.3e at (%d,%d): got %.6f want %.6f\n", what, maxe,
         bi, bj, out[bi * SA + bj], ref[bi * NB + bj]);
  return 0;
}
