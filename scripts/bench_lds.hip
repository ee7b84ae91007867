// Ground-truth microbenchmarks for the expert-NLL inner-loop patterns.
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void __launch_bounds__(512)
bench(float* out, unsigned long long* cyc, int which, int reps) {
  __shared__ float A[101 * 104];
  const int tid = threadIdx.x;
  for (int i = tid; i < 101 * 104; i += 512)
    A[i] = 1.0f + (i & 7) * 1e-3f;
  __syncthreads();
  unsigned long long t0 = wall_clock64();
  float acc = 0.f;
  const int SA = 101, k = 100;
  if (which == 0) {
    // E-like: k threads, each dot over k with 2 strided LDS reads (2-acc)
    for (int rep = 0; rep < reps; ++rep) {
      for (int a = tid; a < k; a += 512) {
        float s0 = 0, s1 = 0;
        int c = 0;
        for (; c + 1 < k; c += 2) {
          s0 += A[c * SA + a] * A[c];
          s1 += A[(c + 1) * SA + a] * A[c + 1];
        }
        acc += s0 + s1;
      }
    }
  } else if (which == 1) {
    // W-like: all 512 threads, tri elements, 1 LDS read pair each + exp
    for (int rep = 0; rep < reps; ++rep) {
      for (int f = tid; f < k * (k + 1) / 2; f += 512) {
        int a = (int)((sqrtf(8.f * f + 1.f) - 1.f) * 0.5f);
        while ((a + 1) * (a + 2) / 2 <= f) ++a;
        while (a * (a + 1) / 2 > f) --a;
        int b = f - a * (a + 1) / 2;
        acc += A[a * SA + b] * __expf(-A[b * SA + a]);
      }
    }
  } else if (which == 2) {
    // H-like: k*32 elements, each a k-dot with 2 LDS reads (4-acc)
    for (int rep = 0; rep < reps; ++rep) {
      for (int f = tid; f < k * 32; f += 512) {
        int a = f / 32, j = f & 31;
        float s0 = 0, s1 = 0, s2 = 0, s3 = 0;
        int b = 0;
        for (; b + 3 < k; b += 4) {
          s0 += A[a * SA + b] * A[b * SA + j];
          s1 += A[a * SA + b + 1] * A[(b + 1) * SA + j];
          s2 += A[a * SA + b + 2] * A[(b + 2) * SA + j];
          s3 += A[a * SA + b + 3] * A[(b + 3) * SA + j];
        }
        acc += (s0 + s1) + (s2 + s3);
      }
    }
  } else if (which == 3) {
    // pure dependent LDS chain, 1 thread per wave
    int idx = tid & 63;
    for (int rep = 0; rep < reps * 100; ++rep)
      idx = (int)A[idx & 1023] + (idx & 255);
    acc = idx;
  }
  __syncthreads();
  unsigned long long t1 = wall_clock64();
  if (tid == 0) cyc[blockIdx.x] = t1 - t0;
  out[tid] = acc;
}

int main(int argc, char** argv) {
  int blocks = argc > 1 ? atoi(argv[1]) : 1;
  float* out; unsigned long long* cyc;
  hipMalloc(&out, 512 * 4); hipMalloc(&cyc, blocks * 8);
  const int reps = 50;
  for (int w = 0; w < 4; ++w) {
    hipLaunchKernelGGL(bench, dim3(blocks), dim3(512), 0, 0, out, cyc, w, reps);
    hipDeviceSynchronize();
    std::vector<unsigned long long> h(blocks);
    hipMemcpy(h.data(), cyc, blocks * 8, hipMemcpyDeviceToHost);
    double us = (double)h[0] / 1e8 * 1e6 / reps;
    printf("pattern %d: %.3f us per rep (blocks=%d)\n", w, us, blocks);
  }
  return 0;
}
