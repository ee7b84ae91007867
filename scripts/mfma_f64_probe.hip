// Empirical lane->element mapping probe for v_mfma_f64_16x16x4_f64 on
// gfx950.  Three runs print D per (lane, reg):
//   run A: a[l] = 1+l, b[l] = (l==0)  -> nonzeros reveal the A i-map (for
//          the k of lane 0) and the D row/col map;
//   run B: a[l] = (l==0), b[l] = 1+l  -> reveals the B j/k-map;
//   run K: a[l] = (l==16), b[l] = 1+l -> k of lane 16 (is k l>>4 or l&3?);
//   run F: a[l] = 1+l, b[l] = 101+l   -> full product for validation.
// Compile ON the GPU box: hipcc --offload-arch=gfx950 -O1 this -o probe
#include <hip/hip_runtime.h>
#include <stdio.h>

typedef __attribute__((ext_vector_type(4))) double f64x4;

__global__ void probe(const double* a, const double* b, double* d) {
  const int l = threadIdx.x;
  f64x4 acc = {0.0, 0.0, 0.0, 0.0};
  acc = __builtin_amdgcn_mfma_f64_16x16x4f64(a[l], b[l], acc, 0, 0, 0);
  for (int r = 0; r < 4; ++r) d[l * 4 + r] = acc[r];
}

static void run(const char* name, const double* ha, const double* hb) {
  double *da, *db, *dd;
  double hd[256];
  hipMalloc(&da, 512);
  hipMalloc(&db, 512);
  hipMalloc(&dd, 2048);
  hipMemcpy(da, ha, 512, hipMemcpyHostToDevice);
  hipMemcpy(db, hb, 512, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, da, db, dd);
  hipMemcpy(hd, dd, 2048, hipMemcpyDeviceToHost);
  hipDeviceSynchronize();
  printf("RUN %s\n", name);
  for (int l = 0; l < 64; ++l)
    for (int r = 0; r < 4; ++r)
      if (hd[l * 4 + r] != 0.0)
        printf("%s l=%d r=%d v=%.1f\n", name, l, r, hd[l * 4 + r]);
  hipFree(da); hipFree(db); hipFree(dd);
}

int main() {
  double ha[64], hb[64];
  for (int l = 0; l < 64; ++l) { ha[l] = 1 + l; hb[l] = (l == 0); }
  run("A", ha, hb);
  for (int l = 0; l < 64; ++l) { ha[l] = (l == 0); hb[l] = 1 + l; }
  run("B", ha, hb);
  for (int l = 0; l < 64; ++l) { ha[l] = (l == 16); hb[l] = 1 + l; }
  run("K", ha, hb);
  for (int l = 0; l < 64; ++l) { ha[l] = 1 + l; hb[l] = 101 + l; }
  run("F", ha, hb);
  printf("done\n");
  return 0;
}
