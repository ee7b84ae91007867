// PyTorch bindings for the spark_gp_amd CDNA4 HIP kernels.

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>

#include <vector>

extern "C" hipError_t launch_fused_expert_nll(
    const float* X, const float* y, const float* scale, float amp,
    float noise, int E, int k, int d, double* out_nll, double* out_sumW0,
    double* out_trG, double* out_contr, int* out_bad,
    unsigned long long* out_clk, hipStream_t stream, size_t* lds_used);

extern "C" hipError_t launch_cross_kernel_tile(
    const float* X, const float* A, const float* s2v, float amp, int c,
    int m, int d, void* out, void* out_lo, void* out_t, void* out_lo_t,
    int out_is_bf16, const float* yv, double* Ky, hipStream_t stream);

extern "C" hipError_t launch_syrk_bf16(const void* KcT, const void* KlT,
                                       int c, int m, int cpitch, int split_k,
                                       float* KK, hipStream_t stream);

extern "C" hipError_t launch_syrk_bf16_sync(const void* KcT, const void* KlT,
                                            int c, int m, int cpitch,
                                            const int* tiles, int nb,
                                            int kpb, int* phase_ctr,
                                            int nactive, float* KK,
                                            hipStream_t stream);

extern "C" hipError_t launch_cross_mfma(const float* Xs, const float* As,
                                         const float* nx, const float* na,
                                         float amp, int c, int m, int d,
                                         void* out_bfT, void* out_loT,
                                         const float* yv, double* Ky,
                                         hipStream_t stream);

extern "C" hipError_t launch_colsum_gemv(const void* Kc, const float* y,
                                         int c, int m, double* Ky,
                                         hipStream_t stream);

extern "C" hipError_t launch_fused_laplace_evidence(
    const float* X, const float* y, float* f, const float* scale, float amp,
    float noise, int E, int k, int d, double tol, int max_newton,
    double* out_psi, double* out_sumlogl, int* out_iters, int* out_bad,
    double* out_logz, double* out_grad, hipStream_t stream,
    size_t* lds_used);

extern "C" hipError_t launch_synth_regression(float* X, float* y, long n,
                                              int d, unsigned long long seed,
                                              float noise_sd,
                                              hipStream_t stream);

extern "C" hipError_t launch_dpotrf_diag(double* A, long m, int jb,
                                         double* Vout, int* bad,
                                         hipStream_t stream);

extern "C" hipError_t launch_dgemm64(int ta, int tb, int sub, int syrk,
                                     const double* A, const double* B,
                                     double* C, int M, int N, int K,
                                     long lda, long ldb, long ldc,
                                     hipStream_t stream);

extern "C" hipError_t launch_fused_laplace_newton(
    const float* X, const float* y, float* f, const float* scale, float amp,
    float noise, int E, int k, int d, double tol, int max_newton,
    double* out_psi, double* out_sumlogl, int* out_iters, int* out_bad,
    hipStream_t stream, size_t* lds_used);

namespace {

void check_hip(hipError_t err, const char* what) {
  TORCH_CHECK(err == hipSuccess, what, ": ", hipGetErrorString(err));
}

hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

}  // namespace

// Returns (nll[E] f64, sumW0[E] f64, trG[E] f64, contr[E,d] f64, bad[E] i32)
std::vector<torch::Tensor> fused_expert_nll_impl(torch::Tensor X,
                                                 torch::Tensor y,
                                                 torch::Tensor scale,
                                                 double amp, double noise,
                                                 bool profile) {
  TORCH_CHECK(X.is_cuda() && X.dtype() == torch::kFloat32 && X.dim() == 3,
              "X must be [E, k, d] float32 on GPU");
  TORCH_CHECK(y.is_cuda() && y.dtype() == torch::kFloat32 && y.dim() == 2);
  TORCH_CHECK(scale.is_cuda() && scale.dtype() == torch::kFloat32);
  auto Xc = X.contiguous();
  auto yc = y.contiguous();
  auto sc = scale.contiguous();
  const int E = X.size(0), k = X.size(1), d = X.size(2);
  TORCH_CHECK(k <= 128 && d <= 128,
              "fused_expert_nll requires k<=128, d<=128");
  auto opts64 = torch::TensorOptions().dtype(torch::kFloat64).device(X.device());
  auto opts32i = torch::TensorOptions().dtype(torch::kInt32).device(X.device());
  auto nll = torch::empty({E}, opts64);
  auto sumW0 = torch::empty({E}, opts64);
  auto trG = torch::empty({E}, opts64);
  auto contr = torch::empty({E, d}, opts64);
  auto bad = torch::empty({E}, opts32i);
  torch::Tensor clk;
  unsigned long long* clk_ptr = nullptr;
  if (profile) {
    clk = torch::zeros({E, 20},
                       torch::TensorOptions().dtype(torch::kInt64)
                           .device(X.device()));
    clk_ptr = (unsigned long long*)clk.data_ptr<int64_t>();
  }
  size_t lds = 0;
  check_hip(launch_fused_expert_nll(
                Xc.data_ptr<float>(), yc.data_ptr<float>(),
                sc.data_ptr<float>(), (float)amp, (float)noise, E, k, d,
                nll.data_ptr<double>(), sumW0.data_ptr<double>(),
                trG.data_ptr<double>(), contr.data_ptr<double>(),
                bad.data_ptr<int>(), clk_ptr, current_stream(), &lds),
            "fused_expert_nll");
  if (profile) return {nll, sumW0, trG, contr, bad, clk};
  return {nll, sumW0, trG, contr, bad};
}

std::vector<torch::Tensor> fused_expert_nll(torch::Tensor X, torch::Tensor y,
                                            torch::Tensor scale, double amp,
                                            double noise) {
  return fused_expert_nll_impl(X, y, scale, amp, noise, false);
}

namespace {
inline int64_t a16i(int64_t n) { return (n + 15) & ~(int64_t)15; }
inline int64_t sa_of(int64_t k) { return (k + 4) & ~(int64_t)3; }
inline int64_t tsz_of(int64_t k) {
  return std::max<int64_t>(std::max<int64_t>(k * 36, 32 * sa_of(k)), 448);
}
}  // namespace

bool fused_expert_nll_supported(int64_t k, int64_t d) {
  if (k > 128 || d > 128 || k < 1) return false;
  // LDS budget: EXACT mirror of nll_lds_bytes2 in expert_nll.hip
  const int64_t dp4 = (d + 4) & ~(int64_t)3;
  int64_t bytes = a16i(8 * 10) + a16i(4 * k * sa_of(k)) +
                  a16i(4 * tsz_of(k)) + a16i(4 * k * dp4) +
                  4 * a16i(4 * k) + a16i(4 * d) + 16;
  return bytes <= 160 * 1024;
}

std::vector<torch::Tensor> cross_kernel_tile(torch::Tensor X, torch::Tensor A,
                                             torch::Tensor s2v, double amp,
                                             bool bf16_out, bool hilo,
                                             bool want_t) {
  TORCH_CHECK(X.is_cuda() && X.dtype() == torch::kFloat32 && X.dim() == 2);
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kFloat32 && A.dim() == 2);
  TORCH_CHECK(X.size(1) == A.size(1), "feature dims differ");
  TORCH_CHECK(!hilo || bf16_out, "hilo requires bf16 output");
  TORCH_CHECK(!want_t || hilo, "transposed outputs require hilo bf16 mode");
  auto Xc = X.contiguous();
  auto Ac = A.contiguous();
  auto sc = s2v.contiguous();
  const int c = X.size(0), m = A.size(0), d = X.size(1);
  auto opts = torch::TensorOptions()
                  .dtype(bf16_out ? torch::kBFloat16 : torch::kFloat32)
                  .device(X.device());
  auto out = torch::empty({c, m}, opts);
  torch::Tensor lo, outT, loT;
  if (hilo) lo = torch::empty({c, m}, opts);
  if (want_t) {
    outT = torch::empty({m, c}, opts);
    loT = torch::empty({m, c}, opts);
  }
  check_hip(launch_cross_kernel_tile(Xc.data_ptr<float>(), Ac.data_ptr<float>(),
                                     sc.data_ptr<float>(), (float)amp, c, m, d,
                                     out.data_ptr(),
                                     hilo ? lo.data_ptr() : nullptr,
                                     want_t ? outT.data_ptr() : nullptr,
                                     want_t ? loT.data_ptr() : nullptr,
                                     bf16_out ? 1 : 0, nullptr, nullptr,
                                     current_stream()),
            "cross_kernel_tile");
  if (want_t) return {out, lo, outT, loT};
  if (hilo) return {out, lo};
  return {out};
}

// MFMA PPA tile: pre-scaled inputs + norms; K1's sqdist-via-GEMM plan.
std::vector<torch::Tensor> cross_mfma_ppa(torch::Tensor Xs, torch::Tensor As,
                                          torch::Tensor nx, torch::Tensor na,
                                          double amp, torch::Tensor y,
                                          torch::Tensor Ky) {
  TORCH_CHECK(Xs.is_cuda() && Xs.dtype() == torch::kFloat32 && Xs.dim() == 2);
  TORCH_CHECK(As.is_cuda() && As.dtype() == torch::kFloat32 && As.dim() == 2);
  TORCH_CHECK(Xs.size(1) == As.size(1));
  const int c = Xs.size(0), m = As.size(0), d = Xs.size(1);
  TORCH_CHECK(nx.numel() == c && na.numel() == m && y.numel() == c);
  TORCH_CHECK(Ky.is_cuda() && Ky.dtype() == torch::kFloat64 &&
              Ky.numel() == m);
  auto Xc = Xs.contiguous();
  auto Ac = As.contiguous();
  auto nxc = nx.contiguous();
  auto nac = na.contiguous();
  auto yc = y.contiguous();
  auto opts = torch::TensorOptions().dtype(torch::kBFloat16)
                  .device(Xs.device());
  auto outT = torch::empty({m, c}, opts);
  auto loT = torch::empty({m, c}, opts);
  check_hip(launch_cross_mfma(Xc.data_ptr<float>(), Ac.data_ptr<float>(),
                              nxc.data_ptr<float>(), nac.data_ptr<float>(),
                              (float)amp, c, m, d, outT.data_ptr(),
                              loT.data_ptr(), yc.data_ptr<float>(),
                              Ky.data_ptr<double>(), current_stream()),
            "cross_mfma_ppa");
  return {outT, loT};
}

// PPA fast path: transposed hi/lo tiles ONLY (no [c, m] copies written)
// plus the fused column-sum Ky += K^T y accumulated from the fp32
// register values (replaces the colsum_gemv re-read pass).
std::vector<torch::Tensor> cross_kernel_tile_ppa(torch::Tensor X,
                                                 torch::Tensor A,
                                                 torch::Tensor s2v,
                                                 double amp, torch::Tensor y,
                                                 torch::Tensor Ky) {
  TORCH_CHECK(X.is_cuda() && X.dtype() == torch::kFloat32 && X.dim() == 2);
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kFloat32 && A.dim() == 2);
  TORCH_CHECK(X.size(1) == A.size(1), "feature dims differ");
  TORCH_CHECK(y.is_cuda() && y.dtype() == torch::kFloat32 &&
              y.numel() == X.size(0));
  TORCH_CHECK(Ky.is_cuda() && Ky.dtype() == torch::kFloat64 &&
              Ky.numel() == A.size(0));
  auto Xc = X.contiguous();
  auto Ac = A.contiguous();
  auto sc = s2v.contiguous();
  auto yc = y.contiguous();
  const int c = X.size(0), m = A.size(0), d = X.size(1);
  auto opts = torch::TensorOptions().dtype(torch::kBFloat16)
                  .device(X.device());
  auto outT = torch::empty({m, c}, opts);
  auto loT = torch::empty({m, c}, opts);
  check_hip(launch_cross_kernel_tile(Xc.data_ptr<float>(), Ac.data_ptr<float>(),
                                     sc.data_ptr<float>(), (float)amp, c, m, d,
                                     nullptr, nullptr, outT.data_ptr(),
                                     loT.data_ptr(), 1,
                                     yc.data_ptr<float>(),
                                     Ky.data_ptr<double>(), current_stream()),
            "cross_kernel_tile_ppa");
  return {outT, loT};
}

// KcT/KlT are the TRANSPOSED kernel blocks [m, c] (k along the contiguous
// axis; see syrk_bf16_kernel).  KK [m, m] is accumulated in place.
void syrk_bf16_acc(torch::Tensor KcT, c10::optional<torch::Tensor> KlT,
                   torch::Tensor KK, int64_t split_k) {
  TORCH_CHECK(KcT.is_cuda() && KcT.dtype() == torch::kBFloat16 &&
              KcT.dim() == 2);
  TORCH_CHECK(KK.is_cuda() && KK.dtype() == torch::kFloat32 && KK.dim() == 2);
  auto Kcc = KcT.contiguous();
  const int m = KcT.size(0), c = KcT.size(1);
  TORCH_CHECK(KK.size(0) == m && KK.size(1) == m,
              "KK must be [m, m] with m = KcT.size(0) (transposed operand)");
  torch::Tensor Klc;
  if (KlT.has_value()) {
    TORCH_CHECK(KlT->sizes() == KcT.sizes() &&
                KlT->dtype() == torch::kBFloat16);
    Klc = KlT->contiguous();
  }
  check_hip(launch_syrk_bf16(Kcc.data_ptr(),
                             KlT.has_value() ? Klc.data_ptr() : nullptr, c, m,
                             c, (int)split_k, KK.data_ptr<float>(),
                             current_stream()),
            "syrk_bf16");
}

// k-synchronized SYRK (large m): tiles [nb, 2] int32 (ti, tj; -1 pads),
// one block per tile, accumulators held across all k; see cross_syrk.hip.
void syrk_bf16_sync_acc(torch::Tensor KcT, c10::optional<torch::Tensor> KlT,
                        torch::Tensor KK, torch::Tensor tiles,
                        int64_t kpb, int64_t nactive) {
  TORCH_CHECK(KcT.is_cuda() && KcT.dtype() == torch::kBFloat16 &&
              KcT.dim() == 2 && KcT.is_contiguous());
  TORCH_CHECK(KK.is_cuda() && KK.dtype() == torch::kFloat32 && KK.dim() == 2);
  TORCH_CHECK(tiles.is_cuda() && tiles.dtype() == torch::kInt32 &&
              tiles.dim() == 2 && tiles.size(1) == 2 && tiles.is_contiguous());
  const int m = KcT.size(0), c = KcT.size(1);
  TORCH_CHECK(KK.size(0) == m && KK.size(1) == m);
  torch::Tensor Klc;
  if (KlT.has_value()) {
    TORCH_CHECK(KlT->sizes() == KcT.sizes() &&
                KlT->dtype() == torch::kBFloat16);
    Klc = KlT->contiguous();
  }
  const int kblocks = (c + 31) / 32;
  const int nphases = (kblocks + (int)kpb - 1) / (int)kpb;
  auto ctr = torch::zeros({nphases},
                          torch::TensorOptions().dtype(torch::kInt32)
                              .device(KK.device()));
  check_hip(launch_syrk_bf16_sync(
                KcT.data_ptr(), KlT.has_value() ? Klc.data_ptr() : nullptr,
                c, m, c, tiles.data_ptr<int>(), (int)tiles.size(0),
                (int)kpb, ctr.data_ptr<int>(), (int)nactive,
                KK.data_ptr<float>(), current_stream()),
            "syrk_bf16_sync");
}

void colsum_gemv_acc(torch::Tensor Kc, torch::Tensor y, torch::Tensor Ky) {
  TORCH_CHECK(Kc.is_cuda() && Kc.dtype() == torch::kBFloat16 && Kc.dim() == 2);
  TORCH_CHECK(y.is_cuda() && y.dtype() == torch::kFloat32);
  TORCH_CHECK(Ky.is_cuda() && Ky.dtype() == torch::kFloat64);
  auto Kcc = Kc.contiguous();
  const int c = Kc.size(0), m = Kc.size(1);
  check_hip(launch_colsum_gemv(Kcc.data_ptr(), y.contiguous().data_ptr<float>(),
                               c, m, Ky.data_ptr<double>(), current_stream()),
            "colsum_gemv");
}

// Runs the per-expert Laplace Newton loop in place on f.
// Returns (psi[E] f64, sumlogl[E] f64, iters[E] i32, bad[E] i32).
std::vector<torch::Tensor> fused_laplace_newton(torch::Tensor X,
                                                torch::Tensor y,
                                                torch::Tensor f,
                                                torch::Tensor scale,
                                                double amp, double noise,
                                                double tol,
                                                int64_t max_newton) {
  TORCH_CHECK(X.is_cuda() && X.dtype() == torch::kFloat32 && X.dim() == 3);
  TORCH_CHECK(y.is_cuda() && y.dtype() == torch::kFloat32 && y.dim() == 2);
  TORCH_CHECK(f.is_cuda() && f.dtype() == torch::kFloat32 && f.dim() == 2
              && f.is_contiguous(), "f must be contiguous fp32 (updated "
              "in place)");
  auto Xc = X.contiguous();
  auto yc = y.contiguous();
  auto sc = scale.contiguous();
  const int E = X.size(0), k = X.size(1), d = X.size(2);
  TORCH_CHECK(k <= 128 && d <= k,
              "fused_laplace_newton requires k<=128, d<=k");
  auto opts64 = torch::TensorOptions().dtype(torch::kFloat64).device(X.device());
  auto opts32i = torch::TensorOptions().dtype(torch::kInt32).device(X.device());
  auto psi = torch::empty({E}, opts64);
  auto sll = torch::empty({E}, opts64);
  auto iters = torch::empty({E}, opts32i);
  auto bad = torch::empty({E}, opts32i);
  size_t lds = 0;
  check_hip(launch_fused_laplace_newton(
                Xc.data_ptr<float>(), yc.data_ptr<float>(),
                f.data_ptr<float>(), sc.data_ptr<float>(), (float)amp,
                (float)noise, E, k, d, tol, (int)max_newton,
                psi.data_ptr<double>(), sll.data_ptr<double>(),
                iters.data_ptr<int>(), bad.data_ptr<int>(), current_stream(),
                &lds),
            "fused_laplace_newton");
  return {psi, sll, iters, bad};
}

bool fused_laplace_newton_supported(int64_t k, int64_t d) {
  if (k > 128 || d > k || k < 1) return false;
  // EXACT mirror of lap_lds_bytes in laplace.hip (single-buffer design)
  const int64_t dp4 = (d + 4) & ~(int64_t)3;
  int64_t bytes = a16i(8 * 10) + a16i(4 * k * sa_of(k)) +
                  a16i(4 * k * dp4) +
                  a16i(4 * tsz_of(k)) + 8 * a16i(4 * k) + a16i(4 * d) + 16;
  return bytes <= 160 * 1024;
}

// Fused Newton + Algorithm 5.1 evidence/gradient (K11).  Runs the Newton
// loop to convergence (f updated in place), then the evidence pass in the
// same launch.  Returns (logz[E] f64, grad[E, d+2] f64 — beta grads then
// amp then noise, all d(logZ)/d(theta) un-negated —, iters[E] i32,
// bad[E] i32).
std::vector<torch::Tensor> fused_laplace_evidence(torch::Tensor X,
                                                  torch::Tensor y,
                                                  torch::Tensor f,
                                                  torch::Tensor scale,
                                                  double amp, double noise,
                                                  double tol,
                                                  int64_t max_newton) {
  TORCH_CHECK(X.is_cuda() && X.dtype() == torch::kFloat32 && X.dim() == 3);
  TORCH_CHECK(y.is_cuda() && y.dtype() == torch::kFloat32 && y.dim() == 2);
  TORCH_CHECK(f.is_cuda() && f.dtype() == torch::kFloat32 && f.dim() == 2
              && f.is_contiguous(), "f must be contiguous fp32");
  auto Xc = X.contiguous();
  auto yc = y.contiguous();
  auto sc = scale.contiguous();
  const int E = X.size(0), k = X.size(1), d = X.size(2);
  auto opts64 = torch::TensorOptions().dtype(torch::kFloat64).device(X.device());
  auto opts32i = torch::TensorOptions().dtype(torch::kInt32).device(X.device());
  auto psi = torch::empty({E}, opts64);
  auto sll = torch::empty({E}, opts64);
  auto logz = torch::empty({E}, opts64);
  auto grad = torch::empty({E, d + 2}, opts64);
  auto iters = torch::empty({E}, opts32i);
  auto bad = torch::empty({E}, opts32i);
  size_t lds = 0;
  check_hip(launch_fused_laplace_evidence(
                Xc.data_ptr<float>(), yc.data_ptr<float>(),
                f.data_ptr<float>(), sc.data_ptr<float>(), (float)amp,
                (float)noise, E, k, d, tol, (int)max_newton,
                psi.data_ptr<double>(), sll.data_ptr<double>(),
                iters.data_ptr<int>(), bad.data_ptr<int>(),
                logz.data_ptr<double>(), grad.data_ptr<double>(),
                current_stream(), &lds),
            "fused_laplace_evidence");
  return {logz, grad, iters, bad};
}

bool fused_laplace_evidence_supported(int64_t k, int64_t d) {
  if (k > 128 || d > k || k < 1) return false;
  // EXACT mirror of lap_lds_bytes(k, d, ev=1) in laplace.hip
  const int64_t dp4 = (d + 4) & ~(int64_t)3;
  int64_t bytes = a16i(8 * 10) + a16i(4 * k * sa_of(k)) +
                  a16i(4 * k * dp4) +
                  a16i(4 * tsz_of(k)) + 8 * a16i(4 * k) + a16i(4 * d) + 16 +
                  2 * a16i(4 * k) + a16i(4 * d) + a16i(8 * (d + 2));
  return bytes <= 160 * 1024;
}

// ---------------------------------------------------------------------------
// K13: blocked fp64 Cholesky path (big_chol.hip) — host orchestration
// ---------------------------------------------------------------------------

// In-place blocked Cholesky of the padded [mp, mp] fp64 matrix A (lower
// triangle; strict upper left untouched).  V [nb, 64, 64] receives the
// explicit inverse of each diagonal block; bad (int32 [1], zeroed by the
// caller) is set on a non-PD pivot.  mp must be a multiple of 64.
void dpotrf64(torch::Tensor A, torch::Tensor V, torch::Tensor bad) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kFloat64 && A.dim() == 2 &&
              A.size(0) == A.size(1) && A.is_contiguous());
  const long mp = A.size(0);
  TORCH_CHECK(mp % 64 == 0, "dpotrf64 requires 64-padded matrices");
  const long nb = mp / 64;
  TORCH_CHECK(V.is_cuda() && V.dtype() == torch::kFloat64 &&
              V.numel() == nb * 64 * 64 && V.is_contiguous());
  TORCH_CHECK(bad.is_cuda() && bad.dtype() == torch::kInt32);
  auto stream = current_stream();
  double* a = A.data_ptr<double>();
  double* v = V.data_ptr<double>();
  int* bd = bad.data_ptr<int>();
  for (long J = 0; J < nb; ++J) {
    const long jb = J * 64;
    check_hip(launch_dpotrf_diag(a, mp, (int)jb, v + J * 4096, bd, stream),
              "dpotrf_diag");
    const long nr = mp - jb - 64;
    if (nr > 0) {
      double* A21 = a + (size_t)(jb + 64) * mp + jb;
      // panel solve L21 = A21 V_J^T (in place)
      check_hip(launch_dgemm64(0, 1, 0, 0, A21, v + J * 4096, A21,
                               (int)nr, 64, 64, mp, 64, mp, stream),
                "dtrsm_panel");
      // trailing A22 -= L21 L21^T (lower tiles only)
      double* A22 = a + (size_t)(jb + 64) * mp + jb + 64;
      check_hip(launch_dgemm64(0, 1, 1, 1, A21, A21, A22,
                               (int)nr, (int)nr, 64, mp, mp, mp, stream),
                "dsyrk_trailing");
    }
  }
}

// Solve (L L^T) X = B in place on B [mp, r] given the factored A (lower L)
// and the diagonal-block inverses V from dpotrf64.  rhs_identity=true
// declares B to be the identity (the explicit-inverse path): the forward
// pass then skips the provably-zero column range of each block step —
// Y = L^{-1} is lower triangular, so block row I has nonzeros only in
// columns < (I+1)*64 — cutting the forward GEMMs from m^3 to m^3/3.
void dchol_solve64(torch::Tensor A, torch::Tensor V, torch::Tensor B,
                   bool rhs_identity) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kFloat64 &&
              A.is_contiguous() && A.dim() == 2);
  TORCH_CHECK(B.is_cuda() && B.dtype() == torch::kFloat64 &&
              B.is_contiguous() && B.dim() == 2 && B.size(0) == A.size(0));
  const long mp = A.size(0), r = B.size(1);
  TORCH_CHECK(mp % 64 == 0);
  const long nb = mp / 64;
  TORCH_CHECK(V.numel() == nb * 64 * 64);
  auto stream = current_stream();
  double* a = A.data_ptr<double>();
  double* v = V.data_ptr<double>();
  double* b = B.data_ptr<double>();
  for (long I = 0; I < nb; ++I) {            // forward: L Y = B
    const long jb = I * 64;
    const long rc = rhs_identity ? std::min<long>(r, jb + 64) : r;
    check_hip(launch_dgemm64(0, 0, 0, 0, v + I * 4096, b + jb * r,
                             b + jb * r, 64, (int)rc, 64, 64, r, r, stream),
              "fwd_diag");
    const long nr = mp - jb - 64;
    if (nr > 0)
      check_hip(launch_dgemm64(0, 0, 1, 0, a + (size_t)(jb + 64) * mp + jb,
                               b + jb * r, b + (jb + 64) * r,
                               (int)nr, (int)rc, 64, mp, r, r, stream),
                "fwd_update");
  }
  for (long I = nb - 1; I >= 0; --I) {       // backward: L^T X = Y
    const long jb = I * 64;
    check_hip(launch_dgemm64(1, 0, 0, 0, v + I * 4096, b + jb * r,
                             b + jb * r, 64, (int)r, 64, 64, r, r, stream),
              "bwd_diag");
    if (jb > 0)
      check_hip(launch_dgemm64(1, 0, 1, 0, a + (size_t)jb * mp, b + jb * r,
                               b, (int)jb, (int)r, 64, mp, r, r, stream),
                "bwd_update");
  }
}

// Generic fp64 MFMA GEMM (layout verification + utility): C = op(A) op(B).
torch::Tensor dgemm64(torch::Tensor A, torch::Tensor B, bool ta, bool tb) {
  TORCH_CHECK(A.is_cuda() && A.dtype() == torch::kFloat64 && A.dim() == 2);
  TORCH_CHECK(B.is_cuda() && B.dtype() == torch::kFloat64 && B.dim() == 2);
  auto Ac = A.contiguous();
  auto Bc = B.contiguous();
  const long M = ta ? A.size(1) : A.size(0);
  const long K = ta ? A.size(0) : A.size(1);
  const long Kb = tb ? B.size(1) : B.size(0);
  const long N = tb ? B.size(0) : B.size(1);
  TORCH_CHECK(K == Kb, "inner dims differ");
  auto C = torch::empty({M, N}, A.options());
  check_hip(launch_dgemm64(ta ? 1 : 0, tb ? 1 : 0, 0, 0,
                           Ac.data_ptr<double>(), Bc.data_ptr<double>(),
                           C.data_ptr<double>(), (int)M, (int)N, (int)K,
                           A.size(1), B.size(1), N, current_stream()),
            "dgemm64");
  return C;
}

// K19: device-side synthetic benchmark data (Philox4x32-10)
std::vector<torch::Tensor> synth_regression(int64_t n, int64_t d,
                                            int64_t seed, double noise_sd) {
  auto dev = torch::TensorOptions().dtype(torch::kFloat32)
                 .device(torch::kCUDA);
  auto X = torch::empty({n, d}, dev);
  auto y = torch::empty({n}, dev);
  check_hip(launch_synth_regression(X.data_ptr<float>(), y.data_ptr<float>(),
                                    n, (int)d, (unsigned long long)seed,
                                    (float)noise_sd, current_stream()),
            "synth_regression");
  return {X, y};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("synth_regression", &synth_regression,
          "device-side Philox benchmark data (K19)");
  mod.def("dpotrf64", &dpotrf64,
          "blocked fp64 Cholesky (K13), in place, 64-padded");
  mod.def("dchol_solve64", &dchol_solve64,
          "blocked fp64 triangular solves (L L^T) X = B, in place on B",
          pybind11::arg("A"), pybind11::arg("V"), pybind11::arg("B"),
          pybind11::arg("rhs_identity") = false);
  mod.def("dgemm64", &dgemm64, "fp64 MFMA GEMM (test/utility)");
  mod.def("fused_laplace_newton", &fused_laplace_newton,
          "per-expert Laplace Newton loop to convergence (CDNA4)");
  mod.def("fused_laplace_newton_supported", &fused_laplace_newton_supported);
  mod.def("fused_laplace_evidence", &fused_laplace_evidence,
          "fused Newton + Algorithm 5.1 evidence/gradient (K11, CDNA4)");
  mod.def("fused_laplace_evidence_supported",
          &fused_laplace_evidence_supported);
  mod.def("fused_expert_nll", &fused_expert_nll,
          "fused per-expert BCM nll+gradient primitives (CDNA4)");
  mod.def("fused_expert_nll_profile", &fused_expert_nll_impl,
          "same, with per-phase wall_clock64 boundaries appended");
  mod.def("fused_expert_nll_supported", &fused_expert_nll_supported);
  mod.def("cross_mfma_ppa", &cross_mfma_ppa,
          "MFMA sqdist cross tile + fused K^T y (K1 plan, CDNA4)");
  mod.def("cross_kernel_tile_ppa", &cross_kernel_tile_ppa,
          "transposed hi/lo tiles + fused K^T y accumulation (CDNA4)");
  mod.def("cross_kernel_tile", &cross_kernel_tile,
          "rectangular RBF/ARD kernel block (CDNA4)",
          pybind11::arg("X"), pybind11::arg("A"), pybind11::arg("s2v"),
          pybind11::arg("amp"), pybind11::arg("bf16_out"),
          pybind11::arg("hilo"), pybind11::arg("want_t") = false);
  mod.def("syrk_bf16_sync_acc", &syrk_bf16_sync_acc,
          "k-synchronized persistent SYRK for large m (CDNA4)");
  mod.def("syrk_bf16_acc", &syrk_bf16_acc,
          "KK += K K^T of the transposed block KcT [m, c]; bf16 MFMA, "
          "fp32 accumulate (CDNA4)");
  mod.def("colsum_gemv_acc", &colsum_gemv_acc,
          "Ky += Kc^T y, fp64 accumulate (CDNA4)");
}
