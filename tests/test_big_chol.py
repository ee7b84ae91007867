"""K13 numerics on MI355X: the hand-written blocked fp64 Cholesky path
(big_chol.hip) against the torch/rocSOLVER fp64 oracle.

Covers: the raw MFMA-f64 GEMM layout in every transpose combination
(asymmetric operands so a row/col swap cannot pass), the blocked factor
at expert (m=1000-class) and big-m (m=8192, marked slow) sizes including
non-multiples of 64, the blocked triangular solves and explicit inverse,
the non-PD breakdown flag, and the full magic_vector_matrix parity with
the CPU oracle path.
"""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _ext():
    from spark_gp_amd import _hip_ext
    return _hip_ext


def _spd(m, seed=0, device="cuda"):
    g = torch.Generator(device="cpu").manual_seed(seed)
    A = torch.randn(m, m, generator=g, dtype=torch.float64)
    M = (A @ A.T + m * torch.eye(m, dtype=torch.float64)).to(device)
    return M


@pytest.mark.parametrize("ta", [False, True])
@pytest.mark.parametrize("tb", [False, True])
def test_dgemm64_all_transposes(ta, tb):
    ext = _ext()
    g = torch.Generator().manual_seed(3)
    M, K, N = 200, 130, 70          # deliberately unequal + non-64-multiples
    A = torch.randn((K, M) if ta else (M, K), generator=g,
                    dtype=torch.float64).cuda()
    B = torch.randn((N, K) if tb else (K, N), generator=g,
                    dtype=torch.float64).cuda()
    C = ext.dgemm64(A, B, ta, tb)
    ref = (A.T if ta else A) @ (B.T if tb else B)
    torch.testing.assert_close(C, ref, rtol=1e-13, atol=1e-10)


@pytest.mark.parametrize("m", [64, 100, 256, 1000])
def test_dpotrf64_matches_torch(m):
    from spark_gp_amd.ops import hip_backend as hb
    M = _spd(m, seed=m)
    L_ref = torch.linalg.cholesky(M)
    Lp, V = hb.chol_factor64(M)
    L = torch.tril(Lp[:m, :m])
    torch.testing.assert_close(L, L_ref, rtol=1e-11, atol=1e-9)
    # diag-block inverses: V_J @ L_JJ = I
    nb = Lp.shape[0] // 64
    for J in range(nb):
        blk = Lp[J * 64:(J + 1) * 64, J * 64:(J + 1) * 64].tril()
        eye = V[J] @ blk
        torch.testing.assert_close(
            eye, torch.eye(64, dtype=torch.float64, device="cuda"),
            rtol=0, atol=1e-10)


@pytest.mark.parametrize("m,r", [(100, 1), (1000, 7), (320, 320)])
def test_chol_solve_and_inverse(m, r):
    from spark_gp_amd.ops import hip_backend as hb
    M = _spd(m, seed=7)
    B = torch.randn(m, r, dtype=torch.float64).cuda()
    Lp, V = hb.chol_factor64(M)
    X = hb.chol_solve64(Lp, V, B, m)
    L_ref = torch.linalg.cholesky(M)
    X_ref = torch.cholesky_solve(B, L_ref)
    torch.testing.assert_close(X, X_ref, rtol=1e-9, atol=1e-9)
    Minv = hb.chol_inverse64(Lp, V, m)
    Minv_ref = torch.cholesky_inverse(L_ref)
    torch.testing.assert_close(Minv, Minv_ref, rtol=1e-9, atol=1e-9)


def test_dpotrf64_non_pd_sets_bad_and_ladder_raises():
    from spark_gp_amd.ops import hip_backend as hb
    from spark_gp_amd.ppa import NotPositiveDefiniteError
    m = 128
    M = _spd(m, seed=1)
    M[5, 5] = -1e6                       # clearly not PD, beyond any jitter
    M[70, 70] = -1e6
    with pytest.raises(NotPositiveDefiniteError):
        hb.chol_factor64(M, max_tries=2)


def test_jitter_ladder_recovers_semidefinite():
    from spark_gp_amd.ops import hip_backend as hb
    m = 96
    # rank-deficient PSD: numerically indefinite under fp64 factorization
    g = torch.Generator().manual_seed(2)
    A = torch.randn(m, 10, generator=g, dtype=torch.float64).cuda()
    M = A @ A.T
    Lp, V = hb.chol_factor64(M)
    L = torch.tril(Lp[:m, :m])
    resid = (L @ L.T - M).abs().max().item()
    assert resid < 1e-6 * M.abs().max().item()


def test_magic_vector_matrix_hip_matches_cpu_oracle():
    # sigma2 = 0.1 keeps cond(PD) ~ 1e7: the comparison then reflects the
    # K13 implementation, not conditioning-amplified 1e-16 input noise
    # (Kmm is rebuilt per device; sqdist GEMM order differs CPU vs GPU)
    import spark_gp_amd.ppa as ppa
    from spark_gp_amd.kernels import ARDRBFKernel, EyeKernel, Scalar
    torch.manual_seed(0)
    m, d, n = 200, 4, 1000
    kernel = 1 * ARDRBFKernel(d) + Scalar(0.1).const * EyeKernel()
    kernel.set_hyperparameters(
        np.array([1.1] + [0.9] * d))
    X = torch.rand(n, d, dtype=torch.float64)
    y = torch.sin(3 * X.sum(-1))
    active = X[:m].clone()
    from spark_gp_amd.ops import torch_backend as tb
    KK, Ky = tb.kmn_knm_and_kmny(kernel, active, X, y)
    # quantize the active set identically on both paths (the GPU pipeline
    # holds X in fp32; Kmm must be built from the SAME points bit-for-bit)
    active_q = active.float().double()
    mv_cpu, mm_cpu = ppa.magic_vector_matrix(kernel, KK, Ky, active_q)
    mv_gpu, mm_gpu = ppa.magic_vector_matrix(
        kernel, KK.cuda(), Ky.cuda(), active_q.cuda().float())
    # remaining delta: Kmm is rebuilt per device (sqdist GEMM reduction
    # order differs CPU vs GPU at ~1e-16/entry), amplified by cond(PD);
    # the same-input linalg parity is pinned at 1e-11 by the tests above
    torch.testing.assert_close(mv_gpu.cpu(), mv_cpu, rtol=1e-6, atol=1e-8)
    torch.testing.assert_close(mm_gpu.cpu(), mm_cpu, rtol=1e-6, atol=1e-8)


@pytest.mark.slow
def test_dpotrf64_big_m():
    """m=8192 (BASELINE config 5 scale): factor + solve correctness via
    residuals (no torch factor of the same size to keep the test fast)."""
    from spark_gp_amd.ops import hip_backend as hb
    m = 8192
    g = torch.Generator(device="cuda").manual_seed(0)
    A = torch.randn(m, 64, generator=g, dtype=torch.float64, device="cuda")
    M = A @ A.T + m * torch.eye(m, dtype=torch.float64, device="cuda")
    Lp, V = hb.chol_factor64(M)
    L = torch.tril(Lp[:m, :m])
    resid = (L @ L.T - M).abs().max().item() / M.abs().max().item()
    assert resid < 1e-12, resid
    b = torch.randn(m, 1, generator=g, dtype=torch.float64, device="cuda")
    x = hb.chol_solve64(Lp, V, b, m)
    err = (M @ x - b).abs().max().item() / b.abs().max().item()
    assert err < 1e-9, err


@pytest.mark.parametrize("m", [1, 63, 64, 65])
def test_dpotrf64_boundary_sizes(m):
    """Padding boundaries: m below/at/above one 64-block."""
    from spark_gp_amd.ops import hip_backend as hb
    M = _spd(m, seed=m + 40)
    Lp, V = hb.chol_factor64(M)
    L_ref = torch.linalg.cholesky(M)
    torch.testing.assert_close(torch.tril(Lp[:m, :m]), L_ref,
                               rtol=1e-11, atol=1e-9)
    b = torch.randn(m, 2, dtype=torch.float64).cuda()
    X = hb.chol_solve64(Lp, V, b, m)
    torch.testing.assert_close(X, torch.cholesky_solve(b, L_ref),
                               rtol=1e-9, atol=1e-9)
