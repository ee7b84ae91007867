"""Python wrapper over the hand-written CDNA4 HIP kernels (_hip_ext).

Gradient assembly happens here (host-side chain rule over the kernel's raw
contraction outputs), mirroring exactly the formulas of
``torch_backend.nll_grad_compiled`` so the two backends are drop-in
interchangeable and unit-diffable.
"""

from __future__ import annotations

import math
from typing import Tuple

import numpy as np
import torch

from ..kernels.base import Kernel
from ..kernels.compiled import CompiledKernel, compile_kernel
from .. import _hip_ext as ext
from . import torch_backend


def _scale_vector(cs: CompiledKernel, theta: np.ndarray, d: int,
                  device, dtype=torch.float32) -> torch.Tensor:
    if cs.base == 'ard':
        s = np.asarray(theta[cs.base_idx], dtype=np.float64)
    else:  # rbf
        sigma = float(theta[cs.base_idx][0])
        s = np.full(d, 1.0 / (math.sqrt(2.0) * sigma))
    return torch.as_tensor(s, dtype=dtype, device=device)


def supports_nll(cs: CompiledKernel, X: torch.Tensor) -> bool:
    if cs.base not in ("ard", "rbf"):
        return False
    if X.dtype != torch.float32 or X.dim() != 3:
        return False
    E, k, d = X.shape
    return bool(ext.fused_expert_nll_supported(k, d))


def nll_grad_compiled(cs: CompiledKernel, theta: np.ndarray,
                      X: torch.Tensor, y: torch.Tensor
                      ) -> Tuple[float, np.ndarray]:
    E, k, d = X.shape
    C = cs.amp(theta)
    nu = cs.noise(theta)
    scale = _scale_vector(cs, theta, d, X.device)
    nll, sumW0, trG, contr, bad = ext.fused_expert_nll(
        X, y.to(torch.float32), scale, float(C), float(nu))

    # single device->host transfer for all reductions (one sync per eval)
    stats = torch.cat([nll.sum().reshape(1), sumW0.sum().reshape(1),
                       trG.sum().reshape(1), bad.sum().double().reshape(1),
                       bad.max().double().reshape(1),
                       contr.sum(0)]).cpu().numpy()
    nll_total, sumW0_t, trG_t = float(stats[0]), float(stats[1]), float(stats[2])
    n_bad = int(stats[3])
    if stats[4] >= 2.0:
        # a non-finite kernel matrix (overflowing line-search iterate): the
        # reference's LAPACK would propagate NaN and L-BFGS-B backtracks;
        # return +inf instead of recomputing every expert on the fallback
        return float("inf"), np.zeros(cs.p)
    contr_t = stats[5:]                               # [d]

    grad = np.zeros(cs.p)
    if cs.amp_idx is not None:
        grad[cs.amp_idx] = -0.5 * sumW0_t
    if cs.base == 'ard':
        beta = np.asarray(theta[cs.base_idx], dtype=np.float64)
        grad[cs.base_idx] = C * beta * contr_t
    else:
        sigma = float(theta[cs.base_idx][0])
        grad[cs.base_idx.start] = -C / (2.0 * sigma ** 3) * float(contr_t.sum())
    if cs.noise_idx:
        for i in cs.noise_idx:
            grad[i] += -0.5 * trG_t

    if n_bad:
        # fp32 Cholesky broke down for these experts (huge-amplitude
        # iterates); recompute them on the torch path (LU fallback inside)
        idx = (bad != 0).nonzero(as_tuple=True)[0]
        nll_b, grad_b = torch_backend.nll_grad_compiled(
            cs, theta, X[idx], y[idx], force_lu=True)
        nll_total += nll_b
        grad += grad_b
    return nll_total, grad


# ---------------------------------------------------------------------------
# Laplace Newton pre-pass (GPC)
# ---------------------------------------------------------------------------

# Tolerance floor for the fused Newton/evidence kernels.  The kernel's
# psi accumulates in fp64 (block_sum doubles), so the fp32 matrix noise,
# not the summation, sets the floor: measured oracle parity IMPROVES with
# tighter tol (grad max-rel 2.6e-4 at 1e-5, 1.9e-6 at 1e-6, 9.8e-7 at
# 1e-7 — BASELINE.md round 2), so the default floor matches the
# reference's default tol.  Below it, a fp64 torch polish finishes from
# the warm latent (see ops.__init__.laplace_nll_grad).
import os as _os
LAPLACE_MIN_TOL = float(_os.environ.get("SPARK_GP_AMD_LAPLACE_MIN_TOL",
                                        "1e-6"))


def supports_laplace(cs: CompiledKernel, X: torch.Tensor) -> bool:
    if cs is None or cs.base not in ("ard", "rbf"):
        return False
    if X.dtype != torch.float32 or X.dim() != 3:
        return False
    E, k, d = X.shape
    return bool(ext.fused_laplace_newton_supported(k, d))


def laplace_newton(cs: CompiledKernel, theta: np.ndarray, X: torch.Tensor,
                   y: torch.Tensor, f: torch.Tensor, tol: float,
                   max_newton: int) -> int:
    """Run the fused per-expert Newton loop to convergence, updating the
    latent ``f`` IN PLACE.  Returns the number of experts the fp32 kernel
    could not handle (their f is left unchanged; the torch evidence pass
    converges them from their warm state)."""
    C = cs.amp(theta)
    nu = cs.noise(theta)
    scale = _scale_vector(cs, theta, X.shape[-1], X.device)
    if not f.is_contiguous():
        raise ValueError("latent f must be contiguous")
    # The fused loop is a warm-starter: the torch evidence pass finishes
    # convergence with reference semantics, so cap the fp32 iterations (the
    # fp32 objective noise floor can sit above a tight tol) and loosen tol
    # to the fp32-representable level.
    eff_tol = max(float(tol), LAPLACE_MIN_TOL)
    psi, sll, iters, bad = ext.fused_laplace_newton(
        X, y.to(torch.float32), f, scale, float(C), float(nu), eff_tol,
        min(int(max_newton), 40))
    return int((bad != 0).sum())


def supports_laplace_evidence(cs: CompiledKernel, X: torch.Tensor) -> bool:
    if cs is None or cs.base not in ("ard", "rbf"):
        return False
    if X.dtype != torch.float32 or X.dim() != 3:
        return False
    E, k, d = X.shape
    return bool(ext.fused_laplace_evidence_supported(k, d))


def laplace_evidence(cs: CompiledKernel, theta: np.ndarray, X: torch.Tensor,
                     y: torch.Tensor, f: torch.Tensor, tol: float,
                     max_newton: int):
    """Fully fused K10+K11: Newton to convergence AND the Algorithm 5.1
    evidence/gradient in one launch (laplace.hip evidence tail).  Updates
    f in place.  Returns (nll, grad) with the torch-path sign convention,
    or None when any expert's fp32 factor broke down (caller falls back
    to the warm torch Newton)."""
    C = cs.amp(theta)
    nu = cs.noise(theta)
    d = X.shape[-1]
    scale = _scale_vector(cs, theta, d, X.device)
    if not f.is_contiguous():
        raise ValueError("latent f must be contiguous")
    eff_tol = max(float(tol), LAPLACE_MIN_TOL)
    logz, grad, iters, bad = ext.fused_laplace_evidence(
        X, y.to(torch.float32), f, scale, float(C), float(nu), eff_tol,
        min(int(max_newton), 40))
    # one device->host transfer for everything
    stats = torch.cat([logz.sum().reshape(1), bad.sum().double().reshape(1),
                       grad.sum(0)]).cpu().numpy()
    if int(stats[1]):
        return None
    logZ = float(stats[0])
    g = stats[2:]
    gb, gamp, gnoise = g[:d], float(g[d]), float(g[d + 1])
    out = np.zeros(cs.p)
    if cs.amp_idx is not None:
        out[cs.amp_idx] = gamp
    if cs.base == "ard":
        out[cs.base_idx] = gb
    else:
        sigma = float(theta[cs.base_idx][0])
        # the kernel differentiates w.r.t. the uniform scale beta =
        # 1/(sqrt(2) sigma); chain rule: dbeta/dsigma = -1/(sqrt(2) s^2)
        out[cs.base_idx.start] = (-float(gb.sum())
                                  / (math.sqrt(2.0) * sigma * sigma))
    for i in cs.noise_idx:
        out[i] += gnoise
    return float(-logZ), -out


# ---------------------------------------------------------------------------
# PPA path
# ---------------------------------------------------------------------------

def supports_ppa(kernel: Kernel, X: torch.Tensor) -> bool:
    cs = compile_kernel(kernel)
    return (cs is not None and cs.base in ("ard", "rbf")
            and X.dtype == torch.float32)


def _s2_vector(cs: CompiledKernel, theta: np.ndarray, d: int, device):
    s = _scale_vector(cs, theta, d, device, dtype=torch.float64)
    return (s * s).to(torch.float32)


_SYNC_TILE_CACHE: dict = {}


def _syrk_sync_tiles(m: int, device):
    """XCD-clustered tile lists for the k-synchronized SYRK.

    One block owns one 256x256 output tile for a whole launch, so a launch
    carries at most 224 tiles (blocks beyond the resident set would break
    the k-cohort).  The dispatcher places block b on XCD b%8 (observed,
    speed-only), so each launch list interleaves 8 per-XCD sequences whose
    tiles are sorted by 2x4-tile patches — an XCD's resident tiles then
    share a handful of 256-row/column operand windows that fit its 4 MB L2
    at the 256-k phase width."""
    import os
    pr, pc = (int(v) for v in
              os.environ.get("SPARK_GP_AMD_SYRK_PATCH", "2x4").split("x"))
    key = (m, str(device), pr, pc)
    if key in _SYNC_TILE_CACHE:
        return _SYNC_TILE_CACHE[key]
    ntile = (m + 255) // 256
    tiles = [(ti, tj) for ti in range(ntile) for tj in range(ti + 1)]
    tiles.sort(key=lambda t: (t[0] // pr, t[1] // pc))
    launches = []
    CAP = 224
    for s in range(0, len(tiles), CAP):
        grp = tiles[s:s + CAP]
        per = (len(grp) + 7) // 8
        L = [(-1, -1)] * (per * 8)
        for x in range(8):
            seg = grp[x * per:(x + 1) * per]
            for j, t in enumerate(seg):
                L[j * 8 + x] = t
        tt = torch.tensor(L, dtype=torch.int32, device=device)
        launches.append((tt, len(grp)))
    _SYNC_TILE_CACHE[key] = launches
    return launches


def _syrk_dispatch(KcT, KlT, KK, m: int):
    import os
    ntile = (m + 255) // 256
    tiles = ntile * (ntile + 1) // 2
    # k-synchronized persistent path wins once the tile set exceeds the
    # resident-block set (measured: m=8192 627 vs 494 TF at kpb=128;
    # m=4096 the split-k kernel is still ahead, 526 vs ~494 —
    # PROFILES.md round-2 SYRK section)
    if tiles > 256 and os.environ.get("SPARK_GP_AMD_SYRK_SYNC", "1") == "1":
        for tt, nact in _syrk_sync_tiles(m, KK.device):
            ext.syrk_bf16_sync_acc(KcT, KlT, KK, tt, 128, nact)
        return
    # split_k: measured sweep (scripts/bench_syrk.py --sweep, r2) — see
    # PROFILES.md; shorter per-block k-ranges keep the drifting column
    # windows closer to L3 residency
    split_k = max(1, min(16, round(4096 / tiles)))
    ext.syrk_bf16_acc(KcT, KlT, KK, split_k)


def kmn_knm_and_kmny(kernel: Kernel, active: torch.Tensor,
                     X: torch.Tensor, y: torch.Tensor,
                     chunk_rows: int = 131072, precision: str = "fp64"
                     ) -> Tuple[torch.Tensor, torch.Tensor]:
    cs = compile_kernel(kernel)
    theta = kernel.get_hyperparameters()
    n, d = X.shape
    m = active.shape[0]
    C = cs.amp(theta)
    s2 = _s2_vector(cs, theta, d, X.device)
    act32 = active.to(torch.float32).contiguous()
    y32 = y.to(torch.float32)

    Ky = torch.zeros(m, dtype=torch.float64, device=X.device)
    if precision == "fp64":
        # reference-parity numerics: K_nm values from the HIP cross kernel
        # (fp32, ~6e-8 quantization), products and accumulation in fp64
        KK64 = torch.zeros(m, m, dtype=torch.float64, device=X.device)
        for s in range(0, n, chunk_rows):
            e = min(n, s + chunk_rows)
            Kc = ext.cross_kernel_tile(X[s:e].contiguous(), act32, s2,
                                       float(C), False, False)[0]
            K64 = Kc.double()
            KK64 += K64.transpose(0, 1) @ K64
            Ky += K64.transpose(0, 1) @ y[s:e].double()
        return KK64, Ky

    # 'mixed': hi/lo bf16 split — KK += hi^T hi + hi^T lo + lo^T hi on the
    # MFMA SYRK keeps input-quantization error at fp32 class while running
    # at bf16 matrix-core rate
    KK = torch.zeros(m, m, dtype=torch.float32, device=X.device)
    # MFMA cross tile (K1 plan): sqdist via ||x'||^2 + ||a'||^2 - 2 x'.a'
    # on f32 matrix cores with pre-scaled coordinates, writing ONLY the
    # transposed hi/lo copies the SYRK stages from and accumulating
    # Ky += K^T y in the same launch.  The elementwise kernel (~3 VALU
    # issues per (element, dim)) remains as the fallback/AB path.
    import os
    use_mfma = os.environ.get("SPARK_GP_AMD_CROSS_MFMA", "1") == "1"
    if use_mfma:
        svec = _scale_vector(cs, theta, d, X.device)
        As = (act32 * svec).contiguous()
        na = (As * As).sum(-1).contiguous()
    for s in range(0, n, chunk_rows):
        e = min(n, s + chunk_rows)
        if use_mfma:
            Xs = (X[s:e] * svec).contiguous()
            nx = (Xs * Xs).sum(-1).contiguous()
            KcT, KlT = ext.cross_mfma_ppa(Xs, As, nx, na, float(C),
                                          y32[s:e].contiguous(), Ky)
        else:
            KcT, KlT = ext.cross_kernel_tile_ppa(X[s:e].contiguous(), act32,
                                                 s2, float(C),
                                                 y32[s:e].contiguous(), Ky)
        _syrk_dispatch(KcT, KlT, KK, m)
    return KK.double(), Ky


# ---------------------------------------------------------------------------
# K13: hand-written blocked fp64 Cholesky path for the magic solves
# ---------------------------------------------------------------------------
# Replaces torch.linalg (rocSOLVER) in ppa.magic_vector_matrix on GPU:
# big_chol.hip's one-workgroup 64x64 diagonal factors + MFMA-f64 GEMM
# tiles for panel solve / trailing SYRK / blocked triangular solves.
# Matrices are padded to a multiple of 64 with an identity block (its
# factor and inverse are exact, so the un-padded region is unaffected).


def _pad64_spd(M: torch.Tensor) -> torch.Tensor:
    m = M.shape[0]
    mp = (m + 63) & ~63
    A = torch.zeros(mp, mp, dtype=torch.float64, device=M.device)
    A[:m, :m] = M
    if mp > m:
        A.diagonal()[m:] = 1.0
    return A


def _dpotrf(Apad: torch.Tensor):
    mp = Apad.shape[0]
    V = torch.empty(mp // 64, 64, 64, dtype=torch.float64,
                    device=Apad.device)
    bad = torch.zeros(1, dtype=torch.int32, device=Apad.device)
    ext.dpotrf64(Apad, V, bad)
    return V, bad


def chol_factor64(M: torch.Tensor, max_tries: int = 6):
    """Blocked fp64 Cholesky with the escalating-jitter ladder of
    ppa._chol_with_jitter (PD check = the factor's breakdown flag, not an
    eigSym pass).  Returns (L_padded [mp,mp], diag-block inverses V)."""
    from ..ppa import NotPositiveDefiniteError
    Apad = _pad64_spd(M)
    V, bad = _dpotrf(Apad)
    if int(bad.item()) == 0:
        return Apad, V
    scale = float(M.diagonal().abs().mean())
    eps = 1e-12
    m = M.shape[0]
    for _ in range(max_tries):
        jit = eps * scale * torch.eye(m, dtype=M.dtype, device=M.device)
        Apad = _pad64_spd(M + jit)
        V, bad = _dpotrf(Apad)
        if int(bad.item()) == 0:
            return Apad, V
        eps *= 100.0
    raise NotPositiveDefiniteError()


def chol_solve64(L: torch.Tensor, V: torch.Tensor, B: torch.Tensor,
                 m: int) -> torch.Tensor:
    """X = (L L^T)^{-1} B for B [m, r]; returns [m, r]."""
    mp = L.shape[0]
    Bp = torch.zeros(mp, B.shape[1], dtype=torch.float64, device=B.device)
    Bp[:m] = B
    ext.dchol_solve64(L, V, Bp)
    return Bp[:m]


def chol_inverse64(L: torch.Tensor, V: torch.Tensor, m: int) -> torch.Tensor:
    mp = L.shape[0]
    B = torch.eye(mp, dtype=torch.float64, device=L.device)
    ext.dchol_solve64(L, V, B, rhs_identity=True)
    return B[:m, :m]


def magic_vector_matrix(kernel: Kernel, KK: torch.Tensor, Ky: torch.Tensor,
                        active: torch.Tensor):
    """GPU magic quantities, entirely on the hand-written K13 kernels
    (ProjectedGaussianProcessHelper.scala:49-60 semantics; fp64)."""
    active64 = active.double()
    Kmm = kernel.training_kernel(active64)
    nu = kernel.white_noise_var()
    PD = nu * Kmm + KK

    Lp, Vp = chol_factor64(PD)
    m = KK.shape[0]
    mv = chol_solve64(Lp, Vp, Ky.unsqueeze(-1), m).squeeze(-1)
    PDinv = chol_inverse64(Lp, Vp, m)
    Lm, Vm = chol_factor64(Kmm)
    Kmminv = chol_inverse64(Lm, Vm, m)
    return mv, PDinv * nu - Kmminv


def cross_kernel(kernel: Kernel, Xtest: torch.Tensor,
                 Xtrain: torch.Tensor) -> torch.Tensor:
    cs = compile_kernel(kernel)
    theta = kernel.get_hyperparameters()
    d = Xtest.shape[-1]
    C = cs.amp(theta)
    s2 = _s2_vector(cs, theta, d, Xtest.device)
    return ext.cross_kernel_tile(Xtest.to(torch.float32).contiguous(),
                                 Xtrain.to(torch.float32).contiguous(),
                                 s2, float(C), False, False)[0].to(Xtest.dtype)
