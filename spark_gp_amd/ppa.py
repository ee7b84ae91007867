"""Projected Process Approximation assembly.

Mirrors ``commons/ProjectedGaussianProcessHelper.scala``:

* ``accumulate_ppa_stats`` — distributed (K_mn K_nm, K_mn y) accumulation
  (C2 collective: one allreduce of an [m,m]+[m] fp64 payload);
* ``magic_vector_matrix`` — PD = sigma2_white * K_mm + K_mn K_nm, the "magic"
  vector PD^-1 K_mn y and matrix inv(PD)*sigma2_white - inv(K_mm).

Deviation from the reference (recorded in SURVEY.md §7): the PD assertion is
the Cholesky info flag instead of a full symmetric eigendecomposition
(``ProjectedGaussianProcessHelper.scala:62-65`` does an O(m^3) eigSym purely
as a validity check).
"""

from __future__ import annotations

from typing import Tuple

import torch

from .kernels.base import Kernel
from .parallel.dist import Comm
from . import ops


class NotPositiveDefiniteError(RuntimeError):
    def __init__(self) -> None:
        super().__init__(
            "a matrix expected to be positive definite is not; this usually "
            "means the `sigma2` parameter is too small — increase it "
            "gradually until the error goes away")


def accumulate_ppa_stats(kernel: Kernel, active: torch.Tensor,
                         X: torch.Tensor, y: torch.Tensor,
                         comm: Comm, precision: str = "fp64"
                         ) -> Tuple[torch.Tensor, torch.Tensor]:
    KK, Ky = ops.kmn_knm_and_kmny(kernel, active, X, y, precision=precision)
    comm.allreduce_(KK)
    comm.allreduce_(Ky)
    return KK, Ky


def magic_vector_matrix(kernel: Kernel, KK: torch.Tensor, Ky: torch.Tensor,
                        active: torch.Tensor
                        ) -> Tuple[torch.Tensor, torch.Tensor]:
    """(magic_vector [m], magic_matrix [m,m]), float64.

    Computed redundantly on every rank (cheaper than broadcasting at m<=8192,
    C4 in SURVEY.md §2.5); all ranks hold identical KK/Ky after allreduce so
    results are identical.

    On GPU the factorization/solves/inverses run on the hand-written K13
    kernels (big_chol.hip: blocked fp64 Cholesky + MFMA-f64 GEMM tiles);
    the torch path below is the CPU oracle."""
    if KK.is_cuda and not ops._force_torch():
        hip = ops._load_hip()
        if ops._require_hip_or_fallback("magic_vector_matrix"):
            return hip.magic_vector_matrix(kernel, KK, Ky, active)
    active64 = active.double()
    Kmm = kernel.training_kernel(active64)            # includes noise diag
    nu = kernel.white_noise_var()
    PD = nu * Kmm + KK

    Lpd = _chol_with_jitter(PD)
    magic_vector = torch.cholesky_solve(Ky.unsqueeze(-1), Lpd).squeeze(-1)
    Lmm = _chol_with_jitter(Kmm)
    magic_matrix = (torch.cholesky_inverse(Lpd) * nu
                    - torch.cholesky_inverse(Lmm))
    return magic_vector, magic_matrix


def _chol_with_jitter(M: torch.Tensor, max_tries: int = 6) -> torch.Tensor:
    """Cholesky with an escalating-jitter ladder.

    The reference asserts PD via eigSym and throws immediately
    (``ProjectedGaussianProcessHelper.scala:62-65``).  Here, matrices that are
    PD in exact arithmetic can be numerically indefinite (fp32/bf16
    accumulation of K_mn K_nm on the GPU path), so failures retry with
    jitter = eps_rel * mean(diag) escalating 1e-12 .. 1e-2 before raising —
    a strict robustness superset of the reference behavior."""
    L, info = torch.linalg.cholesky_ex(M)
    if int(info) == 0:
        return L
    scale = float(M.diagonal().abs().mean())
    eps = 1e-12
    for _ in range(max_tries):
        jit = eps * scale * torch.eye(M.shape[-1], dtype=M.dtype,
                                      device=M.device)
        L, info = torch.linalg.cholesky_ex(M + jit)
        if int(info) == 0:
            return L
        eps *= 100.0
    raise NotPositiveDefiniteError()
