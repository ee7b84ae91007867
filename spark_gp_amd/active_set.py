"""Active-set (inducing point) providers.

Mirrors ``commons/ActiveSetProvider.scala``:
* ``RandomActiveSetProvider``   — uniform global sample (:48-56)
* ``KMeansActiveSetProvider``   — k-means centroids as synthetic inducing
                                  points (:26-43; Spark MLlib KMeans there,
                                  GPU Lloyd iterations with allreduce here)
* ``GreedilyOptimizingActiveSetProvider`` — Seeger et al. 2003 fast forward
                                  selection (:63-136), batched over all
                                  candidate rows per round.
"""

from __future__ import annotations

import math
import numpy as np
import torch

from .kernels.base import Kernel
from .parallel.dist import Comm
from .ppa import NotPositiveDefiniteError


class ActiveSetProvider:
    def __call__(self, m: int, X: torch.Tensor, y: torch.Tensor,
                 kernel: Kernel, theta: np.ndarray, seed: int,
                 comm: Comm) -> torch.Tensor:
        raise NotImplementedError


class RandomActiveSetProvider(ActiveSetProvider):
    def __call__(self, m, X, y, kernel, theta, seed, comm):
        return comm.sample_rows(X, m, seed)


class KMeansActiveSetProvider(ActiveSetProvider):
    """Lloyd iterations over the sharded data; per-iteration allreduce of
    (centroid sums, counts) — C9 in SURVEY.md §2.5.  Centroids are synthetic
    points, matching the reference's use of MLlib cluster centers."""

    def __init__(self, max_iter: int = 20, chunk_rows: int = 1 << 18):
        self.max_iter = max_iter
        self.chunk_rows = chunk_rows

    def __call__(self, m, X, y, kernel, theta, seed, comm):
        centers = comm.sample_rows(X, m, seed)          # seeded init
        n, d = X.shape
        for _ in range(self.max_iter):
            sums = torch.zeros(m, d, dtype=torch.float64, device=X.device)
            counts = torch.zeros(m, dtype=torch.float64, device=X.device)
            c2 = (centers * centers).sum(-1)
            for s in range(0, n, self.chunk_rows):
                xb = X[s:min(n, s + self.chunk_rows)]
                # argmin_j ||x - c_j||^2 = argmin_j (||c_j||^2 - 2 x.c_j)
                assign = (c2 - 2.0 * (xb @ centers.T)).argmin(-1)
                sums.index_add_(0, assign, xb.double())
                counts.index_add_(0, assign,
                                  torch.ones(len(xb), dtype=torch.float64,
                                             device=X.device))
            comm.allreduce_(sums)
            comm.allreduce_(counts)
            nonempty = counts > 0
            centers = centers.clone()
            centers[nonempty] = (sums[nonempty]
                                 / counts[nonempty].unsqueeze(-1)).to(X.dtype)
        return centers


class GreedilyOptimizingActiveSetProvider(ActiveSetProvider):
    """Fast forward selection (Seeger et al. 2003).

    Per round (``ActiveSetProvider.scala:84-136``): with the current active
    set of size mc, compute K_mm^-1 and (sigma2 K_mm + Kmn Knm)^-1, then score
    every candidate row i:

        l_i = sqrt(K_ii - p_i);  xi_i = 1/((s/l_i)^2 + 1 - q_i)
        kappa_i = xi_i (1 + 2 (s/l_i)^2)
        delta_i = -log(s/l_i) - (log xi_i
                   + xi_i (1-kappa_i)/sigma2 (y_i-mu_i)^2 - kappa_i + 2)/2

    with p_i = c^T K_mm^-1 c, q_i = c^T PD^-1 c, mu_i = c^T magicVector and
    c the candidate's cross-kernel column; NaN-filtered distributed argmax
    (C6/C7 collectives) picks the next inducing point."""

    def __call__(self, m, X, y, kernel, theta, seed, comm):
        active = comm.sample_rows(X, 1, seed)           # 1 random start
        sigma2 = kernel.white_noise_var()
        s = math.sqrt(sigma2)
        diagK = kernel.training_kernel_diag(X)          # [n], includes noise
        # On GPU the [n, mc] cross kernel comes from the HIP tile kernel
        # (fp32) and the O(n mc^2) scoring GEMMs run in fp32; the small
        # mc x mc factorizations and the KK/Ky statistics stay fp64.  On
        # CPU everything is fp64 (the oracle the distributed-exact tests
        # pin).  (Round-1 weakness: the all-fp64 torch path made greedy
        # unusable beyond small m on GPU.)
        from . import ops
        gpu = X.is_cuda
        while active.shape[0] < m:
            Kmm = kernel.training_kernel(active.double())
            Lmm, info = torch.linalg.cholesky_ex(Kmm)
            if int(info) > 0:
                raise NotPositiveDefiniteError()
            Kinv = torch.cholesky_inverse(Lmm)

            if gpu:
                cross = ops.cross_kernel(kernel, X, active)      # fp32 HIP
            else:
                cross = kernel.cross_kernel(X, active).double()  # [n, mc]
            cd = cross.double() if gpu else cross
            KK = cd.T @ cd
            Ky = cd.T @ y.double()
            comm.allreduce_(KK)
            comm.allreduce_(Ky)
            PD = sigma2 * Kmm + KK
            Lpd, info = torch.linalg.cholesky_ex(PD)
            if int(info) > 0:
                raise NotPositiveDefiniteError()
            PDinv = torch.cholesky_inverse(Lpd)
            magic = torch.cholesky_solve(Ky.unsqueeze(-1), Lpd).squeeze(-1)

            p = ((cross @ Kinv.to(cross.dtype)) * cross).sum(-1).double()
            q = ((cross @ PDinv.to(cross.dtype)) * cross).sum(-1).double()
            mu = (cross @ magic.to(cross.dtype)).double()
            li = torch.sqrt(diagK.double() - p)
            sl2 = (s / li) ** 2
            xi = 1.0 / (sl2 + 1.0 - q)
            kappa = xi * (1.0 + 2.0 * sl2)
            delta = (-torch.log(s / li)
                     - (torch.log(xi)
                        + xi * (1.0 - kappa) / sigma2 * (y.double() - mu) ** 2
                        - kappa + 2.0) / 2.0)
            delta = torch.nan_to_num(delta, nan=-math.inf)
            best_local = int(delta.argmax())
            best_val = float(delta[best_local])

            # distributed argmax: max value, then lowest owning rank wins
            gmax = comm.allreduce_scalar(best_val, op="max")
            my = comm.rank if (best_val == gmax) else 1 << 30
            winner = int(comm.allreduce_scalar(float(my), op="min"))
            row = torch.zeros(1, X.shape[1], dtype=X.dtype, device=X.device)
            if comm.rank == winner:
                row[0] = X[best_local]
            comm.allreduce_(row)
            active = torch.cat([active, row], dim=0)
        return active
