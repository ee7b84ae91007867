"""Gaussian Process Poisson (count) regression — an additive model family
the reference does not have.

Same BCM + Laplace architecture as binary classification
(``classification/GaussianProcessClassifier.scala``) with the Poisson
log-link likelihood plugged into Algorithms 3.1/5.1
(``spark_gp_amd/likelihoods.py``): per-expert Newton mode finding on the
latent log-rate f, approximate evidence for the hyperparameter
optimization, and the PPA fit on the converged latent.  Prediction returns
the posterior-expected rate E[exp(f*)] = exp(mu + var/2) (lognormal mean),
using the PPA latent variance.
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

from .. import ops
from ..likelihoods import PoissonLikelihood
from ..parallel.dist import get_comm
from ..utils.instrumentation import Instrumentation
from .base import GaussianProcessCommons, group_experts
from .predictor import GaussianProjectedProcessRawPredictor


class GaussianProcessPoissonRegression(GaussianProcessCommons):
    _PARAMS = dict(GaussianProcessCommons._PARAMS,
                   max_newton_iter="_max_newton_iter")

    def __init__(self, **params):
        self._max_newton_iter = 200
        super().__init__(**params)

    def setMaxNewtonIter(self, v: int):
        self._max_newton_iter = int(v)
        return self

    def fit(self, X, y) -> "GaussianProcessPoissonModel":
        instr = Instrumentation("GaussianProcessPoissonRegression")
        comm = get_comm()
        Xt, yt = self._prepare(X, y)
        lik = PoissonLikelihood()

        ok = lik.validate_targets(yt)
        ok = comm.allreduce_scalar(1.0 if ok else 0.0, op="min") > 0.5
        if not ok:
            raise ValueError("targets must be non-negative integer counts")

        groups = group_experts(Xt, yt, self._dataset_size_for_expert)
        kernel = self._get_kernel()
        # latent log-rate per expert; log(1+y) is a cheap feasible start
        fs = [torch.log1p(yg) for _, _, yg in groups]

        def local_obj(theta: np.ndarray) -> Tuple[float, np.ndarray]:
            nll_total, grad_total = 0.0, np.zeros_like(theta)
            for (idx, Xg, yg), fg in zip(groups, fs):
                nll, grad = ops.laplace_nll_grad(
                    kernel, theta, Xg, yg, fg, self._tol,
                    self._max_newton_iter, likelihood=lik)
                nll_total += nll
                grad_total += grad
            return nll_total, grad_total

        theta = self._optimize_hypers(instr, comm, local_obj)
        local_obj(theta)               # refresh f at theta*

        f_flat = torch.empty_like(yt)
        for (idx, _, _), fg in zip(groups, fs):
            f_flat[idx] = fg.reshape(-1)
        raw = self._produce_predictor(instr, comm, Xt, f_flat, theta)
        instr.log_success()
        model = GaussianProcessPoissonModel(raw)
        model._instr = instr
        self.model_ = model
        return model


class GaussianProcessPoissonModel:
    # training clips the latent at PoissonLikelihood.fmax (=30); clamp the
    # extrapolated PPA latent mean to the same ceiling so an
    # out-of-distribution query cannot overflow exp() to inf rate/std
    fmax = 30.0

    def __init__(self, raw: GaussianProjectedProcessRawPredictor):
        self.raw = raw
        self._instr: Optional[Instrumentation] = None

    def _latent(self, X) -> Tuple[torch.Tensor, torch.Tensor]:
        Xt = torch.as_tensor(X, dtype=self.raw.active_set.dtype,
                             device=self.raw.active_set.device)
        if Xt.dim() == 1:
            Xt = Xt.unsqueeze(0)
        return self.raw.predict(Xt, with_var=True)

    def predict(self, X, return_std: bool = False):
        """Posterior-expected count rate E[exp(f*)] = exp(mu + var/2).

        With ``return_std``: the posterior-PREDICTIVE count std via the law
        of total variance, Var[y*] = E[lambda] + Var[lambda] with lognormal
        latent moments (Var[lambda] = (exp(var) - 1) exp(2 mu + var))."""
        mu, var = self._latent(X)
        mu = mu.clamp(max=self.fmax)
        var = var.clamp_min(0.0)
        rate = torch.exp(mu + 0.5 * var)
        if not return_std:
            return rate.cpu().numpy()
        var_lam = torch.expm1(var) * torch.exp(2.0 * mu + var)
        std = torch.sqrt(rate + var_lam)
        return rate.cpu().numpy(), std.cpu().numpy()

    def predict_latent(self, X):
        """(mean, var) of the latent log-rate."""
        mu, var = self._latent(X)
        return mu.cpu().numpy(), var.cpu().numpy()

    def transform(self, X):
        return self.predict(X)
