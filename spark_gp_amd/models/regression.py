"""Gaussian Process Regression (BCM training + PPA prediction).

Mirrors ``regression/GaussianProcessRegression.scala``: the marginal
log-likelihood is approximated as the sum of independent per-expert
likelihoods (Bayesian Committee Machine, Deisenroth & Ng 2015); prediction is
the Projected Process Approximation (Rasmussen & Williams ch. 8.3.4).

The per-expert objective (``GaussianProcessRegression.scala:55-68``)
  L = 1/2 y^T K^-1 y + 1/2 log|K|,  grad_i = -1/2 sum(dK_i o (aa^T - K^-1))
runs batched over all local experts via the compiled fused path
(``spark_gp_amd.ops``) whenever the kernel tree canonicalizes; otherwise the
generic materialized-derivative path is used.
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

from .. import ops
from ..kernels.compiled import compile_kernel
from ..parallel.dist import get_comm
from ..utils.instrumentation import Instrumentation
from .base import GaussianProcessCommons, group_experts
from .predictor import GaussianProjectedProcessRawPredictor


class GaussianProcessRegression(GaussianProcessCommons):
    def fit(self, X, y) -> "GaussianProcessRegressionModel":
        instr = Instrumentation("GaussianProcessRegression")
        comm = get_comm()
        Xt, yt = self._prepare(X, y)
        groups = group_experts(Xt, yt, self._dataset_size_for_expert)

        kernel = self._get_kernel()
        cs = compile_kernel(kernel)

        def local_obj(theta: np.ndarray) -> Tuple[float, np.ndarray]:
            nll_total = 0.0
            grad_total = np.zeros_like(theta)
            for _, Xg, yg in groups:
                if cs is not None:
                    nll, grad = ops.nll_grad_compiled(cs, theta, Xg, yg)
                else:
                    nll, grad = ops.nll_grad_generic(kernel, theta, Xg, yg)
                nll_total += nll
                grad_total += grad
            return nll_total, grad_total

        theta = self._optimize_hypers(instr, comm, local_obj)
        raw = self._produce_predictor(instr, comm, Xt, yt, theta)
        instr.log_success()
        model = GaussianProcessRegressionModel(raw)
        model._instr = instr
        self.model_ = model           # sklearn-interop handle
        return model


class GaussianProcessRegressionModel:
    def __init__(self, raw: GaussianProjectedProcessRawPredictor):
        self.raw = raw
        self._instr: Optional[Instrumentation] = None

    def predict(self, X, return_std: bool = False):
        """PPA mean (and optionally the predictive std).

        The reference's public ``predict`` keeps only the mean
        (``GaussianProcessRegression.scala:79-81``); the variance is exposed
        here as an option since the raw predictor computes it anyway."""
        Xt = torch.as_tensor(X, dtype=self.raw.active_set.dtype,
                             device=self.raw.active_set.device)
        if Xt.dim() == 1:
            Xt = Xt.unsqueeze(0)
        mean, var = self.raw.predict(Xt, with_var=return_std)
        if return_std:
            return (mean.cpu().numpy(),
                    torch.sqrt(var.clamp_min(0.0)).cpu().numpy())
        return mean.cpu().numpy()

    def transform(self, X):
        """DataFrame-style alias of ``predict`` (Spark ``transform`` parity)."""
        return self.predict(X)
