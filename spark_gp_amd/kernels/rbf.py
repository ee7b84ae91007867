"""Stationary RBF-family kernels, batch-evaluated.

Semantics match the reference:

* ``RBFKernel`` — k(a,b) = exp(-||a-b||^2 / (2 sigma^2)), one hyper sigma,
  dK/dsigma = sqd .* K / sigma^3  (``kernel/RBFKernel.scala:50-64``).
* ``ARDRBFKernel`` — k(a,b) = exp(-||(a-b) .* beta||^2), p = d hypers,
  dK/dbeta_j = -2 beta_j (a_j-b_j)^2 .* K  (``kernel/ARDRBFKernel.scala:44-79``).
"""

from __future__ import annotations

import math
from typing import Tuple

import numpy as np
import torch

from .base import Kernel, sqdist, _as_f64


class RBFKernel(Kernel):
    def __init__(self, sigma: float = 1.0, lower: float = 1e-6,
                 upper: float = math.inf):
        self.sigma = float(sigma)
        self.lower = float(lower)
        self.upper = float(upper)

    # hypers
    def get_hyperparameters(self):
        return np.array([self.sigma])

    def set_hyperparameters(self, value):
        self.sigma = float(_as_f64(value)[0])
        return self

    @property
    def num_hyperparameters(self):
        return 1

    def hyperparameter_bounds(self):
        return np.array([self.lower]), np.array([self.upper])

    def white_noise_var(self):
        return 0.0

    # eval
    def training_kernel(self, X):
        sq = sqdist(X, X)
        return torch.exp(sq * (-1.0 / (2.0 * self.sigma ** 2)))

    def training_kernel_diag(self, X):
        return torch.ones(X.shape[:-1], dtype=X.dtype, device=X.device)

    def training_kernel_and_derivative(self, X) -> Tuple[torch.Tensor, torch.Tensor]:
        sq = sqdist(X, X)
        K = torch.exp(sq * (-1.0 / (2.0 * self.sigma ** 2)))
        dK = (sq * K / self.sigma ** 3).unsqueeze(-3)
        return K, dK

    def cross_kernel(self, Xtest, Xtrain):
        sq = sqdist(Xtest, Xtrain)
        return torch.exp(sq * (-1.0 / (2.0 * self.sigma ** 2)))

    def self_kernel(self, Xtest):
        return torch.ones(Xtest.shape[:-1], dtype=Xtest.dtype, device=Xtest.device)

    def __repr__(self):
        return f"RBFKernel(sigma={self.sigma:.1e})"


class ARDRBFKernel(Kernel):
    def __init__(self, p_or_beta, beta: float = 1.0, lower=0.0, upper=math.inf):
        if np.isscalar(p_or_beta):
            p = int(p_or_beta)
            self.beta = np.full(p, float(beta))
            self.lower = np.full(p, float(lower)) if np.isscalar(lower) else _as_f64(lower)
            self.upper = np.full(p, float(upper)) if np.isscalar(upper) else _as_f64(upper)
        else:
            self.beta = _as_f64(p_or_beta).copy()
            p = self.beta.size
            self.lower = (np.full(p, float(lower)) if np.isscalar(lower)
                          else _as_f64(lower))
            self.upper = (np.full(p, float(upper)) if np.isscalar(upper)
                          else _as_f64(upper))

    # hypers
    def get_hyperparameters(self):
        return self.beta.copy()

    def set_hyperparameters(self, value):
        self.beta = _as_f64(value).copy()
        return self

    @property
    def num_hyperparameters(self):
        return self.beta.size

    def hyperparameter_bounds(self):
        return self.lower.copy(), self.upper.copy()

    def white_noise_var(self):
        return 0.0

    def _beta_t(self, ref: torch.Tensor) -> torch.Tensor:
        return torch.as_tensor(self.beta, dtype=ref.dtype, device=ref.device)

    # eval
    def training_kernel(self, X):
        Xs = X * self._beta_t(X)
        return torch.exp(-sqdist(Xs, Xs))

    def training_kernel_diag(self, X):
        return torch.ones(X.shape[:-1], dtype=X.dtype, device=X.device)

    def training_kernel_and_derivative(self, X):
        beta = self._beta_t(X)
        Xs = X * beta
        K = torch.exp(-sqdist(Xs, Xs))
        # dK/dbeta_j = -2 beta_j (X_aj - X_bj)^2 * K — materialized [p, n, n]
        diff = X.unsqueeze(-2) - X.unsqueeze(-3)          # [..., n, n, d]
        d2 = diff * diff                                  # [..., n, n, d]
        dK = (-2.0 * beta) * d2                           # [..., n, n, d]
        dK = dK.permute(*range(dK.dim() - 3), -1, -3, -2) # [..., d, n, n]
        return K, dK * K.unsqueeze(-3)

    def cross_kernel(self, Xtest, Xtrain):
        beta = self._beta_t(Xtrain)
        return torch.exp(-sqdist(Xtest * beta, Xtrain * beta))

    def self_kernel(self, Xtest):
        return torch.ones(Xtest.shape[:-1], dtype=Xtest.dtype, device=Xtest.device)

    def __repr__(self):
        vals = ", ".join(f"{b:.1e}" for b in self.beta)
        return f"ARDRBFKernel(beta=[{vals}])"
