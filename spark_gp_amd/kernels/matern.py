"""Matérn-family stationary kernels (additive capability — the reference
ships only the RBF family, ``kernel/RBFKernel.scala`` /
``kernel/ARDRBFKernel.scala``; Matérn 3/2 and 5/2 are the standard
rough-process complements, Rasmussen & Williams ch. 4.2).

Both take one trainable lengthscale ``l``:

* ``Matern32Kernel`` — k(r) = (1 + s) exp(-s),            s = sqrt(3) r / l,
  dk/dl = s^2 exp(-s) / l.
* ``Matern52Kernel`` — k(r) = (1 + s + s^2/3) exp(-s),    s = sqrt(5) r / l,
  dk/dl = s^2 (1 + s) exp(-s) / (3 l).

They plug into the same DSL (`+`, scalar `*`, ``TrainableScalar``) and run
through the generic objective path (``ops.nll_grad_generic``); there is no
fused HIP fast path for them (the canonicalizer in ``kernels/compiled.py``
recognizes the RBF family only).
"""

from __future__ import annotations

import math
from typing import Tuple

import numpy as np
import torch

from .base import Kernel, sqdist, _as_f64

_EPS = 1e-30  # guards r=0 in d/dl (the derivative -> 0 there anyway)


class _MaternBase(Kernel):
    def __init__(self, l: float = 1.0, lower: float = 1e-6,
                 upper: float = math.inf):
        self.l = float(l)
        self.lower = float(lower)
        self.upper = float(upper)

    def get_hyperparameters(self):
        return np.array([self.l])

    def set_hyperparameters(self, value):
        self.l = float(_as_f64(value)[0])
        return self

    @property
    def num_hyperparameters(self):
        return 1

    def hyperparameter_bounds(self):
        return np.array([self.lower]), np.array([self.upper])

    def white_noise_var(self):
        return 0.0

    def training_kernel_diag(self, X):
        return torch.ones(X.shape[:-1], dtype=X.dtype, device=X.device)

    def self_kernel(self, Xtest):
        return torch.ones(Xtest.shape[:-1], dtype=Xtest.dtype,
                          device=Xtest.device)

    def _r(self, A, B):
        return torch.sqrt(sqdist(A, B).clamp_min(0.0) + _EPS)


class Matern32Kernel(_MaternBase):
    def _k_of_s(self, s):
        return (1.0 + s) * torch.exp(-s)

    def training_kernel(self, X):
        s = self._r(X, X) * (math.sqrt(3.0) / self.l)
        return self._k_of_s(s)

    def training_kernel_and_derivative(self, X) -> Tuple[torch.Tensor, torch.Tensor]:
        s = self._r(X, X) * (math.sqrt(3.0) / self.l)
        e = torch.exp(-s)
        K = (1.0 + s) * e
        dK = (s * s * e / self.l).unsqueeze(-3)          # [..., 1, n, n]
        return K, dK

    def cross_kernel(self, Xtest, Xtrain):
        s = self._r(Xtest, Xtrain) * (math.sqrt(3.0) / self.l)
        return self._k_of_s(s)

    def __repr__(self):
        return f"Matern32Kernel(l={self.l:.1e})"


class Matern52Kernel(_MaternBase):
    def _k_of_s(self, s):
        return (1.0 + s + s * s / 3.0) * torch.exp(-s)

    def training_kernel(self, X):
        s = self._r(X, X) * (math.sqrt(5.0) / self.l)
        return self._k_of_s(s)

    def training_kernel_and_derivative(self, X) -> Tuple[torch.Tensor, torch.Tensor]:
        s = self._r(X, X) * (math.sqrt(5.0) / self.l)
        e = torch.exp(-s)
        K = (1.0 + s + s * s / 3.0) * e
        dK = (s * s * (1.0 + s) * e / (3.0 * self.l)).unsqueeze(-3)
        return K, dK

    def cross_kernel(self, Xtest, Xtrain):
        s = self._r(Xtest, Xtrain) * (math.sqrt(5.0) / self.l)
        return self._k_of_s(s)

    def __repr__(self):
        return f"Matern52Kernel(l={self.l:.1e})"
