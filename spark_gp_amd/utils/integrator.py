"""Gauss-Hermite expectation of a function of a normal variable.

Mirrors ``commons/util/Integrator.scala`` (dead code in the reference's main
path; here it powers ``predict_proba(averaged=True)``):

    E[f(Z)], Z ~ N(mu, var) = 1/sqrt(pi) * sum_i w_i f(sqrt(2) sd x_i + mu)
"""

from __future__ import annotations

import math

import numpy as np


class Integrator:
    def __init__(self, n: int):
        self.nodes, self.weights = np.polynomial.hermite.hermgauss(n)

    def expected_of_function_of_normal(self, mean: float, variance: float,
                                       f) -> float:
        sd = math.sqrt(variance)
        vals = f(math.sqrt(2.0) * sd * self.nodes + mean)
        return float((self.weights * vals).sum() / math.sqrt(math.pi))

    def expected_of_function_of_normal_batch(self, mean: np.ndarray,
                                             variance: np.ndarray, f
                                             ) -> np.ndarray:
        sd = np.sqrt(variance)[..., None]
        z = math.sqrt(2.0) * sd * self.nodes + mean[..., None]
        return (self.weights * f(z)).sum(-1) / math.sqrt(math.pi)
