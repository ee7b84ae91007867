"""Distributed z-score standardization.

Mirrors ``commons/util/Scaling.scala:10-25``: population mean/variance via
two reductions (C11: two allreduce(SUM) of 2*d doubles), zero-variance
dimensions mapped to scale 1.  Used by the Airfoil/MNIST examples; not called
inside the estimators (same as the reference).
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

from ..parallel.dist import Comm, get_comm


class StandardScaler:
    def __init__(self):
        self.mean: Optional[np.ndarray] = None
        self.scale: Optional[np.ndarray] = None

    def fit(self, X, comm: Optional[Comm] = None) -> "StandardScaler":
        comm = comm or get_comm()
        Xt = torch.as_tensor(X, dtype=torch.float64)
        n = comm.allreduce_scalar(float(Xt.shape[0]))
        s = Xt.sum(0)
        comm.allreduce_(s)
        mean = s / n
        v = ((Xt - mean) ** 2).sum(0)
        comm.allreduce_(v)
        var = (v / n).numpy()
        self.mean = mean.numpy()
        self.scale = np.sqrt(np.where(var > 0.0, var, 1.0))
        return self

    def transform(self, X):
        Xn = np.asarray(X, dtype=np.float64)
        return (Xn - self.mean) / self.scale

    def fit_transform(self, X, comm: Optional[Comm] = None):
        return self.fit(X, comm).transform(X)

    def inverse_transform(self, X):
        Xn = np.asarray(X, dtype=np.float64)
        return Xn * self.scale + self.mean


def scale(X, y, comm: Optional[Comm] = None) -> Tuple[np.ndarray, np.ndarray]:
    """Convenience mirroring the reference's ``scale(RDD[LabeledPoint])``."""
    return StandardScaler().fit_transform(X, comm), np.asarray(y, dtype=np.float64)
