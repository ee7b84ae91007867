"""Box-constrained L-BFGS-B driver for hyperparameter optimization.

Replaces Breeze's ``LBFGSB`` (``commons/GaussianProcessCommons.scala:84-86``)
with scipy's reference Fortran L-BFGS-B, wrapped with:

* an exact-key memo cache mirroring ``commons/util/DiffFunctionMemoized.scala``
  (absorbs re-evaluations at identical points);
* deterministic replication: in distributed mode EVERY rank runs the same
  optimizer over the same allreduced objective, so iterates stay bit-identical
  across ranks and no per-iteration broadcast of theta is needed (C10 in
  SURVEY.md §2.5).
"""

from __future__ import annotations

from typing import Callable, Dict, Tuple

import numpy as np
from scipy.optimize import minimize


class MemoizedObjective:
    """f(x) -> (value, grad) with a hash-map cache keyed on the exact vector
    bytes (``DiffFunctionMemoized.scala:8-15``)."""

    def __init__(self, fn: Callable[[np.ndarray], Tuple[float, np.ndarray]]):
        self.fn = fn
        self.cache: Dict[bytes, Tuple[float, np.ndarray]] = {}
        self.evals = 0          # objective evaluations actually computed

    def __call__(self, x: np.ndarray) -> Tuple[float, np.ndarray]:
        key = np.asarray(x, dtype=np.float64).tobytes()
        hit = self.cache.get(key)
        if hit is not None:
            return hit
        self.evals += 1
        val = self.fn(np.asarray(x, dtype=np.float64))
        self.cache[key] = val
        return val


def lbfgsb(fn: Callable[[np.ndarray], Tuple[float, np.ndarray]],
           x0: np.ndarray, lower: np.ndarray, upper: np.ndarray,
           max_iter: int = 100, tol: float = 1e-6,
           restart_on_bound_collapse: bool = True) -> np.ndarray:
    """Minimize fn subject to lower <= x <= upper; returns the argmin.

    ``tol`` maps to both the relative-f tolerance (ftol) and the projected
    gradient tolerance, approximating Breeze LBFGSB's convergence test.

    ``restart_on_bound_collapse``: the BCM marginal likelihood has a
    degenerate local optimum where length-scale-type hyperparameters collapse
    to their lower bound and a huge constant amplitude absorbs the label mean
    (the model then predicts a constant).  When phase 1 ends with
    coordinates parked at their lower bound, one restart is performed from
    the phase-1 solution with those coordinates nudged slightly inward, and
    the better of the two results is kept.  This leaves non-degenerate
    solves untouched and is a pure objective-value improvement."""
    memo = MemoizedObjective(fn)
    x0 = np.asarray(x0, dtype=np.float64)
    lower = np.asarray(lower, dtype=np.float64)
    upper = np.asarray(upper, dtype=np.float64)
    bounds = [(float(l) if np.isfinite(l) else None,
               float(u) if np.isfinite(u) else None)
              for l, u in zip(lower, upper)]
    opts = {"maxiter": max_iter, "ftol": tol, "gtol": tol, "maxls": 40}
    res = minimize(memo, x0, jac=True, method="L-BFGS-B", bounds=bounds,
                   options=opts)
    x_best, f_best = np.asarray(res.x, dtype=np.float64), float(res.fun)

    if restart_on_bound_collapse:
        at_lower = np.isclose(x_best, lower) & (x0 > lower)
        if at_lower.any():
            x1 = x_best.copy()
            nudge = np.where(np.isfinite(x0),
                             lower + 0.01 * np.maximum(x0 - lower, 1.0),
                             lower + 0.01)
            x1[at_lower] = nudge[at_lower]
            res2 = minimize(memo, x1, jac=True, method="L-BFGS-B",
                            bounds=bounds, options=opts)
            if float(res2.fun) < f_best:
                x_best, f_best = np.asarray(res2.x, dtype=np.float64), float(res2.fun)
    return x_best
