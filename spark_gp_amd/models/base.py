"""Shared estimator machinery.

``GaussianProcessParams`` mirrors the Spark ML param set and defaults
(``commons/GaussianProcessParams.scala:8-54``); ``GaussianProcessCommons``
mirrors the shared training flow (``commons/GaussianProcessCommons.scala``):
expert grouping, the effective kernel ``user + sigma2.const * Eye``, the
distributed L-BFGS-B hyperparameter optimization, and PPA model construction.

MI355X-first deviations (recorded in SURVEY.md §7):
* experts are formed locally per rank from contiguous shards — no shuffle
  (C5); round-robin within the shard keeps sizes equal to +-1 row;
* expert problems execute as at most two *batched* tensor groups (one per
  expert size), not one task per expert.
"""

from __future__ import annotations

import time
from typing import Callable, List, Optional, Tuple

import numpy as np
import torch

from ..active_set import ActiveSetProvider, RandomActiveSetProvider
from ..kernels.base import EyeKernel, Kernel, Scalar
from ..kernels.rbf import RBFKernel
from ..optimize import lbfgsb
from ..parallel.dist import Comm
from ..ppa import accumulate_ppa_stats, magic_vector_matrix
from ..utils.instrumentation import Instrumentation
from .predictor import GaussianProjectedProcessRawPredictor


def group_experts(X: torch.Tensor, y: torch.Tensor, k_target: int
                  ) -> List[Tuple[torch.Tensor, torch.Tensor, torch.Tensor]]:
    """Round-robin expert assignment within the local shard.

    numberOfExperts = round(n / k_target) and expert(row i) = i % E, matching
    ``GaussianProcessCommons.scala:26-31``; experts are regrouped by size into
    at most two uniformly-shaped batches [E_g, k_g, d].  Each group carries
    its flat row-index tensor so per-row state (e.g. GPC latent f) can be
    scattered back to shard order."""
    n, d = X.shape
    E = max(1, int(round(n / k_target)))
    lo = n // E
    r = n - lo * E            # first r experts get lo+1 rows
    groups: List[Tuple[torch.Tensor, torch.Tensor, torch.Tensor]] = []
    dev = X.device
    if r > 0:
        j = torch.arange(lo + 1, device=dev).unsqueeze(-1)    # [k_hi, 1]
        e = torch.arange(r, device=dev).unsqueeze(0)          # [1, r]
        idx = (j * E + e).T.reshape(-1)                       # [r*(lo+1)]
        groups.append((idx, X[idx].reshape(r, lo + 1, d),
                       y[idx].reshape(r, lo + 1)))
    if E - r > 0 and lo > 0:
        j = torch.arange(lo, device=dev).unsqueeze(-1)
        e = torch.arange(r, E, device=dev).unsqueeze(0)
        idx = (j * E + e).T.reshape(-1)
        groups.append((idx, X[idx].reshape(E - r, lo, d),
                       y[idx].reshape(E - r, lo)))
    return groups


class GaussianProcessParams:
    """Fluent parameter mixin with the reference's defaults.

    Also sklearn-compatible: ``get_params``/``set_params``/keyword
    construction follow the scikit-learn estimator contract, so
    ``sklearn.base.clone`` and tools like ``cross_val_score`` work (the
    estimator additionally keeps its last fitted model in ``model_`` and
    delegates ``predict`` to it for those tools; the primary API remains
    Spark-style ``fit(X, y) -> Model``)."""

    # sklearn param name -> backing attribute
    _PARAMS = {
        "kernel": "_kernel_factory",
        "dataset_size_for_expert": "_dataset_size_for_expert",
        "sigma2": "_sigma2",
        "active_set_size": "_active_set_size",
        "active_set_provider": "_active_set_provider",
        "max_iter": "_max_iter",
        "tol": "_tol",
        "seed": "_seed",
        "device": "_device",
        "dtype": "_dtype",
        "ppa_precision": "_ppa_precision",
        "optimizer_restart": "_optimizer_restart",
    }

    def __init__(self, **params):
        self._kernel_factory: Callable[[], Kernel] = lambda: RBFKernel()
        self._dataset_size_for_expert = 100
        self._sigma2 = 1e-3
        self._active_set_size = 100
        self._active_set_provider: ActiveSetProvider = RandomActiveSetProvider()
        self._max_iter = 100
        self._tol = 1e-6
        self._seed = 0
        self._device: Optional[str] = None        # None -> auto
        self._dtype: Optional[torch.dtype] = None  # None -> f64 CPU / f32 GPU
        self._ppa_precision = "fp64"  # 'fp64' (reference-parity) | 'mixed'
                                      # (hi/lo bf16 MFMA SYRK, fastest)
        self._optimizer_restart = True  # restart-on-bound-collapse guard
        self.model_ = None            # last fitted model (sklearn interop)
        if params:
            self.set_params(**params)

    # sklearn estimator contract --------------------------------------
    def get_params(self, deep: bool = True):
        return {name: getattr(self, attr) for name, attr in self._PARAMS.items()}

    def set_params(self, **params):
        for name, value in params.items():
            attr = self._PARAMS.get(name)
            if attr is None:
                raise ValueError(f"unknown parameter {name!r}; valid: "
                                 f"{sorted(self._PARAMS)}")
            setattr(self, attr, value)
        return self

    def predict(self, X, **kw):
        """Delegate to the last fitted model (sklearn tools call
        ``est.fit(...); est.predict(...)`` on the estimator itself)."""
        if self.model_ is None:
            raise RuntimeError("estimator is not fitted; call fit(X, y)")
        return self.model_.predict(X, **kw)

    def score(self, X, y):
        """Default sklearn scorer: R^2 for regression, accuracy for
        classification (decided by the fitted model's kind)."""
        import numpy as _np
        pred = self.predict(X)
        y = _np.asarray(y, dtype=_np.float64).reshape(-1)
        from .classification import GaussianProcessClassificationModel
        if isinstance(self.model_, GaussianProcessClassificationModel):
            return float((pred == y).mean())
        ss_res = float(((y - pred) ** 2).sum())
        ss_tot = float(((y - y.mean()) ** 2).sum())
        return 1.0 - ss_res / ss_tot if ss_tot > 0 else 0.0

    # Reference-parity camelCase setters -------------------------------
    def setKernel(self, factory: Callable[[], Kernel]):
        self._kernel_factory = factory
        return self

    def setDatasetSizeForExpert(self, v: int):
        self._dataset_size_for_expert = int(v)
        return self

    def setSigma2(self, v: float):
        self._sigma2 = float(v)
        return self

    def setActiveSetSize(self, v: int):
        self._active_set_size = int(v)
        return self

    def setActiveSetProvider(self, p: ActiveSetProvider):
        self._active_set_provider = p
        return self

    def setMaxIter(self, v: int):
        self._max_iter = int(v)
        return self

    def setTol(self, v: float):
        """Convergence tolerance for L-BFGS-B and the Laplace Newton loops.
        Honored exactly on every path: on GPU the fused Newton kernel
        (fp32 matrices, fp64 objective accumulation) converges to
        max(tol, 1e-6) — measured at oracle parity there — and a float64
        torch polish finishes any tighter request from the warm latent."""
        self._tol = float(v)
        return self

    def setSeed(self, v: int):
        self._seed = int(v)
        return self

    # Additive (no reference analog)
    def setDevice(self, device: str):
        self._device = device
        return self

    def setOptimizerRestart(self, v: bool):
        """Enable/disable the restart-on-bound-collapse guard (an additive
        robustness feature; the reference runs exactly one L-BFGS-B solve)."""
        self._optimizer_restart = bool(v)
        return self

    def setPpaPrecision(self, p: str):
        """'fp64' (default; matches the reference's double-precision PPA) or
        'mixed' (hi/lo-split bf16 MFMA SYRK with fp32 accumulation — the
        fastest path; fine when sigma2 is not tiny)."""
        if p not in ("fp64", "mixed"):
            raise ValueError("ppa precision must be 'fp64' or 'mixed'")
        self._ppa_precision = p
        return self

    def setDtype(self, dtype: torch.dtype):
        self._dtype = dtype
        return self

    # snake_case aliases
    set_kernel = setKernel
    set_dataset_size_for_expert = setDatasetSizeForExpert
    set_sigma2 = setSigma2
    set_active_set_size = setActiveSetSize
    set_active_set_provider = setActiveSetProvider
    set_max_iter = setMaxIter
    set_tol = setTol
    set_seed = setSeed


class GaussianProcessCommons(GaussianProcessParams):
    def _resolve_device(self) -> torch.device:
        if self._device is not None:
            return torch.device(self._device)
        return torch.device("cuda") if torch.cuda.is_available() \
            else torch.device("cpu")

    def _resolve_dtype(self, device: torch.device) -> torch.dtype:
        if self._dtype is not None:
            return self._dtype
        return torch.float32 if device.type == "cuda" else torch.float64

    def _get_kernel(self) -> Kernel:
        """Effective kernel = user kernel + sigma2.const * Eye
        (``GaussianProcessCommons.scala:18``)."""
        return self._kernel_factory() + Scalar(self._sigma2).const * EyeKernel()

    def _prepare(self, X, y) -> Tuple[torch.Tensor, torch.Tensor]:
        device = self._resolve_device()
        dtype = self._resolve_dtype(device)
        Xt = torch.as_tensor(X, dtype=dtype, device=device)
        yt = torch.as_tensor(y, dtype=dtype, device=device).reshape(-1)
        if Xt.dim() == 1:
            Xt = Xt.unsqueeze(-1)
        if Xt.shape[0] != yt.shape[0]:
            raise ValueError("X and y row counts differ")
        return Xt, yt

    def _optimize_hypers(self, instr: Instrumentation, comm: Comm,
                         local_obj: Callable[[np.ndarray], Tuple[float, np.ndarray]]
                         ) -> np.ndarray:
        """Distributed L-BFGS-B (``GaussianProcessCommons.scala:66-92``).

        Each evaluation allreduces a (1+p) float64 payload (C1); the optimizer
        itself runs replicated and deterministically on every rank."""
        instr.log("Optimising the kernel hyperparameters")
        kernel = self._get_kernel()
        x0 = kernel.get_hyperparameters()
        lower, upper = kernel.hyperparameter_bounds()

        evals = [0, 0.0]

        import os as _os
        trace = _os.environ.get("SPARK_GP_AMD_TRACE_OBJECTIVE") == "1"

        def objective(theta: np.ndarray) -> Tuple[float, np.ndarray]:
            te = time.perf_counter()
            nll, grad = local_obj(theta)
            buf = np.concatenate([[nll], grad])
            buf = comm.allreduce_np(buf)
            evals[0] += 1
            evals[1] += time.perf_counter() - te
            if trace and comm.rank == 0:
                print(f"[obj] eval={evals[0]} nll={float(buf[0]):.10g}",
                      flush=True)
            return float(buf[0]), buf[1:]

        t0 = time.perf_counter()
        opt = lbfgsb(objective, x0, lower, upper,
                     max_iter=self._max_iter, tol=self._tol,
                     restart_on_bound_collapse=self._optimizer_restart)
        instr.log_timing("optimize_hypers", time.perf_counter() - t0)
        instr.timings["objective_evals"] = evals[0]
        instr.timings["objective_time"] = evals[1]
        optimal = self._get_kernel().set_hyperparameters(opt)
        instr.log(f"Optimal kernel: {optimal!r}")
        return opt

    def _produce_predictor(self, instr: Instrumentation, comm: Comm,
                           X: torch.Tensor, y: torch.Tensor,
                           theta: np.ndarray
                           ) -> GaussianProjectedProcessRawPredictor:
        """Active set -> PPA stats (C2 allreduce) -> magic quantities
        (``GaussianProcessCommons.scala:40-59``)."""
        kernel = self._get_kernel().set_hyperparameters(theta)

        def stage_clock():
            # device sync at every stage boundary so the published per-stage
            # attribution is true wall time, not enqueue time (async GPU
            # work would otherwise be billed to whichever later stage first
            # blocks on it)
            if X.is_cuda:
                torch.cuda.synchronize(X.device)
            return time.perf_counter()

        t0 = stage_clock()
        active = self._active_set_provider(
            self._active_set_size, X, y, kernel, theta, self._seed, comm)
        t1 = stage_clock()
        instr.log_timing("active_set", t1 - t0)

        KK, Ky = accumulate_ppa_stats(kernel, active, X, y, comm,
                                      precision=self._ppa_precision)
        t2 = stage_clock()
        instr.log_timing("ppa_accumulate", t2 - t1)

        mv, mm = magic_vector_matrix(kernel, KK, Ky, active)
        t3 = stage_clock()
        instr.log_timing("magic_solve", t3 - t2)
        return GaussianProjectedProcessRawPredictor(mv, mm, kernel, active)
