"""Compute-op dispatch: hand-written HIP/CDNA4 kernels on MI355X, batched
PyTorch everywhere else.

Policy (see repo README):
* On a CUDA/ROCm device, ops covered by the HIP extension MUST run through it;
  if the extension is missing on a GPU machine the op raises instead of
  silently falling back to eager PyTorch (set
  ``SPARK_GP_AMD_ALLOW_TORCH_FALLBACK=1`` to override, e.g. for A/B
  comparisons in tests).
* On CPU the torch backend is the float64 oracle.
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import numpy as np
import torch

from ..kernels.base import Kernel
from ..kernels.compiled import CompiledKernel
from . import torch_backend

_hip = None
_hip_err: Optional[str] = None


def _load_hip():
    global _hip, _hip_err
    if _hip is not None or _hip_err is not None:
        return _hip
    try:
        from . import hip_backend
        _hip = hip_backend
    except Exception as e:  # extension not built / not importable
        _hip_err = f"{type(e).__name__}: {e}"
    return _hip


def _allow_fallback() -> bool:
    return os.environ.get("SPARK_GP_AMD_ALLOW_TORCH_FALLBACK", "0") == "1"


def _force_torch() -> bool:
    """Debug/A-B switch: run the eager torch path even when HIP is present."""
    return os.environ.get("SPARK_GP_AMD_FORCE_TORCH", "0") == "1"


def hip_available() -> bool:
    return _load_hip() is not None


def _require_hip_or_fallback(what: str) -> bool:
    """True -> use HIP; False -> torch fallback allowed; raises otherwise."""
    if _load_hip() is not None:
        return True
    if _allow_fallback():
        return False
    raise RuntimeError(
        f"{what}: running on a GPU but the spark_gp_amd HIP extension is not "
        f"loadable ({_hip_err}); build it with `python setup.py build_ext "
        f"--inplace` (PYTORCH_ROCM_ARCH=gfx950) or set "
        f"SPARK_GP_AMD_ALLOW_TORCH_FALLBACK=1 to explicitly allow the eager "
        f"PyTorch path")


def nll_grad_compiled(cs: CompiledKernel, theta: np.ndarray,
                      X: torch.Tensor, y: torch.Tensor
                      ) -> Tuple[float, np.ndarray]:
    if X.is_cuda and cs.base in ("ard", "rbf") and not _force_torch():
        hip = _load_hip()
        if _require_hip_or_fallback("nll_grad_compiled") and \
                hip.supports_nll(cs, X):
            return hip.nll_grad_compiled(cs, theta, X, y)
    return torch_backend.nll_grad_compiled(cs, theta, X, y)


def nll_grad_generic(kernel: Kernel, theta: np.ndarray,
                     X: torch.Tensor, y: torch.Tensor):
    return torch_backend.nll_grad_generic(kernel, theta, X, y)


def laplace_nll_grad(kernel: Kernel, theta: np.ndarray, X: torch.Tensor,
                     y: torch.Tensor, f: torch.Tensor, tol: float,
                     max_newton_iter: int = 200, likelihood=None):
    from ..likelihoods import LogisticLikelihood
    logistic = likelihood is None or isinstance(likelihood,
                                                LogisticLikelihood)
    # the fused HIP Newton kernel implements the logistic link; other
    # likelihoods run on the batched torch path
    if X.is_cuda and logistic and not _force_torch():
        from ..kernels.compiled import compile_kernel
        cs = compile_kernel(kernel)
        hip = _load_hip()
        if (cs is not None
                and _require_hip_or_fallback("laplace_nll_grad")
                and tol >= hip.LAPLACE_MIN_TOL
                and hip.supports_laplace_evidence(cs, X)):
            # fully fused K10+K11: Newton AND the Algorithm 5.1 evidence
            # in one launch; None -> fp32 breakdown, warm torch fallback
            res = hip.laplace_evidence(cs, theta, X, y, f, tol,
                                       max_newton_iter)
            if res is not None:
                return res
            return torch_backend.laplace_nll_grad(
                kernel, theta, X, y, f, tol, max_newton_iter)
        if (cs is not None
                and _require_hip_or_fallback("laplace_nll_grad")
                and hip.supports_laplace(cs, X)):
            # fused Newton loop runs each expert to convergence on the GPU
            # (updates f in place); the torch pass below then converges in
            # 2-3 cheap iterations and computes the Algorithm 5.1 evidence
            # with the reference's exact semantics.
            n_bad = hip.laplace_newton(cs, theta, X, y, f, tol,
                                       max_newton_iter)
            if n_bad == 0 and tol >= hip.LAPLACE_MIN_TOL:
                # Algorithm 5.1 at the converged f via the contraction
                # form — no [E, p, k, k] derivative tensor
                return torch_backend.laplace_evidence_compiled(
                    cs, theta, X, y, f)
            # Some experts fell back, or the user asked for a tol tighter
            # than the fp32 kernel's clamp (LAPLACE_MIN_TOL): the torch
            # Newton polish continues from the warm f with the exact
            # requested tolerance (a few cheap iterations).
            return torch_backend.laplace_nll_grad(
                kernel, theta, X, y, f, tol, max_newton_iter)
    return torch_backend.laplace_nll_grad(kernel, theta, X, y, f, tol,
                                          max_newton_iter,
                                          likelihood=likelihood)


def kmn_knm_and_kmny(kernel: Kernel, active: torch.Tensor,
                     X: torch.Tensor, y: torch.Tensor,
                     chunk_rows: int = 262144, precision: str = "fp64"):
    if X.is_cuda and not _force_torch():
        hip = _load_hip()
        if _require_hip_or_fallback("kmn_knm_and_kmny") and \
                hip.supports_ppa(kernel, X):
            return hip.kmn_knm_and_kmny(kernel, active, X, y, chunk_rows,
                                        precision)
    return torch_backend.kmn_knm_and_kmny(kernel, active, X, y, chunk_rows)


def cross_kernel(kernel: Kernel, Xtest: torch.Tensor, Xtrain: torch.Tensor):
    """Cross-kernel dispatch for prediction paths."""
    if Xtest.is_cuda:
        hip = _load_hip()
        if hip is not None and hip.supports_ppa(kernel, Xtest):
            return hip.cross_kernel(kernel, Xtest, Xtrain)
    return kernel.cross_kernel(Xtest, Xtrain)
