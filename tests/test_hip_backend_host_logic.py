"""CPU tests of ops/hip_backend.py's HOST-side logic (gradient assembly,
bad-expert handling): the HIP extension is monkeypatched with a fake that
produces the same raw per-expert statistics from fp64 torch math, so the
host chain rule must reproduce torch_backend's full result exactly.
(The real-kernel equivalents run under @gpu in test_hip_kernels.py.)"""

import numpy as np
import pytest
import torch

from spark_gp_amd.kernels import (ARDRBFKernel, EyeKernel, Scalar,
                                  compile_kernel, sqdist)
from spark_gp_amd.ops import hip_backend, torch_backend


def _raw_stats(cs, theta, X, y):
    """Per-expert raw statistics exactly as the fused kernel defines them
    (expert_nll.hip outputs), computed in fp64 torch."""
    C = cs.amp(theta)
    nu = cs.noise(theta)
    beta = torch.as_tensor(theta[cs.base_idx], dtype=X.dtype)
    Xs = X * beta
    Kb = torch.exp(-sqdist(Xs, Xs))
    k = X.shape[-2]
    K = C * Kb + nu * torch.eye(k, dtype=X.dtype)
    L = torch.linalg.cholesky(K)
    logdet = 2.0 * torch.log(L.diagonal(dim1=-2, dim2=-1)).sum(-1)
    Kinv = torch.cholesky_inverse(L)
    alpha = (Kinv @ y.unsqueeze(-1)).squeeze(-1)
    nll = 0.5 * (y * alpha).sum(-1) + 0.5 * logdet
    G = alpha.unsqueeze(-1) * alpha.unsqueeze(-2) - Kinv
    W0 = G * Kb
    sumW0 = W0.sum((-1, -2))
    trG = G.diagonal(dim1=-2, dim2=-1).sum(-1)
    r = W0.sum(-1)
    contr = (2.0 * (X * X * r.unsqueeze(-1)).sum(-2)
             - 2.0 * (X * (W0 @ X)).sum(-2))
    bad = torch.zeros(X.shape[0], dtype=torch.int32)
    return nll, sumW0, trG, contr, bad


class _FakeExt:
    """Stands in for the HIP extension; carries the exact fp64 theta (the
    real kernel receives an fp32 scale vector, so reconstructing theta from
    the call would add fp32 noise the host-logic comparison doesn't want)."""

    def __init__(self, cs, theta, bad_mask=None):
        self.cs, self.theta, self.bad_mask = cs, theta, bad_mask

    def fused_expert_nll(self, X, y, scale, amp, noise):
        out = list(_raw_stats(self.cs, self.theta, X.double(), y.double()))
        if self.bad_mask is not None:
            bad = out[-1].clone()
            bad[self.bad_mask] = 1
            # the kernel zeroes a bad expert's outputs
            for t in out[:3]:
                t[self.bad_mask] = 0.0
            out[3][self.bad_mask] = 0.0
            out[-1] = bad
        return tuple(out)


@pytest.fixture
def problem():
    g = torch.Generator().manual_seed(9)
    E, k, d = 5, 24, 4
    X = torch.rand(E, k, d, generator=g, dtype=torch.float64)
    y = torch.sin(3 * X.sum(-1))
    cs = compile_kernel(1 * ARDRBFKernel(d) + Scalar(1e-3).const * EyeKernel())
    rng = np.random.default_rng(3)
    theta = np.concatenate([[1.1], rng.uniform(0.5, 2.0, d)])
    return cs, theta, X, y


def test_host_gradient_assembly_matches_torch_backend(problem, monkeypatch):
    cs, theta, X, y = problem
    monkeypatch.setattr(hip_backend, "ext", _FakeExt(cs, theta))
    nll_h, grad_h = hip_backend.nll_grad_compiled(cs, theta, X, y)
    nll_t, grad_t = torch_backend.nll_grad_compiled(cs, theta, X, y)
    # hip_backend quantizes y to fp32 at the extension boundary (the real
    # kernel's input dtype), so agreement is fp32-level; assembly-formula
    # mistakes would be orders of magnitude larger
    assert nll_h == pytest.approx(nll_t, rel=1e-6)
    np.testing.assert_allclose(grad_h, grad_t, rtol=1e-4)


def test_host_bad_expert_fallback_recomputes_exactly(problem, monkeypatch):
    """Experts flagged bad must be recomputed on the torch force_lu path and
    the combined result must equal the all-torch answer."""
    cs, theta, X, y = problem
    monkeypatch.setattr(hip_backend, "ext",
                        _FakeExt(cs, theta, bad_mask=[1, 3]))
    nll_h, grad_h = hip_backend.nll_grad_compiled(cs, theta, X, y)
    nll_t, grad_t = torch_backend.nll_grad_compiled(cs, theta, X, y)
    assert nll_h == pytest.approx(nll_t, rel=1e-6)
    np.testing.assert_allclose(grad_h, grad_t, rtol=1e-4)
