"""Property-based invariants of the kernel DSL (hypothesis):

* every kernel's training matrix is symmetric PSD (+ noise diag where
  applicable);
* hyperparameter get/set round-trips through the composite tree layout;
* ``compile_kernel``'s canonical (C, base, nu) evaluation agrees with the
  direct DSL tree evaluation for every canonicalizable tree;
* deterministic refit: identical seeds produce identical models.
"""

import numpy as np
import pytest
import torch
from hypothesis import given, settings, strategies as st

from spark_gp_amd.kernels import (ARDRBFKernel, EyeKernel, Matern32Kernel,
                                  Matern52Kernel, RBFKernel, Scalar,
                                  compile_kernel)

DIM = 3


def _data(n, seed):
    rng = np.random.default_rng(seed)
    return torch.as_tensor(rng.normal(size=(n, DIM)))


kernel_factories = st.sampled_from([
    lambda hp: RBFKernel(hp),
    lambda hp: ARDRBFKernel(np.full(DIM, hp)),
    lambda hp: Matern32Kernel(hp),
    lambda hp: Matern52Kernel(hp),
])


@settings(max_examples=30, deadline=None)
@given(factory=kernel_factories,
       hp=st.floats(0.05, 5.0),
       amp=st.floats(0.1, 10.0),
       noise=st.floats(1e-6, 1e-1),
       n=st.integers(2, 12),
       seed=st.integers(0, 10))
def test_training_kernel_symmetric_psd(factory, hp, amp, noise, n, seed):
    k = amp * factory(hp) + Scalar(noise).const * EyeKernel()
    X = _data(n, seed)
    K = k.training_kernel(X).numpy()
    np.testing.assert_allclose(K, K.T, atol=1e-12)
    w = np.linalg.eigvalsh(K)
    assert w.min() > -1e-9          # PSD up to fp noise (noise diag added)


@settings(max_examples=30, deadline=None)
@given(a=st.floats(0.05, 5.0), b=st.floats(0.05, 5.0),
       c=st.floats(0.1, 10.0))
def test_hyperparameter_roundtrip_composite(a, b, c):
    k = (c * ARDRBFKernel(np.full(DIM, a))
         + Matern52Kernel(b)
         + Scalar(1e-3).const * EyeKernel())
    theta = k.get_hyperparameters()
    assert theta.shape == (1 + DIM + 1,)          # C prepended, then betas, l
    theta2 = theta * 1.7 + 0.01
    k.set_hyperparameters(theta2)
    np.testing.assert_allclose(k.get_hyperparameters(), theta2)
    lo, hi = k.hyperparameter_bounds()
    assert lo.shape == theta.shape == hi.shape
    assert (lo <= hi).all()


@settings(max_examples=25, deadline=None)
@given(amp=st.floats(0.1, 10.0), beta=st.floats(0.05, 3.0),
       noise=st.floats(1e-6, 1e-1), seed=st.integers(0, 5))
def test_compiled_canonical_matches_tree(amp, beta, noise, seed):
    """C*Kb + nu*I from the canonicalizer == direct DSL evaluation."""
    k = amp * ARDRBFKernel(np.full(DIM, beta)) \
        + Scalar(noise).const * EyeKernel()
    cs = compile_kernel(k)
    assert cs is not None
    theta = k.get_hyperparameters()
    X = _data(6, seed)
    direct = k.training_kernel(X).numpy()
    C = cs.amp(theta)
    nu = cs.noise(theta)
    s = torch.as_tensor(theta[cs.base_idx])
    Xs = X * s
    from spark_gp_amd.kernels import sqdist
    canon = (C * torch.exp(-sqdist(Xs, Xs))
             + nu * torch.eye(6, dtype=X.dtype)).numpy()
    np.testing.assert_allclose(direct, canon, rtol=1e-12, atol=1e-14)


def test_refit_is_deterministic():
    from spark_gp_amd import GaussianProcessRegression
    rng = np.random.default_rng(3)
    X = rng.uniform(size=(300, 2))
    y = np.sin(3 * X.sum(-1)) + 0.05 * rng.normal(size=300)

    def fit():
        return (GaussianProcessRegression()
                .setKernel(lambda: 1 * ARDRBFKernel(2))
                .setDatasetSizeForExpert(50).setActiveSetSize(60)
                .setSigma2(1e-2).setMaxIter(25).setSeed(11)
                .setDevice("cpu").fit(X, y))

    p1 = fit().predict(X[:40])
    p2 = fit().predict(X[:40])
    np.testing.assert_array_equal(p1, p2)
