"""Kernel DSL unit tests.

Mirrors the reference's test strategy (``src/test/scala/.../kernel/``):
brute-force oracles, finite-difference gradient checks, stateful-API errors,
plus hyperparameter-layout tests (layout = model-format compatibility).
"""

import math

import numpy as np
import pytest
import torch

from spark_gp_amd.kernels import (ARDRBFKernel, EyeKernel, RBFKernel, Scalar,
                                  SumOfKernels, TrainableScalarTimesKernel,
                                  TrainingVectorsNotInitializedError,
                                  WhiteNoiseKernel, compile_kernel, sqdist)

TD = torch.float64


def brute_rbf(X, Z, sigma):
    X, Z = np.asarray(X), np.asarray(Z)
    out = np.zeros((len(X), len(Z)))
    for i in range(len(X)):
        for j in range(len(Z)):
            d2 = ((X[i] - Z[j]) ** 2).sum()
            out[i, j] = math.exp(-d2 / (2 * sigma ** 2))
    return out


def brute_ard(X, Z, beta):
    X, Z = np.asarray(X), np.asarray(Z)
    out = np.zeros((len(X), len(Z)))
    for i in range(len(X)):
        for j in range(len(Z)):
            w = ((X[i] - Z[j]) * beta)
            out[i, j] = math.exp(-(w * w).sum())
    return out


@pytest.fixture
def X3():
    return torch.tensor([[0.1, 0.2], [1.0, -0.5], [0.3, 0.9]], dtype=TD)


def test_rbf_training_kernel_matches_bruteforce(X3):
    k = RBFKernel(0.7)
    K = k.training_kernel(X3).numpy()
    np.testing.assert_allclose(K, brute_rbf(X3, X3, 0.7), atol=1e-12)
    assert np.allclose(np.diag(K), 1.0)


def test_ard_training_kernel_matches_bruteforce(X3):
    beta = np.array([0.5, 2.0])
    k = ARDRBFKernel(beta)
    K = k.training_kernel(X3).numpy()
    np.testing.assert_allclose(K, brute_ard(X3, X3, beta), atol=1e-12)


def test_cross_kernel_and_single_vector(X3):
    k = RBFKernel(0.9).set_training_vectors(X3)
    Xt = torch.tensor([[0.0, 0.0], [0.5, 0.5]], dtype=TD)
    C = k.cross_kernel(Xt, X3).numpy()
    np.testing.assert_allclose(C, brute_rbf(Xt, X3, 0.9), atol=1e-12)
    row = k.cross_kernel_(torch.tensor([0.0, 0.0], dtype=TD)).numpy()
    np.testing.assert_allclose(row[0], C[0], atol=1e-12)


def test_stateful_api_raises_before_set():
    k = RBFKernel(1.0)
    with pytest.raises(TrainingVectorsNotInitializedError):
        k.training_kernel_()
    with pytest.raises(TrainingVectorsNotInitializedError):
        k.training_kernel_and_derivative_()


def fd_gradient_check(kernel, X, h=1e-6, tol=1e-5):
    """Central finite differences on every hyperparameter, like
    ``RBFKernelTest.scala:41-60`` but tighter (fp64)."""
    theta0 = kernel.get_hyperparameters()
    K, dK = kernel.training_kernel_and_derivative(X)
    for i in range(len(theta0)):
        tp = theta0.copy(); tp[i] += h
        tm = theta0.copy(); tm[i] -= h
        Kp = kernel.set_hyperparameters(tp).training_kernel(X)
        Km = kernel.set_hyperparameters(tm).training_kernel(X)
        fd = (Kp - Km) / (2 * h)
        np.testing.assert_allclose(dK[i].numpy(), fd.numpy(), atol=tol,
                                   err_msg=f"hyper {i}")
    kernel.set_hyperparameters(theta0)


def test_rbf_derivative_fd(X3):
    fd_gradient_check(RBFKernel(0.7), X3)


def test_ard_derivative_fd(X3):
    fd_gradient_check(ARDRBFKernel(np.array([0.5, 2.0])), X3)


def test_composite_derivative_fd(X3):
    k = (Scalar(1.5).between(0, 30) * ARDRBFKernel(np.array([0.5, 2.0]))
         + WhiteNoiseKernel(0.3, 0, 1) + Scalar(1e-2).const * EyeKernel())
    fd_gradient_check(k, X3)


def test_hyperparameter_layout_prepend_and_concat():
    # TrainableScalar prepends C; Sum concatenates (reference layout).
    k = 1 * ARDRBFKernel(np.array([2.0, 3.0])) + WhiteNoiseKernel(0.5, 0.1, 1)
    theta = k.get_hyperparameters()
    np.testing.assert_allclose(theta, [1.0, 2.0, 3.0, 0.5])
    lo, up = k.hyperparameter_bounds()
    np.testing.assert_allclose(lo, [0.0, 0.0, 0.0, 0.1])
    np.testing.assert_allclose(up[3], 1.0)
    k.set_hyperparameters([4.0, 5.0, 6.0, 0.7])
    np.testing.assert_allclose(k.get_hyperparameters(), [4.0, 5.0, 6.0, 0.7])


def test_white_noise_and_self_kernel(X3):
    k = 1 * RBFKernel(0.5) + Scalar(1e-3).const * EyeKernel()
    assert k.white_noise_var() == pytest.approx(1e-3)
    K = k.training_kernel(X3)
    assert K[0, 0].item() == pytest.approx(1.0 + 1e-3)
    s = k.self_kernel(X3)
    np.testing.assert_allclose(s.numpy(), [1.001] * 3)
    # Eye cross kernel is zero -> composite cross has no noise term
    C = k.cross_kernel(X3, X3)
    assert C[0, 0].item() == pytest.approx(1.0)


def test_batched_equals_loop(X3):
    k = ARDRBFKernel(np.array([0.5, 2.0]))
    Xb = torch.stack([X3, X3 + 1.0])          # [2, 3, 2]
    Kb = k.training_kernel(Xb)
    for e in range(2):
        np.testing.assert_allclose(Kb[e].numpy(),
                                   k.training_kernel(Xb[e]).numpy(),
                                   atol=1e-12)


def test_compile_kernel_flagship_patterns():
    cs = compile_kernel(1 * ARDRBFKernel(3) + Scalar(1e-4).const * EyeKernel())
    assert cs is not None and cs.base == "ard"
    assert cs.amp_idx == 0 and cs.base_idx == slice(1, 4)
    assert cs.noise_const == pytest.approx(1e-4) and cs.noise_idx == []

    cs2 = compile_kernel(1 * RBFKernel(0.1, 1e-6, 10) + WhiteNoiseKernel(0.5, 0, 1)
                         + Scalar(1e-3).const * EyeKernel())
    assert cs2 is not None and cs2.base == "rbf"
    assert cs2.amp_idx == 0 and cs2.base_idx == slice(1, 2)
    assert cs2.noise_idx == [2] and cs2.noise_const == pytest.approx(1e-3)

    # unsupported: two stationary bases -> generic fallback
    assert compile_kernel(1 * RBFKernel(1.0) + 1 * RBFKernel(2.0)) is None


def test_sqdist_nonnegative_and_symmetric():
    X = torch.randn(50, 8, dtype=TD)
    sq = sqdist(X, X)
    assert (sq >= 0).all()
    np.testing.assert_allclose(sq.numpy(), sq.T.numpy(), atol=1e-12)
    assert np.allclose(np.diag(sq.numpy()), 0.0, atol=1e-12)


def test_rbf_reference_fixture_parity():
    """The reference's own unit-test fixture (``RBFKernelTest.scala:27-39``):
    dataset [(1,2), (2,3), (5,7)], sigma = sqrt(0.2), expected K to 1e-4 —
    numeric parity with the reference's asserted values."""
    X = torch.tensor([[1.0, 2.0], [2.0, 3.0], [5.0, 7.0]], dtype=torch.float64)
    k = RBFKernel(math.sqrt(0.2))
    K = k.training_kernel(X).numpy()
    expected = np.array([[1.000000e+00, 6.737947e-03, 3.053624e-45],
                         [6.737947e-03, 1.000000e+00, 7.187782e-28],
                         [3.053624e-45, 7.187782e-28, 1.000000e+00]])
    np.testing.assert_allclose(K, expected, atol=1e-4)
    # and much tighter on the representable entries
    np.testing.assert_allclose(K[0, 1], expected[0, 1], rtol=1e-6)
