"""Probit-link GP classification (BASELINE config 3).

The probit likelihood enters Algorithms 3.1/5.1 only through log p and its
first three f-derivatives, so the tests pin down exactly that chain:
finite differences link each derivative to the one below it (including deep
tails where naive Phi would underflow), the evidence gradient is FD-checked
at the objective level, and an end-to-end fit must classify.
"""

import numpy as np
import pytest
import torch

from spark_gp_amd import GaussianProcessClassifier, ProbitLikelihood
from spark_gp_amd.kernels import ARDRBFKernel, EyeKernel, Scalar
from spark_gp_amd.ops import torch_backend as tb

TD = torch.float64


def test_probit_derivative_chain_fd():
    lik = ProbitLikelihood()
    f = torch.tensor([-8.0, -3.0, -0.7, 0.0, 0.4, 2.5, 8.0], dtype=TD)
    y = torch.tensor([0.0, 1.0, 0.0, 1.0, 1.0, 0.0, 1.0], dtype=TD)
    h = 1e-6
    fp, fm = f + h, f - h
    d1_fd = (lik.log_lik(fp, y) - lik.log_lik(fm, y)) / (2 * h)
    np.testing.assert_allclose(lik.d1(f, y).numpy(), d1_fd.numpy(),
                               rtol=1e-6, atol=1e-8)
    w_fd = -(lik.d1(fp, y) - lik.d1(fm, y)) / (2 * h)
    np.testing.assert_allclose(lik.w(f, y).numpy(), w_fd.numpy(),
                               rtol=1e-5, atol=1e-8)
    d3_fd = (lik.w(fp, y) - lik.w(fm, y)) / (2 * h)
    np.testing.assert_allclose(lik.d3(f, y).numpy(), d3_fd.numpy(),
                               rtol=1e-4, atol=1e-6)


def test_probit_tails_finite_and_bounded():
    lik = ProbitLikelihood()
    f = torch.tensor([-40.0, -20.0, 20.0, 40.0], dtype=TD)
    y = torch.tensor([1.0, 0.0, 1.0, 0.0], dtype=TD)
    for q in (lik.log_lik(f, y), lik.d1(f, y), lik.w(f, y), lik.d3(f, y)):
        assert torch.isfinite(q).all(), q
    w = lik.w(f, y)
    assert bool((w > 0).all()) and bool((w <= 1.0 + 1e-12).all()), w


def test_probit_evidence_gradient_fd():
    g = torch.Generator().manual_seed(3)
    X = torch.randn(2, 12, 2, generator=g, dtype=TD)
    y = (X.sum(-1) > 0).to(TD)
    factory = lambda: 1 * ARDRBFKernel(2) + Scalar(1e-2).const * EyeKernel()
    theta = np.array([1.0, 0.8, 1.2])
    tol = 1e-10

    def ev(t):
        f = torch.zeros_like(y)
        return tb.laplace_nll_grad(factory(), t, X, y, f, tol,
                                   likelihood=ProbitLikelihood())

    nll, grad = ev(theta)
    assert np.isfinite(nll)
    h = 1e-5
    for i in range(len(theta)):
        tp, tm = theta.copy(), theta.copy()
        tp[i] += h
        tm[i] -= h
        fp, _ = ev(tp)
        fm, _ = ev(tm)
        assert grad[i] == pytest.approx((fp - fm) / (2 * h), rel=2e-4,
                                        abs=1e-6)


def _blobs(n=240, seed=0):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(n, 2))
    X[: n // 2] += 2.2
    X[n // 2:] -= 2.2
    y = np.zeros(n)
    y[: n // 2] = 1.0
    return X, y


def test_probit_classifier_end_to_end():
    X, y = _blobs()
    model = (GaussianProcessClassifier()
             .setKernel(lambda: 1 * ARDRBFKernel(2))
             .setLink("probit")
             .setDatasetSizeForExpert(40).setActiveSetSize(50)
             .setSigma2(1e-3).setMaxIter(30).setSeed(0).setDevice("cpu")
             .fit(X, y))
    assert model.link == "probit"
    acc = float((model.predict(X) == y).mean())
    assert acc > 0.97, acc
    proba = model.predict_proba(X)
    assert proba.shape == (len(y), 2)
    np.testing.assert_allclose(proba.sum(-1), 1.0, atol=1e-12)
    # probabilities use the Phi link, not sigmoid
    f = model.predict_raw(X)[:, 1]
    from scipy.special import ndtr
    np.testing.assert_allclose(proba[:, 1], ndtr(f), atol=1e-6)
    # averaged predictive probabilities shrink toward 1/2
    pa = model.predict_proba(X, averaged=True)[:, 1]
    assert float(np.max(np.abs(pa - 0.5))) <= \
        float(np.max(np.abs(proba[:, 1] - 0.5))) + 1e-12


def test_probit_model_io_roundtrip(tmp_path):
    from spark_gp_amd.models.model_io import load_model, save_model
    X, y = _blobs(n=120, seed=2)
    model = (GaussianProcessClassifier()
             .setKernel(lambda: 1 * ARDRBFKernel(2))
             .setLink("probit")
             .setDatasetSizeForExpert(30).setActiveSetSize(40)
             .setSigma2(1e-3).setMaxIter(15).setSeed(0).setDevice("cpu")
             .fit(X, y))
    save_model(model, str(tmp_path / "m"))
    loaded = load_model(str(tmp_path / "m"))
    assert loaded.link == "probit"
    np.testing.assert_allclose(loaded.predict_proba(X[:20]),
                               model.predict_proba(X[:20]), atol=1e-10)
