"""Poisson (count) GP regression — additive model family (no reference
analog): likelihood derivative checks, objective-level finite-difference
gradient of the Laplace evidence (mirrors the GPC test strategy), rate
recovery end-to-end, target validation, and persistence."""

import numpy as np
import pytest
import torch

from spark_gp_amd import (GaussianProcessPoissonRegression, PoissonLikelihood,
                          load_model, save_model)
from spark_gp_amd.kernels import ARDRBFKernel, EyeKernel, RBFKernel, Scalar
from spark_gp_amd.ops import torch_backend as tb


def test_poisson_likelihood_derivatives_fd():
    lik = PoissonLikelihood()
    g = torch.Generator().manual_seed(0)
    f = torch.randn(4, 9, generator=g, dtype=torch.float64)
    y = torch.poisson(torch.exp(f), generator=g)
    h = 1e-6
    for got, fd in [
        (lik.d1(f, y), (lik.log_lik(f + h, y) - lik.log_lik(f - h, y)) / (2 * h)),
        (-lik.w(f, y), (lik.d1(f + h, y) - lik.d1(f - h, y)) / (2 * h)),
        # d3 is dW/df by the pipeline's convention (see likelihoods.py)
        (lik.d3(f, y), (lik.w(f + h, y) - lik.w(f - h, y)) / (2 * h)),
    ]:
        np.testing.assert_allclose(got.numpy(), fd.numpy(), rtol=1e-4,
                                   atol=1e-6)


def test_poisson_evidence_gradient_fd():
    """d(-logZ)/dtheta from Algorithm 5.1 with the Poisson likelihood vs
    central finite differences of the evidence itself."""
    g = torch.Generator().manual_seed(1)
    E, k, d = 3, 14, 2
    X = torch.rand(E, k, d, generator=g, dtype=torch.float64)
    rate = torch.exp(1.0 + torch.sin(4 * X.sum(-1)))
    y = torch.poisson(rate, generator=g).double()
    kernel = 1 * ARDRBFKernel(d) + Scalar(1e-2).const * EyeKernel()
    theta0 = np.array([1.2, 1.4, 0.9])
    lik = PoissonLikelihood()
    tol = 1e-10

    def ev(theta):
        f = torch.log1p(y.clone())
        nll, grad = tb.laplace_nll_grad(kernel, theta, X, y, f, tol,
                                        likelihood=lik)
        return nll, grad

    nll0, grad0 = ev(theta0)
    h = 1e-5
    for i in range(len(theta0)):
        tp, tm = theta0.copy(), theta0.copy()
        tp[i] += h
        tm[i] -= h
        fd = (ev(tp)[0] - ev(tm)[0]) / (2 * h)
        assert grad0[i] == pytest.approx(fd, rel=2e-3, abs=1e-5), i


def test_poisson_fit_recovers_rates():
    rng = np.random.default_rng(5)
    n = 3000
    X = rng.uniform(size=(n, 2))
    true_log_rate = 1.5 + np.sin(3 * X.sum(-1))
    y = rng.poisson(np.exp(true_log_rate)).astype(np.float64)
    model = (GaussianProcessPoissonRegression()
             .setKernel(lambda: 1 * ARDRBFKernel(2))
             .setDatasetSizeForExpert(60).setActiveSetSize(100)
             .setSigma2(1e-2).setMaxIter(25).setSeed(0).setDevice("cpu")
             .fit(X, y))
    mu, var = model.predict_latent(X[:500])
    # latent log-rate recovered to ~0.15 RMSE over a range of ~2
    err = float(np.sqrt(np.mean((mu - true_log_rate[:500]) ** 2)))
    assert err < 0.2, err
    rate = model.predict(X[:500])
    assert (rate > 0).all()
    rel = np.abs(rate - np.exp(true_log_rate[:500])) / np.exp(true_log_rate[:500])
    assert np.median(rel) < 0.2


def test_poisson_rejects_bad_targets():
    X = np.random.default_rng(0).uniform(size=(50, 2))
    gp = (GaussianProcessPoissonRegression()
          .setKernel(lambda: 1 * RBFKernel(1.0))
          .setDatasetSizeForExpert(25).setActiveSetSize(10).setDevice("cpu"))
    with pytest.raises(ValueError, match="counts"):
        gp.fit(X, np.full(50, -1.0))
    with pytest.raises(ValueError, match="counts"):
        gp.fit(X, np.full(50, 0.5))


def test_poisson_model_persistence(tmp_path):
    rng = np.random.default_rng(1)
    X = rng.uniform(size=(400, 2))
    y = rng.poisson(np.exp(1.0 + np.sin(3 * X.sum(-1)))).astype(np.float64)
    model = (GaussianProcessPoissonRegression()
             .setKernel(lambda: 1 * ARDRBFKernel(2))
             .setDatasetSizeForExpert(50).setActiveSetSize(60)
             .setSigma2(1e-2).setMaxIter(15).setSeed(0).setDevice("cpu")
             .fit(X, y))
    save_model(model, str(tmp_path))
    loaded = load_model(str(tmp_path))
    assert type(loaded).__name__ == "GaussianProcessPoissonModel"
    np.testing.assert_allclose(loaded.predict(X[:20]), model.predict(X[:20]),
                               rtol=1e-12)


def test_poisson_predictive_std_covers_counts():
    """Var[y*] = E[lambda] + Var[lambda]: the predictive std must be at
    least sqrt(rate) (Poisson floor) and the empirical counts should fall
    within ~3 predictive stds of the predicted rate."""
    rng = np.random.default_rng(9)
    X = rng.uniform(size=(2000, 2))
    true_rate = np.exp(1.0 + np.sin(3 * X.sum(-1)))
    y = rng.poisson(true_rate).astype(np.float64)
    model = (GaussianProcessPoissonRegression()
             .setKernel(lambda: 1 * ARDRBFKernel(2))
             .setDatasetSizeForExpert(60).setActiveSetSize(100)
             .setSigma2(1e-2).setMaxIter(20).setSeed(0).setDevice("cpu")
             .fit(X, y))
    rate, std = model.predict(X[:500], return_std=True)
    assert (std >= np.sqrt(rate) - 1e-9).all()
    cover = np.mean(np.abs(y[:500] - rate) <= 3.0 * std)
    assert cover > 0.97, cover
