"""GP classification tests: batched Laplace vs a naive per-expert
transliteration, FD gradient of the evidence, and end-to-end accuracy."""

import math

import numpy as np
import pytest
import torch

from spark_gp_amd import (GaussianProcessClassifier, OneVsRest, RBFKernel,
                          Scalar, accuracy)
from spark_gp_amd.kernels import ARDRBFKernel, EyeKernel
from spark_gp_amd.ops import torch_backend as tb

TD = torch.float64


def naive_laplace(kernel, theta, X1, y1, f1, tol):
    """One-expert transliteration of
    ``classification/GaussianProcessClassifier.scala:74-129`` (fp64)."""
    kernel.set_hyperparameters(theta)
    K, dK = kernel.training_kernel_and_derivative(X1)
    K = K.numpy()
    dK = dK.numpy()
    y = y1.numpy().copy()
    f = f1.numpy().copy()
    n = len(y)
    old_obj, new_obj = -math.inf, -np.finfo(np.float64).max
    step = 1.0
    sig = lambda v: 1.0 / (1.0 + np.exp(-v))
    while abs(old_obj - new_obj) > tol and step > tol:
        pi = sig(f)
        w = pi * (1 - pi)
        sqw = np.sqrt(w)
        B = np.eye(n) + (sqw[:, None] * K) * sqw[None, :]
        L = np.linalg.cholesky(B)
        grad_logp = y - pi
        b = w * f + grad_logp
        t = np.linalg.solve(L, sqw * (K @ b))
        a = b - sqw * np.linalg.solve(L.T, t)
        f_cand = (1 - step) * f + step * (K @ a)
        with np.errstate(over="ignore"):
            obj_cand = (-0.5 * a @ f_cand
                        + np.sum(np.log(sig((2 * y - 1) * f_cand))))
        if obj_cand > old_obj:
            f = f_cand
            old_obj, new_obj = new_obj, obj_cand
        else:
            step /= 2
    logZ = new_obj - np.sum(np.log(np.diag(L)))
    R = sqw[:, None] * np.linalg.solve(L.T, np.linalg.solve(L, np.diag(sqw)))
    C = np.linalg.solve(L, sqw[:, None] * K)
    d3 = -(2 * pi - 1) * pi * pi * np.exp(-f)
    s2 = -0.5 * (np.diag(K) - np.diag(C.T @ C)) * d3
    grads = []
    for i in range(dK.shape[0]):
        Di = dK[i]
        s1 = 0.5 * a @ Di @ a - 0.5 * np.sum(R * Di)
        bb = Di @ grad_logp
        s3 = bb - K @ R @ bb
        grads.append(s1 + s2 @ s3)
    return -logZ, -np.array(grads), f


def make_cls_batch(E=3, k=18, d=2, seed=0):
    g = torch.Generator().manual_seed(seed)
    X = torch.randn(E, k, d, generator=g, dtype=TD)
    y = (X.sum(-1) > 0).to(TD)
    return X, y


def test_batched_laplace_matches_naive():
    X, y = make_cls_batch()
    factory = lambda: 1 * RBFKernel(0.8) + Scalar(1e-3).const * EyeKernel()
    theta = np.array([1.1, 0.9])
    tol = 1e-6
    f = torch.zeros_like(y)
    nll, grad = tb.laplace_nll_grad(factory(), theta, X, y, f, tol)
    nll_ref, grad_ref = 0.0, np.zeros_like(theta)
    for e in range(X.shape[0]):
        nz, gz, f_ref = naive_laplace(factory(), theta, X[e], y[e],
                                      torch.zeros(X.shape[1], dtype=TD), tol)
        nll_ref += nz
        grad_ref += gz
        np.testing.assert_allclose(f[e].numpy(), f_ref, atol=1e-8)
    assert nll == pytest.approx(nll_ref, rel=1e-8)
    np.testing.assert_allclose(grad, grad_ref, rtol=1e-6, atol=1e-9)


def test_laplace_evidence_gradient_fd():
    X, y = make_cls_batch(E=2, k=12, seed=5)
    factory = lambda: 1 * ARDRBFKernel(2) + Scalar(1e-2).const * EyeKernel()
    theta = np.array([1.0, 0.8, 1.2])
    tol = 1e-10      # tight Newton tolerance so FD of the evidence is clean

    def ev(t):
        f = torch.zeros_like(y)     # cold start each eval for determinism
        return tb.laplace_nll_grad(factory(), t, X, y, f, tol)

    nll, grad = ev(theta)
    h = 1e-5
    for i in range(len(theta)):
        tp, tm = theta.copy(), theta.copy()
        tp[i] += h
        tm[i] -= h
        fp, _ = ev(tp)
        fm, _ = ev(tm)
        assert grad[i] == pytest.approx((fp - fm) / (2 * h), rel=2e-4,
                                        abs=1e-6)


def test_classifier_rejects_bad_labels():
    X = np.random.default_rng(0).random((40, 2))
    y = np.full(40, 2.0)
    with pytest.raises(ValueError):
        GaussianProcessClassifier().setDevice("cpu").fit(X, y)


def _toy_classifier():
    return (GaussianProcessClassifier()
            .setKernel(lambda: 1 * RBFKernel(1.0, 1e-3, 10))
            .setDatasetSizeForExpert(40)
            .setActiveSetSize(30)
            .setSigma2(1e-3)
            .setMaxIter(20)
            .setSeed(7)
            .setDevice("cpu"))


def test_classifier_separable_blobs():
    rng = np.random.default_rng(0)
    n = 160
    X = np.concatenate([rng.normal(-2, 0.7, (n // 2, 2)),
                        rng.normal(2, 0.7, (n // 2, 2))])
    y = np.concatenate([np.zeros(n // 2), np.ones(n // 2)])
    model = _toy_classifier().fit(X, y)
    acc = accuracy(y, model.predict(X))
    assert acc > 0.97
    proba = model.predict_proba(X)
    assert proba.shape == (n, 2)
    np.testing.assert_allclose(proba.sum(-1), 1.0, atol=1e-9)
    raw = model.predict_raw(X)
    np.testing.assert_allclose(raw[:, 0], -raw[:, 1], atol=1e-12)
    # averaged predictive probabilities (Gauss-Hermite) stay calibrated
    proba_avg = model.predict_proba(X, averaged=True)
    np.testing.assert_allclose(proba_avg.sum(-1), 1.0, atol=1e-6)
    assert accuracy(y, (proba_avg[:, 1] > 0.5).astype(float)) > 0.97


def test_one_vs_rest_iris():
    from sklearn.datasets import load_iris
    data = load_iris()
    X, y = data.data, data.target.astype(np.float64)
    ovr = OneVsRest(lambda: (GaussianProcessClassifier()
                             .setKernel(lambda: 1 * RBFKernel(1.0, 1e-3, 10))
                             .setDatasetSizeForExpert(20)
                             .setActiveSetSize(30)
                             .setSigma2(1e-3)
                             .setMaxIter(20)
                             .setSeed(7)
                             .setDevice("cpu")))
    model = ovr.fit(X, y)
    acc = accuracy(y, model.predict(X))
    assert acc > 0.9


def test_laplace_evidence_compiled_matches_generic():
    """K11 contraction-form evidence (no [E,p,k,k] tensor) vs the generic
    materialized-derivative path, at a converged latent f."""
    from spark_gp_amd.kernels import compile_kernel, WhiteNoiseKernel
    g = torch.Generator().manual_seed(0)
    for factory, theta in [
        (lambda: 1 * ARDRBFKernel(3) + Scalar(1e-2).const * EyeKernel(),
         np.array([1.1, 0.8, 1.2, 0.6])),
        (lambda: 1 * RBFKernel(0.7) + WhiteNoiseKernel(0.1, 0, 1)
         + Scalar(1e-3).const * EyeKernel(),
         np.array([0.9, 0.75, 0.12])),
    ]:
        X = torch.randn(4, 18, 3, generator=g, dtype=TD)
        y = (X.sum(-1) > 0).double()
        kern = factory()
        cs = compile_kernel(kern)
        f = torch.zeros(4, 18, dtype=TD)
        nll_ref, grad_ref = tb.laplace_nll_grad(kern, theta, X, y, f, 1e-10)
        nll_c, grad_c = tb.laplace_evidence_compiled(cs, theta, X, y, f)
        assert nll_c == pytest.approx(nll_ref, rel=1e-6)
        np.testing.assert_allclose(grad_c, grad_ref, rtol=1e-5,
                                   atol=1e-6 * np.abs(grad_ref).max())


def test_averaged_proba_shrinks_toward_half():
    """Gauss-Hermite averaging over the latent posterior must pull
    probabilities toward 0.5 relative to the plain sigmoid-of-mean
    (Jensen: E[sigmoid(Z)] is closer to 1/2 than sigmoid(E[Z]) for the
    logistic link), and agree with it where the posterior is confident."""
    rng = np.random.default_rng(4)
    X = np.concatenate([rng.normal(-1.0, 0.6, (200, 2)),
                        rng.normal(1.0, 0.6, (200, 2))])
    y = np.concatenate([np.zeros(200), np.ones(200)])
    model = (GaussianProcessClassifier()
             .setKernel(lambda: 1 * RBFKernel(1.0, 1e-3, 10))
             .setDatasetSizeForExpert(50).setActiveSetSize(60)
             .setSigma2(1e-3).setMaxIter(30).setSeed(0).setDevice("cpu")
             .fit(X, y))
    # points far outside the data: high posterior variance
    Xq = np.concatenate([X[:50], rng.normal(0.0, 6.0, (50, 2))])
    p_plain = model.predict_proba(Xq)[:, 1]
    p_avg = model.predict_proba(Xq, averaged=True)[:, 1]
    # same labels either way
    assert ((p_plain > 0.5) == (p_avg > 0.5)).all()
    # averaging never moves AWAY from 1/2 (up to fp noise)
    assert (np.abs(p_avg - 0.5) <= np.abs(p_plain - 0.5) + 1e-12).all()
    # and strictly shrinks somewhere on the uncertain points
    assert (np.abs(p_avg[50:] - 0.5) < np.abs(p_plain[50:] - 0.5) - 1e-6).any()
