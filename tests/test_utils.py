"""Utility tests: scaling, integrator, model IO, active-set providers,
memoized objective."""

import math

import numpy as np
import pytest
import torch

from spark_gp_amd import (GaussianProcessRegression,
                          GreedilyOptimizingActiveSetProvider, Integrator,
                          KMeansActiveSetProvider, RandomActiveSetProvider,
                          RBFKernel, Scalar, StandardScaler, load_model,
                          save_model)
from spark_gp_amd.kernels import ARDRBFKernel, EyeKernel, WhiteNoiseKernel
from spark_gp_amd.models.model_io import kernel_from_spec, kernel_to_spec
from spark_gp_amd.optimize import MemoizedObjective
from spark_gp_amd.parallel.dist import Comm

TD = torch.float64


def test_standard_scaler_zscore_and_zero_variance():
    X = np.array([[1.0, 5.0], [3.0, 5.0], [5.0, 5.0]])
    s = StandardScaler().fit(X)
    Xs = s.transform(X)
    np.testing.assert_allclose(Xs.mean(0), 0.0, atol=1e-12)
    # population variance; zero-variance dim -> scale 1 (Scaling.scala:10-25)
    np.testing.assert_allclose(Xs[:, 0].std(), 1.0, atol=1e-12)
    np.testing.assert_allclose(Xs[:, 1], 0.0, atol=1e-12)
    assert s.scale[1] == 1.0


def test_integrator_expected_sigmoid_vs_monte_carlo():
    integ = Integrator(64)
    rng = np.random.default_rng(0)
    for mean, var in [(0.3, 0.5), (-1.2, 2.0), (2.0, 0.1)]:
        got = integ.expected_of_function_of_normal(
            mean, var, lambda z: 1.0 / (1.0 + np.exp(-z)))
        z = rng.normal(mean, math.sqrt(var), 200000)
        mc = float((1.0 / (1.0 + np.exp(-z))).mean())
        se = float((1.0 / (1.0 + np.exp(-z))).std() / math.sqrt(len(z)))
        assert abs(got - mc) < 3 * se + 1e-4


def test_kernel_spec_roundtrip():
    k = (1 * ARDRBFKernel(np.array([0.5, 2.0]))
         + WhiteNoiseKernel(0.3, 0.1, 1.0)
         + Scalar(1e-3).const * EyeKernel())
    k2 = kernel_from_spec(kernel_to_spec(k))
    np.testing.assert_allclose(k2.get_hyperparameters(),
                               k.get_hyperparameters())
    lo1, up1 = k.hyperparameter_bounds()
    lo2, up2 = k2.hyperparameter_bounds()
    np.testing.assert_allclose(lo1, lo2)
    np.testing.assert_allclose(up1, up2)
    X = torch.randn(6, 2, dtype=TD)
    np.testing.assert_allclose(k.training_kernel(X).numpy(),
                               k2.training_kernel(X).numpy(), atol=1e-14)


def test_model_save_load_roundtrip(tmp_path):
    rng = np.random.default_rng(0)
    X = rng.random((200, 2))
    y = np.sin(3 * X[:, 0]) + X[:, 1]
    gp = (GaussianProcessRegression()
          .setKernel(lambda: 1 * ARDRBFKernel(2))
          .setDatasetSizeForExpert(50)
          .setActiveSetSize(40)
          .setSigma2(1e-3)
          .setMaxIter(20)
          .setSeed(3)
          .setDevice("cpu"))
    model = gp.fit(X, y)
    p1 = model.predict(X[:20])
    save_model(model, str(tmp_path / "m"))
    loaded = load_model(str(tmp_path / "m"))
    p2 = loaded.predict(X[:20])
    np.testing.assert_allclose(p1, p2, rtol=1e-12)
    _, s1 = model.predict(X[:20], return_std=True)
    _, s2 = loaded.predict(X[:20], return_std=True)
    np.testing.assert_allclose(s1, s2, rtol=1e-10)


def _provider_setup(n=300, d=2, seed=0):
    rng = np.random.default_rng(seed)
    X = torch.tensor(rng.random((n, d)), dtype=TD)
    y = torch.sin(3 * X[:, 0]) + X[:, 1]
    kernel = (1 * RBFKernel(0.5) + Scalar(1e-2).const * EyeKernel())
    theta = kernel.get_hyperparameters()
    return X, y, kernel, theta


@pytest.mark.parametrize("provider,m", [
    (RandomActiveSetProvider(), 25),
    (KMeansActiveSetProvider(max_iter=5), 25),
    (GreedilyOptimizingActiveSetProvider(), 12),
])
def test_active_set_providers_shapes(provider, m):
    X, y, kernel, theta = _provider_setup()
    active = provider(m, X, y, kernel, theta, seed=3, comm=Comm())
    assert active.shape == (m, X.shape[1])
    assert torch.isfinite(active).all()


def test_random_provider_rows_come_from_data():
    X, y, kernel, theta = _provider_setup()
    active = RandomActiveSetProvider()(10, X, y, kernel, theta, 3, Comm())
    # every sampled row must be an actual data row
    d = (active.unsqueeze(1) - X.unsqueeze(0)).abs().sum(-1).min(-1).values
    assert float(d.max()) < 1e-12


def test_greedy_provider_picks_informative_points():
    # two separated clusters: greedy selection must cover both
    rng = np.random.default_rng(1)
    X = torch.tensor(np.concatenate([rng.normal(-3, 0.2, (100, 1)),
                                     rng.normal(3, 0.2, (100, 1))]), dtype=TD)
    y = torch.cat([torch.full((100,), -1.0, dtype=TD),
                   torch.full((100,), 1.0, dtype=TD)])
    kernel = (1 * RBFKernel(1.0) + Scalar(1e-2).const * EyeKernel())
    active = GreedilyOptimizingActiveSetProvider()(
        8, X, y, kernel, kernel.get_hyperparameters(), 0, Comm())
    assert (active < 0).any() and (active > 0).any()


def test_memoized_objective_caches():
    calls = []

    def fn(x):
        calls.append(x.copy())
        return float((x ** 2).sum()), 2 * x

    memo = MemoizedObjective(fn)
    x = np.array([1.0, 2.0])
    a = memo(x)
    b = memo(x.copy())
    assert len(calls) == 1 and a[0] == b[0]
    memo(np.array([3.0, 4.0]))
    assert len(calls) == 2


def test_classifier_model_save_load_roundtrip(tmp_path):
    from spark_gp_amd import GaussianProcessClassifier
    rng = np.random.default_rng(0)
    n = 120
    X = np.concatenate([rng.normal(-2, 0.7, (n // 2, 2)),
                        rng.normal(2, 0.7, (n // 2, 2))])
    y = np.concatenate([np.zeros(n // 2), np.ones(n // 2)])
    model = (GaussianProcessClassifier()
             .setKernel(lambda: 1 * RBFKernel(1.0, 1e-3, 10))
             .setDatasetSizeForExpert(40)
             .setActiveSetSize(30)
             .setSigma2(1e-3)
             .setMaxIter(15)
             .setSeed(7)
             .setDevice("cpu")).fit(X, y)
    p1 = model.predict_proba(X[:20])
    save_model(model, str(tmp_path / "clf"))
    loaded = load_model(str(tmp_path / "clf"))
    assert type(loaded).__name__ == "GaussianProcessClassificationModel"
    p2 = loaded.predict_proba(X[:20])
    np.testing.assert_allclose(p1, p2, rtol=1e-10)


def test_integrator_averaged_proba_single_matches_batch():
    integ = Integrator(32)
    f = np.array([0.5, -1.0])
    v = np.array([0.3, 1.5])
    batch = integ.expected_of_function_of_normal_batch(
        f, v, lambda z: 1.0 / (1.0 + np.exp(-z)))
    for i in range(2):
        single = integ.expected_of_function_of_normal(
            f[i], v[i], lambda z: 1.0 / (1.0 + np.exp(-z)))
        assert single == pytest.approx(batch[i], rel=1e-12)


def test_kmeans_empty_clusters_keep_previous_centers():
    """With heavy duplicates, Lloyd iterations empty some clusters; their
    centers must survive (no NaN division) and all centers stay finite."""
    import torch
    from spark_gp_amd import KMeansActiveSetProvider
    from spark_gp_amd.parallel.dist import Comm
    X = torch.tensor([[0.0, 0.0]] * 50 + [[10.0, 10.0]] * 50,
                     dtype=torch.float64)
    y = torch.zeros(100, dtype=torch.float64)
    centers = KMeansActiveSetProvider(max_iter=8)(
        8, X, y, None, None, 3, Comm())
    assert centers.shape == (8, 2)
    assert torch.isfinite(centers).all()


def test_serve_missing_model_dir_fails_cleanly():
    import pytest as _pytest
    _pytest.importorskip("fastapi")
    from spark_gp_amd.serve import create_app
    with _pytest.raises(FileNotFoundError):
        create_app("/nonexistent/model/dir")


def test_scaler_inverse_transform_roundtrip():
    from spark_gp_amd import StandardScaler
    rng = np.random.default_rng(8)
    X = rng.normal(3.0, [1.0, 5.0, 0.0], size=(100, 3))   # one constant dim
    s = StandardScaler().fit(X)
    np.testing.assert_allclose(s.inverse_transform(s.transform(X)), X,
                               rtol=1e-12, atol=1e-12)


def test_synthetic_generators_deterministic():
    from spark_gp_amd.data import (benchmark_regression_data,
                                   performance_benchmark_data,
                                   shard_performance_benchmark_data)
    X1, y1 = performance_benchmark_data(100, 3, seed=7)
    X2, y2 = performance_benchmark_data(100, 3, seed=7)
    np.testing.assert_array_equal(X1, X2)
    np.testing.assert_array_equal(y1, y2)
    # shards partition the total row count
    tot = sum(shard_performance_benchmark_data(103, 4, r, 4)[0].shape[0]
              for r in range(4))
    assert tot == 103
    Xb, yb = benchmark_regression_data(50, 4, seed=1)
    assert Xb.shape == (50, 4) and yb.shape == (50,) and np.isfinite(yb).all()
