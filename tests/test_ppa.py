"""PPA assembly tests: flattened accumulation vs a naive per-expert
transliteration of ``ProjectedGaussianProcessHelper.scala``, magic-quantity
math, and the PD failure path."""

import numpy as np
import pytest
import torch

from spark_gp_amd.kernels import ARDRBFKernel, EyeKernel, Scalar
from spark_gp_amd.models.base import group_experts
from spark_gp_amd.parallel.dist import Comm
from spark_gp_amd.ppa import (NotPositiveDefiniteError, accumulate_ppa_stats,
                              magic_vector_matrix)

TD = torch.float64


def setup_data(n=120, d=3, seed=0):
    g = torch.Generator().manual_seed(seed)
    X = torch.randn(n, d, generator=g, dtype=TD)
    y = torch.sin(X.sum(-1))
    return X, y


def naive_stats(kernel, active, X, y, k_expert=30):
    """Per-expert treeAggregate transliteration (:20-36)."""
    m = active.shape[0]
    KK = torch.zeros(m, m, dtype=TD)
    Ky = torch.zeros(m, dtype=TD)
    groups = group_experts(X, y, k_expert)
    for _, Xg, yg in groups:
        for e in range(Xg.shape[0]):
            C = kernel.cross_kernel(active, Xg[e])     # [m, k]
            KK += C @ C.T
            Ky += C @ yg[e]
    return KK, Ky


def test_flattened_accumulation_equals_per_expert():
    X, y = setup_data()
    kernel = 1 * ARDRBFKernel(3) + Scalar(1e-2).const * EyeKernel()
    active = X[:10].clone()
    KK, Ky = accumulate_ppa_stats(kernel, active, X, y, Comm())
    KKn, Kyn = naive_stats(kernel, active, X, y)
    np.testing.assert_allclose(KK.numpy(), KKn.numpy(), rtol=1e-10)
    np.testing.assert_allclose(Ky.numpy(), Kyn.numpy(), rtol=1e-10)


def test_magic_quantities_match_direct_solves():
    X, y = setup_data()
    kernel = 1 * ARDRBFKernel(3) + Scalar(1e-2).const * EyeKernel()
    active = X[:10].clone()
    KK, Ky = accumulate_ppa_stats(kernel, active, X, y, Comm())
    mv, mm = magic_vector_matrix(kernel, KK, Ky, active)
    Kmm = kernel.training_kernel(active)
    nu = kernel.white_noise_var()
    PD = nu * Kmm + KK
    np.testing.assert_allclose(mv.numpy(),
                               torch.linalg.solve(PD, Ky).numpy(), rtol=1e-8)
    ref_mm = torch.linalg.inv(PD) * nu - torch.linalg.inv(Kmm)
    np.testing.assert_allclose(mm.numpy(), ref_mm.numpy(),
                               rtol=1e-6, atol=1e-10)


def test_not_positive_definite_raises():
    X, y = setup_data(n=30)
    # duplicate active-set rows + zero noise -> singular K_mm
    kernel = 1 * ARDRBFKernel(3) + Scalar(0.0).const * EyeKernel()
    active = torch.cat([X[:5], X[:5]])
    KK = torch.zeros(10, 10, dtype=TD)
    Ky = torch.zeros(10, dtype=TD)
    with pytest.raises(NotPositiveDefiniteError):
        magic_vector_matrix(kernel, KK, Ky, active)


def test_ppa_with_full_active_set_interpolates():
    """With the active set = the training set and small noise, the PPA mean
    must closely reproduce smooth training targets."""
    from spark_gp_amd import GaussianProcessRegression, RBFKernel

    g = torch.Generator().manual_seed(1)
    X = torch.rand(200, 1, generator=g, dtype=TD).numpy()
    y = np.sin(4.0 * X[:, 0])
    gp = (GaussianProcessRegression()
          .setKernel(lambda: 1 * RBFKernel(0.5, 1e-3, 10))
          .setDatasetSizeForExpert(50)
          .setActiveSetSize(200)
          .setSigma2(1e-5)
          .setMaxIter(30)
          .setSeed(0)
          .setDevice("cpu"))
    model = gp.fit(X, y)
    pred = model.predict(X)
    assert np.sqrt(np.mean((pred - y) ** 2)) < 5e-3


def test_group_experts_partition():
    X = torch.arange(23, dtype=TD).unsqueeze(-1)
    y = torch.arange(23, dtype=TD)
    groups = group_experts(X, y, 5)
    # E = round(23/5) = 5 experts -> 3 of size 5, wait: 23 = 4*5+3 -> lo=4,
    # r=3: 3 experts of 5 rows + 2 of 4 rows
    sizes = sorted([int(g[1].shape[0]) * int(g[1].shape[1]) for g in groups])
    total = sum(int(g[1].shape[0]) * int(g[1].shape[1]) for g in groups)
    assert total == 23
    all_idx = torch.cat([g[0] for g in groups])
    assert sorted(all_idx.tolist()) == list(range(23))
    # round-robin: expert e = idx % E
    for idx, Xg, yg in groups:
        E = 5
        k = Xg.shape[1]
        mat = idx.reshape(Xg.shape[0], k)
        assert ((mat % E) == mat[:, :1] % E).all()


def test_predictive_variance_shrinks_at_training_points():
    """PPA predictive variance must be ~sigma2-level at training inputs and
    grow far from the data (Rasmussen & Williams eq. 8.27 behavior)."""
    import torch
    from spark_gp_amd import GaussianProcessRegression
    from spark_gp_amd.kernels import ARDRBFKernel

    rng = np.random.default_rng(5)
    X = rng.uniform(size=(600, 2))
    y = np.sin(4 * X.sum(-1)) + 0.05 * rng.normal(size=600)
    model = (GaussianProcessRegression()
             .setKernel(lambda: 1 * ARDRBFKernel(2))
             .setDatasetSizeForExpert(60).setActiveSetSize(200)
             .setSigma2(1e-2).setMaxIter(30).setSeed(0).setDevice("cpu")
             .fit(X, y))
    _, std_in = model.predict(X[:50], return_std=True)
    far = np.full((10, 2), 25.0)          # far outside the unit square
    mean_far, std_far = model.predict(far, return_std=True)
    assert std_in.mean() < 0.35
    assert std_far.min() > 3 * std_in.mean()
    # far from data the PPA mean falls back toward the prior mean 0
    assert np.abs(mean_far).max() < 0.1


def test_group_experts_partition_properties():
    """Every row lands in exactly one expert; sizes differ by at most 1;
    assignment is the reference's round-robin (row i -> expert i % E)
    re-grouped by size (``GaussianProcessCommons.scala:26-31``)."""
    import torch
    from spark_gp_amd.models.base import group_experts
    for n, kt in [(101, 10), (100, 100), (7, 3), (250, 100), (5, 100)]:
        X = torch.arange(n, dtype=torch.float64).unsqueeze(-1).repeat(1, 2)
        y = torch.arange(n, dtype=torch.float64)
        groups = group_experts(X, y, kt)
        E = max(1, int(round(n / kt)))
        all_idx = torch.cat([idx for idx, _, _ in groups])
        assert sorted(all_idx.tolist()) == list(range(n))
        sizes = [Xg.shape[1] for _, Xg, _ in groups for _ in range(Xg.shape[0])]
        assert max(sizes) - min(sizes) <= 1
        assert sum(sizes) == n
        # row -> expert is i % E: rows of one expert are congruent mod E
        for idx, Xg, yg in groups:
            flat = idx.reshape(Xg.shape[0], Xg.shape[1])
            for row_ids in flat:
                assert len(set(int(i) % E for i in row_ids)) == 1
        # X/y rows carried intact
        for idx, Xg, yg in groups:
            assert torch.equal(Xg[..., 0].reshape(-1).to(torch.int64), idx)
            assert torch.equal(yg.reshape(-1).to(torch.int64), idx)
