"""scikit-learn estimator-contract interop: get_params/set_params, clone,
and cross_val_score over the estimator (the primary API stays Spark-style
``fit -> Model``; sklearn tools use the estimator's ``model_`` delegate)."""

import numpy as np
import pytest

sklearn = pytest.importorskip("sklearn")

from sklearn.base import clone
from sklearn.model_selection import cross_val_score

from spark_gp_amd import GaussianProcessRegression
from spark_gp_amd.kernels import ARDRBFKernel
from spark_gp_amd.models.classification import GaussianProcessClassifier


def _est():
    return (GaussianProcessRegression()
            .setKernel(lambda: 1 * ARDRBFKernel(2))
            .setDatasetSizeForExpert(60).setActiveSetSize(60)
            .setSigma2(1e-2).setMaxIter(20).setSeed(0).setDevice("cpu"))


def test_get_set_params_roundtrip():
    est = _est()
    p = est.get_params()
    assert p["sigma2"] == 1e-2 and p["max_iter"] == 20
    est.set_params(sigma2=5e-3, seed=4)
    assert est._sigma2 == 5e-3 and est._seed == 4
    with pytest.raises(ValueError, match="unknown parameter"):
        est.set_params(bogus=1)


def test_clone_preserves_params_and_is_unfitted():
    est = _est()
    rng = np.random.default_rng(0)
    X = rng.uniform(size=(200, 2))
    y = np.sin(3 * X.sum(-1))
    est.fit(X, y)
    c = clone(est)
    pe, pc = est.get_params(), c.get_params()
    # the provider object is legitimately re-instantiated by clone
    assert type(pc.pop("active_set_provider")) \
        is type(pe.pop("active_set_provider"))
    assert pc == pe
    assert c.model_ is None                  # clone is unfitted
    with pytest.raises(RuntimeError, match="not fitted"):
        c.predict(X[:2])
    # the original delegates predict to its fitted model
    np.testing.assert_allclose(est.predict(X[:5]), est.model_.predict(X[:5]))


def test_cross_val_score_regression():
    rng = np.random.default_rng(1)
    X = rng.uniform(size=(400, 2))
    y = np.sin(3 * X.sum(-1)) + 0.05 * rng.normal(size=400)
    scores = cross_val_score(_est(), X, y, cv=3,
                             scoring="neg_root_mean_squared_error")
    assert scores.shape == (3,)
    assert (-scores < 0.2).all(), scores


def test_classifier_clone():
    c = GaussianProcessClassifier(max_newton_iter=50, sigma2=1e-2)
    assert c._max_newton_iter == 50 and c._sigma2 == 1e-2
    c2 = clone(c)
    assert c2.get_params()["max_newton_iter"] == 50


def test_default_score_r2_and_cross_val_without_scoring():
    rng = np.random.default_rng(2)
    X = rng.uniform(size=(300, 2))
    y = np.sin(3 * X.sum(-1)) + 0.05 * rng.normal(size=300)
    est = _est()
    est.fit(X, y)
    assert est.score(X, y) > 0.9          # R^2 on train
    scores = cross_val_score(_est(), X, y, cv=3)   # default scorer
    assert (scores > 0.8).all(), scores
