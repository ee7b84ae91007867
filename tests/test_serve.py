"""Model-serving layer (serve.py): save -> serve -> HTTP predict round
trips for both model kinds, via fastapi's in-process TestClient."""

import numpy as np
import pytest

fastapi = pytest.importorskip("fastapi")

from spark_gp_amd import GaussianProcessRegression, save_model
from spark_gp_amd.kernels import ARDRBFKernel
from spark_gp_amd.models.classification import GaussianProcessClassifier
from spark_gp_amd.serve import create_app


@pytest.fixture(scope="module")
def reg_model_dir(tmp_path_factory):
    rng = np.random.default_rng(0)
    X = rng.uniform(size=(400, 3))
    y = np.sin(3 * X.sum(-1)) + 0.05 * rng.normal(size=400)
    model = (GaussianProcessRegression()
             .setKernel(lambda: 1 * ARDRBFKernel(3))
             .setDatasetSizeForExpert(50).setActiveSetSize(80)
             .setSigma2(1e-2).setMaxIter(25).setSeed(0).setDevice("cpu")
             .fit(X, y))
    d = tmp_path_factory.mktemp("reg_model")
    save_model(model, str(d))
    return str(d), model, X


def test_serve_regression_roundtrip(reg_model_dir):
    from fastapi.testclient import TestClient
    path, model, X = reg_model_dir
    client = TestClient(create_app(path))

    h = client.get("/health").json()
    assert h["status"] == "ok" and h["kind"] == "regression" and h["d"] == 3

    r = client.post("/predict", json={"X": X[:10].tolist()})
    assert r.status_code == 200
    np.testing.assert_allclose(r.json()["mean"], model.predict(X[:10]),
                               rtol=1e-10)

    r = client.post("/predict", json={"X": X[:5].tolist(),
                                      "return_std": True})
    body = r.json()
    assert len(body["std"]) == 5 and min(body["std"]) >= 0

    # wrong feature dim -> 422 with a helpful message
    r = client.post("/predict", json={"X": [[1.0, 2.0]]})
    assert r.status_code == 422


def test_serve_classification_roundtrip(tmp_path):
    from fastapi.testclient import TestClient
    rng = np.random.default_rng(1)
    X = np.concatenate([rng.normal(-1.0, 0.5, (150, 2)),
                        rng.normal(1.0, 0.5, (150, 2))])
    y = np.concatenate([np.zeros(150), np.ones(150)])
    model = (GaussianProcessClassifier()
             .setKernel(lambda: 1 * ARDRBFKernel(2))
             .setDatasetSizeForExpert(50).setActiveSetSize(60)
             .setSigma2(1e-3).setMaxIter(20).setSeed(0).setDevice("cpu")
             .fit(X, y))
    save_model(model, str(tmp_path))
    client = TestClient(create_app(str(tmp_path)))

    assert client.get("/health").json()["kind"] == "classification"
    r = client.post("/predict", json={"X": X[:20].tolist()}).json()
    assert (np.array(r["label"]) == y[:20]).mean() >= 0.9
    assert all(0.0 <= p <= 1.0 for p in r["proba"])


def test_serve_poisson_roundtrip(tmp_path):
    from fastapi.testclient import TestClient
    from spark_gp_amd import GaussianProcessPoissonRegression, save_model
    rng = np.random.default_rng(3)
    X = rng.uniform(size=(400, 2))
    y = rng.poisson(np.exp(1.0 + np.sin(3 * X.sum(-1)))).astype(np.float64)
    model = (GaussianProcessPoissonRegression()
             .setKernel(lambda: 1 * ARDRBFKernel(2))
             .setDatasetSizeForExpert(50).setActiveSetSize(60)
             .setSigma2(1e-2).setMaxIter(15).setSeed(0).setDevice("cpu")
             .fit(X, y))
    save_model(model, str(tmp_path))
    client = TestClient(create_app(str(tmp_path)))
    assert client.get("/health").json()["kind"] == "poisson"
    r = client.post("/predict", json={"X": X[:10].tolist()}).json()
    np.testing.assert_allclose(r["rate"], model.predict(X[:10]), rtol=1e-10)
    assert all(v >= 0 for v in r["latent_var"])
