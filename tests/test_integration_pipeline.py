"""End-to-end user workflow: standardize -> fit -> evaluate -> persist ->
reload -> serve over HTTP.  Exercises the public API exactly as
MIGRATING.md documents it."""

import numpy as np
import pytest

from spark_gp_amd import (GaussianProcessRegression, StandardScaler,
                          load_model, rmse, save_model)
from spark_gp_amd.kernels import ARDRBFKernel


def test_full_regression_pipeline(tmp_path):
    rng = np.random.default_rng(42)
    X_raw = rng.normal(loc=5.0, scale=[1.0, 10.0, 0.1],
                       size=(800, 3))          # wildly different scales
    y = np.sin(X_raw[:, 0]) + 0.1 * X_raw[:, 1] / 10.0 \
        + 0.05 * rng.normal(size=800)

    scaler = StandardScaler().fit(X_raw)
    X = scaler.transform(X_raw)
    assert abs(X.mean()) < 1e-8 and abs(X.std() - 1.0) < 0.1

    model = (GaussianProcessRegression()
             .setKernel(lambda: 1 * ARDRBFKernel(3))
             .setDatasetSizeForExpert(80)
             .setActiveSetSize(150)
             .setSigma2(1e-2)
             .setMaxIter(30)
             .setSeed(1)
             .setDevice("cpu")
             .fit(X, y))
    err = rmse(y, model.predict(X))
    assert err < 0.12, err

    save_model(model, str(tmp_path))
    reloaded = load_model(str(tmp_path))
    np.testing.assert_allclose(reloaded.predict(X[:50]),
                               model.predict(X[:50]), rtol=1e-12)

    fastapi = pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient
    from spark_gp_amd.serve import create_app
    client = TestClient(create_app(str(tmp_path)))
    served = client.post("/predict",
                         json={"X": X[:10].tolist()}).json()["mean"]
    np.testing.assert_allclose(served, model.predict(X[:10]), rtol=1e-10)
