from .classification import (GaussianProcessClassificationModel,
                             GaussianProcessClassifier)
from .model_io import load_model, save_model
from .poisson import (GaussianProcessPoissonModel,
                      GaussianProcessPoissonRegression)
from .regression import GaussianProcessRegression, GaussianProcessRegressionModel

__all__ = [
    "GaussianProcessRegression", "GaussianProcessRegressionModel",
    "GaussianProcessClassifier", "GaussianProcessClassificationModel",
    "GaussianProcessPoissonRegression", "GaussianProcessPoissonModel",
    "save_model", "load_model",
]
