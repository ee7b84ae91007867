from .classification import (GaussianProcessClassificationModel,
                             GaussianProcessClassifier)
from .model_io import load_model, save_model
from .regression import GaussianProcessRegression, GaussianProcessRegressionModel

__all__ = [
    "GaussianProcessRegression", "GaussianProcessRegressionModel",
    "GaussianProcessClassifier", "GaussianProcessClassificationModel",
    "save_model", "load_model",
]
