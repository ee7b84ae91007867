from .evaluation import (OneVsRest, accuracy, cross_validate, rmse,
                         train_validation_split)
from .instrumentation import Instrumentation
from .integrator import Integrator
from .scaling import StandardScaler, scale

__all__ = [
    "rmse", "accuracy", "cross_validate", "train_validation_split",
    "OneVsRest", "Integrator", "StandardScaler", "scale", "Instrumentation",
]
