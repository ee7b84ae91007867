"""Evaluation utilities replacing the Spark ML services the reference's
examples consume: k-fold cross-validation, train/validation
split, RMSE / accuracy evaluators and a one-vs-rest multiclass wrapper.
"""

from __future__ import annotations

from typing import Callable, List

import numpy as np


def rmse(y_true, y_pred) -> float:
    y_true = np.asarray(y_true, dtype=np.float64).reshape(-1)
    y_pred = np.asarray(y_pred, dtype=np.float64).reshape(-1)
    return float(np.sqrt(np.mean((y_true - y_pred) ** 2)))


def accuracy(y_true, y_pred) -> float:
    y_true = np.asarray(y_true).reshape(-1)
    y_pred = np.asarray(y_pred).reshape(-1)
    return float(np.mean(y_true == y_pred))


def _fold_indices(n: int, num_folds: int, seed: int) -> List[np.ndarray]:
    rng = np.random.default_rng(seed)
    perm = rng.permutation(n)
    return [perm[i::num_folds] for i in range(num_folds)]


def cross_validate(estimator_factory: Callable[[], object], X, y,
                   num_folds: int = 10, metric=rmse, seed: int = 0) -> float:
    """Mean metric over k folds (Spark ``CrossValidator`` analog,
    ``regression/examples/GPExample.scala:17-27``)."""
    X = np.asarray(X, dtype=np.float64)
    y = np.asarray(y, dtype=np.float64).reshape(-1)
    folds = _fold_indices(len(y), num_folds, seed)
    scores = []
    for i in range(num_folds):
        test_idx = folds[i]
        train_idx = np.concatenate([folds[j] for j in range(num_folds)
                                    if j != i])
        est = estimator_factory()
        model = est.fit(X[train_idx], y[train_idx])
        scores.append(metric(y[test_idx], model.predict(X[test_idx])))
    return float(np.mean(scores))


def train_validation_split(estimator, X, y, train_ratio: float = 0.8,
                           metric=rmse, seed: int = 0) -> float:
    """Spark ``TrainValidationSplit`` analog (``classification/examples/MNIST.scala:34-40``)."""
    X = np.asarray(X, dtype=np.float64)
    y = np.asarray(y, dtype=np.float64).reshape(-1)
    rng = np.random.default_rng(seed)
    perm = rng.permutation(len(y))
    cut = int(train_ratio * len(y))
    tr, te = perm[:cut], perm[cut:]
    model = estimator.fit(X[tr], y[tr])
    return metric(y[te], model.predict(X[te]))


class OneVsRest:
    """Multiclass via one binary GP classifier per class
    (Spark ``OneVsRest`` analog, ``classification/examples/Iris.scala:27-35``)."""

    def __init__(self, estimator_factory: Callable[[], object]):
        self.factory = estimator_factory
        self.models = []
        self.classes_: np.ndarray = np.zeros(0)

    def fit(self, X, y) -> "OneVsRest":
        y = np.asarray(y).reshape(-1)
        self.classes_ = np.unique(y)
        self.models = []
        for c in self.classes_:
            est = self.factory()
            self.models.append(est.fit(X, (y == c).astype(np.float64)))
        return self

    def predict(self, X) -> np.ndarray:
        scores = np.stack([m.predict_raw(X)[:, 1] for m in self.models],
                          axis=-1)
        return self.classes_[np.argmax(scores, axis=-1)]
