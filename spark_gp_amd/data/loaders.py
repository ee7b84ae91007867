"""Dataset loaders for the example/acceptance suites.

``airfoil.csv`` is the public-domain UCI "Airfoil Self-Noise" dataset
(NASA; https://archive.ics.uci.edu/dataset/291/airfoil+self+noise) — the same
CSV the reference ships for its asserted RMSE gate
(``regression/examples/Airfoil.scala``).  It is a data file, not code.
"""

from __future__ import annotations

import os
from typing import Tuple

import numpy as np

_HERE = os.path.dirname(__file__)


def load_airfoil(path: str = None) -> Tuple[np.ndarray, np.ndarray]:
    """1503 rows, 5 features, label = scaled sound pressure level (dB)."""
    path = path or os.path.join(_HERE, "airfoil.csv")
    raw = np.loadtxt(path, delimiter=",")
    return raw[:, :5].copy(), raw[:, 5].copy()


def load_digits_pair(a: int = 6, b: int = 8) -> Tuple[np.ndarray, np.ndarray]:
    """Binary digit-classification stand-in for the reference's MNIST 6-vs-8
    example (its mnist68.csv is a missing blob in the repo; sklearn's bundled
    8x8 digits need no network)."""
    from sklearn.datasets import load_digits
    data = load_digits()
    mask = (data.target == a) | (data.target == b)
    X = data.data[mask].astype(np.float64)
    y = (data.target[mask] == b).astype(np.float64)
    return X, y
