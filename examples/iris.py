"""Iris 3-class classification via one-vs-rest binary GP classifiers —
mirrors ``classification/examples/Iris.scala`` (expert=20, active=30,
10-fold CV accuracy printed)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from spark_gp_amd import (GaussianProcessClassifier, OneVsRest, RBFKernel,
                          accuracy)
from spark_gp_amd.utils.evaluation import _fold_indices


def main():
    from sklearn.datasets import load_iris
    data = load_iris()
    X, y = data.data, data.target.astype(np.float64)

    def binary_gp():
        return (GaussianProcessClassifier()
                .setKernel(lambda: 1 * RBFKernel(1.0, 1e-3, 10))
                .setDatasetSizeForExpert(20)
                .setActiveSetSize(30)
                .setSigma2(1e-3)
                .setMaxIter(50)
                .setSeed(7))

    folds = _fold_indices(len(y), 10, seed=13)
    accs = []
    for i in range(10):
        test = folds[i]
        train = np.concatenate([folds[j] for j in range(10) if j != i])
        model = OneVsRest(binary_gp).fit(X[train], y[train])
        accs.append(accuracy(y[test], model.predict(X[test])))
    print("accuracy:", float(np.mean(accs)))
    return float(np.mean(accs))


if __name__ == "__main__":
    main()
