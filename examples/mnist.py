"""Binary digit classification — mirrors
``classification/examples/MNIST.scala`` (CLI-parameterized expert/active-set
sizes, feature scaling, 80/20 train/validation split, accuracy printed).

The reference's mnist68.csv is a missing blob in its repo; sklearn's bundled
8x8 digits (no network needed) stand in: digits 6 vs 8 by default.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse


from spark_gp_amd import (GaussianProcessClassifier, RBFKernel, StandardScaler,
                          accuracy, train_validation_split)
from spark_gp_amd.data import load_digits_pair


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--expert-size", type=int, default=100)
    p.add_argument("--active-set", type=int, default=100)
    p.add_argument("--digits", type=int, nargs=2, default=(6, 8))
    args = p.parse_args()

    X, y = load_digits_pair(*args.digits)
    Xs = StandardScaler().fit_transform(X)

    gp = (GaussianProcessClassifier()
          .setKernel(lambda: 1 * RBFKernel(10.0))
          .setDatasetSizeForExpert(args.expert_size)
          .setActiveSetSize(args.active_set)
          .setSigma2(1e-3)
          .setTol(1e-3)
          .setSeed(13))

    acc = train_validation_split(gp, Xs, y, train_ratio=0.8,
                                 metric=accuracy, seed=13)
    print("accuracy:", acc)
    return acc


if __name__ == "__main__":
    main()
