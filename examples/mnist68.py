"""BASELINE config 3: MNIST-scale 6-vs-8 binary GP classification with the
probit link, m=1000 active set, on 1 MI355X.

Mirrors ``classification/examples/MNIST.scala:15-45`` (feature scaling,
RBF(10) kernel, tol=1e-3, 80/20 train/validation split, accuracy printed) at
the full stated scale: 11,769 rows x 784 features (the MNIST train-set
6-vs-8 count).  The reference repo's mnist68.csv is a missing blob and this
environment has no network, so the data is a synthetic 784-dim MNIST-shaped
stand-in (``spark_gp_amd.data.mnist_like_binary``) — recorded as such in
BASELINE.md.
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import argparse

from spark_gp_amd import (GaussianProcessClassifier, RBFKernel, StandardScaler,
                          accuracy, train_validation_split)
from spark_gp_amd.data import mnist_like_binary


def main(argv=None):
    p = argparse.ArgumentParser()
    p.add_argument("--rows", type=int, default=11769)
    p.add_argument("--expert-size", type=int, default=100)
    p.add_argument("--active-set", type=int, default=1000)
    p.add_argument("--link", type=str, default="probit",
                   choices=["probit", "logistic"])
    p.add_argument("--max-iter", type=int, default=100)
    p.add_argument("--device", type=str, default=None)
    args = p.parse_args(argv)

    X, y = mnist_like_binary(args.rows, seed=13)
    Xs = StandardScaler().fit_transform(X)

    gp = (GaussianProcessClassifier()
          .setKernel(lambda: 1 * RBFKernel(10.0))
          .setLink(args.link)
          .setDatasetSizeForExpert(args.expert_size)
          .setActiveSetSize(args.active_set)
          .setSigma2(1e-3)
          .setTol(1e-3)
          .setMaxIter(args.max_iter)
          .setSeed(13))
    if args.device:
        gp.setDevice(args.device)

    t0 = time.perf_counter()
    acc = train_validation_split(gp, Xs, y, train_ratio=0.8,
                                 metric=accuracy, seed=13)
    dt = time.perf_counter() - t0
    print(f"rows={args.rows} d=784 m={args.active_set} link={args.link} "
          f"accuracy: {acc:.4f}  fit+eval: {dt:.2f}s")
    return acc


if __name__ == "__main__":
    main()
