"""Minimal multi-process (one process per GPU / CPU rank) fit — the
pattern described in MIGRATING.md "Running distributed".

Launch:
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
        --master-addr 127.0.0.1 examples/distributed_fit.py

Works on CPU (gloo) and on MI355X nodes (RCCL over xGMI) with the same
code; every rank ends up holding the identical model.
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from spark_gp_amd import GaussianProcessRegression, get_comm, init_from_env
from spark_gp_amd.kernels import ARDRBFKernel


def main():
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    init_from_env(device)
    comm = get_comm()

    # each rank generates/loads ONLY its shard (any contiguous split works)
    rng = np.random.default_rng(100 + comm.rank)
    n_local = 50_000
    X = rng.uniform(size=(n_local, 8))
    y = np.sin(2 * X.sum(-1)) + 0.05 * rng.normal(size=n_local)

    model = (GaussianProcessRegression()
             .setKernel(lambda: 1 * ARDRBFKernel(8))
             .setDatasetSizeForExpert(100)
             .setActiveSetSize(500)
             .setSigma2(1e-3)
             .setMaxIter(30)
             .setSeed(0)
             .setDevice(str(device))
             .fit(X, y))

    # the model is replicated: predictions need no gathering
    Xq = np.random.default_rng(7).uniform(size=(1000, 8))
    pred = model.predict(Xq)
    err = float(np.sqrt(np.mean((pred - np.sin(2 * Xq.sum(-1))) ** 2)))
    print(f"rank {comm.rank}/{comm.world_size}: holdout rmse {err:.4f}")
    return err


if __name__ == "__main__":
    main()
