"""Synthetic data generators for tests and benchmarks.

``performance_benchmark_data`` mirrors the reference's timing-harness data:
y = sin(sum(x)/1000) with x ~ U[0,1)^d
(``regression/benchmark/PerformanceBenchmark.scala:24-39``).

``sin_wave`` mirrors the Synthetics example: y = sin(x) + N(0, 0.01) on a
linspace grid (``regression/examples/Synthetics.scala:17-23``).
"""

from __future__ import annotations

from typing import Tuple

import numpy as np


def performance_benchmark_data(n: int, d: int = 3, seed: int = 13,
                               dtype=np.float32) -> Tuple[np.ndarray, np.ndarray]:
    rng = np.random.default_rng(seed)
    X = rng.random((n, d), dtype=np.float64)
    y = np.sin(X.sum(-1) / 1000.0)
    return X.astype(dtype), y.astype(dtype)


def benchmark_regression_data(n: int, d: int = 32, seed: int = 13,
                              noise_sd: float = 0.1, dtype=np.float32
                              ) -> Tuple[np.ndarray, np.ndarray]:
    """Non-degenerate synthetic regression target for the flagship bench:
    y = sin(2 sum x) + noise.  (The reference harness's y = sin(sum x/1000)
    is nearly constant, which drives the ARD length-scales to the boundary
    and makes the fit degenerate — a constant-predictor corner case, not a
    representative GP training workload.)"""
    rng = np.random.default_rng(seed)
    X = rng.random((n, d), dtype=np.float64)
    y = np.sin(2.0 * X.sum(-1)) + noise_sd * rng.standard_normal(n)
    return X.astype(dtype), y.astype(dtype)


def shard_performance_benchmark_data(n_total: int, d: int, rank: int,
                                     world_size: int, seed: int = 13,
                                     dtype=np.float32):
    """Each rank generates only its contiguous shard (no network, no
    broadcast); seeds differ per rank so shards are independent draws."""
    base = n_total // world_size
    rem = n_total % world_size
    n_local = base + (1 if rank < rem else 0)
    return benchmark_regression_data(n_local, d, seed=seed + 1009 * rank,
                                     dtype=dtype)


def sin_wave(n: int = 2000, noise_var: float = 0.01, seed: int = 13
             ) -> Tuple[np.ndarray, np.ndarray]:
    rng = np.random.default_rng(seed)
    x = np.linspace(0.0, 1.0, n)
    y = np.sin(x) + rng.normal(0.0, np.sqrt(noise_var), size=n)
    return x.reshape(-1, 1), y
