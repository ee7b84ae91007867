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


def mnist_like_binary(n_rows: int = 11769, side: int = 28, modes: int = 8,
                      strokes: int = 6, seed: int = 13,
                      dtype=np.float64) -> Tuple[np.ndarray, np.ndarray]:
    """A 784-dim MNIST-shaped binary stand-in at MNIST 6-vs-8 row counts.

    There is no network in this environment, so BASELINE config 3 ("MNIST
    6-vs-8, probit, m=1000") runs on synthetic images with the same shape
    statistics: each class is a mixture of ``modes`` prototype "digits"
    (writing styles), a prototype being ``strokes`` Gaussian strokes on a
    ``side`` x ``side`` canvas; every sample re-renders its prototype with
    per-sample stroke jitter (position/width/intensity), pixel noise and
    clipping to [0, 1].  Classes differ in stroke layout; within-class
    variation is nonlinear in pixel space — qualitatively the difficulty
    profile of real MNIST pairs, and honestly labeled as a stand-in in
    BASELINE.md."""
    rng = np.random.default_rng(seed)
    d = side * side
    ys = rng.integers(0, 2, size=n_rows).astype(np.float64)
    gy, gx = np.mgrid[0:side, 0:side]
    grid = np.stack([gy.ravel(), gx.ravel()], -1).astype(np.float64)  # [d,2]

    # per class, per mode: stroke centers [G,2], widths [G], amps [G]
    centers = rng.uniform(side * 0.15, side * 0.85, (2, modes, strokes, 2))
    widths = rng.uniform(1.2, 2.8, (2, modes, strokes))
    amps = rng.uniform(0.6, 1.0, (2, modes, strokes))

    mode_of = rng.integers(0, modes, size=n_rows)
    X = np.empty((n_rows, d), dtype=np.float64)
    chunk = 2048
    for s in range(0, n_rows, chunk):
        e = min(n_rows, s + chunk)
        n = e - s
        cls = ys[s:e].astype(int)
        md = mode_of[s:e]
        c = centers[cls, md] + rng.normal(0.0, 0.8, (n, strokes, 2))
        w = widths[cls, md] * np.exp(rng.normal(0.0, 0.12, (n, strokes)))
        a = amps[cls, md] * np.exp(rng.normal(0.0, 0.15, (n, strokes)))
        # [n, G, d] squared distances stroke-center -> pixel
        dist2 = ((c[:, :, None, :] - grid[None, None, :, :]) ** 2).sum(-1)
        img = (a[:, :, None]
               * np.exp(-dist2 / (2.0 * w[:, :, None] ** 2))).sum(1)
        img += rng.normal(0.0, 0.05, img.shape)
        X[s:e] = np.clip(img, 0.0, 1.0)
    return X.astype(dtype), ys


def shard_benchmark_data_device(n_total: int, d: int, rank: int,
                                world_size: int, seed: int = 13,
                                noise_sd: float = 0.1):
    """This rank's shard of the flagship benchmark data, generated ON the
    GPU by the K19 Philox4x32-10 kernel (synth.hip) — same distribution as
    ``benchmark_regression_data`` (X ~ U[0,1)^d, y = sin(2 sum x) + noise)
    without the host-side generation + H2D that costs minutes and >100 GB
    of host RAM at the 100M x 128 config."""
    base = n_total // world_size
    rem = n_total % world_size
    n_local = base + (1 if rank < rem else 0)
    from spark_gp_amd import _hip_ext
    return _hip_ext.synth_regression(n_local, d, int(seed + 1009 * rank),
                                     float(noise_sd))


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
