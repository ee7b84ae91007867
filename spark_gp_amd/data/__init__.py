from .loaders import load_airfoil, load_digits_pair
from .synthetic import (benchmark_regression_data, mnist_like_binary,
                        shard_benchmark_data_device,
                        performance_benchmark_data,
                        shard_performance_benchmark_data, sin_wave)

__all__ = ["performance_benchmark_data", "benchmark_regression_data",
           "mnist_like_binary", "shard_benchmark_data_device",
           "shard_performance_benchmark_data",
           "sin_wave", "load_airfoil", "load_digits_pair"]
