from .loaders import load_airfoil, load_digits_pair
from .synthetic import (performance_benchmark_data,
                        shard_performance_benchmark_data, sin_wave)

__all__ = ["performance_benchmark_data", "shard_performance_benchmark_data",
           "sin_wave", "load_airfoil", "load_digits_pair"]
