"""Covariance-function (kernel) DSL.

Re-implements the kernel algebra of the reference library
(``/root/reference/src/main/scala/org/apache/spark/ml/commons/kernel/``) as a
declarative, batch-evaluating expression tree on top of PyTorch tensors.

Design differences from the reference (deliberate, MI355X-first):

* Kernels are *specs + evaluators*: every evaluation method takes the data as a
  batched tensor argument ``X`` with shape ``[..., n, d]`` so thousands of
  expert chunks evaluate in a single batched call (one HIP kernel launch per
  op instead of one JVM task per expert).  The reference instead stores one
  ``Array[Vector]`` per kernel instance (``kernel/Kernel.scala:123-133``).
* For API parity the stateful interface (``set_training_vectors`` and
  zero-argument ``training_kernel()`` etc., see ``kernel/Kernel.scala:12-98``)
  is also provided.
* Hyperparameters are float64 numpy vectors on the host.  The layout (ordering,
  prepend/concat semantics) matches the reference exactly — this is
  model-format compatibility:
  - ``SumOfKernels`` concatenates child vectors (``kernel/SumOfKernels.scala:19-35``)
  - ``TrainableScalarTimesKernel`` *prepends* its scalar C
    (``kernel/ScalarTimesKernel.scala:76-98``)
"""

from __future__ import annotations

import math
from typing import Optional, Tuple

import numpy as np
import torch


class TrainingVectorsNotInitializedError(RuntimeError):
    """Raised when a stateful evaluation method is called before
    ``set_training_vectors`` (parity with ``kernel/Kernel.scala:116-117``)."""

    def __init__(self) -> None:
        super().__init__("set_training_vectors must be called before "
                         "using the stateful kernel evaluation API")


def sqdist(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Pairwise squared euclidean distances.

    a: [..., n, d], b: [..., m, d]  ->  [..., n, m]

    Uses the ||a||^2 + ||b||^2 - 2 a.b^T decomposition so the dominant cost is
    a GEMM (MFMA-friendly); clamps tiny negatives from cancellation to zero.
    """
    a2 = (a * a).sum(-1, keepdim=True)          # [..., n, 1]
    b2 = (b * b).sum(-1, keepdim=True)          # [..., m, 1]
    sq = a2 + b2.transpose(-1, -2) - 2.0 * (a @ b.transpose(-1, -2))
    return sq.clamp_min_(0.0)


def _as_f64(v) -> np.ndarray:
    return np.asarray(v, dtype=np.float64).reshape(-1)


class Kernel:
    """Base covariance function.  See ``kernel/Kernel.scala:12-98``."""

    # ----- hyperparameter interface -------------------------------------
    def get_hyperparameters(self) -> np.ndarray:
        raise NotImplementedError

    def set_hyperparameters(self, value) -> "Kernel":
        raise NotImplementedError

    @property
    def num_hyperparameters(self) -> int:
        raise NotImplementedError

    def hyperparameter_bounds(self) -> Tuple[np.ndarray, np.ndarray]:
        """(lower, upper) element-wise box bounds."""
        raise NotImplementedError

    def white_noise_var(self) -> float:
        """Variance of white noise presumed by the kernel
        (``kernel/Kernel.scala:96-98``); overridden by Eye/Sum/Scalar."""
        raise NotImplementedError

    # ----- batched evaluation (X: [..., n, d] torch tensor) -------------
    def training_kernel(self, X: torch.Tensor) -> torch.Tensor:
        """K with K[..., i, j] = k(X[..., i, :], X[..., j, :])."""
        raise NotImplementedError

    def training_kernel_diag(self, X: torch.Tensor) -> torch.Tensor:
        """Diagonal of ``training_kernel`` — [..., n]."""
        raise NotImplementedError

    def training_kernel_and_derivative(
        self, X: torch.Tensor
    ) -> Tuple[torch.Tensor, torch.Tensor]:
        """(K [..., n, n], dK [..., p, n, n]) with dK[..., i] = dK/dtheta_i.

        Generic path: materializes the p derivative matrices.  The hot
        training path avoids this via the compiled fused objective
        (``spark_gp_amd.kernels.compiled``)."""
        raise NotImplementedError

    def cross_kernel(self, Xtest: torch.Tensor, Xtrain: torch.Tensor) -> torch.Tensor:
        """[..., t, n] with K[i, j] = k(test_i, train_j)."""
        raise NotImplementedError

    def self_kernel(self, Xtest: torch.Tensor) -> torch.Tensor:
        """k(x, x) for each row — [...,] given [..., d]."""
        raise NotImplementedError

    # ----- stateful parity API ------------------------------------------
    _train: Optional[torch.Tensor] = None

    def set_training_vectors(self, X) -> "Kernel":
        self._train = torch.as_tensor(X)
        return self

    def get_training_vectors(self) -> torch.Tensor:
        if self._train is None:
            raise TrainingVectorsNotInitializedError()
        return self._train

    # zero-arg overloads matching the reference call shape
    def training_kernel_(self) -> torch.Tensor:
        return self.training_kernel(self.get_training_vectors())

    def training_kernel_diag_(self) -> torch.Tensor:
        return self.training_kernel_diag(self.get_training_vectors())

    def training_kernel_and_derivative_(self):
        return self.training_kernel_and_derivative(self.get_training_vectors())

    def cross_kernel_(self, Xtest) -> torch.Tensor:
        Xt = torch.as_tensor(Xtest)
        if Xt.dim() == 1:
            Xt = Xt.unsqueeze(0)
        return self.cross_kernel(Xt, self.get_training_vectors())

    # ----- DSL operators -------------------------------------------------
    def __add__(self, other: "Kernel") -> "SumOfKernels":
        return SumOfKernels(self, other)

    def __rmul__(self, c) -> "Kernel":
        # `1 * kernel` — trainable scalar with default bounds [0, inf),
        # matching the implicit toScalar of kernel/package.scala:4.
        return Scalar(float(c)) * self

    def __mul__(self, c):
        if isinstance(c, (int, float)):
            return Scalar(float(c)) * self
        return NotImplemented


class EyeKernel(Kernel):
    """Identity-matrix kernel (``kernel/Kernel.scala:142-164``)."""

    def get_hyperparameters(self) -> np.ndarray:
        return np.zeros(0)

    def set_hyperparameters(self, value) -> "EyeKernel":
        return self

    @property
    def num_hyperparameters(self) -> int:
        return 0

    def hyperparameter_bounds(self):
        return np.zeros(0), np.zeros(0)

    def white_noise_var(self) -> float:
        return 1.0

    def training_kernel(self, X):
        n = X.shape[-2]
        eye = torch.eye(n, dtype=X.dtype, device=X.device)
        return eye.expand(*X.shape[:-2], n, n).clone()

    def training_kernel_diag(self, X):
        return torch.ones(X.shape[:-1], dtype=X.dtype, device=X.device)

    def training_kernel_and_derivative(self, X):
        K = self.training_kernel(X)
        dK = torch.zeros((*X.shape[:-2], 0, X.shape[-2], X.shape[-2]),
                         dtype=X.dtype, device=X.device)
        return K, dK

    def cross_kernel(self, Xtest, Xtrain):
        t, n = Xtest.shape[-2], Xtrain.shape[-2]
        shape = torch.broadcast_shapes(Xtest.shape[:-2], Xtrain.shape[:-2])
        return torch.zeros((*shape, t, n), dtype=Xtest.dtype, device=Xtest.device)

    def self_kernel(self, Xtest):
        return torch.ones(Xtest.shape[:-1], dtype=Xtest.dtype, device=Xtest.device)

    def __repr__(self):
        return "I"


def WhiteNoiseKernel(initial: float, lower: float, upper: float) -> Kernel:
    """Trainable white noise = (initial between lower and upper) * EyeKernel
    (``kernel/Kernel.scala:166-169``)."""
    return Scalar(initial).between(lower, upper) * EyeKernel()


class SumOfKernels(Kernel):
    """k' = k1 + k2 (``kernel/SumOfKernels.scala``).  Hyper vectors concat."""

    def __init__(self, k1: Kernel, k2: Kernel):
        self.k1, self.k2 = k1, k2

    def get_hyperparameters(self):
        return np.concatenate([self.k1.get_hyperparameters(),
                               self.k2.get_hyperparameters()])

    def set_hyperparameters(self, value):
        v = _as_f64(value)
        n1 = self.k1.num_hyperparameters
        self.k1.set_hyperparameters(v[:n1])
        self.k2.set_hyperparameters(v[n1:])
        return self

    @property
    def num_hyperparameters(self):
        return self.k1.num_hyperparameters + self.k2.num_hyperparameters

    def hyperparameter_bounds(self):
        l1, u1 = self.k1.hyperparameter_bounds()
        l2, u2 = self.k2.hyperparameter_bounds()
        return np.concatenate([l1, l2]), np.concatenate([u1, u2])

    def white_noise_var(self):
        return self.k1.white_noise_var() + self.k2.white_noise_var()

    def training_kernel(self, X):
        return self.k1.training_kernel(X) + self.k2.training_kernel(X)

    def training_kernel_diag(self, X):
        return self.k1.training_kernel_diag(X) + self.k2.training_kernel_diag(X)

    def training_kernel_and_derivative(self, X):
        K1, d1 = self.k1.training_kernel_and_derivative(X)
        K2, d2 = self.k2.training_kernel_and_derivative(X)
        return K1 + K2, torch.cat([d1, d2], dim=-3)

    def cross_kernel(self, Xtest, Xtrain):
        return (self.k1.cross_kernel(Xtest, Xtrain)
                + self.k2.cross_kernel(Xtest, Xtrain))

    def self_kernel(self, Xtest):
        return self.k1.self_kernel(Xtest) + self.k2.self_kernel(Xtest)

    def set_training_vectors(self, X):
        super().set_training_vectors(X)
        self.k1.set_training_vectors(X)
        self.k2.set_training_vectors(X)
        return self

    def __repr__(self):
        parts = [repr(k) for k in (self.k1, self.k2)]
        return " + ".join(p for p in parts if p)


class ScalarTimesKernel(Kernel):
    """Base for C * k (``kernel/ScalarTimesKernel.scala:6-31``)."""

    def __init__(self, kernel: Kernel, C: float):
        if C < 0:
            raise ValueError("C should be non-negative")
        self.kernel = kernel
        self.C = float(C)

    def white_noise_var(self):
        return self.C * self.kernel.white_noise_var()

    def training_kernel(self, X):
        return self.kernel.training_kernel(X) * self.C

    def training_kernel_diag(self, X):
        return self.kernel.training_kernel_diag(X) * self.C

    def cross_kernel(self, Xtest, Xtrain):
        return self.kernel.cross_kernel(Xtest, Xtrain) * self.C

    def self_kernel(self, Xtest):
        return self.kernel.self_kernel(Xtest) * self.C

    def set_training_vectors(self, X):
        super().set_training_vectors(X)
        self.kernel.set_training_vectors(X)
        return self

    def __repr__(self):
        return f"{self.C:.1e} * {self.kernel!r}" if self.C != 0 else ""


class ConstantTimesKernel(ScalarTimesKernel):
    """C fixed (``kernel/ScalarTimesKernel.scala:41-59``)."""

    def get_hyperparameters(self):
        return self.kernel.get_hyperparameters()

    def set_hyperparameters(self, value):
        self.kernel.set_hyperparameters(value)
        return self

    @property
    def num_hyperparameters(self):
        return self.kernel.num_hyperparameters

    def hyperparameter_bounds(self):
        return self.kernel.hyperparameter_bounds()

    def training_kernel_and_derivative(self, X):
        K, dK = self.kernel.training_kernel_and_derivative(X)
        return K * self.C, dK * self.C


class TrainableScalarTimesKernel(ScalarTimesKernel):
    """C trainable, *prepended* as hyper 0
    (``kernel/ScalarTimesKernel.scala:71-98``)."""

    def __init__(self, kernel: Kernel, C: float,
                 C_lower: float = 0.0, C_upper: float = math.inf):
        super().__init__(kernel, C)
        self.C_lower = float(C_lower)
        self.C_upper = float(C_upper)

    def get_hyperparameters(self):
        return np.concatenate([[self.C], self.kernel.get_hyperparameters()])

    def set_hyperparameters(self, value):
        v = _as_f64(value)
        self.C = float(v[0])
        self.kernel.set_hyperparameters(v[1:])
        return self

    @property
    def num_hyperparameters(self):
        return 1 + self.kernel.num_hyperparameters

    def hyperparameter_bounds(self):
        lo, up = self.kernel.hyperparameter_bounds()
        return (np.concatenate([[self.C_lower], lo]),
                np.concatenate([[self.C_upper], up]))

    def training_kernel_and_derivative(self, X):
        K, dK = self.kernel.training_kernel_and_derivative(X)
        # dK'/dC = K (unscaled); the child's derivatives scale by C.
        return K * self.C, torch.cat([K.unsqueeze(-3), dK * self.C], dim=-3)


class Scalar:
    """Scalar builder for the DSL (``kernel/ScalarTimesKernel.scala:100-141``).

    >>> Scalar(1).between(0, 30) * ARDRBFKernel(5)
    >>> Scalar(0.5).const * EyeKernel()
    >>> 1 * kernel            # trainable, bounds [0, inf)
    """

    def __init__(self, C: float, lower: float = 0.0, upper: float = math.inf,
                 trainable: bool = True):
        if trainable and not lower < upper:
            raise ValueError("lower bound must be below upper bound "
                             "for a trainable scalar")
        self.C, self.lower, self.upper, self.trainable = C, lower, upper, trainable

    def __mul__(self, kernel: Kernel) -> Kernel:
        if self.trainable:
            return TrainableScalarTimesKernel(kernel, self.C, self.lower, self.upper)
        return ConstantTimesKernel(kernel, self.C)

    def between(self, lower: float, upper: float) -> "Scalar":
        return Scalar(self.C, lower, upper, self.trainable)

    def below(self, upper: float) -> "Scalar":
        return Scalar(self.C, self.lower, upper, self.trainable)

    @property
    def const(self) -> "Scalar":
        return Scalar(self.C, self.C, self.C, trainable=False)
