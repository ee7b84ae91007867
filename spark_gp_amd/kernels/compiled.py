"""Kernel-tree canonicalization for the fused training fast path.

The effective training kernel is always
``user_kernel + sigma2.const * EyeKernel`` (see
``commons/GaussianProcessCommons.scala:18``).  The flagship shapes —
``1 * ARDRBFKernel(d) + noise`` and ``1 * RBFKernel(s) + noise`` — canonicalize
to

    K(theta) = C * Kb(theta_base)  +  nu(theta) * I

with Kb a single stationary base (RBF or ARD-RBF), C either a constant or one
trainable hyper, and nu = const + sum of trainable white-noise hypers.  The
fused objective (torch batched or the hand-written HIP kernel) consumes this
``CompiledKernel`` and computes the per-expert negative log marginal
likelihood *and its full gradient* without ever materializing the
``[p, k, k]`` derivative tensor the reference builds per expert
(``kernel/ARDRBFKernel.scala:61-79``).

Trees that do not match the pattern fall back to the generic batched path
(materialized derivatives) — full API generality is preserved.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

from .base import (ConstantTimesKernel, EyeKernel, Kernel, SumOfKernels,
                   TrainableScalarTimesKernel)
from .rbf import ARDRBFKernel, RBFKernel


@dataclass
class CompiledKernel:
    """Canonical form  C * base(theta_base) + nu * I.

    Index fields refer to positions in the *global* hyperparameter vector of
    the kernel tree (reference layout: Sum concatenates, TrainableScalar
    prepends)."""

    p: int                               # total number of hyperparameters
    base: str                            # 'rbf' | 'ard' | 'none'
    base_idx: slice                      # slice of theta for the base kernel
    amp_idx: Optional[int]               # index of trainable amplitude C, or None
    amp_const: float                     # amplitude when amp_idx is None
    noise_const: float                   # constant white-noise variance
    noise_idx: List[int] = field(default_factory=list)  # trainable noise hypers

    def amp(self, theta) -> float:
        return float(theta[self.amp_idx]) if self.amp_idx is not None else self.amp_const

    def noise(self, theta) -> float:
        return self.noise_const + sum(float(theta[i]) for i in self.noise_idx)


class _Acc:
    def __init__(self):
        self.base = None          # ('rbf'|'ard', base_idx_slice, amp_idx, amp_const)
        self.noise_const = 0.0
        self.noise_idx: List[int] = []
        self.ok = True


def _walk(k: Kernel, offset: int, scale_const: float,
          scale_idx: Optional[int], acc: _Acc) -> int:
    """Walk the tree accumulating canonical terms.  Returns hypers consumed.

    ``scale_const``/``scale_idx``: the product of enclosing scalar factors —
    at most one may be trainable for the pattern to hold."""
    if isinstance(k, SumOfKernels):
        n1 = _walk(k.k1, offset, scale_const, scale_idx, acc)
        n2 = _walk(k.k2, offset + n1, scale_const, scale_idx, acc)
        return n1 + n2
    if isinstance(k, TrainableScalarTimesKernel):
        if scale_idx is not None:
            acc.ok = False          # nested trainable scalars: not canonical
            return k.num_hyperparameters
        _walk(k.kernel, offset + 1, scale_const, offset, acc)
        return k.num_hyperparameters
    if isinstance(k, ConstantTimesKernel):
        _walk(k.kernel, offset, scale_const * k.C, scale_idx, acc)
        return k.num_hyperparameters
    if isinstance(k, EyeKernel):
        if scale_idx is not None:
            acc.noise_idx.append(scale_idx)
            if scale_const != 1.0:
                acc.ok = False      # const * trainable * I: not canonical
        else:
            acc.noise_const += scale_const
        return 0
    if isinstance(k, (RBFKernel, ARDRBFKernel)):
        if acc.base is not None:
            acc.ok = False          # two stationary bases: fall back
            return k.num_hyperparameters
        kind = 'rbf' if isinstance(k, RBFKernel) else 'ard'
        p = k.num_hyperparameters
        acc.base = (kind, slice(offset, offset + p), scale_idx,
                    scale_const if scale_idx is None else 1.0)
        if scale_idx is not None and scale_const != 1.0:
            acc.ok = False
        return p
    acc.ok = False                  # unknown kernel type
    return k.num_hyperparameters


def compile_kernel(kernel: Kernel) -> Optional[CompiledKernel]:
    """Return the canonical form, or None if the tree does not match."""
    acc = _Acc()
    p = _walk(kernel, 0, 1.0, None, acc)
    if not acc.ok or acc.base is None:
        return None
    kind, base_idx, amp_idx, amp_const = acc.base
    return CompiledKernel(p=p, base=kind, base_idx=base_idx, amp_idx=amp_idx,
                          amp_const=amp_const, noise_const=acc.noise_const,
                          noise_idx=acc.noise_idx)
