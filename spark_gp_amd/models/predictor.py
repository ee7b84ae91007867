"""The size-m projected-process predictor shared by GPR and GPC.

Mirrors ``GaussianProjectedProcessRawPredictor``
(``commons/GaussianProcessCommons.scala:118-126``): the model is the magic
vector (m), magic matrix (m x m), the active set (m x d) and the kernel with
optimal hyperparameters; prediction cost is independent of the training set
size.  Prediction here is batched: an [t, m] cross-kernel GEMM + GEMV for the
mean and a rowwise quadratic form for the variance (K14 in SURVEY.md §2.4).
"""

from __future__ import annotations

from typing import Tuple

import torch

from ..kernels.base import Kernel
from .. import ops


class GaussianProjectedProcessRawPredictor:
    def __init__(self, magic_vector: torch.Tensor, magic_matrix: torch.Tensor,
                 kernel: Kernel, active_set: torch.Tensor):
        self.magic_vector = magic_vector.double()
        self.magic_matrix = magic_matrix.double()
        self.kernel = kernel
        self.active_set = active_set

    def predict(self, X: torch.Tensor, with_var: bool = True
                ) -> Tuple[torch.Tensor, torch.Tensor]:
        """X: [t, d] -> (mean [t], var [t]).

        mean = cross . magicVector
        var  = k(x,x) + cross . magicMatrix . cross^T   (rowwise)
        """
        act = self.active_set.to(X.dtype).to(X.device)
        cross = ops.cross_kernel(self.kernel, X, act).double()
        mv = self.magic_vector.to(cross.device)
        mean = cross @ mv
        if not with_var:
            return mean, None
        mm = self.magic_matrix.to(cross.device)
        var = (self.kernel.self_kernel(X).double()
               + ((cross @ mm) * cross).sum(-1))
        return mean, var
