"""Saved-model format (additive capability — the reference has no model
persistence at all, SURVEY.md §5 Checkpoint/resume).

Layout (C12 in SURVEY.md §2.5): one ``model.safetensors`` file with
``magic_vector [m]``, ``magic_matrix [m, m]``, ``active_set [m, d]`` plus a
JSON ``spec.json`` holding the kernel expression tree, its optimal
hyperparameters and the model kind.
"""

from __future__ import annotations

import json
import math
import os
from typing import Dict

import numpy as np

from ..kernels import (ARDRBFKernel, ConstantTimesKernel, EyeKernel, Kernel,
                       Matern32Kernel, Matern52Kernel, RBFKernel, SumOfKernels,
                       TrainableScalarTimesKernel)
from .classification import GaussianProcessClassificationModel
from .poisson import GaussianProcessPoissonModel
from .predictor import GaussianProjectedProcessRawPredictor
from .regression import GaussianProcessRegressionModel


def _enc(x):
    """JSON-safe float (inf round-trips as a string)."""
    if isinstance(x, float) and math.isinf(x):
        return "inf" if x > 0 else "-inf"
    return x


def _dec(x):
    if x == "inf":
        return math.inf
    if x == "-inf":
        return -math.inf
    return x


def kernel_to_spec(k: Kernel) -> Dict:
    if isinstance(k, SumOfKernels):
        return {"op": "sum", "args": [kernel_to_spec(k.k1), kernel_to_spec(k.k2)]}
    if isinstance(k, TrainableScalarTimesKernel):
        return {"op": "tscale", "C": k.C, "lower": _enc(k.C_lower),
                "upper": _enc(k.C_upper), "arg": kernel_to_spec(k.kernel)}
    if isinstance(k, ConstantTimesKernel):
        return {"op": "cscale", "C": k.C, "arg": kernel_to_spec(k.kernel)}
    if isinstance(k, EyeKernel):
        return {"op": "eye"}
    if isinstance(k, RBFKernel):
        return {"op": "rbf", "sigma": k.sigma, "lower": _enc(k.lower),
                "upper": _enc(k.upper)}
    if isinstance(k, (Matern32Kernel, Matern52Kernel)):
        return {"op": "m32" if isinstance(k, Matern32Kernel) else "m52",
                "l": k.l, "lower": _enc(k.lower), "upper": _enc(k.upper)}
    if isinstance(k, ARDRBFKernel):
        return {"op": "ard", "beta": k.beta.tolist(),
                "lower": [_enc(float(v)) for v in k.lower],
                "upper": [_enc(float(v)) for v in k.upper]}
    raise TypeError(f"cannot serialize kernel type {type(k).__name__}")


def kernel_from_spec(spec: Dict) -> Kernel:
    op = spec["op"]
    if op == "sum":
        return SumOfKernels(kernel_from_spec(spec["args"][0]),
                            kernel_from_spec(spec["args"][1]))
    if op == "tscale":
        return TrainableScalarTimesKernel(kernel_from_spec(spec["arg"]),
                                          spec["C"], _dec(spec["lower"]),
                                          _dec(spec["upper"]))
    if op == "cscale":
        return ConstantTimesKernel(kernel_from_spec(spec["arg"]), spec["C"])
    if op == "eye":
        return EyeKernel()
    if op == "rbf":
        return RBFKernel(spec["sigma"], _dec(spec["lower"]), _dec(spec["upper"]))
    if op in ("m32", "m52"):
        cls = Matern32Kernel if op == "m32" else Matern52Kernel
        return cls(spec["l"], _dec(spec["lower"]), _dec(spec["upper"]))
    if op == "ard":
        return ARDRBFKernel(np.array(spec["beta"]),
                            beta=1.0,
                            lower=np.array([_dec(v) for v in spec["lower"]]),
                            upper=np.array([_dec(v) for v in spec["upper"]]))
    raise ValueError(f"unknown kernel spec op {op!r}")


def save_model(model, path: str) -> None:
    from safetensors.torch import save_file
    os.makedirs(path, exist_ok=True)
    raw: GaussianProjectedProcessRawPredictor = model.raw
    if isinstance(model, GaussianProcessClassificationModel):
        kind = "classification"
    elif isinstance(model, GaussianProcessPoissonModel):
        kind = "poisson"
    else:
        kind = "regression"
    save_file({
        "magic_vector": raw.magic_vector.cpu().contiguous(),
        "magic_matrix": raw.magic_matrix.cpu().contiguous(),
        "active_set": raw.active_set.cpu().contiguous(),
    }, os.path.join(path, "model.safetensors"))
    spec = {"kind": kind,
            "kernel": kernel_to_spec(raw.kernel),
            "format_version": 1}
    if kind == "classification":
        spec["link"] = getattr(model, "link", "logistic")
    with open(os.path.join(path, "spec.json"), "w") as fh:
        json.dump(spec, fh, indent=2)


def load_model(path: str, device: str = "cpu"):
    from safetensors.torch import load_file
    tensors = load_file(os.path.join(path, "model.safetensors"), device=device)
    with open(os.path.join(path, "spec.json")) as fh:
        spec = json.load(fh)
    kernel = kernel_from_spec(spec["kernel"])
    raw = GaussianProjectedProcessRawPredictor(
        tensors["magic_vector"], tensors["magic_matrix"], kernel,
        tensors["active_set"])
    if spec["kind"] == "classification":
        return GaussianProcessClassificationModel(
            raw, link=spec.get("link", "logistic"))
    if spec["kind"] == "poisson":
        return GaussianProcessPoissonModel(raw)
    return GaussianProcessRegressionModel(raw)
