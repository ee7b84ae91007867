"""Distributed runtime: one process per GPU, RCCL (``nccl`` backend on ROCm)
over xGMI; ``gloo`` for CPU tests.

Replaces the reference's Spark driver-executor collectives:

* C1  treeAggregate of (nll, grad)          -> allreduce of a (1+p) fp64 tensor
* C2  treeAggregate of (K_mn K_nm, K_mn y)  -> allreduce of [m,m]+[m] fp64
* C3  takeSample of the active set          -> seeded global index sampling +
                                               disjoint-fill + allreduce(SUM)
* C5  groupByKey expert shuffle             -> none: contiguous per-rank shards,
                                               experts formed locally
* C8  aggregate AND over labels             -> allreduce(MIN) of a byte
* C10 hyperparameter push                   -> none: L-BFGS-B runs replicated
                                               and deterministically on every
                                               rank (identical allreduced
                                               objective => identical iterates)

All collective payloads here are tiny relative to the GEMMs producing them
(SURVEY.md §2.5 topology note), so plain RCCL defaults are used.
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

try:
    import torch.distributed as dist
except Exception:  # pragma: no cover
    dist = None


class Comm:
    """Thin wrapper over torch.distributed; degenerates to no-ops when
    uninitialized (single-process)."""

    def __init__(self):
        self._active = dist is not None and dist.is_available() \
            and dist.is_initialized()

    @property
    def active(self) -> bool:
        return self._active

    @property
    def rank(self) -> int:
        return dist.get_rank() if self._active else 0

    @property
    def world_size(self) -> int:
        return dist.get_world_size() if self._active else 1

    def _backend_device(self, t: torch.Tensor) -> torch.Tensor:
        """gloo cannot reduce CUDA tensors; nccl cannot reduce CPU tensors."""
        if not self._active:
            return t
        backend = dist.get_backend()
        if backend == "gloo" and t.is_cuda:
            return t.cpu()
        if backend == "nccl" and not t.is_cuda:
            return t.cuda()
        return t

    def allreduce_(self, t: torch.Tensor, op: str = "sum") -> torch.Tensor:
        if not self._active:
            return t
        buf = self._backend_device(t)
        red = {"sum": dist.ReduceOp.SUM, "min": dist.ReduceOp.MIN,
               "max": dist.ReduceOp.MAX}[op]
        dist.all_reduce(buf, op=red)
        if buf.data_ptr() != t.data_ptr():
            t.copy_(buf.to(t.device))
        return t

    # persistent staging buffer for host-side payloads under the nccl
    # backend: avoids a fresh allocation + H2D/D2H pair per objective
    # evaluation (the (1+p) C1 allreduce fires once per L-BFGS eval)
    _np_stage: Optional[torch.Tensor] = None

    def allreduce_np(self, arr: np.ndarray, op: str = "sum") -> np.ndarray:
        if not self._active:
            return arr
        t = torch.from_numpy(np.ascontiguousarray(arr))
        if dist.get_backend() == "nccl":
            cls = type(self)
            if (cls._np_stage is None or cls._np_stage.numel() < t.numel()
                    or cls._np_stage.dtype != t.dtype):
                cls._np_stage = torch.empty(t.numel(), dtype=t.dtype,
                                            device="cuda")
            buf = cls._np_stage[:t.numel()].view_as(t)
            buf.copy_(t, non_blocking=False)
            red = {"sum": dist.ReduceOp.SUM, "min": dist.ReduceOp.MIN,
                   "max": dist.ReduceOp.MAX}[op]
            dist.all_reduce(buf, op=red)
            t.copy_(buf.cpu())
            return t.numpy()
        self.allreduce_(t, op)
        return t.numpy()

    def allreduce_scalar(self, x: float, op: str = "sum") -> float:
        return float(self.allreduce_np(np.array([x], dtype=np.float64), op)[0])

    def barrier(self):
        if self._active:
            dist.barrier()

    # ----- sharding helpers ---------------------------------------------
    def global_extent(self, n_local: int) -> Tuple[int, int]:
        """(N_global, my global row offset) for contiguous rank shards."""
        if not self._active:
            return n_local, 0
        counts = torch.zeros(self.world_size, dtype=torch.int64)
        counts[self.rank] = n_local
        self.allreduce_(counts)
        offset = int(counts[: self.rank].sum())
        return int(counts.sum()), offset

    def sample_rows(self, X_local: torch.Tensor, m: int, seed: int,
                    y_local: Optional[torch.Tensor] = None):
        """Uniform global sample of m rows without replacement (C3).

        Every rank draws the SAME m global indices from a seeded RNG, fills
        the rows it owns into a zero buffer, and an allreduce(SUM) over the
        disjoint fills assembles the sample on all ranks.  Replaces
        ``RDD.takeSample`` (``commons/ActiveSetProvider.scala:55``)."""
        n_local, d = X_local.shape
        N, off = self.global_extent(n_local)
        if m > N:
            raise ValueError(f"cannot sample {m} rows from {N}")
        rng = np.random.default_rng(seed)
        idx = rng.choice(N, size=m, replace=False)
        buf = torch.zeros(m, d, dtype=X_local.dtype, device=X_local.device)
        ybuf = (torch.zeros(m, dtype=y_local.dtype, device=y_local.device)
                if y_local is not None else None)
        local_mask = (idx >= off) & (idx < off + n_local)
        pos = np.nonzero(local_mask)[0]
        if pos.size:
            src = torch.as_tensor(idx[pos] - off, device=X_local.device)
            buf[torch.as_tensor(pos, device=X_local.device)] = X_local[src]
            if ybuf is not None:
                ybuf[torch.as_tensor(pos, device=X_local.device)] = y_local[src]
        self.allreduce_(buf)
        if ybuf is not None:
            self.allreduce_(ybuf)
            return (buf, ybuf)
        return buf


def get_comm() -> Comm:
    return Comm()


def init_from_env(device: Optional[torch.device] = None) -> Comm:
    """Initialize torch.distributed from torchrun env vars if present.

    Uses the nccl (=RCCL on ROCm) backend when the target device is CUDA,
    gloo otherwise.  Safe to call when WORLD_SIZE is absent (no-op).

    Failure semantics (SURVEY.md §5 failure-detection row): fail fast, no
    elastic recovery — a rank that desyncs or dies must take the job down
    instead of hanging the collective forever.  Concretely:

    * every collective carries a timeout (default 600 s, override with
      ``SPARK_GP_AMD_COMM_TIMEOUT_S``) enforced by the RCCL watchdog thread;
    * ``TORCH_NCCL_ASYNC_ERROR_HANDLING=1`` (set here unless the user chose
      a value) makes the watchdog ABORT the process on a timed-out or
      errored collective rather than logging and hanging;
    * recovery is restart-the-job, mirroring the reference's
      delegate-to-Spark stance (there is no partial-world continue: the BCM
      objective is a fixed-order sum over all ranks)."""
    import datetime
    import os
    if dist is None or dist.is_initialized():
        return Comm()
    if "WORLD_SIZE" not in os.environ or int(os.environ["WORLD_SIZE"]) <= 1:
        return Comm()
    use_cuda = (device is not None and device.type == "cuda") or \
        (device is None and torch.cuda.is_available())
    backend = "nccl" if use_cuda else "gloo"
    if use_cuda:
        os.environ.setdefault("TORCH_NCCL_ASYNC_ERROR_HANDLING", "1")
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
    timeout = datetime.timedelta(
        seconds=float(os.environ.get("SPARK_GP_AMD_COMM_TIMEOUT_S", "600")))
    dist.init_process_group(backend=backend, timeout=timeout)
    return Comm()
