"""Real-RCCL validation on MI355X hardware (marked gpu).

The CPU suite exercises the distributed logic over gloo (test_distributed.py);
these tests create actual RCCL communicators so the nccl-backend code path of
``parallel/dist.py`` (C1 objective allreduce, C2 PPA stat allreduce, C3
sampling) has run on real hardware before the driver's 8-GPU scaling bench:

* world_size=1 RCCL group: every collective goes through RCCL self-reduce;
  full fit must match the non-distributed fit bit-for-bit.
* world_size=2 with BOTH ranks on the one leased GPU: a true multi-rank RCCL
  communicator over a single device.  RCCL (unlike Spark) may refuse
  duplicate devices in one communicator; if it does, the test records that
  and skips — the world=1 test above still covers the RCCL code path.
"""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from spark_gp_amd.data import performance_benchmark_data

pytestmark = pytest.mark.gpu


def _shard(X, y, rank, world):
    n = len(y)
    base, rem = divmod(n, world)
    start = rank * base + min(rank, rem)
    stop = start + base + (1 if rank < rem else 0)
    return X[start:stop], y[start:stop]


def _fit_params(gp):
    return (gp.setDatasetSizeForExpert(50).setActiveSetSize(60)
            .setSigma2(1e-3).setMaxIter(10).setSeed(3).setDevice("cuda"))


def _data():
    X, _ = performance_benchmark_data(600, 3, seed=11, dtype=np.float32)
    y = np.sin(X.sum(-1) * 3.0).astype(np.float32)   # non-trivial target
    return X, y


def _reference_fit():
    """Single-process (no process group) fit on the full data."""
    from spark_gp_amd import GaussianProcessRegression
    from spark_gp_amd.kernels import ARDRBFKernel
    X, y = _data()
    gp = _fit_params(GaussianProcessRegression()
                     .setKernel(lambda: 1 * ARDRBFKernel(3)))
    model = gp.fit(torch.tensor(X, device="cuda"),
                   torch.tensor(y, device="cuda"))
    Xq = torch.tensor(X[:64], device="cuda")
    return model.predict(Xq)


def _rccl_worker(rank, world, port, out_q):
    import traceback
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist
        torch.cuda.set_device(0)      # both ranks share the one leased GPU
        dist.init_process_group("nccl", rank=rank, world_size=world)
        from spark_gp_amd import GaussianProcessRegression
        from spark_gp_amd.kernels import ARDRBFKernel
        from spark_gp_amd.parallel.dist import get_comm
        comm = get_comm()
        # C1-shaped smoke: allreduce a (1+p) fp64 payload over RCCL
        buf = comm.allreduce_np(np.arange(4, dtype=np.float64) + rank)
        X, _ = performance_benchmark_data(600, 3, seed=11, dtype=np.float32)
        y = np.sin(X.sum(-1) * 3.0).astype(np.float32)
        Xl, yl = _shard(X, y, rank, world)
        gp = _fit_params(GaussianProcessRegression()
                         .setKernel(lambda: 1 * ARDRBFKernel(3)))
        model = gp.fit(torch.tensor(Xl, device="cuda"),
                       torch.tensor(yl, device="cuda"))
        pred = model.predict(torch.tensor(X[:64], device="cuda"))
        dist.destroy_process_group()
        out_q.put((rank, "ok", (buf, pred)))
    except Exception as e:
        out_q.put((rank, "error", f"{type(e).__name__}: {e}\n"
                   + traceback.format_exc()))


def _spawn_rccl(world, port):
    ctx = mp.get_context("spawn")
    out_q = ctx.Queue()
    procs = [ctx.Process(target=_rccl_worker, args=(r, world, port, out_q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    try:
        for _ in range(world):
            rank, status, payload = out_q.get(timeout=240)
            results[rank] = (status, payload)
    finally:
        for p in procs:
            p.join(timeout=60)
            if p.is_alive():
                p.terminate()
    return results


def test_rccl_world1_matches_plain_fit():
    """world_size=1 nccl group: collectives run through a real RCCL
    communicator; fit must agree with the group-less fit."""
    ref = _reference_fit()
    results = _spawn_rccl(1, 29711)
    status, payload = results[0]
    assert status == "ok", payload
    buf, pred = payload
    np.testing.assert_allclose(buf, np.arange(4, dtype=np.float64))
    np.testing.assert_allclose(pred, ref, rtol=1e-4, atol=1e-4)


def test_rccl_two_ranks_one_gpu():
    """First multi-rank RCCL communicator: 2 ranks sharing cuda:0, tiny
    distributed fit.  The replicated optimizer + allreduced stats must give
    the identical model on both ranks (expert grouping differs from the
    1-process run, so cross-rank identity + fit quality is the check, as in
    the gloo suite)."""
    results = _spawn_rccl(2, 29713)
    errors = {r: p for r, (s, p) in results.items() if s != "ok"}
    if errors:
        msg = "; ".join(str(p).splitlines()[0] for p in errors.values())
        if "uplicate" in msg or "invalid" in msg.lower():
            pytest.skip(f"RCCL refuses 2 ranks on one device: {msg}")
        raise AssertionError(f"rank failures: {errors}")
    buf0, pred0 = results[0][1]
    buf1, pred1 = results[1][1]
    # sum over ranks of (arange + rank) = 2*arange + 1
    np.testing.assert_allclose(buf0, 2.0 * np.arange(4, dtype=np.float64) + 1.0)
    np.testing.assert_allclose(buf1, buf0, rtol=0)
    np.testing.assert_allclose(pred0, pred1, rtol=1e-6, atol=1e-6)
    X, y = _data()
    rmse = float(np.sqrt(np.mean((pred0 - y[:64]) ** 2)))
    assert rmse < 0.2, rmse
