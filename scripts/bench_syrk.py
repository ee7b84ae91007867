"""SYRK microbenchmark: numerics vs fp64 and sustained throughput.

Usage (GPU box):  python scripts/bench_syrk.py [rows] [m]
Prints achieved bf16 TFLOP/s (counting all 3 hi/lo MFMA products) and the
max relative error of KK against a torch fp64 reference.
"""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch

from spark_gp_amd import _hip_ext as ext  # fail loudly if missing

rows = int(sys.argv[1]) if len(sys.argv) > 1 else 131072
m = int(sys.argv[2]) if len(sys.argv) > 2 else 1000

torch.manual_seed(0)
dev = "cuda"
# hi/lo split of a synthetic fp32 kernel block in (0, 1]
V = torch.rand(rows, m, device=dev, dtype=torch.float32)
Kc = V.to(torch.bfloat16)
Kl = (V - Kc.float()).to(torch.bfloat16)
KcT = Kc.T.contiguous()              # SYRK takes transposed operands [m, c]
KlT = Kl.T.contiguous()

ntile = (m + 255) // 256
tiles = ntile * (ntile + 1) // 2
split_k = max(1, min(64, 256 // tiles))
print(f"rows={rows} m={m} split_k={split_k}")

# numerics: KK vs fp64 of the REPRESENTED value (hi+lo), which is what the
# 3-product SYRK computes (lo x lo term ~2^-32, below fp32 resolution)
KK = torch.zeros(m, m, device=dev, dtype=torch.float32)
ext.syrk_bf16_acc(KcT, KlT, KK, split_k)
torch.cuda.synchronize()
Vr = Kc.double() + Kl.double()
ref = Vr.T @ Vr
err = (KK.double() - ref).abs().max().item()
rel = err / ref.abs().max().item()
print(f"max abs err {err:.3e}  rel {rel:.3e}")
assert rel < 3e-5, "SYRK numerics out of tolerance"

# throughput (sustained; caller should pin clocks)
flops = 3 * 2.0 * rows * m * m          # 3 MFMA products
for _ in range(3):
    ext.syrk_bf16_acc(KcT, KlT, KK, split_k)
torch.cuda.synchronize()
reps = 20
t0 = time.perf_counter()
for _ in range(reps):
    ext.syrk_bf16_acc(KcT, KlT, KK, split_k)
torch.cuda.synchronize()
dt = (time.perf_counter() - t0) / reps
print(f"{dt * 1e3:.3f} ms/call  ->  {flops / dt / 1e12:.1f} TF bf16 "
      f"({flops / 3 / dt / 1e12:.1f} TF fp32-equivalent)")

# optional split_k sweep: shorter per-block k-ranges shrink the live
# column-window working set toward L3/L2 residency (the round-1 TCC
# analysis: 38% hit, ~6x compulsory traffic from drifting tiles) at the
# price of split_k-fold output atomic traffic — measure the trade.
if "--sweep" in sys.argv:
    for sk in (split_k, 8, 16, 32, 64, 128, 256):
        for _ in range(2):
            ext.syrk_bf16_acc(KcT, KlT, KK, sk)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            ext.syrk_bf16_acc(KcT, KlT, KK, sk)
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 10
        print(f"  split_k={sk:4d}: {dt * 1e3:8.3f} ms  "
              f"{flops / dt / 1e12:6.1f} TF bf16")

# A/B the k-synchronized persistent path (m >= 4096)
if "--sync" in sys.argv and m >= 4096:
    import os as _os
    from spark_gp_amd.ops import hip_backend as hb
    launches = hb._syrk_sync_tiles(m, "cuda")
    print(f"sync patch={_os.environ.get('SPARK_GP_AMD_SYRK_PATCH','2x4')}: "
          f"{len(launches)} launches, tiles {[n for _, n in launches]}")

    for kpb in (128,):
        def run_sync():
            for tt, nact in launches:
                ext.syrk_bf16_sync_acc(KcT, KlT, KK, tt, kpb, nact)
        for _ in range(2):
            run_sync()
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(8):
            run_sync()
        torch.cuda.synchronize()
        dt = (time.perf_counter() - t0) / 8
        print(f"sync kpb={kpb:3d}: {dt * 1e3:8.3f} ms/call  ->  "
              f"{flops / dt / 1e12:6.1f} TF bf16")
