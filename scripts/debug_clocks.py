import sys, os, time, threading, subprocess
sys.path.insert(0, "/root/repo")
import numpy as np, torch

def sclk():
    try:
        out = subprocess.run(["rocm-smi", "--showgpuclocks"], capture_output=True,
                             text=True, timeout=10).stdout
        for line in out.splitlines():
            if "sclk" in line:
                return line.split("(")[-1].split(")")[0]
    except Exception as e:
        return str(e)
    return "?"

r = subprocess.run(["rocm-smi", "--setperfdeterminism", "2100"],
                   capture_output=True, text=True)
print("setperfdeterminism rc:", r.returncode, (r.stdout + r.stderr).strip()[-200:])

from spark_gp_amd import _hip_ext as ext
E = 100000
g = torch.Generator().manual_seed(0)
X = torch.rand(E, 100, 32, generator=g).cuda()
y = torch.sin(3*X.sum(-1)).cuda()
sc = torch.rand(32, generator=g).add(0.5).cuda()

stop = False
def sampler():
    while not stop:
        print("  sclk:", sclk(), flush=True)
        time.sleep(1.5)
th = threading.Thread(target=sampler); th.start()

for rep in range(4):
    t0 = time.perf_counter()
    for _ in range(3):
        ext.fused_expert_nll(X, y, sc, 1.0, 1e-3)
    torch.cuda.synchronize()
    print(f"rep {rep}: {(time.perf_counter()-t0)/3*1e3:.2f} ms/launch", flush=True)
stop = True; th.join()
