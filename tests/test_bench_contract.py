"""The driver depends on bench.py's CLI and output contract: one JSON line
from rank 0 with specific fields.  Run the real script on CPU with a tiny
config and validate the line so contract regressions are caught here, not
at round end on a GPU box."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--rows", "6000", "--dim", "4", "--active-set", "60",
         "--steps", "2", "--warmup", "1", "--max-iter", "5",
         "--device", "cpu"],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    rec = json.loads(lines[0])

    # fields the driver and judge read
    assert rec["metric"].startswith("rows/sec")
    assert isinstance(rec["value"], float) and rec["value"] > 0
    assert rec["unit"] == "rows/s"
    assert rec["n_gpus"] == 1
    assert rec["steps"] == 2 and rec["warmup"] == 1
    assert rec["ms_per_step"] > 0
    assert rec["higher_is_better"] is True
    assert rec["scaling"] == "strong"
    assert rec["vs_baseline"] is None      # reference publishes no numbers
    assert rec["data"] == "synthetic"
    assert rec["dtype"] in ("fp32", "fp64")
    cfg = rec["config"]
    assert cfg["rows"] == 6000 and cfg["dim"] == 4
    assert "parallelism" in cfg and cfg["parallelism"].endswith("dp1")
    # value must be whole-job rows/sec consistent with ms_per_step
    assert abs(rec["value"] - 6000 / (rec["ms_per_step"] / 1000.0)) \
        / rec["value"] < 1e-6
