#!/bin/bash
# Standard GPU validation batch (run via gpurun from the repo root):
#   /usr/local/graft/bin/gpurun --timeout 1200 -- 'bash scripts/ci_gpu.sh'
# Produces logs under gpurun_out/ci/.
set -u
mkdir -p gpurun_out/ci
rocm-smi --setperfdeterminism 2100 > /dev/null 2>&1

echo "== pytest -m gpu =="
timeout 600 python -m pytest tests -m gpu -q 2>&1 | tail -2

echo "== smoke =="
timeout 90 python -c "import __graft_entry__; __graft_entry__.smoke()" 2>&1 | tail -1

echo "== bench (10M default config) =="
timeout 240 python bench.py --steps 3 --warmup 1 2>&1 | grep -E "^\{|stage timings" \
    | tee gpurun_out/ci/bench.json

echo "== kernel profile (2M short run) =="
( cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT" && \
  timeout 150 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/ci/prof -o ci \
      -- python bench.py --rows 2000000 --steps 1 --warmup 1 \
         --min-warmup-seconds 4 > gpurun_out/ci/rocprof.log 2>&1 )
ls gpurun_out/ci/prof 2>/dev/null
