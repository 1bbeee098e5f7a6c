#!/bin/bash
set -x
mkdir -p gpurun_out

echo "=== pytest -m gpu ==="
python -m pytest tests -m gpu -q --timeout 900 2>&1 | tail -8

echo "=== bench native bf16 +graph ==="
timeout 600 python bench.py --gpus 1 --steps 10 --warmup 3 | tail -2

echo "=== rocprof kernel stats (2 steps) ==="
export TMPDIR=/tmp
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof2 -o bench2 \
  -- python bench.py --gpus 1 --steps 2 --warmup 1 --graph 0 > gpurun_out/prof2.log 2>&1
ls gpurun_out/prof2/ 2>/dev/null
tail -5 gpurun_out/prof2.log
