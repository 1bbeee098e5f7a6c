#!/bin/bash
# Round-1 GPU validation: full gpu test suite, native-vs-eager bench A/B,
# and a rocprofv3 kernel-stats profile of the flagship step.
set -x
mkdir -p gpurun_out

echo "=== pytest -m gpu ==="
python -m pytest tests -m gpu -q --timeout 900 2>&1 | tail -8

echo "=== bench native bf16 +graph ==="
timeout 600 python bench.py --gpus 1 --steps 10 --warmup 3 | tail -2

echo "=== bench native bf16 no-graph ==="
timeout 600 python bench.py --gpus 1 --steps 10 --warmup 3 --graph 0 | tail -2

echo "=== bench eager-fallback bf16 (stock PyTorch ops) ==="
SEIST_AMD_ALLOW_FALLBACK=1 timeout 600 python bench.py --gpus 1 --steps 10 --warmup 3 --graph 0 | tail -2

echo "=== bench eager-fallback fp32 (reference numerics) ==="
SEIST_AMD_ALLOW_FALLBACK=1 timeout 600 python bench.py --gpus 1 --steps 10 --warmup 3 --graph 0 --dtype fp32 | tail -2

echo "=== rocprof kernel stats (native, no graph) ==="
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 600 rocprofv3 --kernel-trace --stats -d gpurun_out/prof -o bench_native \
  -- python bench.py --gpus 1 --steps 3 --warmup 2 --graph 0 > gpurun_out/prof_native.log 2>&1
tail -40 $(ls gpurun_out/prof/*bench_native*stats* 2>/dev/null | head -1) 2>/dev/null || \
  grep -A40 "KernelName\|NAME" gpurun_out/prof_native.log | head -50 || tail -20 gpurun_out/prof_native.log
