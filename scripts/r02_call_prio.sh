#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 400 python -m pytest tests/test_gpu_parity.py tests/test_property_large.py -x -q -m gpu > gpurun_out/pr_tests.log 2>&1
echo "rc=$?" >> gpurun_out/pr_tests.log
timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/pr_a.json 2> gpurun_out/pr_a.err
timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/pr_b.json 2> gpurun_out/pr_b.err
timeout 280 python bench.py --skip-cpu-baseline --steps 40000 --warmup 200 > gpurun_out/pr_soak.json 2> gpurun_out/pr_soak.err
tail -n 2 gpurun_out/pr_tests.log
for f in gpurun_out/pr_*.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
true
