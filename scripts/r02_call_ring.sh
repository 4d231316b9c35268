#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 400 python -m pytest tests/test_gpu_parity.py tests/test_property_large.py -x -q -m gpu > gpurun_out/rg_tests.log 2>&1
echo "rc=$?" >> gpurun_out/rg_tests.log
run() {
  BENCH_RING_PANES=$1 BENCH_WM_FUSE=$2 timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/rg_r$1_f$2.json 2>&1
}
run 16 4
run 24 6
run 32 8
run 32 4
tail -n 2 gpurun_out/rg_tests.log
for f in gpurun_out/rg_*.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
true
