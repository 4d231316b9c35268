#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
run() {
  ARROYO_AMD_MF_SLOTS=$1 ARROYO_AMD_MF_RANGE=$2 timeout 240 \
    python bench.py --skip-cpu-baseline --steps 400 --warmup 120 \
    > gpurun_out/fa_s$1_r$2.json 2> gpurun_out/fa_s$1_r$2.err
}
run 2048 1024
run 4096 1024
run 4096 2048
timeout 300 python -m pytest tests/test_gpu_parity.py -x -q -m gpu > gpurun_out/fa_tests.log 2>&1
echo "rc=$?" >> gpurun_out/fa_tests.log
tail -n 2 gpurun_out/fa_tests.log
for f in gpurun_out/fa_*.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
true
