#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 400 python -m pytest tests/test_gpu_parity.py tests/test_property_large.py -x -q -m gpu > gpurun_out/f5_tests.log 2>&1
echo "rc=$?" >> gpurun_out/f5_tests.log
run() {
  ARROYO_AMD_MF_RANGE=$1 ARROYO_AMD_MF_SLOTS=$2 timeout 240 \
    python bench.py --skip-cpu-baseline --steps 400 --warmup 120 \
    > gpurun_out/f5_r$1_s$2.json 2> gpurun_out/f5_r$1_s$2.err
}
run 1024 2048
run 512 2048
run 512 1024
run 256 1024
tail -n 2 gpurun_out/f5_tests.log
for f in gpurun_out/f5_*.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
for f in gpurun_out/f5_*.err; do e=$(tail -n 1 $f); case "$e" in *Error*) echo "ERR $f: $e";; esac; done
true
