#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
run() {
  ARROYO_AMD_FIRE_LAG=$1 ARROYO_AMD_MF_RANGE=$2 timeout 240 \
    python bench.py --skip-cpu-baseline --steps 400 --warmup 120 \
    > gpurun_out/lg_l$1_r$2.json 2> gpurun_out/lg_l$1_r$2.err
}
run 1 1024
run 2 1024
run 3 1024
run 2 512
run 3 512
timeout 400 python -m pytest tests/test_gpu_parity.py tests/test_property_large.py -x -q -m gpu > gpurun_out/lg_tests.log 2>&1
echo "rc=$?" >> gpurun_out/lg_tests.log
tail -n 2 gpurun_out/lg_tests.log
for f in gpurun_out/lg_*.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
for f in gpurun_out/lg_*.err; do e=$(tail -n 1 $f); case "$e" in *Error*) echo "ERR $f: $e";; esac; done
true
