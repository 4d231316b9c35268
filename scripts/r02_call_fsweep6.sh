#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 400 python -m pytest tests/test_gpu_parity.py tests/test_property_large.py -x -q -m gpu > gpurun_out/fw6_tests.log 2>&1
echo "rc=$?" >> gpurun_out/fw6_tests.log
run() {
  BENCH_LOG2_CAP=$1 BENCH_WM_FUSE=$2 ARROYO_AMD_MF_RANGE=$3 timeout 240 \
    python bench.py --skip-cpu-baseline --steps 400 --warmup 120 \
    > gpurun_out/fw6_c$1_f$2_r$3.json 2> gpurun_out/fw6_c$1_f$2_r$3.err
}
run 19 4 1024
run 19 4 1024
run 20 4 1024
run 19 8 1024
timeout 280 python bench.py --skip-cpu-baseline --steps 40000 --warmup 200 > gpurun_out/fw6_soak.json 2> gpurun_out/fw6_soak.err
tail -n 2 gpurun_out/fw6_tests.log
for f in gpurun_out/fw6_*.json; do echo "== $f"; grep -o '"value": [0-9.]*\|"timed_s": [0-9.]*' $f | head -2; done
true
