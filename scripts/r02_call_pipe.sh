#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 500 python -m pytest tests/test_gpu_parity.py tests/test_property_large.py -x -q -m gpu > gpurun_out/pp_tests.log 2>&1
echo "rc=$?" >> gpurun_out/pp_tests.log
timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/pp_on.json 2> gpurun_out/pp_on.err
BENCH_PIPELINE=0 timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/pp_off.json 2> gpurun_out/pp_off.err
ARROYO_AMD_SPIN=0 timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/pp_nospin.json 2> gpurun_out/pp_nospin.err
timeout 280 python bench.py --skip-cpu-baseline --steps 40000 --warmup 200 > gpurun_out/pp_soak.json 2> gpurun_out/pp_soak.err
tail -n 2 gpurun_out/pp_tests.log
for f in gpurun_out/pp_*.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
for f in gpurun_out/pp_*.err; do e=$(tail -n 1 $f); case "$e" in *Error*) echo "ERR $f: $e";; esac; done
true
