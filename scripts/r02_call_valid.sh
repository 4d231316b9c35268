#!/bin/bash
# Robustness: full suite x2, serial-mode sanity, long soak
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
for i in 1 2; do
  timeout 700 python -m pytest tests -x -q -m gpu >> gpurun_out/vd_pytest.log 2>&1
  echo "run$i rc=$?" >> gpurun_out/vd_pytest.log
done
ARROYO_AMD_FIRE_STREAM=0 timeout 400 python -m pytest tests/test_gpu_parity.py -x -q -m gpu >> gpurun_out/vd_pytest.log 2>&1
echo "serial rc=$?" >> gpurun_out/vd_pytest.log
ARROYO_AMD_FIRE_STREAM=0 timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/vd_serial.json 2> gpurun_out/vd_serial.err
timeout 380 python bench.py --skip-cpu-baseline --steps 120000 --warmup 200 > gpurun_out/vd_soak.json 2> gpurun_out/vd_soak.err
grep "rc=" gpurun_out/vd_pytest.log
for f in gpurun_out/vd_serial.json gpurun_out/vd_soak.json; do echo "== $f"; grep -o '"value": [0-9.]*\|"timed_s": [0-9.]*' $f | head -2; done
true
