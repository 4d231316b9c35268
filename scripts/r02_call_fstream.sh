#!/bin/bash
# Dual-stream fire path: parity + soak + bench A/B vs serialized mode
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 500 python -m pytest tests/test_gpu_parity.py tests/test_property_large.py tests/test_multikey.py -x -q -m gpu > gpurun_out/fs_tests.log 2>&1
echo "tests rc=$?" >> gpurun_out/fs_tests.log
# repeat the randomized parity tests to catch rare interleavings
for i in 1 2 3; do
  timeout 300 python -m pytest tests/test_gpu_parity.py -x -q -m gpu >> gpurun_out/fs_tests.log 2>&1
  echo "rep$i rc=$?" >> gpurun_out/fs_tests.log
done
timeout 300 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/fs_on.json 2> gpurun_out/fs_on.err
ARROYO_AMD_FIRE_STREAM=0 timeout 300 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/fs_off.json 2> gpurun_out/fs_off.err
tail -6 gpurun_out/fs_tests.log
for f in gpurun_out/fs_on.json gpurun_out/fs_off.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
