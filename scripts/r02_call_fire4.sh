#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 400 python -m pytest tests/test_gpu_parity.py tests/test_property_large.py -x -q -m gpu > gpurun_out/f4_tests.log 2>&1
echo "rc=$?" >> gpurun_out/f4_tests.log
run() {
  ARROYO_AMD_UPD=$1 ARROYO_AMD_BQ=$2 timeout 240 \
    python bench.py --skip-cpu-baseline --steps 400 --warmup 120 \
    > gpurun_out/f4_u$1_q$2.json 2> gpurun_out/f4_u$1_q$2.err
}
run batch 8
run batch 16
run batch 4
run lds 8
timeout 280 python bench.py --skip-cpu-baseline --steps 40000 --warmup 200 > gpurun_out/f4_soak.json 2> gpurun_out/f4_soak.err
tail -n 2 gpurun_out/f4_tests.log
for f in gpurun_out/f4_*.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
true
