#!/bin/bash
# refine around c19/f4/r1024 + validate new defaults with the full suite
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 700 python -m pytest tests -x -q -m gpu > gpurun_out/fw3_pytest.log 2>&1
echo "pytest rc=$?" >> gpurun_out/fw3_pytest.log
run() {
  BENCH_LOG2_CAP=$1 BENCH_WM_FUSE=$2 ARROYO_AMD_MF_RANGE=$3 timeout 240 \
    python bench.py --skip-cpu-baseline --steps 400 --warmup 120 \
    > gpurun_out/fw3_c$1_f$2_r$3.json 2> gpurun_out/fw3_c$1_f$2_r$3.err
}
run 19 4 1024
run 20 4 1024
run 19 4 2048
run 20 4 2048
run 19 6 1024
tail -n 3 gpurun_out/fw3_pytest.log
for f in gpurun_out/fw3_*.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
