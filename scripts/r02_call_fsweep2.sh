#!/bin/bash
# With the fire path overlapped, re-sweep pane-table capacity and wm fusion
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
run() {
  BENCH_LOG2_CAP=$1 BENCH_WM_FUSE=$2 ARROYO_AMD_MF_RANGE=$3 timeout 240 \
    python bench.py --skip-cpu-baseline --steps 400 --warmup 120 \
    > gpurun_out/fw_c$1_f$2_r$3.json 2> gpurun_out/fw_c$1_f$2_r$3.err
}
run 18 4 512
run 19 4 512
run 19 4 1024
run 18 8 512
run 19 8 1024
run 17 4 512
for f in gpurun_out/fw_*.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
