#!/bin/bash
# Post-CPI merge-range sweep: dense entry reads are latency-bound at 512
# WGs (2/CU); smaller home ranges raise grid-level occupancy.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
run() {
  ARROYO_AMD_MF_RANGE=$1 ARROYO_AMD_MF_SLOTS=$2 timeout 240 \
    python bench.py --skip-cpu-baseline --steps 400 --warmup 120 \
    > gpurun_out/mf_r$1_s$2.json 2> gpurun_out/mf_r$1_s$2.err
}
run 512 2048
run 256 2048
run 256 1024
run 128 1024
run 64 1024
for f in gpurun_out/mf_r*.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
