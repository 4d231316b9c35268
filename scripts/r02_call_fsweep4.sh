#!/bin/bash
# Validate race fixes: full suite + soak + re-sweep the failed configs
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 700 python -m pytest tests -x -q -m gpu > gpurun_out/fw4_pytest.log 2>&1
echo "pytest rc=$?" >> gpurun_out/fw4_pytest.log
run() {
  BENCH_LOG2_CAP=$1 BENCH_WM_FUSE=$2 ARROYO_AMD_MF_RANGE=$3 timeout 240 \
    python bench.py --skip-cpu-baseline --steps 400 --warmup 120 \
    > gpurun_out/fw4_c$1_f$2_r$3.json 2> gpurun_out/fw4_c$1_f$2_r$3.err
}
run 19 4 1024
run 20 4 1024
run 19 6 1024
run 19 8 1024
run 18 4 512
# soak the default config for sustained-rate + stability evidence
timeout 280 python bench.py --skip-cpu-baseline --steps 40000 --warmup 200 > gpurun_out/fw4_soak.json 2> gpurun_out/fw4_soak.err
tail -n 3 gpurun_out/fw4_pytest.log
for f in gpurun_out/fw4_*.json; do echo "== $f"; grep -o '"value": [0-9.]*\|"timed_s": [0-9.]*' $f | head -2; done
for f in gpurun_out/fw4_*.err; do e=$(tail -n 1 $f); [ -n "$e" ] && echo "ERR $f: $e"; done
true
