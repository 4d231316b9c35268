#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 700 python -m pytest tests -x -q -m gpu > gpurun_out/du_pytest.log 2>&1
echo "rc=$?" >> gpurun_out/du_pytest.log
timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/du_a.json 2> gpurun_out/du_a.err
timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/du_b.json 2> gpurun_out/du_b.err
timeout 300 env BENCH_MIN_TIMED_S=10 python bench.py --skip-cpu-baseline > gpurun_out/du_soak.json 2> gpurun_out/du_soak.err
tail -n 2 gpurun_out/du_pytest.log
for f in gpurun_out/du_*.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
for f in gpurun_out/du_*.err; do e=$(tail -n 1 $f); case "$e" in *Error*) echo "ERR $f: $e";; esac; done
true
