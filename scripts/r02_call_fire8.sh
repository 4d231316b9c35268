#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 500 python -m pytest tests/test_gpu_parity.py tests/test_property_large.py tests/test_edge_cases.py -x -q -m gpu > gpurun_out/f8_tests.log 2>&1
echo "rc=$?" >> gpurun_out/f8_tests.log
timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/f8_on.json 2> gpurun_out/f8_on.err
timeout 280 python bench.py --skip-cpu-baseline --steps 40000 --warmup 200 > gpurun_out/f8_soak.json 2> gpurun_out/f8_soak.err
cd /tmp && export TMPDIR=/tmp
timeout 240 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/f8_prof -o f8 -- python /root/repo/bench.py --skip-cpu-baseline --steps 300 --warmup 80 > /root/repo/gpurun_out/f8_prof.json 2>/dev/null
tail -n 2 /root/repo/gpurun_out/f8_tests.log
for f in /root/repo/gpurun_out/f8_on.json /root/repo/gpurun_out/f8_soak.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
head -8 /root/repo/gpurun_out/f8_prof/f8_kernel_stats.csv
true
