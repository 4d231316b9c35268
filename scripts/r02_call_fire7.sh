#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 700 python -m pytest tests -x -q -m gpu > gpurun_out/f7_pytest.log 2>&1
echo "rc=$?" >> gpurun_out/f7_pytest.log
timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/f7_on.json 2> gpurun_out/f7_on.err
ARROYO_AMD_CPI_RANGE=128 timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/f7_cr128.json 2> gpurun_out/f7_cr128.err
ARROYO_AMD_CPI_RANGE=512 timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 > gpurun_out/f7_cr512.json 2> gpurun_out/f7_cr512.err
timeout 280 python bench.py --skip-cpu-baseline --steps 40000 --warmup 200 > gpurun_out/f7_soak.json 2> gpurun_out/f7_soak.err
cd /tmp && export TMPDIR=/tmp
timeout 240 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/f7_prof -o f7 -- python /root/repo/bench.py --skip-cpu-baseline --steps 300 --warmup 80 > /root/repo/gpurun_out/f7_prof.json 2>/dev/null
tail -n 2 /root/repo/gpurun_out/f7_pytest.log
for f in /root/repo/gpurun_out/f7_on.json /root/repo/gpurun_out/f7_cr128.json /root/repo/gpurun_out/f7_cr512.json /root/repo/gpurun_out/f7_soak.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
head -7 /root/repo/gpurun_out/f7_prof/f7_kernel_stats.csv
true
