#!/bin/bash
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
run() {
  ARROYO_AMD_PBLOCKS=$1 timeout 240 python bench.py --skip-cpu-baseline --steps 400 --warmup 120 \
    > gpurun_out/f6_p$1.json 2> gpurun_out/f6_p$1.err
}
run 1024
run 2048
run 4096
run 3072
cd /tmp && export TMPDIR=/tmp
timeout 240 rocprofv3 --kernel-trace --stats --output-format csv -d /root/repo/gpurun_out/f6_prof -o f6 -- python /root/repo/bench.py --skip-cpu-baseline --steps 300 --warmup 80 > /root/repo/gpurun_out/f6_prof.json 2>/dev/null
for f in /root/repo/gpurun_out/f6_p*.json; do echo "== $f"; grep -o '"value": [0-9.]*' $f | head -1; done
head -8 /root/repo/gpurun_out/f6_prof/f6_kernel_stats.csv
true
