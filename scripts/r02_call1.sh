#!/bin/bash
# Round-2 call 1: validate the packed update path (parity), then measure it.
# Run via gpurun from the repo root; outputs under gpurun_out/.
set -x
mkdir -p gpurun_out
cd /root/repo

# 1. GPU parity suite (packed path is default for COUNT shapes)
timeout 420 python -m pytest tests -m gpu -x -q > gpurun_out/r02a_pytest.log 2>&1
echo "pytest rc=$?" | tee -a gpurun_out/r02a_pytest.log
tail -5 gpurun_out/r02a_pytest.log

# 2. Default bench (full: cpu_baseline 1+N cores, host boundary leg)
timeout 300 python bench.py > gpurun_out/r02a_bench_default.json 2> gpurun_out/r02a_bench_default.err
tail -2 gpurun_out/r02a_bench_default.json

# 3. Sweep: packed grid size, watermark fusion, and the round-1 baseline
sweep() {
  name=$1; shift
  env "$@" timeout 120 python bench.py --skip-cpu-baseline --steps 300 --warmup 100 \
    > gpurun_out/r02a_sweep_${name}.json 2> gpurun_out/r02a_sweep_${name}.err
  echo "== ${name}: $(cat gpurun_out/r02a_sweep_${name}.json)"
}
sweep pb512   ARROYO_AMD_PBLOCKS=512
sweep pb1024  ARROYO_AMD_PBLOCKS=1024
sweep pb2048  ARROYO_AMD_PBLOCKS=2048
sweep pb4096  ARROYO_AMD_PBLOCKS=4096
sweep pb1024_wf1 ARROYO_AMD_PBLOCKS=1024 BENCH_WM_FUSE=1
sweep pb1024_wf4 ARROYO_AMD_PBLOCKS=1024 BENCH_WM_FUSE=4
sweep pb1024_wf8 ARROYO_AMD_PBLOCKS=1024 BENCH_WM_FUSE=8
sweep unpacked ARROYO_AMD_PACKED=0

# 4. rocprofv3 kernel stats for the packed kernel (default knobs)
cd /tmp && export TMPDIR=/tmp
timeout 240 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/r02a_prof -o r02a -- \
  python /root/repo/bench.py --skip-cpu-baseline --steps 200 --warmup 50 \
  > /root/repo/gpurun_out/r02a_prof_bench.json 2> /root/repo/gpurun_out/r02a_prof.err
ls /root/repo/gpurun_out/r02a_prof* 2>/dev/null
grep -l stats /root/repo/gpurun_out/r02a_prof/* 2>/dev/null | head
echo DONE
