#!/bin/bash
# Batched-watermark verification: parity subset + bench A/B + fuse sweep
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
mkdir -p gpurun_out
timeout 420 python -m pytest tests/test_gpu_parity.py tests/test_property_large.py -x -q -m gpu > gpurun_out/wmb_tests.log 2>&1
echo "tests rc=$?" >> gpurun_out/wmb_tests.log
for fuse in 4 8 16; do
  BENCH_WM_FUSE=$fuse timeout 300 python bench.py --steps 400 --warmup 120 > gpurun_out/wmb_fuse${fuse}.json 2> gpurun_out/wmb_fuse${fuse}.err
done
tail -1 gpurun_out/wmb_tests.log
for f in gpurun_out/wmb_fuse*.json; do echo "== $f"; tail -c 600 $f; echo; done
