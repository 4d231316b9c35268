#!/bin/bash
# Round-2 call 2: split-layout wave-combine kernel — parity + sweep vs lds.
set -x
cd /root/repo
mkdir -p gpurun_out

timeout 240 python -m pytest tests/test_gpu_parity.py tests/test_property_large.py -x -q -m gpu \
  > gpurun_out/r02b_pytest.log 2>&1
echo "pytest rc=$?"; tail -3 gpurun_out/r02b_pytest.log

sweep() {
  name=$1; shift
  env "$@" timeout 120 python bench.py --skip-cpu-baseline --steps 300 --warmup 100 \
    > gpurun_out/r02b_${name}.json 2> gpurun_out/r02b_${name}.err
  python - "$name" <<'EOF'
import json,sys
try:
  d=json.load(open(f"gpurun_out/r02b_{sys.argv[1]}.json"))
  r=d["roofline"]
  print(f"== {sys.argv[1]}: value={d['value']/1e9:.2f}G launch={r['avg_launch_us']:.1f}us rows/launch={r['rows_per_launch']/1e6:.2f}M frac={r['frac']:.4f}")
except Exception as e:
  print(f"== {sys.argv[1]}: FAILED {e}")
EOF
}
sweep split_pb512       ARROYO_AMD_UPD=split ARROYO_AMD_PBLOCKS=512
sweep split_pb1024      ARROYO_AMD_UPD=split ARROYO_AMD_PBLOCKS=1024
sweep split_pb2048      ARROYO_AMD_UPD=split ARROYO_AMD_PBLOCKS=2048
sweep split_pb1024_wf8  ARROYO_AMD_UPD=split ARROYO_AMD_PBLOCKS=1024 BENCH_WM_FUSE=8
sweep split_pb2048_wf8  ARROYO_AMD_UPD=split ARROYO_AMD_PBLOCKS=2048 BENCH_WM_FUSE=8
sweep lds_wf4           ARROYO_AMD_UPD=lds BENCH_WM_FUSE=4
sweep lds_wf8           ARROYO_AMD_UPD=lds BENCH_WM_FUSE=8
echo DONE
