#!/bin/bash
# Round-2 call 6: wide-quad (MLP) sweep of the batched kernel.
set -x
cd /root/repo
mkdir -p gpurun_out

timeout 180 python -m pytest tests/test_gpu_parity.py -x -q -m gpu > gpurun_out/r02f_pytest.log 2>&1
echo "pytest rc=$?"; tail -2 gpurun_out/r02f_pytest.log

sweep() {
  name=$1; shift
  env "$@" timeout 120 python bench.py --skip-cpu-baseline --steps 300 --warmup 100 \
    > gpurun_out/r02f_${name}.json 2> gpurun_out/r02f_${name}.err
  python - "$name" <<'EOF'
import json,sys
try:
  d=json.load(open(f"gpurun_out/r02f_{sys.argv[1]}.json"))
  r=d["roofline"]
  print(f"== {sys.argv[1]}: value={d['value']/1e9:.2f}G launch={r['avg_launch_us']:.1f}us rows/launch={r['rows_per_launch']/1e6:.2f}M frac={r['frac']:.4f}")
except Exception as e:
  print(f"== {sys.argv[1]}: FAILED {e}")
EOF
}
sweep q8_pb1024   ARROYO_AMD_BQ=8  ARROYO_AMD_PBLOCKS=1024
sweep q8_pb2048   ARROYO_AMD_BQ=8  ARROYO_AMD_PBLOCKS=2048
sweep q8_pb4096   ARROYO_AMD_BQ=8  ARROYO_AMD_PBLOCKS=4096
sweep q16_pb1024  ARROYO_AMD_BQ=16 ARROYO_AMD_PBLOCKS=1024
sweep q16_pb2048  ARROYO_AMD_BQ=16 ARROYO_AMD_PBLOCKS=2048
sweep q16_pb512   ARROYO_AMD_BQ=16 ARROYO_AMD_PBLOCKS=512
sweep q8_pb2048_wf4 ARROYO_AMD_BQ=8 ARROYO_AMD_PBLOCKS=2048 BENCH_WM_FUSE=4
sweep lds_base    ARROYO_AMD_UPD=lds
echo DONE
