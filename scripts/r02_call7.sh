#!/bin/bash
# Round-2 call 7: radix-regroup vs batch vs lds; parity for the new path.
set -x
cd /root/repo
mkdir -p gpurun_out

ARROYO_AMD_UPD=rdx2 timeout 180 python -m pytest tests/test_gpu_parity.py tests/test_property_large.py -x -q -m gpu > gpurun_out/r02g_pytest.log 2>&1
echo "pytest(rdx2) rc=$?"; tail -2 gpurun_out/r02g_pytest.log

sweep() {
  name=$1; shift
  env "$@" timeout 120 python bench.py --skip-cpu-baseline --steps 300 --warmup 100 \
    > gpurun_out/r02g_${name}.json 2> gpurun_out/r02g_${name}.err
  python - "$name" <<'EOF'
import json,sys
try:
  d=json.load(open(f"gpurun_out/r02g_{sys.argv[1]}.json"))
  r=d["roofline"]
  print(f"== {sys.argv[1]}: value={d['value']/1e9:.2f}G launch={r['avg_launch_us']:.1f}us rows/launch={r['rows_per_launch']/1e6:.2f}M frac={r['frac']:.4f}")
except Exception as e:
  print(f"== {sys.argv[1]}: FAILED {e}")
EOF
}
sweep rdx2        ARROYO_AMD_UPD=rdx2
sweep rdx2_wf4    ARROYO_AMD_UPD=rdx2 BENCH_WM_FUSE=4
sweep rdx2_wf8    ARROYO_AMD_UPD=rdx2 BENCH_WM_FUSE=8
sweep q8_wf4      ARROYO_AMD_BQ=8 ARROYO_AMD_PBLOCKS=2048 BENCH_WM_FUSE=4
sweep lds_base    ARROYO_AMD_UPD=lds
echo DONE
