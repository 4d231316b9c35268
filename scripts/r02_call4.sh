#!/bin/bash
# Round-2 call 4: batched-probe kernel parity + sweep (batch vs lds), plus
# fire-path knobs (MF_RANGE) now that accum/retire launches are fused.
set -x
cd /root/repo
mkdir -p gpurun_out

timeout 300 python -m pytest tests/test_gpu_parity.py tests/test_property_large.py tests/test_edge_cases.py -x -q -m gpu \
  > gpurun_out/r02d_pytest.log 2>&1
echo "pytest rc=$?"; tail -3 gpurun_out/r02d_pytest.log

sweep() {
  name=$1; shift
  env "$@" timeout 120 python bench.py --skip-cpu-baseline --steps 300 --warmup 100 \
    > gpurun_out/r02d_${name}.json 2> gpurun_out/r02d_${name}.err
  python - "$name" <<'EOF'
import json,sys
try:
  d=json.load(open(f"gpurun_out/r02d_{sys.argv[1]}.json"))
  r=d["roofline"]
  print(f"== {sys.argv[1]}: value={d['value']/1e9:.2f}G launch={r['avg_launch_us']:.1f}us rows/launch={r['rows_per_launch']/1e6:.2f}M frac={r['frac']:.4f}")
except Exception as e:
  print(f"== {sys.argv[1]}: FAILED {e}")
EOF
}
sweep batch4_pb512      ARROYO_AMD_PBLOCKS=512
sweep batch4_pb1024     ARROYO_AMD_PBLOCKS=1024
sweep batch4_pb2048     ARROYO_AMD_PBLOCKS=2048
sweep batch2_pb1024     ARROYO_AMD_BQ=2 ARROYO_AMD_PBLOCKS=1024
sweep batch4_pb1024_wf4 ARROYO_AMD_PBLOCKS=1024 BENCH_WM_FUSE=4
sweep batch4_mfr1024    ARROYO_AMD_PBLOCKS=1024 ARROYO_AMD_MF_RANGE=1024 ARROYO_AMD_MF_SLOTS=2048
sweep batch4_mfr512     ARROYO_AMD_PBLOCKS=1024 ARROYO_AMD_MF_RANGE=512 ARROYO_AMD_MF_SLOTS=2048
sweep lds_base          ARROYO_AMD_UPD=lds
sweep lds_mfr1024       ARROYO_AMD_UPD=lds ARROYO_AMD_MF_RANGE=1024 ARROYO_AMD_MF_SLOTS=2048
echo DONE
