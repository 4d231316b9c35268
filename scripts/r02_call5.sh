#!/bin/bash
# Round-2 call 5: find where the LDS update kernel's 68us actually goes.
# Stage isolation + streaming ceiling + SQ wait/issue counters + TCC traffic.
set -x
cd /root/repo
mkdir -p gpurun_out

timeout 300 python scripts/isolate_update.py > gpurun_out/r02e_isolate.log 2>&1
tail -30 gpurun_out/r02e_isolate.log

cd /tmp && export TMPDIR=/tmp
PB="python /root/repo/bench.py --skip-cpu-baseline --steps 150 --warmup 50"
timeout 240 rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_WAIT_ANY,SQ_WAIT_INST_ANY,SQ_ACTIVE_INST_ANY \
  --output-format csv -d /root/repo/gpurun_out/r02e_sq -o sq -- $PB \
  > /root/repo/gpurun_out/r02e_sq.log 2>&1
timeout 240 rocprofv3 --pmc FETCH_SIZE \
  --output-format csv -d /root/repo/gpurun_out/r02e_fetch -o fetch -- $PB \
  > /root/repo/gpurun_out/r02e_fetch.log 2>&1
timeout 240 rocprofv3 --pmc WRITE_SIZE,TCC_HIT_sum,TCC_MISS_sum \
  --output-format csv -d /root/repo/gpurun_out/r02e_write -o write -- $PB \
  > /root/repo/gpurun_out/r02e_write.log 2>&1
ls /root/repo/gpurun_out/r02e_*/ 2>/dev/null | head

python - <<'EOF'
import csv, glob, collections
for tag in ("sq", "fetch", "write"):
    files = glob.glob(f"/root/repo/gpurun_out/r02e_{tag}/**/*counter_collection.csv", recursive=True)
    if not files:
        print(f"{tag}: no csv"); continue
    agg = collections.defaultdict(lambda: collections.defaultdict(float))
    n = collections.Counter()
    for f in files:
        for row in csv.DictReader(open(f)):
            k = row.get("Kernel_Name", "?")[:50]
            agg[k][row["Counter_Name"]] += float(row["Counter_Value"])
            n[(k, row["Counter_Name"])] += 1
    print(f"=== {tag}")
    for k, c in sorted(agg.items()):
        if "k_update" not in k and "k_merge" not in k and "retire" not in k:
            continue
        parts = []
        for cn, v in sorted(c.items()):
            cnt = n[(k, cn)]
            parts.append(f"{cn}={v/max(cnt,1):.3e}(x{cnt})")
        print(f"  {k}: " + " ".join(parts))
EOF
echo DONE
