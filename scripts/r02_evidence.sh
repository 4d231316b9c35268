#!/bin/bash
# Round-2 final-config evidence: full GPU suite, default bench (with CPU
# baselines + boundary leg), rocprof kernel stats, PMC traffic.
set -x
cd /root/repo
mkdir -p gpurun_out

timeout 700 python -m pytest tests -x -q -m gpu > gpurun_out/r02z_pytest.log 2>&1
echo "pytest rc=$?"; tail -3 gpurun_out/r02z_pytest.log

timeout 300 python bench.py > gpurun_out/r02z_bench_final.json 2> gpurun_out/r02z_bench.err
tail -1 gpurun_out/r02z_bench_final.json

cd /tmp && export TMPDIR=/tmp
PB="python /root/repo/bench.py --skip-cpu-baseline --steps 200 --warmup 50"
timeout 240 rocprofv3 --kernel-trace --stats --output-format csv \
  -d /root/repo/gpurun_out/r02z_prof -o r02z -- $PB \
  > /root/repo/gpurun_out/r02z_prof_bench.json 2> /root/repo/gpurun_out/r02z_prof.err
timeout 240 rocprofv3 --pmc FETCH_SIZE --output-format csv \
  -d /root/repo/gpurun_out/r02z_fetch -o fetch -- $PB > /dev/null 2>&1
timeout 240 rocprofv3 --pmc WRITE_SIZE,TCC_HIT_sum,TCC_MISS_sum --output-format csv \
  -d /root/repo/gpurun_out/r02z_write -o write -- $PB > /dev/null 2>&1
ls /root/repo/gpurun_out/r02z_prof/ /root/repo/gpurun_out/r02z_fetch/ 2>/dev/null

python - <<'EOF'
import csv, glob, collections
# kernel stats summary
fs = glob.glob("/root/repo/gpurun_out/r02z_prof/**/*kernel_stats.csv", recursive=True) or \
     glob.glob("/root/repo/gpurun_out/r02z_prof/**/*stats*.csv", recursive=True)
print("stats files:", fs)
for f in fs[:1]:
    for row in list(csv.DictReader(open(f)))[:10]:
        print(row)
for tag in ("fetch", "write"):
    files = glob.glob(f"/root/repo/gpurun_out/r02z_{tag}/**/*counter_collection.csv", recursive=True)
    if not files:
        print(f"{tag}: none"); continue
    agg = collections.defaultdict(lambda: collections.defaultdict(float)); n = collections.Counter()
    for f in files:
        for row in csv.DictReader(open(f)):
            k = row.get("Kernel_Name", "?")[:46]
            agg[k][row["Counter_Name"]] += float(row["Counter_Value"]); n[(k, row["Counter_Name"])] += 1
    print(f"=== {tag}")
    for k, c in sorted(agg.items()):
        if "update" not in k and "merge" not in k and "retire" not in k: continue
        print(" ", k, {cn: f"{v/max(n[(k,cn)],1):.3e}x{n[(k,cn)]}" for cn, v in c.items()})
EOF
echo DONE
# genuine sustained soak (MIN_TIMED_S stretches the timed region)
BENCH_MIN_TIMED_S=30 timeout 300 python /root/repo/bench.py --skip-cpu-baseline > /root/repo/gpurun_out/r02z_soak30.json 2>/dev/null
tail -c 400 /root/repo/gpurun_out/r02z_soak30.json
