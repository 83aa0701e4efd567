#!/bin/bash
# Round-2 conv-perf profiling. Traces go to /tmp (NOT gpurun_out — the
# copy-back limit is 64 MiB); only CSV summaries are returned.
export TMPDIR=/tmp
export PYTHONPATH=/root/repo
cd /root/repo
mkdir -p gpurun_out
rm -rf /tmp/prof_r2 && mkdir -p /tmp/prof_r2
echo "=== kernel stats (rocprofv3 --kernel-trace --stats) ==="
(cd /tmp && timeout 500 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof_r2 \
  -o r2stats -- python /root/repo/bench.py --steps 40 --warmup 15) \
  > gpurun_out/prof_bench.log 2>&1
echo "rocprof rc=$?"
tail -3 gpurun_out/prof_bench.log
find /tmp/prof_r2 -name "*kernel_stats*" -exec cp {} gpurun_out/r2_kernel_stats.csv \;
echo "--- top kernels ---"
head -14 gpurun_out/r2_kernel_stats.csv 2>/dev/null | cut -c1-140
echo "=== PMC counters ==="
rm -rf /tmp/prof_pmc && mkdir -p /tmp/prof_pmc
(cd /tmp && timeout 500 rocprofv3 --output-format csv --pmc SQ_BUSY_CYCLES SQ_VALU_MFMA_BUSY_CYCLES \
  SQ_LDS_BANK_CONFLICT SQ_INSTS_VALU SQ_INSTS_MFMA -d /tmp/prof_pmc -o r2pmc \
  -- python /root/repo/bench.py --steps 15 --warmup 8 --no-autotune) \
  > gpurun_out/prof_pmc.log 2>&1
echo "pmc rc=$?"
tail -3 gpurun_out/prof_pmc.log
find /tmp/prof_pmc -name "*counter_collection*" -o -name "*counter*csv" | head -3
find /tmp/prof_pmc -name "*counter*" -exec cp {} gpurun_out/r2_pmc_counters.csv \;
python - << 'PYEOF'
import csv, collections
try:
    agg = collections.defaultdict(lambda: collections.defaultdict(float))
    with open("/root/repo/gpurun_out/r2_pmc_counters.csv") as f:
        for row in csv.DictReader(f):
            agg[row["Kernel_Name"][:70]][row["Counter_Name"]] += float(row["Counter_Value"])
    print(f"{'kernel':70} {'mfma/busy':>10} {'conflicts':>10} {'valu':>12}")
    for k, c in sorted(agg.items(), key=lambda kv: -kv[1].get("SQ_BUSY_CYCLES", 0))[:12]:
        busy = c.get("SQ_BUSY_CYCLES", 1)
        print(f"{k:70} {c.get('SQ_VALU_MFMA_BUSY_CYCLES',0)/busy:10.3f} "
              f"{c.get('SQ_LDS_BANK_CONFLICT',0):10.0f} "
              f"{c.get('SQ_INSTS_VALU',0):12.0f}")
except Exception as e:
    print("pmc summary failed:", e)
PYEOF
du -sh gpurun_out 2>/dev/null
echo "=== done ==="
