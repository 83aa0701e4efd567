#!/bin/bash
# Round-2 conv-perf profiling: kernel stats + PMC (separate runs; gpurun
# refuses pmc+trace combinations).
cd /tmp && export TMPDIR=/tmp
cd /root/repo
mkdir -p gpurun_out/prof_r2
echo "=== kernel stats (rocprofv3 --kernel-trace --stats) ==="
timeout 500 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_r2 -o r2stats \
  -- python bench.py --steps 60 --warmup 20 > gpurun_out/prof_bench.log 2>&1
find gpurun_out/prof_r2 -name "*kernel_stats*" -exec cp {} gpurun_out/r2_kernel_stats.csv \;
head -14 gpurun_out/r2_kernel_stats.csv 2>/dev/null | cut -c1-150
echo "=== PMC counters ==="
timeout 500 rocprofv3 --pmc SQ_BUSY_CYCLES SQ_VALU_MFMA_BUSY_CYCLES \
  SQ_LDS_BANK_CONFLICT SQ_WAIT_ANY SQ_INSTS_VALU SQ_INSTS_MFMA \
  -d gpurun_out/prof_r2 -o r2pmc -- python bench.py --steps 20 --warmup 10 \
  --no-autotune > gpurun_out/prof_pmc.log 2>&1 || \
timeout 500 rocprofv3 --pmc SQ_BUSY_CYCLES SQ_VALU_MFMA_BUSY_CYCLES \
  SQ_LDS_BANK_CONFLICT -d gpurun_out/prof_r2 -o r2pmc -- python bench.py \
  --steps 20 --warmup 10 --no-autotune > gpurun_out/prof_pmc.log 2>&1
find gpurun_out/prof_r2 -name "*counter*" -exec cp {} gpurun_out/r2_pmc_counters.csv \;
python - << 'PYEOF'
import csv, collections
try:
    agg = collections.defaultdict(lambda: collections.defaultdict(float))
    with open("gpurun_out/r2_pmc_counters.csv") as f:
        for row in csv.DictReader(f):
            agg[row["Kernel_Name"][:80]][row["Counter_Name"]] += float(row["Counter_Value"])
    print(f"{'kernel':80} {'mfma/busy':>10} {'conflicts':>10}")
    for k, c in sorted(agg.items(), key=lambda kv: -kv[1].get("SQ_BUSY_CYCLES", 0))[:12]:
        busy = c.get("SQ_BUSY_CYCLES", 1)
        print(f"{k:80} {c.get('SQ_VALU_MFMA_BUSY_CYCLES',0)/busy:10.3f} "
              f"{c.get('SQ_LDS_BANK_CONFLICT',0):10.0f}")
except Exception as e:
    print("pmc summary failed:", e)
PYEOF
echo "=== done ==="
