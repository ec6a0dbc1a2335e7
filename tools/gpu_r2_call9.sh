#!/bin/bash
# PMC counters on the final packed chain kernels (in-bench, mtsac).
# Per gpurun rules: --pmc runs alone (no trace domains).
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
cd /tmp && export TMPDIR=/tmp
mkdir -p "$GRAFT_REPO_ROOT/gpurun_out"

timeout 420 rocprofv3 \
  --pmc GRBM_GUI_ACTIVE GRBM_TA_BUSY SQ_WAVE_CYCLES SQ_WAIT_INST_ANY SQ_INSTS_MFMA \
  --output-format csv -d /tmp/pmc_r20 -o pmc \
  -- python "$GRAFT_REPO_ROOT/bench.py" --steps 200 --warmup 50 \
     --min-timed-seconds 0 --skip-fp32-probe --skip-rollout-probe \
     > /tmp/pmc_bench.log 2>&1
echo "bench rc=$?"
tail -2 /tmp/pmc_bench.log
find /tmp/pmc_r20 -name "*.csv" | head
# aggregate per-kernel counter means for the big kernels
python - <<'EOF'
import csv, glob, collections
f = sorted(glob.glob("/tmp/pmc_r20/**/*counter_collection.csv", recursive=True))
print("files:", f)
if f:
    agg = collections.defaultdict(lambda: collections.defaultdict(float))
    cnt = collections.defaultdict(int)
    for row in csv.DictReader(open(f[-1])):
        k = row["Kernel_Name"].split("(")[0]
        agg[k][row["Counter_Name"]] += float(row["Counter_Value"])
        cnt[(k, row["Counter_Name"])] += 1
    out = open("/root/repo/gpurun_out/r20_pmc_summary.txt", "w")
    for k in sorted(agg, key=lambda k: -agg[k].get("SQ_WAVE_CYCLES", 0)):
        lines = [f"{k}"]
        for c, v in sorted(agg[k].items()):
            n = cnt[(k, c)]
            lines.append(f"    {c}: total {v:.3e}  mean/launch {v/max(n,1):.3e} (n={n})")
        print("\n".join(lines))
        out.write("\n".join(lines) + "\n")
    out.close()
EOF
echo DONE
