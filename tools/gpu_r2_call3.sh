#!/bin/bash
# Round-2 GPU call 3: rocprofv3 kernel stats, copying back ONLY the stats
# CSVs (kernel traces stay on the box — they blew the 64 MiB merge cap).
cd /root/repo
rm -rf gpurun_out/prof_* 2>/dev/null
mkdir -p gpurun_out
export TMPDIR=/tmp
cd /tmp
{
  for cfg in "mtsac:" "care:--config care" "dp1:--force-ddp"; do
    name="${cfg%%:*}"; extra="${cfg#*:}"
    echo "== rocprof $name =="
    timeout 400 rocprofv3 --kernel-trace --stats --output-format csv -d /tmp/prof_$name -- \
      python /root/repo/bench.py --steps 200 --warmup 30 \
        --min-timed-seconds 0.1 --max-windows 2 \
        --skip-rollout-probe --skip-fp32-probe $extra \
        > /root/repo/gpurun_out/prof_$name.log 2>&1
    echo "RC=$?"; grep -m1 value /root/repo/gpurun_out/prof_$name.log | head -c 400; echo
    find /tmp/prof_$name -name "*kernel_stats*" -exec cp {} /root/repo/gpurun_out/${name}_kernel_stats.csv \;
    find /tmp/prof_$name -name "*domain_stats*" -exec cp {} /root/repo/gpurun_out/${name}_domain_stats.csv \;
  done
  ls -la /root/repo/gpurun_out/*.csv
  echo "== ALL DONE =="
} 2>&1 | tee /root/repo/gpurun_out/call3_summary.log
