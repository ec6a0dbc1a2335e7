#!/bin/bash
# Round-2 consolidated evidence run: all variants + async topology + DP.
cd /root/repo
mkdir -p gpurun_out
{
  echo "== suite =="
  timeout 400 python -m pytest tests -m gpu -x -q > gpurun_out/suite4.log 2>&1
  echo "SUITE_RC=$?"; grep -E "passed|failed" gpurun_out/suite4.log | tail -1
  for cfgname in mtsac sac vsac care; do
    echo "== bench $cfgname =="
    timeout 300 python bench.py --config $cfgname --steps 300 --warmup 50 --skip-rollout-probe --skip-fp32-probe 2>/dev/null | head -c 330; echo
  done
  echo "== bench mtsac full (with fp32 + env probe) =="
  timeout 300 python bench.py --steps 300 --warmup 50 2>/dev/null | head -c 500; echo
  echo "== async topology (players->rings->GPU learner, 25 s) =="
  timeout 200 python bench.py --async --async-seconds 25 --players 4 2>/dev/null | head -c 500; echo
  echo "== force-ddp world-1 (segmented DP graphs) =="
  timeout 300 python bench.py --steps 300 --warmup 50 --force-ddp --skip-rollout-probe --skip-fp32-probe 2>/dev/null | head -c 330; echo
  echo "== ALL DONE =="
} 2>&1 | tee gpurun_out/call4_summary.log
