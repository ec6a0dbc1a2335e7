#!/bin/bash
# Round-2 GPU call 2: new GPU tests + fresh rocprofv3 kernel stats
#  for the final MTSAC and CARE paths.
cd /root/repo
mkdir -p gpurun_out
{
  echo "== 1. new GPU tests =="
  timeout 500 python -m pytest tests/test_gpu_kernels.py::test_narrow_bwd_gate_equivalence tests/test_gpu_kernels.py::test_dp_segmented_graphs_world1 -x -q > gpurun_out/newtests.log 2>&1
  echo "NEWTESTS_RC=$?"; tail -5 gpurun_out/newtests.log

  export TMPDIR=/tmp
  cd /tmp
  echo "== 2. rocprof mtsac (graphed) =="
  timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_mtsac -- python /root/repo/bench.py --steps 200 --warmup 30 --min-timed-seconds 0.1 --max-windows 2 --skip-rollout-probe --skip-fp32-probe > /root/repo/gpurun_out/prof_mtsac.log 2>&1
  echo "PROF_MTSAC_RC=$?"; grep -m1 value /root/repo/gpurun_out/prof_mtsac.log

  echo "== 3. rocprof care (graphed, narrow on) =="
  timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_care -- python /root/repo/bench.py --config care --steps 200 --warmup 30 --min-timed-seconds 0.1 --max-windows 2 --skip-rollout-probe --skip-fp32-probe > /root/repo/gpurun_out/prof_care.log 2>&1
  echo "PROF_CARE_RC=$?"; grep -m1 value /root/repo/gpurun_out/prof_care.log

  echo "== 4. rocprof dp world-1 (segmented graphs + RCCL) =="
  timeout 400 rocprofv3 --kernel-trace --stats -d /root/repo/gpurun_out/prof_dp1 -- python /root/repo/bench.py --steps 200 --warmup 30 --min-timed-seconds 0.1 --max-windows 2 --force-ddp --skip-rollout-probe --skip-fp32-probe > /root/repo/gpurun_out/prof_dp1.log 2>&1
  echo "PROF_DP_RC=$?"; grep -m1 value /root/repo/gpurun_out/prof_dp1.log

  cd /root/repo
  find gpurun_out/prof_* -name "*kernel_stats*" | head
  echo "== ALL DONE =="
} 2>&1 | tee gpurun_out/call2_summary.log
