#!/bin/bash
# Round-2 GPU call 1: validate refactors + staged narrow backward + DP path.
cd /root/repo
mkdir -p gpurun_out
{
  echo "== 1. full GPU suite =="
  timeout 400 python -m pytest tests -m gpu -x -q > gpurun_out/gputest.log 2>&1
  echo "SUITE_RC=$?"; tail -3 gpurun_out/gputest.log

  echo "== 2. narrow-bwd validation =="
  timeout 300 python tools/validate_narrow_bwd.py > gpurun_out/narrow.log 2>&1
  echo "NARROW_RC=$?"; tail -15 gpurun_out/narrow.log

  echo "== 3. gated GPU suite (DSAC_NARROW_BWD=1) =="
  timeout 300 env DSAC_NARROW_BWD=1 python -m pytest tests/test_gpu_kernels.py tests/test_care.py -m gpu -x -q > gpurun_out/gputest_narrow.log 2>&1
  echo "NARROW_SUITE_RC=$?"; tail -3 gpurun_out/gputest_narrow.log

  echo "== 4. bench mtsac (default path, bf16+fp32) =="
  timeout 400 python bench.py --steps 300 --warmup 50 > gpurun_out/bench_mtsac.json 2> gpurun_out/bench_mtsac.err
  echo "BENCH_RC=$?"; cat gpurun_out/bench_mtsac.json; tail -5 gpurun_out/bench_mtsac.err

  echo "== 5. force-ddp world-1 (RCCL init + segmented DP graphs) =="
  timeout 400 python bench.py --steps 300 --warmup 50 --force-ddp --skip-rollout-probe --skip-fp32-probe > gpurun_out/bench_dp1.json 2> gpurun_out/bench_dp1.err
  echo "DP_RC=$?"; cat gpurun_out/bench_dp1.json; tail -8 gpurun_out/bench_dp1.err

  echo "== 6. care bench A/B narrow-bwd =="
  timeout 300 env DSAC_NARROW_BWD=0 python bench.py --config care --steps 200 --warmup 30 --skip-rollout-probe --skip-fp32-probe > gpurun_out/bench_care_off.json 2>gpurun_out/bench_care_off.err
  echo "CARE_OFF_RC=$?"; cat gpurun_out/bench_care_off.json
  timeout 300 env DSAC_NARROW_BWD=1 python bench.py --config care --steps 200 --warmup 30 --skip-rollout-probe --skip-fp32-probe > gpurun_out/bench_care_on.json 2>gpurun_out/bench_care_on.err
  echo "CARE_ON_RC=$?"; cat gpurun_out/bench_care_on.json
  echo "== ALL DONE =="
} 2>&1 | tee gpurun_out/call1_summary.log
