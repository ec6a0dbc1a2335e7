#!/bin/bash
# Round-2 call 5: validate the side-stream critic-overlap in seg1
# (SAC + CARE), then A/B the flagship benches with it.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out

echo "=== targeted GPU tests (chain + manual-agreement + DP graphs) ==="
timeout 420 python -m pytest \
  tests/test_gpu_kernels.py tests/test_care.py tests/test_engine.py \
  tests/test_checkpoint.py -x -q -m gpu 2>&1 | tail -5

echo "=== bench mtsac (overlap ON) ==="
timeout 240 python bench.py --steps 2000 --warmup 300 --min-timed-seconds 2 \
  2>gpurun_out/c5_mtsac.err | tail -1
echo "=== bench care ==="
timeout 240 python bench.py --config care --steps 1200 --warmup 200 \
  --min-timed-seconds 2 2>gpurun_out/c5_care.err | tail -1
echo "=== bench mt1_care ==="
timeout 240 python bench.py --config mt1_care --steps 1200 --warmup 200 \
  --min-timed-seconds 2 2>gpurun_out/c5_mt1care.err | tail -1
echo "=== bench mtsac world-1 DP (segmented graphs + overlap) ==="
timeout 240 python bench.py --steps 1500 --warmup 250 --force-ddp \
  --min-timed-seconds 2 2>gpurun_out/c5_dp.err | tail -1
echo "=== bench care world-1 DP ==="
timeout 240 python bench.py --config care --steps 1000 --warmup 150 \
  --force-ddp --min-timed-seconds 2 2>gpurun_out/c5_caredp.err | tail -1
echo DONE
