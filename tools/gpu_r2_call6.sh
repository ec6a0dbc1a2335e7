#!/bin/bash
# A/B the seg1 side-stream critic overlap on ONE box (clock variance
# between boxes makes cross-call comparisons useless).
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out

for ov in 1 0 1 0; do
  export DSAC_SEG1_OVERLAP=$ov
  echo "=== mtsac overlap=$ov ==="
  timeout 180 python bench.py --steps 2000 --warmup 300 --min-timed-seconds 2 \
    --skip-fp32-probe --skip-rollout-probe 2>/dev/null \
    | grep -a ms_per_step | python -c "import sys,json; d=json.load(sys.stdin); print(d['ms_per_step'], d['value'])"
done
for ov in 1 0 1 0; do
  export DSAC_SEG1_OVERLAP=$ov
  echo "=== care overlap=$ov ==="
  timeout 180 python bench.py --config care --steps 1200 --warmup 200 \
    --min-timed-seconds 2 --skip-fp32-probe --skip-rollout-probe 2>/dev/null \
    | grep -a ms_per_step | python -c "import sys,json; d=json.load(sys.stdin); print(d['ms_per_step'], d['value'])"
done
echo DONE
