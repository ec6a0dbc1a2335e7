#!/bin/bash
# Round-2 call 7: consolidated evidence run.
#  - full GPU test suite
#  - benches for all 5 variants + world-1 DP (mtsac, care) + async
#  - legacy per-layer path sanity (DSAC_CHAIN=0)
#  - rocprofv3 kernel-stats CSV of the final packed chain path
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out

J () { grep -a '"metric"'; }

echo "=== full GPU test suite ==="
timeout 900 python -m pytest tests/ -q -m gpu 2>&1 | tail -3

echo "=== bench mtsac ==="
timeout 200 python bench.py --steps 2000 --warmup 300 --min-timed-seconds 2 \
  2>/dev/null | J
echo "=== bench care ==="
timeout 200 python bench.py --config care --steps 1200 --warmup 200 \
  --min-timed-seconds 2 2>/dev/null | J
echo "=== bench mt1_care ==="
timeout 200 python bench.py --config mt1_care --steps 1200 --warmup 200 \
  --min-timed-seconds 2 2>/dev/null | J
echo "=== bench sac (LunarLander) ==="
timeout 200 python bench.py --config sac --steps 4000 --warmup 500 \
  --min-timed-seconds 2 2>/dev/null | J
echo "=== bench vsac ==="
timeout 200 python bench.py --config vsac --steps 2000 --warmup 300 \
  --min-timed-seconds 2 2>/dev/null | J
echo "=== bench mtsac world-1 DP ==="
timeout 200 python bench.py --steps 1500 --warmup 250 --force-ddp \
  --min-timed-seconds 2 --skip-fp32-probe 2>/dev/null | J
echo "=== bench care world-1 DP ==="
timeout 200 python bench.py --config care --steps 1000 --warmup 150 \
  --force-ddp --min-timed-seconds 2 --skip-fp32-probe 2>/dev/null | J
echo "=== bench mtsac DSAC_CHAIN=0 (legacy per-layer path) ==="
DSAC_CHAIN=0 timeout 200 python bench.py --steps 1500 --warmup 250 \
  --min-timed-seconds 2 --skip-fp32-probe 2>/dev/null | J
echo "=== bench async (players -> rings -> learner) ==="
timeout 200 python bench.py --async --async-seconds 45 --chunk-steps 256 \
  2>/dev/null | J

echo "=== rocprof kernel stats (packed chain path) ==="
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv \
  -d /tmp/prof_r19 -o r19 -- \
  python "$GRAFT_REPO_ROOT/bench.py" --steps 300 --warmup 100 \
  --min-timed-seconds 0 --skip-fp32-probe --skip-rollout-probe \
  > /tmp/r19_bench.log 2>&1
tail -1 /tmp/r19_bench.log
find /tmp/prof_r19 -name "*kernel_stats*" -exec cp {} "$GRAFT_REPO_ROOT/gpurun_out/r19_mtsac_kernel_stats.csv" \;
cd "$GRAFT_REPO_ROOT"
python tools/print_kernel_stats.py gpurun_out/r19_mtsac_kernel_stats.csv | head -30
echo DONE
