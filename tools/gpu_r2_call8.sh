#!/bin/bash
# Round-2 call 8: validate the fused single-WG loss kernels + lazy alpha.
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
cd "$GRAFT_REPO_ROOT"
mkdir -p gpurun_out

J () { grep -a '"metric"'; }

echo "=== full GPU test suite ==="
timeout 900 python -m pytest tests/ -q -m gpu > gpurun_out/c8_pytest.log 2>&1
grep -a "passed\|failed\|error" gpurun_out/c8_pytest.log | tail -3

echo "=== bench mtsac ==="
timeout 200 python bench.py --steps 2000 --warmup 300 --min-timed-seconds 2 \
  --skip-fp32-probe --skip-rollout-probe 2>/dev/null | J
echo "=== bench care ==="
timeout 200 python bench.py --config care --steps 1200 --warmup 200 \
  --min-timed-seconds 2 --skip-fp32-probe --skip-rollout-probe 2>/dev/null | J
echo "=== bench sac ==="
timeout 200 python bench.py --config sac --steps 4000 --warmup 500 \
  --min-timed-seconds 2 --skip-fp32-probe --skip-rollout-probe 2>/dev/null | J

echo "=== rocprof kernel stats ==="
cd /tmp && export TMPDIR=/tmp
timeout 300 rocprofv3 --kernel-trace --stats --output-format csv \
  -d /tmp/prof_c8 -o c8 -- \
  python "$GRAFT_REPO_ROOT/bench.py" --steps 300 --warmup 100 \
  --min-timed-seconds 0 --skip-fp32-probe --skip-rollout-probe \
  > /tmp/c8_bench.log 2>&1
find /tmp/prof_c8 -name "*kernel_stats*" -exec cp {} "$GRAFT_REPO_ROOT/gpurun_out/c8_mtsac_kernel_stats.csv" \;
cd "$GRAFT_REPO_ROOT"
python tools/print_kernel_stats.py gpurun_out/c8_mtsac_kernel_stats.csv | head -28
echo DONE
