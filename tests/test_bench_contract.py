"""The driver's bench contract: `python bench.py ...` must print ONE
JSON line with the agreed keys, value = whole-job aggregate, and run the
exact requested step count.  Run on CPU (eager path) via subprocess so
arg parsing, config loading, the measure loop and the JSON emission are
covered end-to-end."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def _run_bench(*extra):
    out = subprocess.run(
        [sys.executable, "bench.py", "--device", "cpu", "--steps", "8",
         "--warmup", "2", "--min-timed-seconds", "0.05",
         "--skip-rollout-probe", "--skip-fp32-probe", *extra],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, f"expected ONE JSON line, got {len(lines)}"
    return json.loads(lines[0])


def test_bench_json_contract_default():
    d = _run_bench()
    assert REQUIRED_KEYS <= set(d)
    assert d["n_gpus"] == 1 and d["steps"] == 8 and d["warmup"] == 2
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert d["unit"] == "grad_steps/s" and d["data"] == "synthetic"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    # value and ms_per_step must be consistent inverses
    assert abs(d["value"] * d["ms_per_step"] / 1000.0 - 1.0) < 0.01
    cfgd = d["config"]
    for k in ("model", "global_batch", "seq_len", "parallelism"):
        assert k in cfgd
    assert cfgd["parallelism"] == "dp1"
    assert "MTSAC" in d["metric"]


def test_bench_json_contract_sac_baseline_ratio():
    d = _run_bench("--config", "sac")
    # the only derivable reference rate is the LunarLander config
    assert d["vs_baseline"] is not None
    assert abs(d["vs_baseline"] - d["value"] / 34.1) < 0.11
