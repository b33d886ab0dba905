"""bench.py driver contract: one JSON line on stdout with the required
fields (the driver parses exactly this), tiny model on CPU."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"]


def test_bench_json_contract():
    proc = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--model", "tiny",
         "--agents-per-gpu", "1", "--steps", "1", "--warmup", "0",
         "--device", "cpu"],
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert proc.returncode == 0, proc.stderr[-2000:]
    lines = [l for l in proc.stdout.strip().splitlines() if l.startswith("{")]
    assert len(lines) == 1, f"expected ONE json line, got: {proc.stdout!r}"
    out = json.loads(lines[0])
    for field in REQUIRED:
        assert field in out, f"missing {field}"
    assert out["metric"] == "consensus agent-steps/sec"
    assert out["unit"] == "agent-steps/sec"
    assert out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    assert out["dtype"] == "bf16"
    assert out["data"] == "synthetic"
    assert out["n_gpus"] == 1 and out["steps"] == 1 and out["warmup"] == 0
    assert out["value"] > 0 and out["ms_per_step"] > 0
    cfg = out["config"]
    assert cfg["model"] == "tiny" and cfg["agents_total"] == 1
    assert cfg["p50_step_latency_ms"] > 0
    assert cfg["decisions_completed"] >= 1
