"""bench.py driver-contract smoke: the JSON line must parse and carry
the agreed fields (the round driver consumes it verbatim)."""

import json
import os
import subprocess
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--cpu",
         "--nodes", "300", "--jobs", "150", "--pods-per-job", "4",
         "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=600, cwd=ROOT)
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.splitlines() if l.startswith("{")][-1]
    j = json.loads(line)
    for field in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling",
                  "vs_baseline", "dtype", "data", "config"):
        assert field in j, f"missing {field}"
    assert j["metric"] == "pods_scheduled_per_sec"
    assert j["n_gpus"] == 1 and j["steps"] == 2 and j["warmup"] == 1
    assert j["higher_is_better"] is True
    assert j["data"] == "synthetic"
    assert j["value"] > 0 and j["vs_baseline"] > 0
    assert "WARNING" not in out.stdout      # all pods bound every step
