"""The driver depends on bench.py's JSON contract: one line from rank 0
with the documented fields. Run a tiny configuration and validate the
shape strictly."""

import json
import os
import subprocess
import sys

from containerpilot_amd import REPO_ROOT


def test_bench_json_contract():
    result = subprocess.run(
        [sys.executable, os.path.join(REPO_ROOT, "bench.py"),
         "--steps", "2", "--warmup", "1", "--jobs", "10", "--watches", "2"],
        capture_output=True, text=True, timeout=300, cwd=REPO_ROOT)
    assert result.returncode == 0, result.stdout + result.stderr
    lines = [ln for ln in result.stdout.splitlines() if ln.startswith("{")]
    assert len(lines) == 1, result.stdout
    doc = json.loads(lines[0])

    for field in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                  "ms_per_step", "higher_is_better", "scaling",
                  "vs_baseline", "dtype", "data", "config"):
        assert field in doc, field
    assert doc["metric"] == "events/sec"
    assert doc["n_gpus"] == 1
    assert doc["steps"] == 2
    assert doc["warmup"] == 1
    assert doc["higher_is_better"] is True
    assert doc["scaling"] == "weak"
    assert doc["value"] > 0
    assert doc["vs_baseline"] == round(doc["value"] / 10000.0, 3)
    assert 900 < doc["ms_per_step"] < 1500  # a step is one second
    assert "p99_dispatch_ms" in doc["config"]
    assert doc["config"]["jobs"] == 10
