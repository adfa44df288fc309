"""The driver's multi-rank bench contract: bench.py under
torch.distributed.run with the gloo backend, world_size 2 (runs on CPU).
Each rank drives an independent daemon; rank 0 prints one JSON line
aggregating all ranks."""

import json
import os
import subprocess
import sys

from containerpilot_amd import REPO_ROOT


def test_torchrun_two_ranks():
    env = dict(os.environ)
    env.setdefault("MASTER_ADDR", "127.0.0.1")
    result = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run",
         "--nnodes=1", "--nproc-per-node", "2",
         "--master-addr", "127.0.0.1", "--master-port", "29537",
         os.path.join(REPO_ROOT, "bench.py"),
         "--gpus", "2", "--steps", "3", "--warmup", "1",
         "--jobs", "20", "--watches", "5"],
        capture_output=True, text=True, timeout=600, env=env,
        cwd=REPO_ROOT)
    assert result.returncode == 0, result.stdout + result.stderr
    line = [ln for ln in result.stdout.splitlines()
            if ln.startswith("{") and '"metric"' in ln]
    assert line, result.stdout + result.stderr
    doc = json.loads(line[-1])
    assert doc["n_gpus"] == 2
    assert doc["steps"] == 3
    assert doc["value"] > 0
    assert doc["scaling"] == "weak"
    assert doc["config"]["parallelism"].startswith("2 ")
