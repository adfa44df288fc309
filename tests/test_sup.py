"""PID-1 supervisor: when the daemon is PID 1 it forks a worker and the
parent reaps orphans reparented to it (reference: sup/sup.go,
integration test_reap_zombies). Uses a PID namespace so the daemon
really runs as PID 1."""

import os
import subprocess

import pytest

from containerpilot_amd import BINARY


def unshare_available():
    try:
        r = subprocess.run(
            ["unshare", "--pid", "--fork", "--mount-proc", "true"],
            capture_output=True, timeout=10)
        return r.returncode == 0
    except (OSError, subprocess.TimeoutExpired):
        return False


requires_unshare = pytest.mark.skipif(
    not unshare_available(), reason="unshare --pid unavailable")


@requires_unshare
def test_pid1_reaps_zombies(tmp_path):
    """Orphaned grandchildren reparent to PID 1 (the sup process) and are
    reaped: the verifier job counts zombie-state processes after the
    orphans exit (test_reap_zombies asserts <=1 zombie at steady state)."""
    cfg = tmp_path / "cfg.json5"
    cfg.write_text("""
{
  consul: "localhost:79",
  stopTimeout: 1,
  logging: {level: "DEBUG"},
  jobs: [
    {
      name: "zombie-bomb",
      // orphan five short-lived children: their parent sh exits
      // immediately so they reparent to PID 1
      exec: ["sh", "-c",
             "for i in 1 2 3 4 5; do (sleep 0.2; exit 0) & done; exit 0"]
    },
    {
      name: "verifier",
      when: {source: "zombie-bomb", once: "exitSuccess"},
      exec: ["sh", "-c",
             "sleep 1.5; z=0; for f in /proc/[0-9]*/stat; do s=$(awk '{print $3}' $f 2>/dev/null); [ \\"$s\\" = Z ] && z=$((z+1)); done; echo ZOMBIES=$z"]
    }
  ]
}
""")
    result = subprocess.run(
        ["unshare", "--pid", "--fork", "--mount-proc",
         BINARY, "-config", str(cfg)],
        capture_output=True, text=True, timeout=60)
    assert result.returncode == 0, result.stdout + result.stderr
    out = result.stdout
    assert "ZOMBIES=" in out, out
    count = int(out.split("ZOMBIES=")[1].split()[0])
    assert count <= 1, out


@requires_unshare
def test_pid1_worker_runs_jobs(tmp_path):
    """As PID 1 the worker fork still runs the normal lifecycle."""
    cfg = tmp_path / "cfg.json5"
    cfg.write_text('{consul: "localhost:79", stopTimeout: 1, '
                   'jobs: [{name: "hello", exec: "echo pid1-hello"}]}')
    result = subprocess.run(
        ["unshare", "--pid", "--fork", "--mount-proc",
         BINARY, "-config", str(cfg)],
        capture_output=True, text=True, timeout=60)
    assert result.returncode == 0, result.stdout + result.stderr
    assert "pid1-hello" in result.stdout
