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


def _run_subreaper_scenario(tmp_path):
    """Shared impl: daemon forced into sup mode (subreaper) reaps
    orphaned grandchildren; orphans visibly reparent to the sup process.
    Runs unprivileged — no PID namespace needed (fallback for hosts
    where unshare --pid is not permitted, e.g. the GPU box)."""
    import time

    sys_path_hack = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    if sys_path_hack not in __import__("sys").path:
        __import__("sys").path.insert(0, sys_path_hack)
    from containerpilot_amd import harness

    pids_file = tmp_path / "orphan_pids"
    d = harness.Daemon(config_dict={
        "consul": "localhost:79",
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [
            {"name": "main-app", "exec": ["sleep", "60"]},
            {
                "name": "zombie-bomb",
                # orphan five children: their parent sh exits right away
                "exec": ["sh", "-c",
                         "for i in 1 2 3 4 5; do (sleep 2; exit 0) & "
                         "echo $! >> %s; done; exit 0" % pids_file],
            },
        ],
    }, env={"CPILOT_FORCE_SUP": "1"}).start()
    try:
        d.wait_for_socket(timeout=15)
        sup_pid = d.proc.pid  # the launched process stays behind as sup

        deadline = time.time() + 10
        orphans = []
        while time.time() < deadline:
            if pids_file.exists():
                orphans = [int(x) for x in
                           pids_file.read_text().split()]
                if len(orphans) == 5:
                    break
            time.sleep(0.1)
        assert len(orphans) == 5, d.log()

        # while alive, the orphans must have reparented to sup (the
        # subreaper), not to the container init
        reparented = 0
        for pid in orphans:
            try:
                with open("/proc/%d/stat" % pid) as f:
                    ppid = int(f.read().split()[3])
                if ppid == sup_pid:
                    reparented += 1
            except (OSError, ValueError):
                pass
        assert reparented >= 4, (
            "orphans did not reparent to sup (pid %d): %d/5"
            % (sup_pid, reparented))

        # after they exit, sup must reap them: no lingering zombies
        time.sleep(3.5)
        zombies = 0
        for pid in orphans:
            try:
                with open("/proc/%d/stat" % pid) as f:
                    if f.read().split()[2] == "Z":
                        zombies += 1
            except OSError:
                pass  # fully gone: reaped
        assert zombies <= 1, ("unreaped zombies under sup: %d" % zombies)
    finally:
        d.terminate()
        assert d.wait(timeout=30) == 0
        d.cleanup()


def test_subreaper_sup_reaps_orphans(tmp_path):
    _run_subreaper_scenario(tmp_path)


@pytest.mark.gpu
def test_subreaper_sup_reaps_orphans_gpu_box(tmp_path):
    """On-target evidence for the PID-1 reap claim:
    the namespace tests skip on the GPU box (unshare unpermitted), so
    the subreaper fallback runs there instead."""
    _run_subreaper_scenario(tmp_path)
