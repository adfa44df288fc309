"""Failure-detection / recovery behaviors: listener bind
retry, Consul agent errors are log-and-continue, and service recovery
after an agent outage."""

import socket
import time


def wait_until(predicate, timeout=25.0, interval=0.1):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if predicate():
            return True
        time.sleep(interval)
    return False


def test_telemetry_bind_retry(daemon_factory, mock_consul):
    """The telemetry listener retries binding (10x1s,
    telemetry/telemetry.go:77-91): hold the port briefly, release it,
    the daemon comes up anyway."""
    blocker = socket.socket()
    blocker.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    blocker.bind(("127.0.0.1", 0))
    port = blocker.getsockname()[1]
    blocker.listen(1)

    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "jobs": [{"name": "main-app", "exec": "sleep 60"}],
        "telemetry": {"port": port, "interfaces": ["static:127.0.0.1"]},
    })
    d.start()
    time.sleep(2.0)  # a couple of failed binds
    blocker.close()

    def telemetry_up():
        try:
            s = socket.create_connection(("127.0.0.1", port), timeout=1)
            s.close()
            return True
        except OSError:
            return False

    assert wait_until(telemetry_up), d.log()
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_consul_errors_log_and_continue(daemon_factory, mock_consul):
    """Agent 500s never take the daemon down: health checks keep
    running, and service state recovers when the agent does
    (discovery/consul.go:90-93 log-and-continue)."""
    mock_consul.set_health("up", [
        {"ID": "u1", "Address": "10.0.0.1", "Port": 1000}])
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [{
            "name": "app", "exec": "sleep 60", "port": 8000,
            "interfaces": ["static:127.0.0.1"],
            "health": {"exec": "true", "interval": 1, "ttl": 5},
        }],
        "watches": [{"name": "up", "interval": 1}],
    }).start()
    d.wait_for_socket()
    assert wait_until(lambda: len(mock_consul.services) > 0)

    # agent starts failing: daemon keeps checking, logs warnings
    mock_consul.fail_mode = True
    marker = len(d.log())
    time.sleep(2.5)
    assert d.proc.poll() is None
    log_tail = d.log()[marker:]
    assert "{StatusHealthy app}" in log_tail  # checks keep passing
    assert "TTL failed" in log_tail or "failed to query" in log_tail

    # agent recovers: TTL updates flow again
    mock_consul.fail_mode = False
    n = len(mock_consul.ttl_updates)
    assert wait_until(lambda: len(mock_consul.ttl_updates) > n)
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_bind_retry_does_not_block_dispatch(daemon_factory):
    """A contended telemetry port retries on a loop timer; event dispatch
    (periodic jobs, control endpoints) continues during the retry window
    and the listener comes up once the port frees (reference retries in a
    goroutine, control/control.go:125-140 / telemetry.go:77-91).
    Regression test for the r1 reactor-blocking sleep_for retry."""
    import socket as socketmod
    import time
    import urllib.request

    # occupy the port the telemetry server wants
    blocker = socketmod.socket()
    blocker.setsockopt(socketmod.SOL_SOCKET, socketmod.SO_REUSEADDR, 1)
    blocker.bind(("127.0.0.1", 0))
    port = blocker.getsockname()[1]
    blocker.listen(1)

    d = daemon_factory({
        # telemetry advertises a service, so discovery must be
        # configured; an unreachable agent only logs warnings
        "consul": "127.0.0.1:1",
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [
            {"name": "main-app", "exec": ["sleep", "60"]},
            # a fast periodic job proves timers keep firing
            {"name": "ticker", "exec": ["true"],
             "when": {"interval": "200ms"}, "restarts": "unlimited"},
        ],
        "telemetry": {"port": port, "interfaces": ["static:127.0.0.1"]},
    }).start()
    d.wait_for_socket(timeout=10)

    # during the retry window the reactor must stay live: control plane
    # answers and the periodic job keeps running
    t0 = time.time()
    status, _ = d.control("GET", "/v3/ping")
    assert status == 200
    assert time.time() - t0 < 1.0, "control response stalled behind retry"
    runs_before = d.log().count("ticker.Run start")
    time.sleep(1.0)
    runs_after = d.log().count("ticker.Run start")
    assert runs_after - runs_before >= 3, (
        "periodic job stalled during bind retry:\n" + d.log()[-2000:])
    assert "telemetry: error listening" in d.log()

    # free the port: the loop-timer retry should bind within ~2s
    blocker.close()
    deadline = time.time() + 5
    body = None
    while time.time() < deadline:
        try:
            with urllib.request.urlopen(
                    "http://127.0.0.1:%d/status" % port, timeout=2) as resp:
                body = resp.read()
                break
        except OSError:
            time.sleep(0.2)
    assert body is not None, "telemetry never bound after port freed:\n" + \
        d.log()[-2000:]
    assert b"ticker" in body
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_spawn_helper_death_recovers(daemon_factory):
    """Killing a spawn helper mid-run is survived: the daemon respawns
    it, in-flight launches fail loudly (ExitFailed -> restart policy),
    and health checks keep completing afterwards."""
    import os
    import signal as signalmod
    import time

    d = daemon_factory({
        "consul": "127.0.0.1:1",
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [
            {"name": "main-app", "exec": ["sleep", "60"],
             "port": 23456, "interfaces": ["static:127.0.0.1"],
             "health": {"exec": ["true"], "interval": "500ms", "ttl": 5}},
        ],
    }).start()
    d.wait_for_socket()
    time.sleep(1.5)

    # helpers are direct children of the worker process named
    # cpilot-spawn-helper; the worker is d.proc.pid (no sup here)
    def helper_pids():
        pids = []
        for pid in os.listdir("/proc"):
            if not pid.isdigit():
                continue
            try:
                with open("/proc/%s/stat" % pid) as f:
                    parts = f.read().split()
                if parts[1] == "(cpilot-spawn-he)" and \
                        int(parts[3]) == d.proc.pid:
                    pids.append(int(pid))
            except (OSError, ValueError):
                pass
        return pids

    before = helper_pids()
    assert before, "no spawn helpers found under the daemon"
    os.kill(before[0], signalmod.SIGKILL)

    # respawn logged, and checks still complete after the death
    deadline = time.time() + 10
    while time.time() < deadline and "died; respawning" not in d.log():
        time.sleep(0.2)
    assert "died; respawning" in d.log(), d.log()[-2000:]
    marker = d.log().count("check.main-app exited without error")
    deadline = time.time() + 10
    while time.time() < deadline and \
            d.log().count("check.main-app exited without error") < marker + 3:
        time.sleep(0.2)
    assert d.log().count("check.main-app exited without error") >= marker + 3, \
        "checks did not resume after helper death:\n" + d.log()[-2000:]
    # pool healed: same helper count, new pid present
    after = helper_pids()
    assert len(after) == len(before), (before, after)
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_no_fd_leak_across_reloads(daemon_factory, mock_consul):
    """Reload churn must not leak fds: consul workers are joined per
    generation and their pooled keep-alive connections must close at
    thread exit (regression test for the thread_local pool leak)."""
    import os
    import time

    mock_consul.set_health("up", [
        {"ID": "u1", "Address": "10.0.0.1", "Port": 1000}])
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "jobs": [{
            "name": "app", "exec": "sleep 300", "port": 8000,
            "interfaces": ["static:127.0.0.1"],
            "health": {"exec": "true", "interval": 1, "ttl": 5},
        }],
        "watches": [{"name": "up", "interval": 1}],
    }).start()
    d.wait_for_socket()
    time.sleep(2)  # let TTL/keep-alive traffic flow

    def fd_count():
        return len(os.listdir("/proc/%d/fd" % d.proc.pid))

    before = fd_count()
    for _ in range(8):
        d.control("POST", "/v3/reload")
        time.sleep(0.6)
        d.wait_for_socket(timeout=15)
        time.sleep(0.6)
    after = fd_count()
    assert after <= before + 4, (
        "fd count grew across reloads: %d -> %d" % (before, after))
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_oversized_environment_fails_loudly(daemon_factory):
    """A spawn whose serialized request exceeds the helper protocol cap
    (256 KiB) fails as ExitFailed instead of wedging the job (E2BIG
    path in the spawner client)."""
    import time

    # ~1.2 MiB of environment split across vars (a single var would
    # exceed the kernel's per-string exec limit before reaching us);
    # over the helper protocol's 1 MiB request cap
    bigenv = {"CPILOT_TEST_BIG_%03d" % i: "x" * 2048 for i in range(600)}
    d = daemon_factory({
        "consul": "localhost:79",
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [
            {"name": "main-app", "exec": ["sleep", "30"]},
            {"name": "bigenv", "exec": ["true"],
             "when": {"interval": "500ms"}, "restarts": "unlimited"},
        ],
    }, env=bigenv).start()
    d.wait_for_socket()
    time.sleep(2)
    log = d.log()
    # the spawns failed loudly (E2BIG = "Argument list too long")...
    assert "unable to start" in log and "{ExitFailed bigenv}" in log, \
        log[-2000:]
    # ...and the daemon stayed healthy
    status, _ = d.control("GET", "/v3/ping")
    assert status == 200
    d.terminate()
    assert d.wait(timeout=30) == 0
