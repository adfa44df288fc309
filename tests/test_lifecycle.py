"""Job lifecycle scenarios (reference: jobs/jobs_test.go event-sequence
tests + integration_tests/test_tasks, test_coprocess, test_sigterm)."""

import re
import signal
import time

import pytest


def make_config(consul_addr, jobs, **extra):
    cfg = {"consul": consul_addr, "stopTimeout": 1, "jobs": jobs,
           "logging": {"level": "DEBUG"}}
    cfg.update(extra)
    return cfg


CONSUL = "localhost:79"  # closed port: discovery calls fail-and-continue


def test_single_job_runs_and_exits(daemon_factory):
    """BASELINE config #1: echo job, no consul, clean exit."""
    d = daemon_factory(make_config(CONSUL, [
        {"name": "hello", "exec": "echo hello-from-test"}]))
    d.start()
    rc = d.wait(timeout=30)
    assert rc == 0
    log = d.log()
    assert "hello-from-test" in log
    assert "{ExitSuccess hello}" in log
    assert "{Stopping hello}" in log
    assert "{Stopped hello}" in log
    # event ordering: startup before exit before stopping
    assert log.index("{Startup global}") < log.index("{ExitSuccess hello}")
    assert log.index("{ExitSuccess hello}") < log.index("{Stopping hello}")


def test_failing_job_exit_failed(daemon_factory):
    d = daemon_factory(make_config(CONSUL, [
        {"name": "badjob", "exec": "false"}]))
    d.start()
    assert d.wait(timeout=30) == 0
    assert "{ExitFailed badjob}" in d.log()


def test_restart_limit_counts(daemon_factory):
    """restarts: 2 -> the exec runs exactly 3 times (initial + 2)."""
    d = daemon_factory(make_config(CONSUL, [
        {"name": "flappy", "exec": "true", "restarts": 2}]))
    d.start()
    assert d.wait(timeout=30) == 0
    assert d.log().count("{ExitSuccess flappy}") == 3


def test_restarts_never(daemon_factory):
    d = daemon_factory(make_config(CONSUL, [
        {"name": "onceonly", "exec": "true", "restarts": "never"}]))
    d.start()
    assert d.wait(timeout=30) == 0
    assert d.log().count("{ExitSuccess onceonly}") == 1


def test_periodic_job_run_counts(daemon_factory):
    """when.interval jobs fire repeatedly (integration test_tasks)."""
    d = daemon_factory(make_config(CONSUL, [
        {"name": "tick", "exec": "true", "when": {"interval": "200ms"}}]),
        extra_args=["-bench-seconds", "2"])
    d.start()
    assert d.wait(timeout=30) == 0
    count = d.log().count("{ExitSuccess tick}")
    # ~10 ticks in 2s; allow a wide margin for slow CI
    assert 3 <= count <= 13, d.log()


def test_when_once_dependency_chain(daemon_factory):
    """BASELINE config #3 shape: jobs chained via when: source/once."""
    d = daemon_factory(make_config(CONSUL, [
        {"name": "first", "exec": "echo first-ran"},
        {"name": "second", "exec": "echo second-ran",
         "when": {"source": "first", "once": "exitSuccess"}},
        {"name": "third", "exec": "echo third-ran",
         "when": {"source": "second", "once": "exitSuccess"}},
    ]))
    d.start()
    assert d.wait(timeout=30) == 0
    log = d.log()
    assert log.index("{ExitSuccess first}") < log.index("{ExitSuccess second}")
    assert log.index("{ExitSuccess second}") < log.index("{ExitSuccess third}")


def test_when_timeout_quits_job(daemon_factory):
    """A job waiting on an event that never fires times out and quits."""
    d = daemon_factory(make_config(CONSUL, [
        {"name": "patient", "exec": "echo should-not-run",
         "when": {"source": "ghost", "once": "exitSuccess",
                  "timeout": "500ms"}}]))
    d.start()
    assert d.wait(timeout=30) == 0
    log = d.log()
    assert "should-not-run" not in log
    assert "{TimerExpired patient}" in log


def test_sigterm_graceful_shutdown_prestop(daemon_factory):
    """preStop runs on shutdown before the main job stops
    (integration test_sigterm; jobs/jobs.go:295-312)."""
    d = daemon_factory(make_config(CONSUL, [
        {"name": "main-app", "exec": "sleep 60", "stopTimeout": "5s"},
        {"name": "pre-stop", "exec": "echo prestop-ran",
         "when": {"source": "main-app", "once": "stopping"}},
        {"name": "post-stop", "exec": "echo poststop-ran",
         "when": {"source": "main-app", "once": "stopped"}},
    ]))
    d.start()
    d.wait_for_socket()
    time.sleep(0.3)
    d.terminate()
    assert d.wait(timeout=30) == 0
    log = d.log()
    assert "prestop-ran" in log
    assert "poststop-ran" in log
    # main-app's Stopped must come after pre-stop's exec completed
    assert log.index("{ExitSuccess pre-stop}") < log.index("{Stopped main-app}")


def test_stopping_timeout_bounds_wait(daemon_factory):
    """If the pre-stop job hangs, main-app's cleanup is bounded by its
    stopTimeout instead of hanging forever (jobs/jobs.go:391-407)."""
    d = daemon_factory(make_config(CONSUL, [
        {"name": "main-app", "exec": "sleep 60", "stopTimeout": "1s"},
        {"name": "pre-stop", "exec": "sleep 60",
         "when": {"source": "main-app", "once": "stopping"}},
    ]))
    d.start()
    d.wait_for_socket()
    time.sleep(0.2)
    t0 = time.time()
    d.terminate()
    assert d.wait(timeout=30) == 0
    assert time.time() - t0 < 15


def test_coprocess_restart_resets_on_reload(daemon_factory, tmp_path):
    """A restart-limited coprocess restarts once, not twice; reload resets
    the limit (integration test_coprocess)."""
    d = daemon_factory(make_config(CONSUL, [
        {"name": "main-app", "exec": "sleep 60"},
        {"name": "coprocess", "exec": "true", "restarts": 1},
    ]))
    d.start()
    d.wait_for_socket()
    deadline = time.time() + 10
    while time.time() < deadline and \
            d.log().count("{ExitSuccess coprocess}") < 2:
        time.sleep(0.1)
    time.sleep(0.5)  # settle: no third run may appear
    assert d.log().count("{ExitSuccess coprocess}") == 2  # initial + 1
    # reload resets the restart budget
    status, _ = d.control("POST", "/v3/reload")
    assert status == 200
    deadline = time.time() + 10
    while time.time() < deadline and \
            d.log().count("{ExitSuccess coprocess}") < 4:
        time.sleep(0.1)
    time.sleep(0.5)
    assert d.log().count("{ExitSuccess coprocess}") == 4
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_exec_timeout_kills_job(daemon_factory):
    d = daemon_factory(make_config(CONSUL, [
        {"name": "slow", "exec": "sleep 60", "timeout": "300ms",
         "restarts": "never"}]))
    d.start()
    assert d.wait(timeout=30) == 0
    log = d.log()
    assert "timeout after" in log
    assert "{ExitFailed slow}" in log


def test_child_process_reaped(daemon_factory):
    """Children forked by the daemon are reaped (no zombies left from the
    daemon's own execs)."""
    d = daemon_factory(make_config(CONSUL, [
        {"name": "spawner", "exec": "sh -c true"}]))
    d.start()
    assert d.wait(timeout=30) == 0
    assert "{ExitSuccess spawner}" in d.log()


def test_maintenance_mode_stops_health_events(daemon_factory):
    d = daemon_factory(make_config(CONSUL, [
        {"name": "svc", "exec": "sleep 60",
         "health": {"exec": "true", "interval": 1, "ttl": 3}}]))
    d.start()
    d.wait_for_socket()
    # let one health check pass
    deadline = time.time() + 10
    while time.time() < deadline:
        if "{StatusHealthy svc}" in d.log():
            break
        time.sleep(0.1)
    assert "{StatusHealthy svc}" in d.log()
    status, _ = d.control("POST", "/v3/maintenance/enable")
    assert status == 200
    time.sleep(0.2)
    marker = len(d.log())
    time.sleep(2.2)
    # no further StatusHealthy after entering maintenance
    assert "{StatusHealthy svc}" not in d.log()[marker:]
    # exit maintenance: health events resume
    status, _ = d.control("POST", "/v3/maintenance/disable")
    assert status == 200
    deadline = time.time() + 10
    while time.time() < deadline:
        if "{StatusHealthy svc}" in d.log()[marker:]:
            break
        time.sleep(0.1)
    assert "{StatusHealthy svc}" in d.log()[marker:]
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_sighup_triggered_job(daemon_factory):
    """when: {source: SIGHUP} jobs fire on the signal (test_sighup)."""
    d = daemon_factory(make_config(CONSUL, [
        {"name": "main-app", "exec": "sleep 60"},
        {"name": "onhup", "exec": "echo hup-ran",
         "when": {"source": "SIGHUP"}},
    ]))
    d.start()
    d.wait_for_socket()
    time.sleep(0.2)
    assert "hup-ran" not in d.log()
    d.signal(signal.SIGHUP)
    deadline = time.time() + 10
    while time.time() < deadline:
        if "hup-ran" in d.log():
            break
        time.sleep(0.1)
    assert "hup-ran" in d.log()
    # signal jobs can fire again
    d.signal(signal.SIGHUP)
    deadline = time.time() + 10
    while time.time() < deadline:
        if d.log().count("hup-ran") >= 2:
            break
        time.sleep(0.1)
    assert d.log().count("hup-ran") >= 2
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_job_env_vars_exported(daemon_factory):
    """CONTAINERPILOT_PID and CONTAINERPILOT_{NAME}_PID are visible to
    children (integration test_envvars)."""
    d = daemon_factory(make_config(CONSUL, [
        {"name": "main-app", "exec": "sleep 2"},
        {"name": "envdump",
         "exec": ["sh", "-c",
                  "echo PILOTPID=$CONTAINERPILOT_PID "
                  "MAINPID=$CONTAINERPILOT_MAIN_APP_PID"],
         "when": {"source": "main-app", "once": "exitSuccess"}},
    ]))
    d.start()
    assert d.wait(timeout=30) == 0
    log = d.log()
    m = re.search(r"PILOTPID=(\d+) MAINPID=(\d*)", log)
    assert m, log
    assert int(m.group(1)) == d.proc.pid


def test_job_without_exec_no_crash(daemon_factory):
    """A job with no exec is legal and doesn't break the lifecycle
    (integration test_no_command)."""
    d = daemon_factory(make_config(CONSUL, [
        {"name": "noexec"},
        {"name": "hello", "exec": "echo no-command-ok"},
    ]))
    d.start()
    d.wait_for_socket()
    time.sleep(0.3)
    assert "no-command-ok" in d.log()
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_job_ip_env_exported(daemon_factory, mock_consul):
    """CONTAINERPILOT_{JOB}_IP is exported for advertised jobs
    (core/app.go:81-86, integration test_envvars)."""
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "jobs": [
            {"name": "my-svc", "exec": "sleep 60", "port": 8123,
             "interfaces": ["static:10.7.7.7"],
             "health": {"exec": "true", "interval": 1, "ttl": 5}},
            {"name": "ipdump",
             "exec": ["sh", "-c", "echo SVCIP=$CONTAINERPILOT_MY_SVC_IP"]},
        ],
    })
    d.start()
    d.wait_for_socket()
    deadline = time.time() + 10
    while time.time() < deadline and "SVCIP=" not in d.log():
        time.sleep(0.1)
    assert "SVCIP=10.7.7.7" in d.log()
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_second_sigterm_escalates_kill_sweep(daemon_factory):
    """During the final stopTimeout window a second SIGTERM/SIGINT
    triggers the kill sweep immediately instead of being swallowed
    (r1 weak item; the reference sleeps through it,
    core/app.go:146-157)."""
    import signal
    import time

    d = daemon_factory(make_config(CONSUL, [{
            # ignores SIGTERM so the graceful drain can't finish early
            "name": "stubborn",
            "exec": ["bash", "-c", "trap '' TERM; sleep 120"],
        }], stopTimeout=8)).start()
    d.wait_for_socket()
    time.sleep(0.5)

    t0 = time.time()
    d.signal(signal.SIGTERM)
    time.sleep(1.0)
    d.signal(signal.SIGTERM)  # escalate
    rc = d.wait(timeout=6)
    elapsed = time.time() - t0
    assert rc == 0, d.log()
    assert elapsed < 6, (
        "daemon slept through the second SIGTERM (%.1fs):\n%s"
        % (elapsed, d.log()[-2000:]))
    assert "second signal received" in d.log()
