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
