"""Blocking-query watches (extension beyond the reference): changes
propagate in milliseconds instead of the poll interval, and in-flight
long-polls are cancelled on teardown so shutdown/reload never stall."""

import time


def wait_until(predicate, timeout=15.0, interval=0.05):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if predicate():
            return True
        time.sleep(interval)
    return False


def make_daemon(daemon_factory, mock_consul):
    return daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [
            {"name": "main-app", "exec": "sleep 60"},
            {"name": "onchange", "exec": "echo fast-change",
             "when": {"source": "watch.backend", "each": "changed"}},
        ],
        # interval 30 means polling would take up to 30s to notice —
        # blocking mode must beat that by orders of magnitude
        "watches": [{"name": "backend", "interval": 30, "blocking": True}],
    })


def test_blocking_watch_fires_fast(daemon_factory, mock_consul):
    mock_consul.set_health("backend", [])
    d = make_daemon(daemon_factory, mock_consul).start()
    d.wait_for_socket()
    time.sleep(1.0)  # first (index=0) query returns + long-poll parks

    t0 = time.time()
    mock_consul.set_health("backend", [
        {"ID": "b-1", "Address": "10.0.0.1", "Port": 9000}])
    assert wait_until(lambda: "{StatusChanged watch.backend}" in d.log(),
                      timeout=10)
    latency = time.time() - t0
    assert latency < 5.0, latency  # worst case incl. re-issue floor
    assert wait_until(lambda: "fast-change" in d.log())

    # a second change also propagates quickly
    t0 = time.time()
    mock_consul.set_health("backend", [
        {"ID": "b-1", "Address": "10.0.0.2", "Port": 9000}])
    assert wait_until(lambda: d.log().count("fast-change") >= 2, timeout=10)
    assert time.time() - t0 < 5.0
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_blocking_watch_teardown_is_fast(daemon_factory, mock_consul):
    """SIGTERM while a 10s long-poll is parked: the cancel token must
    interrupt it — shutdown takes ~stopTimeout, not ~wait."""
    mock_consul.set_health("backend", [])
    d = make_daemon(daemon_factory, mock_consul).start()
    d.wait_for_socket()
    time.sleep(1.5)  # long-poll is now parked on the mock
    t0 = time.time()
    d.terminate()
    assert d.wait(timeout=30) == 0
    assert time.time() - t0 < 8.0


def test_blocking_watch_survives_reload(daemon_factory, mock_consul):
    mock_consul.set_health("backend", [])
    d = make_daemon(daemon_factory, mock_consul).start()
    d.wait_for_socket()
    time.sleep(1.0)
    t0 = time.time()
    status, _ = d.control("POST", "/v3/reload")
    assert status == 200
    time.sleep(0.5)
    d.wait_for_socket(timeout=15)
    assert time.time() - t0 < 10.0
    # watch works in the new generation too
    mock_consul.set_health("backend", [
        {"ID": "b-9", "Address": "10.0.0.9", "Port": 9000}])
    assert wait_until(lambda: "fast-change" in d.log(), timeout=10)
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_many_blocking_watches_dont_starve_heartbeats(daemon_factory,
                                                      mock_consul):
    """Parked long-polls run on their own threads: eight blocking
    watches must not block the worker pool that carries TTL
    heartbeats."""
    for i in range(8):
        mock_consul.set_health("up-%d" % i, [])
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "jobs": [{
            "name": "svc", "exec": "sleep 60", "port": 8300,
            "interfaces": ["static:127.0.0.1"],
            "health": {"exec": "true", "interval": 1, "ttl": 5},
        }],
        "watches": [{"name": "up-%d" % i, "interval": 30, "blocking": True}
                    for i in range(8)],
    }).start()
    d.wait_for_socket()
    time.sleep(2.0)  # all eight long-polls are parked now
    n = len(mock_consul.ttl_updates)
    time.sleep(3.0)
    # heartbeats kept flowing while every blocking query was parked
    assert len(mock_consul.ttl_updates) >= n + 2, mock_consul.ttl_updates
    d.terminate()
    assert d.wait(timeout=30) == 0
