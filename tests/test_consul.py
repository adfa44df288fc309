"""Consul wire behavior against the mock agent: registration payloads,
TTL heartbeats, deregistration on stop/maintenance, initial_status lazy
registration, and watch-triggered jobs.
(reference: discovery/consul.go, discovery/service.go, watches/,
integration test_discovery_consul.)"""

import socket
import time


def wait_until(predicate, timeout=15.0, interval=0.1):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if predicate():
            return True
        time.sleep(interval)
    return False


def test_service_registration_and_ttl(daemon_factory, mock_consul):
    """BASELINE config #2: main job + health check registers and
    heartbeats against the (mock) Consul agent."""
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [{
            "name": "app",
            "exec": "sleep 60",
            "port": 8000,
            "interfaces": ["static:10.1.2.3"],
            "tags": ["prod", "blue"],
            "health": {"exec": "true", "interval": 1, "ttl": 5},
            "consul": {"enableTagOverride": True,
                       "deregisterCriticalServiceAfter": "90m"},
        }],
    }).start()
    d.wait_for_socket()

    assert wait_until(lambda: len(mock_consul.services) > 0), d.log()
    hostname = socket.gethostname()
    svc_id = "app-%s" % hostname
    assert svc_id in mock_consul.services
    reg = mock_consul.services[svc_id]
    # byte-compatible registration payload (discovery/service.go:93-110)
    assert reg["Name"] == "app"
    assert reg["Port"] == 8000
    assert reg["Address"] == "10.1.2.3"
    assert reg["Tags"] == ["prod", "blue"]
    assert reg["EnableTagOverride"] is True
    assert reg["Check"]["TTL"] == "5s"
    assert reg["Check"]["DeregisterCriticalServiceAfter"] == "90m"
    assert "containerpilot" in reg["Check"]["Notes"]

    # TTL updates flow with checkID service:<id> and Status passing
    assert wait_until(lambda: len(mock_consul.ttl_updates) >= 2)
    check_id, payload = mock_consul.ttl_updates[0]
    assert check_id == "service:%s" % svc_id
    assert payload["Status"] == "passing"
    assert payload["Output"] == "ok"

    # deregistered on SIGTERM (jobs/jobs.go:409-411)
    d.terminate()
    assert d.wait(timeout=30) == 0
    assert svc_id in mock_consul.deregistered


def test_initial_status_registration(daemon_factory, mock_consul):
    """initial_status registers immediately (before the first passing
    health check) with the configured status."""
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "jobs": [{
            "name": "warmup",
            "exec": "sleep 60",
            "port": 9001,
            "initial_status": "warning",
            "interfaces": ["static:10.1.2.3"],
            "health": {"exec": "sleep 60", "interval": 30, "ttl": 60},
        }],
    }).start()
    d.wait_for_socket()
    assert wait_until(lambda: len(mock_consul.services) > 0), d.log()
    reg = list(mock_consul.services.values())[0]
    assert reg["Check"]["Status"] == "warning"
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_unhealthy_check_no_ttl_update(daemon_factory, mock_consul):
    """Failing health checks publish StatusUnhealthy and never send a TTL
    pass (the TTL lapses server-side; CP never sends 'fail')."""
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [{
            "name": "sick",
            "exec": "sleep 60",
            "port": 8001,
            "interfaces": ["static:10.1.2.3"],
            "health": {"exec": "false", "interval": 1, "ttl": 2},
        }],
    }).start()
    d.wait_for_socket()
    assert wait_until(lambda: "{StatusUnhealthy sick}" in d.log())
    assert mock_consul.ttl_updates == []
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_watch_change_triggers_job(daemon_factory, mock_consul):
    """BASELINE config #3: watch fires StatusChanged on upstream change
    and a dependent job runs each time (watches/watches.go:85-96)."""
    mock_consul.set_health("backend", [])
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [
            {"name": "main-app", "exec": "sleep 60"},
            {"name": "onchange", "exec": "echo saw-change",
             "when": {"source": "watch.backend", "each": "changed"}},
        ],
        "watches": [{"name": "backend", "interval": 1}],
    }).start()
    d.wait_for_socket()
    time.sleep(1.5)  # a few polls with empty state: no change events
    assert "{StatusChanged watch.backend}" not in d.log()

    mock_consul.set_health("backend", [
        {"ID": "backend-1", "Address": "10.0.0.1", "Port": 9000}])
    assert wait_until(lambda: "{StatusChanged watch.backend}" in d.log())
    assert wait_until(lambda: "{StatusHealthy watch.backend}" in d.log())
    assert wait_until(lambda: "saw-change" in d.log())

    # a second change (new address) fires again
    before = d.log().count("saw-change")
    mock_consul.set_health("backend", [
        {"ID": "backend-1", "Address": "10.0.0.2", "Port": 9000}])
    assert wait_until(lambda: d.log().count("saw-change") > before)

    # removing all instances fires StatusUnhealthy
    mock_consul.set_health("backend", [])
    assert wait_until(lambda: "{StatusUnhealthy watch.backend}" in d.log())
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_watch_no_change_no_events(daemon_factory, mock_consul):
    """Stable upstream set -> exactly one StatusChanged (initial
    appearance), then silence (compareForChange semantics)."""
    mock_consul.set_health("stable", [
        {"ID": "s-1", "Address": "10.0.0.1", "Port": 1000}])
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [{"name": "main-app", "exec": "sleep 60"}],
        "watches": [{"name": "stable", "interval": 1}],
    }).start()
    d.wait_for_socket()
    assert wait_until(lambda: "{StatusChanged watch.stable}" in d.log())
    time.sleep(2.5)
    assert d.log().count("{StatusChanged watch.stable}") == 1
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_watch_tag_and_dc_in_query(daemon_factory, mock_consul):
    """watch tag/dc flow into the health query string
    (discovery/consul.go:87-89: Health().Service(name, tag, passing,
    {Datacenter: dc}))."""
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "jobs": [{"name": "main-app", "exec": "sleep 60"}],
        "watches": [{"name": "tagged", "interval": 1, "tag": "prod",
                     "dc": "us-east-1"}],
    }).start()
    d.wait_for_socket()
    assert wait_until(lambda: any(
        "/v1/health/service/tagged" in path and "passing=1" in path and
        "tag=prod" in path and "dc=us-east-1" in path
        for _, path in mock_consul.requests)), mock_consul.requests[-5:]
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_job_gated_on_watch_healthy(daemon_factory, mock_consul):
    """A job with when: {source: watch.db, once: healthy} starts only
    after the upstream becomes available — the canonical 'wait for the
    database' pattern."""
    mock_consul.set_health("db", [])
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [
            {"name": "main-app", "exec": "sleep 60"},
            {"name": "migrate", "exec": "echo migrated",
             "when": {"source": "watch.db", "once": "healthy"}},
        ],
        "watches": [{"name": "db", "interval": 1}],
    }).start()
    d.wait_for_socket()
    time.sleep(1.5)
    assert "migrated" not in d.log()
    mock_consul.set_health("db", [
        {"ID": "db-1", "Address": "10.0.0.9", "Port": 5432}])
    assert wait_until(lambda: "migrated" in d.log()), d.log()
    assert d.log().count("migrated") == 1
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_watch_tag_filters_instances(daemon_factory, mock_consul):
    """Only instances carrying the watch's tag count as upstream members
    (tag-partitioned pools)."""
    mock_consul.set_health("pool", [
        {"ID": "pool-1", "Address": "10.0.0.1", "Port": 80,
         "Tags": ["blue"]},
        {"ID": "pool-2", "Address": "10.0.0.2", "Port": 80,
         "Tags": ["green"]},
    ])
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [{"name": "main-app", "exec": "sleep 60"}],
        "watches": [{"name": "pool", "interval": 1, "tag": "blue"}],
    }).start()
    d.wait_for_socket()
    # becomes healthy (one blue instance)
    assert wait_until(lambda: "{StatusHealthy watch.pool}" in d.log())
    marker = len(d.log())
    # removing the green instance is invisible to a blue-tag watch
    mock_consul.set_health("pool", [
        {"ID": "pool-1", "Address": "10.0.0.1", "Port": 80,
         "Tags": ["blue"]},
    ])
    time.sleep(2.5)
    assert "{StatusChanged watch.pool}" not in d.log()[marker:]
    # removing the blue instance fires unhealthy
    mock_consul.set_health("pool", [])
    assert wait_until(lambda: "{StatusUnhealthy watch.pool}" in d.log())
    d.terminate()
    assert d.wait(timeout=30) == 0
