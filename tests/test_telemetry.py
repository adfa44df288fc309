"""Telemetry endpoint behavior: Prometheus /metrics (internal + user
collectors), JSON /status, the synthetic containerpilot service, and
metric events via the control plane.
(reference: telemetry/*.go, integration test_telemetry.)"""

import json
import socket
import time
import urllib.request


def wait_until(predicate, timeout=15.0, interval=0.1):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if predicate():
            return True
        time.sleep(interval)
    return False


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def http_get(url):
    with urllib.request.urlopen(url, timeout=5) as resp:
        return resp.status, resp.read().decode()


def make_daemon(daemon_factory, mock_consul, port):
    return daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [
            {"name": "main-app", "exec": "sleep 60"},
            {"name": "svc", "exec": "sleep 60", "port": 7777,
             "interfaces": ["static:10.9.9.9"],
             "health": {"exec": "true", "interval": 1, "ttl": 5}},
        ],
        "watches": [{"name": "upstream", "interval": 1}],
        "telemetry": {
            "port": port,
            "interfaces": ["static:127.0.0.1"],
            "metrics": [
                {"namespace": "app", "subsystem": "worker",
                 "name": "jobs_done", "help": "done", "type": "counter"},
                {"namespace": "app", "subsystem": "worker",
                 "name": "queue_depth", "help": "depth", "type": "gauge"},
            ],
        },
    })


def test_metrics_endpoint(daemon_factory, mock_consul):
    port = free_port()
    d = make_daemon(daemon_factory, mock_consul, port).start()
    d.wait_for_socket()
    url = "http://127.0.0.1:%d/metrics" % port
    assert wait_until(lambda: _up(url))
    _, body = http_get(url)
    # internal collectors (events/bus.go:60-68, control/control.go:25-33,
    # discovery/consul.go:14-22)
    assert "containerpilot_events" in body
    assert 'code="Startup"' in body
    assert "containerpilot_control_http_requests" in body
    assert "containerpilot_watch_instances" in body
    # user collectors registered even before any values
    assert "app_worker_jobs_done" in body
    assert "app_worker_queue_depth" in body

    # push metrics through the control plane (endpoints.go:111-129)
    status, _ = d.control("POST", "/v3/metric", json.dumps({
        "app_worker_jobs_done": 3,
        "app_worker_queue_depth": 12.5,
    }))
    assert status == 200
    assert wait_until(
        lambda: "app_worker_jobs_done 3" in http_get(url)[1])
    _, body = http_get(url)
    assert "app_worker_queue_depth 12.5" in body

    # counters add, gauges set
    d.control("POST", "/v3/metric", json.dumps({
        "app_worker_jobs_done": 2,
        "app_worker_queue_depth": 4,
    }))
    assert wait_until(
        lambda: "app_worker_jobs_done 5" in http_get(url)[1])
    _, body = http_get(url)
    assert "app_worker_queue_depth 4" in body

    # unknown metric names are dropped silently
    d.control("POST", "/v3/metric", json.dumps({"nope": 1}))
    d.terminate()
    assert d.wait(timeout=30) == 0


def _up(url):
    try:
        http_get(url)
        return True
    except OSError:
        return False


def test_status_endpoint(daemon_factory, mock_consul):
    port = free_port()
    d = make_daemon(daemon_factory, mock_consul, port).start()
    d.wait_for_socket()
    url = "http://127.0.0.1:%d/status" % port
    assert wait_until(lambda: _up(url))
    _, body = http_get(url)
    doc = json.loads(body)
    assert doc["Version"]
    # jobs without a service land in Jobs, advertised ones in Services
    job_names = [j["Name"] for j in doc["Jobs"]]
    assert "main-app" in job_names
    svc = [s for s in doc["Services"] if s["Name"] == "svc"][0]
    assert svc["Port"] == 7777
    assert svc["Address"] == "10.9.9.9"
    # the synthetic containerpilot service is advertised too
    cp = [s for s in doc["Services"] if s["Name"] == "containerpilot"]
    assert cp and cp[0]["Port"] == port
    assert doc["Watches"] == ["upstream"]

    # after a passing health check the status flips to healthy
    assert wait_until(
        lambda: [s for s in json.loads(http_get(url)[1])["Services"]
                 if s["Name"] == "svc"][0]["Status"] == "healthy")
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_synthetic_containerpilot_service_registers(daemon_factory,
                                                    mock_consul):
    """The telemetry service heartbeats itself into Consul with TTL 15
    (telemetry/telemetry_config.go:71-86) and deregisters on shutdown
    (integration test_telemetry)."""
    port = free_port()
    d = make_daemon(daemon_factory, mock_consul, port).start()
    d.wait_for_socket()
    hostname = socket.gethostname()
    cp_id = "containerpilot-%s" % hostname
    assert wait_until(lambda: cp_id in mock_consul.services, timeout=20), \
        d.log()
    reg = mock_consul.services[cp_id]
    assert reg["Port"] == port
    assert reg["Check"]["TTL"] == "15s"
    d.terminate()
    assert d.wait(timeout=30) == 0
    assert cp_id in mock_consul.deregistered


def test_histogram_and_summary_metrics(daemon_factory, mock_consul):
    """histogram/summary collectors expose buckets/quantiles after
    observations (telemetry/metrics_config.go:63-80)."""
    port = free_port()
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "jobs": [{"name": "main-app", "exec": "sleep 60"}],
        "telemetry": {
            "port": port,
            "interfaces": ["static:127.0.0.1"],
            "metrics": [
                {"namespace": "app", "subsystem": "rq", "name": "latency",
                 "help": "h", "type": "histogram"},
                {"namespace": "app", "subsystem": "rq", "name": "sizes",
                 "help": "s", "type": "summary"},
            ],
        },
    }).start()
    d.wait_for_socket()
    url = "http://127.0.0.1:%d/metrics" % port
    assert wait_until(lambda: _up(url))
    for v in (0.01, 0.02, 0.3, 2.0):
        d.control("POST", "/v3/metric",
                  json.dumps({"app_rq_latency": v, "app_rq_sizes": v * 100}))
    assert wait_until(
        lambda: "app_rq_latency_count 4" in http_get(url)[1])
    _, body = http_get(url)
    assert 'app_rq_latency_bucket{le="0.025"} 2' in body
    assert "app_rq_latency_sum" in body
    assert 'app_rq_sizes{quantile="0.5"}' in body
    assert "app_rq_sizes_count 4" in body
    d.terminate()
    assert d.wait(timeout=30) == 0
