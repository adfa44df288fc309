"""BASELINE config #4: the nginx-upstream autopilot pattern — a frontend
whose config is re-rendered/reloaded whenever the set of healthy backends
changes, plus a telemetry scrape.
(reference: integration test_discovery_consul + docs/ nginx example.)"""

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


def test_nginx_autopilot_pattern(daemon_factory, mock_consul, tmp_path):
    telemetry_port = free_port()
    upstream_file = tmp_path / "upstreams.txt"
    # "nginx" frontend: long-running job advertised in consul;
    # "onchange-backend" re-renders upstreams on each watch change
    # (standing in for `consul-template` + `nginx -s reload`)
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [
            {"name": "nginx", "exec": "sleep 60", "port": free_port(),
             "interfaces": ["static:127.0.0.1"],
             "health": {"exec": "true", "interval": 1, "ttl": 5}},
            {"name": "onchange-backend",
             "exec": ["sh", "-c",
                      "echo rendered >> " + str(upstream_file)],
             "when": {"source": "watch.backend", "each": "changed"}},
        ],
        "watches": [{"name": "backend", "interval": 1}],
        "telemetry": {"port": telemetry_port,
                      "interfaces": ["static:127.0.0.1"]},
    }).start()
    d.wait_for_socket()

    # two backends come up -> first render
    mock_consul.set_health("backend", [
        {"ID": "backend-1", "Address": "10.0.0.1", "Port": 9000},
        {"ID": "backend-2", "Address": "10.0.0.2", "Port": 9000},
    ])
    assert wait_until(lambda: upstream_file.exists() and
                      len(upstream_file.read_text().splitlines()) >= 1), \
        d.log()

    # one backend dies -> reload again
    n = len(upstream_file.read_text().splitlines())
    mock_consul.set_health("backend", [
        {"ID": "backend-1", "Address": "10.0.0.1", "Port": 9000},
    ])
    assert wait_until(lambda: len(
        upstream_file.read_text().splitlines()) > n)

    # the frontend registered itself and heartbeats
    hostname = socket.gethostname()
    assert wait_until(lambda: ("nginx-%s" % hostname) in mock_consul.services)

    # telemetry scrape shows the watch gauge tracking backend count
    def gauge_value():
        try:
            with urllib.request.urlopen(
                    "http://127.0.0.1:%d/metrics" % telemetry_port,
                    timeout=5) as resp:
                body = resp.read().decode()
        except OSError:
            return None
        for line in body.splitlines():
            if line.startswith(
                    'containerpilot_watch_instances{service="backend"}'):
                return float(line.split()[-1])
        return None

    assert wait_until(lambda: gauge_value() == 1.0)
    d.terminate()
    assert d.wait(timeout=30) == 0
