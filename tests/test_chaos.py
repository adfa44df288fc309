"""Soak/chaos: a loaded daemon survives interleaved reloads, maintenance
toggles, metrics, and signals, then drains cleanly. This is the
'hammering' style of the reference's test_config_reload, widened to the
whole control surface."""

import json
import random
import signal
import time


def test_chaos_control_surface(daemon_factory, mock_consul):
    for i in range(5):
        mock_consul.set_health("up-%d" % i,
                               [{"ID": "u%d" % i, "Address": "10.0.0.1",
                                 "Port": 1000 + i}])
    jobs = [{"name": "main-app", "exec": "sleep 300"}]
    for i in range(10):
        jobs.append({
            "name": "svc-%02d" % i, "exec": "sleep 300",
            "port": 21000 + i, "interfaces": ["static:127.0.0.1"],
            "health": {"exec": "true", "interval": 1, "ttl": 5},
        })
    for i in range(5):
        jobs.append({
            "name": "tick-%d" % i, "exec": "true",
            "when": {"interval": "200ms"},
        })
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "logging": {"level": "INFO"},
        "jobs": jobs,
        "watches": [{"name": "up-%d" % i, "interval": 1}
                    for i in range(5)],
    }).start()
    d.wait_for_socket()

    rng = random.Random(42)
    deadline = time.time() + 12
    actions = 0
    while time.time() < deadline:
        op = rng.randrange(6)
        try:
            if op == 0:
                d.control("POST", "/v3/reload")
                time.sleep(0.4)
                d.wait_for_socket(timeout=15)
            elif op == 1:
                d.control("POST", "/v3/maintenance/enable")
            elif op == 2:
                d.control("POST", "/v3/maintenance/disable")
            elif op == 3:
                d.control("POST", "/v3/metric",
                          json.dumps({"x": rng.random()}))
            elif op == 4:
                d.signal(signal.SIGHUP)
            else:
                status, _ = d.control("GET", "/v3/ping")
                assert status == 200
        except OSError:
            # socket mid-flip during a reload; must come back
            d.wait_for_socket(timeout=15)
        actions += 1
        time.sleep(0.1)

    assert actions > 50
    d.wait_for_socket(timeout=15)
    status, _ = d.control("GET", "/v3/ping")
    assert status == 200
    assert d.proc.poll() is None  # still alive under chaos
    d.terminate()
    assert d.wait(timeout=60) == 0
