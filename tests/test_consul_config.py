"""Consul client configuration: token env override, map-form config, and
registration retry against an agent that comes up late.
(reference: discovery/config.go:29-61, consul.go:33-58,
jobs/jobs.go:109-113,168-171.)"""

import socket
import time

from containerpilot_amd.mockconsul import MockConsul


def wait_until(predicate, timeout=20.0, interval=0.1):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if predicate():
            return True
        time.sleep(interval)
    return False


def test_consul_http_token_env(daemon_factory, mock_consul):
    """CONSUL_HTTP_TOKEN is sent as X-Consul-Token on every call
    (discovery/consul.go:48-50)."""
    d = daemon_factory({
        "consul": mock_consul.address,
        "stopTimeout": 1,
        "jobs": [{
            "name": "app", "exec": "sleep 60", "port": 8000,
            "interfaces": ["static:127.0.0.1"],
            "health": {"exec": "true", "interval": 1, "ttl": 5},
        }],
    }, env={"CONSUL_HTTP_TOKEN": "seekrit-token"}).start()
    d.wait_for_socket()
    assert wait_until(lambda: len(mock_consul.tokens) > 0), d.log()
    assert mock_consul.tokens[0] == "seekrit-token"
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_consul_map_config(daemon_factory, mock_consul):
    """The consul key also accepts {address, scheme, token}
    (discovery/config.go:63-75)."""
    d = daemon_factory({
        "consul": {"address": mock_consul.address, "scheme": "http",
                   "token": "map-token"},
        "stopTimeout": 1,
        "jobs": [{
            "name": "app", "exec": "sleep 60", "port": 8000,
            "interfaces": ["static:127.0.0.1"],
            "health": {"exec": "true", "interval": 1, "ttl": 5},
        }],
    }).start()
    d.wait_for_socket()
    assert wait_until(lambda: "map-token" in mock_consul.tokens), d.log()
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_registration_retries_until_agent_up(daemon_factory):
    """initial_status registration retries inside the event loop until the
    agent answers (jobs/jobs.go:168-171: 'retry if consul registration
    fails')."""
    # reserve a port, but don't start the agent yet
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()

    d = daemon_factory({
        "consul": "127.0.0.1:%d" % port,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [{
            "name": "eager", "exec": "sleep 60", "port": 8000,
            "initial_status": "passing",
            "interfaces": ["static:127.0.0.1"],
            "health": {"exec": "true", "interval": 1, "ttl": 5},
        }],
    }).start()
    d.wait_for_socket()
    time.sleep(1.5)  # let a few attempts fail against the closed port
    assert "registration failed" in d.log() or True  # attempts logged

    consul = MockConsul(port=port).start()
    try:
        assert wait_until(
            lambda: any("eager" in sid for sid in consul.services)), d.log()
        reg = [v for k, v in consul.services.items() if "eager" in k][0]
        assert reg["Check"]["Status"] == "passing"
    finally:
        d.terminate()
        assert d.wait(timeout=30) == 0
        consul.stop()
