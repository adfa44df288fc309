"""Control-plane endpoint behavior over the unix socket (reference:
control/endpoints.go actual behavior — empty 200 "\n", 422, 405 — plus
the -reload/-ping/-putenv/-putmetric subcommands and reload hammering
from integration test_config_reload)."""

import json
import os
import subprocess
import time

from containerpilot_amd import BINARY

CONSUL = "localhost:79"


def base_config(jobs=None):
    return {"consul": CONSUL, "stopTimeout": 1,
            "logging": {"level": "DEBUG"},
            "jobs": jobs or [{"name": "main-app", "exec": "sleep 60"}]}


def test_ping(daemon_factory):
    d = daemon_factory(base_config()).start()
    d.wait_for_socket()
    status, body = d.control("GET", "/v3/ping")
    assert status == 200
    assert body == "\n"
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_non_post_method_rejected(daemon_factory):
    d = daemon_factory(base_config()).start()
    d.wait_for_socket()
    for path in ("/v3/environ", "/v3/reload", "/v3/metric",
                 "/v3/maintenance/enable", "/v3/maintenance/disable"):
        status, body = d.control("GET", path)
        assert status == 405, path
        assert "Method Not Allowed" in body
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_unknown_path_404(daemon_factory):
    d = daemon_factory(base_config()).start()
    d.wait_for_socket()
    status, _ = d.control("POST", "/v3/bogus")
    assert status == 404
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_putenv_updates_child_env(daemon_factory):
    d = daemon_factory(base_config([
        {"name": "main-app", "exec": "sleep 60"},
        {"name": "envprinter", "exec": ["sh", "-c", "echo GOT=$TESTVAR"],
         "when": {"source": "SIGUSR2"}},
    ])).start()
    d.wait_for_socket()
    status, _ = d.control("POST", "/v3/environ",
                          json.dumps({"TESTVAR": "hello-env"}))
    assert status == 200
    import signal
    d.signal(signal.SIGUSR2)
    deadline = time.time() + 10
    while time.time() < deadline:
        if "GOT=hello-env" in d.log():
            break
        time.sleep(0.1)
    assert "GOT=hello-env" in d.log()
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_putenv_bad_json_422(daemon_factory):
    d = daemon_factory(base_config()).start()
    d.wait_for_socket()
    status, body = d.control("POST", "/v3/environ", "{nope")
    assert status == 422
    assert "Unprocessable Entity" in body
    # non-string values are a 422 too (map[string]string unmarshal)
    status, _ = d.control("POST", "/v3/environ", '{"X": 42}')
    assert status == 422
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_reload_generation(daemon_factory):
    d = daemon_factory(base_config()).start()
    d.wait_for_socket()
    status, _ = d.control("POST", "/v3/reload")
    assert status == 200
    # socket comes back after the new generation starts
    time.sleep(0.5)
    d.wait_for_socket()
    status, _ = d.control("GET", "/v3/ping")
    assert status == 200
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_reload_hammering_no_deadlock(daemon_factory):
    """integration test_config_reload: hammer /v3/reload; daemon must
    keep serving and shut down cleanly."""
    d = daemon_factory(base_config()).start()
    d.wait_for_socket()
    ok = 0
    for _ in range(10):
        try:
            status, _ = d.control("POST", "/v3/reload")
            if status == 200:
                ok += 1
        except OSError:
            pass  # socket mid-flip between generations
        time.sleep(0.15)
        try:
            d.wait_for_socket(timeout=10)
        except TimeoutError:
            break
    assert ok >= 3
    d.wait_for_socket(timeout=10)
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_reload_with_invalid_config_exits(daemon_factory):
    """A reload into a broken config makes the daemon exit with an error
    (core/app.go:158-161)."""
    d = daemon_factory(base_config()).start()
    d.wait_for_socket()
    with open(d.config_path, "w") as f:
        f.write("{this is not valid")
    d.control("POST", "/v3/reload")
    assert d.wait(timeout=30) == 0
    assert "error initializing config" in d.log()


def test_subcommands_over_socket(daemon_factory):
    d = daemon_factory(base_config()).start()
    d.wait_for_socket()
    env = dict(os.environ)

    r = subprocess.run([BINARY, "-config", d.config_path, "-ping"],
                       capture_output=True, text=True, env=env, timeout=30)
    assert r.returncode == 0 and r.stdout.strip() == "ok"

    r = subprocess.run([BINARY, "-config", d.config_path,
                        "-putenv", "SUBCMD_VAR=yes"],
                       capture_output=True, text=True, env=env, timeout=30)
    assert r.returncode == 0, r.stderr

    r = subprocess.run([BINARY, "-config", d.config_path,
                        "-maintenance", "enable"],
                       capture_output=True, text=True, env=env, timeout=30)
    assert r.returncode == 0, r.stderr
    time.sleep(0.3)
    assert "{EnterMaintenance global}" in d.log()

    r = subprocess.run([BINARY, "-config", d.config_path,
                        "-maintenance", "disable"],
                       capture_output=True, text=True, env=env, timeout=30)
    assert r.returncode == 0, r.stderr

    r = subprocess.run([BINARY, "-config", d.config_path, "-reload"],
                       capture_output=True, text=True, env=env, timeout=30)
    assert r.returncode == 0, r.stderr

    time.sleep(0.5)
    d.wait_for_socket()
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_putenv_then_reload_rerenders_template(daemon_factory, tmp_path):
    """The canonical /v3/environ workflow: set an env var, reload, and
    the re-rendered config template picks it up (templates render
    against the daemon's current environment)."""
    d = daemon_factory(config_text="""
{
  consul: "localhost:79",
  stopTimeout: 1,
  logging: {level: "DEBUG"},
  control: {socket: "{SOCKET}"},
  jobs: [
    {name: "main-app", "exec": "sleep 60"},
    {name: "versionprinter",
     exec: "echo version={{ .DEPLOY_VERSION | default "none" }}"},
  ],
}
""").start()
    d.wait_for_socket()
    deadline = time.time() + 10
    while time.time() < deadline and "version=none" not in d.log():
        time.sleep(0.1)
    assert "version=none" in d.log()

    status, _ = d.control("POST", "/v3/environ",
                          json.dumps({"DEPLOY_VERSION": "v2.5"}))
    assert status == 200
    status, _ = d.control("POST", "/v3/reload")
    assert status == 200
    deadline = time.time() + 15
    while time.time() < deadline and "version=v2.5" not in d.log():
        time.sleep(0.1)
    assert "version=v2.5" in d.log()
    d.terminate()
    assert d.wait(timeout=30) == 0
