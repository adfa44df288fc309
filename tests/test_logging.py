"""Logging behavior: formats, file output with SIGUSR1 reopen
(logrotate), and per-job raw passthrough.
(reference: config/logger/logging.go, integration test_logging,
test_reopen.)"""

import json
import os
import signal
import time


CONSUL = "localhost:79"


def test_json_log_format(daemon_factory):
    d = daemon_factory({
        "consul": CONSUL, "stopTimeout": 1,
        "logging": {"level": "INFO", "format": "json"},
        "jobs": [{"name": "main-app", "exec": "sleep 60"}],
    }).start()
    d.wait_for_socket()
    d.terminate()
    assert d.wait(timeout=30) == 0
    lines = [ln for ln in d.log().splitlines() if ln.strip()]
    assert lines
    for ln in lines:
        doc = json.loads(ln)
        assert "level" in doc and "msg" in doc and "time" in doc


def test_text_log_format(daemon_factory):
    d = daemon_factory({
        "consul": CONSUL, "stopTimeout": 1,
        "logging": {"level": "INFO", "format": "text"},
        "jobs": [{"name": "hello", "exec": "echo textfmt"}],
    }).start()
    assert d.wait(timeout=30) == 0
    assert 'level=info' in d.log()


def test_invalid_log_config_rejected(daemon_factory):
    d = daemon_factory({
        "consul": CONSUL,
        "logging": {"level": "NOPE"},
        "jobs": [{"name": "hello", "exec": "true"}],
    }).start()
    assert d.wait(timeout=30) != 0
    assert "Unknown log level" in d.log()

    d2 = daemon_factory({
        "consul": CONSUL,
        "logging": {"format": "yaml"},
        "jobs": [{"name": "hello", "exec": "true"}],
    }).start()
    assert d2.wait(timeout=30) != 0
    assert "Unknown log format" in d2.log()


def test_file_output_and_sigusr1_reopen(daemon_factory, tmp_path):
    """Log to a file; after the file is rotated away, SIGUSR1 reopens the
    path so new lines land in the new file (test_reopen)."""
    logfile = tmp_path / "cp.log"
    d = daemon_factory({
        "consul": CONSUL, "stopTimeout": 1,
        "logging": {"level": "INFO", "output": str(logfile)},
        "jobs": [
            {"name": "main-app", "exec": "sleep 60"},
            {"name": "ticker", "exec": "echo tick",
             "when": {"interval": "300ms"}},
        ],
    }).start()
    d.wait_for_socket()
    deadline = time.time() + 10
    while time.time() < deadline and not (
            logfile.exists() and "tick" in logfile.read_text()):
        time.sleep(0.1)
    assert "tick" in logfile.read_text()

    rotated = tmp_path / "cp.log.1"
    os.rename(logfile, rotated)
    d.signal(signal.SIGUSR1)
    deadline = time.time() + 10
    while time.time() < deadline and not (
            logfile.exists() and "tick" in logfile.read_text()):
        time.sleep(0.1)
    assert logfile.exists() and "tick" in logfile.read_text()
    d.terminate()
    assert d.wait(timeout=30) == 0


def test_raw_vs_wrapped_job_output(daemon_factory):
    """Wrapped output is prefixed with timestamp/job/pid; raw output is
    passed through untouched (jobs/config.go:280-283,
    commands/commands.go:97-103)."""
    d = daemon_factory({
        "consul": CONSUL, "stopTimeout": 1,
        "logging": {"level": "INFO"},
        "jobs": [
            {"name": "wrapped", "exec": "echo wrapped-line"},
            {"name": "rawjob", "exec": "echo raw-line",
             "logging": {"raw": True}},
        ],
    }).start()
    assert d.wait(timeout=30) == 0
    lines = d.log().splitlines()
    raw = [ln for ln in lines if "raw-line" in ln]
    wrapped = [ln for ln in lines if "wrapped-line" in ln]
    assert raw and raw[0] == "raw-line"        # exact passthrough
    assert wrapped and wrapped[0] != "wrapped-line"  # has prefix
    assert "wrapped" in wrapped[0]             # job field present
