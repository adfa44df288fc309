"""CLI flag and subcommand behavior (reference: core/flags.go,
subcommands/subcommands.go, config.RenderConfig)."""

import os
import subprocess

from containerpilot_amd import BINARY


def run_cli(*args, env=None, timeout=30):
    e = dict(os.environ)
    if env:
        e.update(env)
    return subprocess.run([BINARY, *args], capture_output=True, text=True,
                          timeout=timeout, env=e)


def test_version_flag():
    result = run_cli("-version")
    assert result.returncode == 0
    assert result.stdout.startswith("Version: ")
    assert "GitHash:" in result.stdout


def test_template_render_to_stdout(tmp_path):
    cfg = tmp_path / "cfg.json5"
    cfg.write_text('{consul: "{{ .TEST_CONSUL | default "localhost:8500" }}"}')
    result = run_cli("-template", "-config", str(cfg))
    assert result.returncode == 0
    assert '"localhost:8500"' in result.stdout

    result = run_cli("-template", "-config", str(cfg),
                     env={"TEST_CONSUL": "consul:8501"})
    assert '"consul:8501"' in result.stdout


def test_template_render_to_file(tmp_path):
    cfg = tmp_path / "cfg.json5"
    out = tmp_path / "out.json5"
    cfg.write_text('{consul: "localhost:8500"}')
    result = run_cli("-template", "-config", str(cfg), "-out", str(out))
    assert result.returncode == 0
    assert out.read_text() == '{consul: "localhost:8500"}'


def test_missing_config_flag():
    result = run_cli(env={"CONTAINERPILOT": ""})
    assert result.returncode != 0
    assert "-config flag is required" in result.stderr


def test_config_via_env_var(tmp_path):
    cfg = tmp_path / "cfg.json5"
    cfg.write_text('{consul: "localhost:8500"}')
    result = run_cli("-template", env={"CONTAINERPILOT": str(cfg)})
    assert result.returncode == 0


def test_unknown_flag():
    result = run_cli("-bogus")
    assert result.returncode == 2
    assert "flag provided but not defined" in result.stderr


def test_bad_putenv_format():
    result = run_cli("-putenv", "novalue")
    assert result.returncode != 0
    assert "was not in the format" in result.stderr


def test_config_parse_error_reporting(tmp_path):
    cfg = tmp_path / "bad.json5"
    cfg.write_text('{consul: "localhost:8500",\n  jobs: [}\n}')
    result = run_cli("-config", str(cfg))
    assert result.returncode != 0
    assert "parse error at line:col" in result.stderr


def test_unknown_config_key(tmp_path):
    cfg = tmp_path / "bad.json5"
    cfg.write_text('{consul: "localhost:8500", frobnicate: 1}')
    result = run_cli("-config", str(cfg))
    assert result.returncode != 0
    assert "unknown config keys" in result.stderr
