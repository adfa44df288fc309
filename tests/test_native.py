"""Exercise the in-tree native library through the ctypes bindings —
the same C++ code paths the daemon executes (template engine, JSON5
parser, duration rules, full config validation)."""

import json

import pytest

from containerpilot_amd import native


def test_version():
    assert native.version()


def test_render_template(monkeypatch):
    monkeypatch.setenv("NATIVE_TEST_VAR", "abc")
    assert native.render_template("v={{ .NATIVE_TEST_VAR }}") == "v=abc"
    assert native.render_template(
        '{{ "x:y:z" | split ":" | join "-" }}') == "x-y-z"
    with pytest.raises(ValueError):
        native.render_template("{{ bogusfunc }}")


def test_parse_duration():
    assert native.parse_duration_ns('60') == 60 * 10**9
    assert native.parse_duration_ns('"1m30s"') == 90 * 10**9
    assert native.parse_duration_ns('"250ms"') == 250 * 10**6
    with pytest.raises(ValueError):
        native.parse_duration_ns('"nope"')


def test_json5():
    doc = json.loads(native.json5_to_json(
        "{a: 1, 'b': [2, 3,], /* c */ d: {e: true}}"))
    assert doc == {"a": 1, "b": [2, 3], "d": {"e": True}}
    with pytest.raises(ValueError) as exc:
        native.json5_to_json("{a: }")
    assert "parse error at line:col" in str(exc.value)


def test_validate_config():
    assert native.validate_config(
        '{consul: "x:8500", jobs: [{name: "j", exec: "true"}]}') is None
    err = native.validate_config('{jobs: []}')
    assert "no discovery backend defined" in err
    err = native.validate_config(
        '{consul: "x:8500", jobs: [{name: "j", exec: "true", junk: 1}]}')
    assert "invalid keys" in err
