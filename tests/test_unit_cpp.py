"""Run the native unit-test binary (covers JSON5, durations, templates,
event codes, arg parsing, IP specs, config validation — the counterpart
of the reference's per-package *_test.go suites)."""

import subprocess

from containerpilot_amd import UNITTEST_BINARY


def test_cpp_unittests():
    result = subprocess.run([UNITTEST_BINARY], capture_output=True, text=True,
                            timeout=60)
    assert result.returncode == 0, result.stdout + result.stderr
    assert "all unit tests passed" in result.stdout
