import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from containerpilot_amd import harness  # noqa: E402


def pytest_configure(config):
    config.addinivalue_line(
        "markers", "gpu: tests that must run on a machine with an AMD GPU")


@pytest.fixture(scope="session", autouse=True)
def built_binary():
    """Build the daemon once per test session."""
    return harness.build()


@pytest.fixture
def mock_consul():
    from containerpilot_amd.mockconsul import MockConsul
    consul = MockConsul().start()
    yield consul
    consul.stop()


@pytest.fixture
def daemon_factory():
    daemons = []

    def factory(config_dict=None, config_text=None, **kwargs):
        d = harness.Daemon(config_dict=config_dict, config_text=config_text,
                           **kwargs)
        daemons.append(d)
        return d

    yield factory
    for d in daemons:
        d.cleanup()
