"""containerpilot_amd: an AMD-node-native rebuild of ContainerPilot.

The framework itself is a native C++17 daemon (``bin/containerpilot``)
that runs as PID 1 inside a container: it supervises jobs, reaps
zombies, runs health checks, registers services in Consul, polls Consul
for upstream changes, exposes Prometheus telemetry and an HTTP control
plane over a unix socket. See SURVEY.md for the full component map of
the reference (TritonDataCenter/containerpilot) this reproduces.

This Python package is the test/bench harness around the daemon:
  - harness:    build + spawn + drive the daemon from tests
  - mockconsul: an in-process Consul agent API mock (register/TTL/
                deregister/health endpoints) for integration tests
"""

import os

__version__ = "3.9.0-amd"

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BINARY = os.path.join(REPO_ROOT, "bin", "containerpilot")
UNITTEST_BINARY = os.path.join(REPO_ROOT, "bin", "cpilot_unittests")
