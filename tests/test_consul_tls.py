"""HTTPS to the Consul agent: scheme https + tls config / env overrides.
(reference: discovery/config.go TLS plumbing via the Go consul client.)"""

import os
import socket
import subprocess
import time

import pytest

from containerpilot_amd.mockconsul import MockConsul


def wait_until(predicate, timeout=20.0, interval=0.1):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if predicate():
            return True
        time.sleep(interval)
    return False


@pytest.fixture(scope="module")
def self_signed_cert(tmp_path_factory):
    d = tmp_path_factory.mktemp("tls")
    cert = d / "cert.pem"
    key = d / "key.pem"
    subprocess.run(
        ["openssl", "req", "-x509", "-newkey", "rsa:2048", "-nodes",
         "-keyout", str(key), "-out", str(cert), "-days", "2",
         "-subj", "/CN=localhost",
         "-addext", "subjectAltName=DNS:localhost,IP:127.0.0.1"],
        check=True, capture_output=True)
    return str(cert), str(key)


def tls_daemon(daemon_factory, consul, extra_env):
    return daemon_factory({
        "consul": "https://%s" % consul.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [{
            "name": "secure-app", "exec": "sleep 60", "port": 8200,
            "interfaces": ["static:127.0.0.1"],
            "health": {"exec": "true", "interval": 1, "ttl": 5},
        }],
    }, env=extra_env)


def test_https_with_ca_env(daemon_factory, self_signed_cert):
    cert, key = self_signed_cert
    consul = MockConsul(certfile=cert, keyfile=key).start()
    try:
        d = tls_daemon(daemon_factory, consul, {
            "CONSUL_CACERT": cert,
            "CONSUL_TLS_SERVER_NAME": "localhost",
        }).start()
        d.wait_for_socket()
        assert wait_until(
            lambda: any("secure-app" in sid for sid in consul.services)), \
            d.log()
        d.terminate()
        assert d.wait(timeout=30) == 0
    finally:
        consul.stop()


def test_https_rejects_untrusted_ca(daemon_factory, self_signed_cert,
                                    tmp_path):
    cert, key = self_signed_cert
    consul = MockConsul(certfile=cert, keyfile=key).start()
    try:
        # no CA configured and verification on: handshake must fail
        d = tls_daemon(daemon_factory, consul, {
            "CONSUL_CACERT": "", "CONSUL_HTTP_SSL_VERIFY": "true",
        }).start()
        d.wait_for_socket()
        time.sleep(2.5)
        assert not consul.services, d.log()
        assert "registration failed" in d.log() or \
               "TLS handshake failed" in d.log()
        d.terminate()
        assert d.wait(timeout=30) == 0
    finally:
        consul.stop()


def test_https_insecure_skip_verify(daemon_factory, self_signed_cert):
    cert, key = self_signed_cert
    consul = MockConsul(certfile=cert, keyfile=key).start()
    try:
        d = tls_daemon(daemon_factory, consul, {
            "CONSUL_HTTP_SSL_VERIFY": "false",
        }).start()
        d.wait_for_socket()
        assert wait_until(
            lambda: any("secure-app" in sid for sid in consul.services)), \
            d.log()
        d.terminate()
        assert d.wait(timeout=30) == 0
    finally:
        consul.stop()
