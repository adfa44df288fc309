"""Build and drive the containerpilot daemon from tests and benchmarks."""

import json
import os
import shutil
import signal
import socket
import subprocess
import tempfile
import time

from . import BINARY, REPO_ROOT


def build(force=False):
    """Build the daemon + unit test binary with cmake/ninja (idempotent)."""
    if not force and os.path.exists(BINARY):
        return BINARY
    build_dir = os.path.join(REPO_ROOT, "build")
    subprocess.run(
        ["cmake", "-S", REPO_ROOT, "-B", build_dir, "-G", "Ninja",
         "-DCMAKE_BUILD_TYPE=Release"],
        check=True, capture_output=True)
    subprocess.run(["ninja", "-C", build_dir], check=True,
                   capture_output=True)
    return BINARY


class Daemon:
    """A running containerpilot instance with a scratch work dir."""

    def __init__(self, config_dict=None, config_text=None, workdir=None,
                 extra_args=(), env=None):
        self.workdir = workdir or tempfile.mkdtemp(prefix="cpilot-test-")
        self.socket_path = os.path.join(self.workdir, "cp.socket")
        if config_dict is not None:
            config_dict = dict(config_dict)
            config_dict.setdefault("control", {"socket": self.socket_path})
            config_text = json.dumps(config_dict)
        else:
            config_text = config_text.replace("{SOCKET}", self.socket_path)
        self.config_path = os.path.join(self.workdir, "containerpilot.json5")
        with open(self.config_path, "w") as f:
            f.write(config_text)
        self.stats_path = os.path.join(self.workdir, "stats.json")
        self.log_path = os.path.join(self.workdir, "daemon.log")
        self.extra_args = list(extra_args)
        self.env = dict(os.environ)
        if env:
            self.env.update(env)
        self.proc = None
        self._logf = None

    def start(self):
        build()
        self._logf = open(self.log_path, "wb")
        self.proc = subprocess.Popen(
            [BINARY, "-config", self.config_path,
             "-stats-out", self.stats_path] + self.extra_args,
            stdout=self._logf, stderr=subprocess.STDOUT, env=self.env,
            start_new_session=True)
        return self

    def wait_for_socket(self, timeout=10.0):
        deadline = time.time() + timeout
        while time.time() < deadline:
            if os.path.exists(self.socket_path):
                try:
                    s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
                    s.settimeout(1.0)
                    s.connect(self.socket_path)
                    s.close()
                    return True
                except OSError:
                    pass
            if self.proc and self.proc.poll() is not None:
                raise RuntimeError(
                    "daemon exited early (rc=%s):\n%s"
                    % (self.proc.returncode, self.log()))
            time.sleep(0.05)
        raise TimeoutError("control socket never came up:\n" + self.log())

    def control(self, method, path, body=None):
        """Issue an HTTP request over the control socket."""
        s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        s.settimeout(5.0)
        s.connect(self.socket_path)
        payload = body.encode() if body else b""
        req = (f"{method} {path} HTTP/1.1\r\nHost: localhost\r\n"
               f"Connection: close\r\n"
               f"Content-Length: {len(payload)}\r\n\r\n").encode() + payload
        s.sendall(req)
        resp = b""
        while True:
            chunk = s.recv(65536)
            if not chunk:
                break
            resp += chunk
        s.close()
        head, _, rbody = resp.partition(b"\r\n\r\n")
        status = int(head.split(b" ")[1])
        return status, rbody.decode(errors="replace")

    def signal(self, sig):
        os.kill(self.proc.pid, sig)

    def terminate(self):
        if self.proc and self.proc.poll() is None:
            self.proc.send_signal(signal.SIGTERM)

    def wait(self, timeout=30):
        return self.proc.wait(timeout=timeout)

    def stop(self, timeout=30):
        """SIGTERM and wait; SIGKILL the process group as a last resort."""
        if self.proc is None:
            return None
        if self.proc.poll() is None:
            self.terminate()
            try:
                return self.proc.wait(timeout=timeout)
            except subprocess.TimeoutExpired:
                os.killpg(os.getpgid(self.proc.pid), signal.SIGKILL)
                return self.proc.wait(timeout=5)
        return self.proc.returncode

    def log(self):
        try:
            with open(self.log_path, errors="replace") as f:
                return f.read()
        except OSError:
            return ""

    def stats(self):
        with open(self.stats_path) as f:
            return json.load(f)

    def cleanup(self):
        self.stop(timeout=10)
        if self._logf:
            self._logf.close()
            self._logf = None
        shutil.rmtree(self.workdir, ignore_errors=True)
