"""Consul wire-contract suite (golden transcripts).

The reference verifies its consul client against a real `consul agent
-dev` (discovery/test_server.go:18-91). This environment has no network
and no consul binary to vendor, so — as the fallback — this module
encodes Consul 1.0.0's DOCUMENTED wire contract for the four endpoints
the daemon uses and verifies both directions against it:

  outbound: every request the daemon makes is validated byte-level
    (method, path, query, headers, exact JSON key set/casing/types that
    api.AgentServiceRegistration et al. accept) by a strict agent that
    records any deviation as a violation;
  inbound: documented agent behaviors are replayed (403 ACL denial,
    500s with bodies, X-Consul-Index blocking semantics including index
    reset, chunked transfer-encoding as Go's net/http may produce) and
    the daemon must behave like the reference's client stack does.

Sources for the golden shapes: consul api AgentServiceRegistration /
AgentCheckRegistration marshaling, /v1/agent and /v1/health endpoint
docs for 1.0.0, and the reference's payloads
(discovery/service.go:93-110, consul.go:87-98).
"""

import json
import re
import socket
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from urllib.parse import parse_qs, urlparse


def wait_until(predicate, timeout=20.0, interval=0.1):
    deadline = time.time() + timeout
    while time.time() < deadline:
        if predicate():
            return True
        time.sleep(interval)
    return False


REGISTER_KEYS = {"ID", "Name", "Tags", "Port", "Address",
                 "EnableTagOverride", "Check", "Checks", "Kind", "Meta"}
CHECK_KEYS = {"CheckID", "Name", "TTL", "Status", "Notes",
              "DeregisterCriticalServiceAfter", "HTTP", "Interval",
              "Timeout", "Script", "TCP", "DockerContainerID", "Shell"}
VALID_STATUSES = {"passing", "warning", "critical"}


class StrictAgent:
    """Consul-1.0-shaped agent that records contract violations."""

    def __init__(self, acl_deny_first=0, chunked_health=False,
                 index_sequence=None):
        self.lock = threading.Lock()
        self.violations = []
        self.services = {}
        self.ttl_updates = []
        self.deregistered = []
        self.health = {}
        self.health_queries = []  # parsed query dicts, in order
        self.acl_deny_remaining = acl_deny_first
        self.chunked_health = chunked_health
        self.index_sequence = list(index_sequence or [])
        self.index = 5
        outer = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, fmt, *args):
                pass

            def violate(self, msg):
                with outer.lock:
                    outer.violations.append(
                        "%s %s: %s" % (self.command, self.path, msg))

            def _respond(self, code, body=b"",
                         extra_headers=(), chunked=False):
                self.send_response(code)
                self.send_header("Content-Type", "application/json")
                for k, v in extra_headers:
                    self.send_header(k, v)
                if chunked:
                    self.send_header("Transfer-Encoding", "chunked")
                    self.end_headers()
                    # split the body into several chunks like Go's
                    # net/http does for streamed responses
                    for i in range(0, len(body), 40):
                        chunk = body[i:i + 40]
                        self.wfile.write(b"%x\r\n%s\r\n"
                                         % (len(chunk), chunk))
                    self.wfile.write(b"0\r\n\r\n")
                else:
                    self.send_header("Content-Length", str(len(body)))
                    self.end_headers()
                    self.wfile.write(body)

            def _check_common(self):
                # HTTP/1.1 requires Host; the api client always sends it
                if not self.headers.get("Host"):
                    self.violate("missing Host header")

            def do_GET(self):
                self._check_common()
                parsed = urlparse(self.path)
                if not parsed.path.startswith("/v1/health/service/"):
                    self.violate("unexpected GET endpoint")
                    return self._respond(404, b"not found")
                qs = parse_qs(parsed.query)
                with outer.lock:
                    outer.health_queries.append(
                        {k: v[-1] for k, v in qs.items()})
                for key in qs:
                    if key not in ("passing", "tag", "dc", "index",
                                   "wait", "near", "stale", "consistent"):
                        self.violate("unknown query param %r" % key)
                if qs.get("passing", ["0"])[0] not in ("1", "true"):
                    self.violate("health query without passing filter")
                if "wait" in qs and not re.fullmatch(
                        r"\d+(s|m)", qs["wait"][0]):
                    self.violate("bad wait duration %r" % qs["wait"][0])
                if "index" in qs and not qs["index"][0].isdigit():
                    self.violate("non-numeric index %r" % qs["index"][0])
                name = parsed.path[len("/v1/health/service/"):]
                with outer.lock:
                    entries = list(outer.health.get(name, []))
                    if outer.index_sequence:
                        outer.index = outer.index_sequence.pop(0)
                    idx = outer.index
                body = json.dumps([
                    {"Node": {"Node": "strict"},
                     "Service": {"ID": e["ID"], "Service": name,
                                 "Address": e.get("Address", ""),
                                 "Port": e.get("Port", 0),
                                 "Tags": e.get("Tags", [])},
                     "Checks": []} for e in entries]).encode()
                self._respond(200, body,
                              extra_headers=[("X-Consul-Index", str(idx))],
                              chunked=outer.chunked_health)

            def do_PUT(self):
                self._check_common()
                parsed = urlparse(self.path)
                length = int(self.headers.get("Content-Length", "0"))
                raw = self.rfile.read(length) if length else b""
                with outer.lock:
                    deny = outer.acl_deny_remaining > 0
                    if deny:
                        outer.acl_deny_remaining -= 1
                if deny:
                    # documented ACL failure shape
                    return self._respond(403, b"Permission denied")

                if parsed.path == "/v1/agent/service/register":
                    return self._register(raw)
                if parsed.path.startswith("/v1/agent/check/update/"):
                    return self._ttl(parsed.path, raw)
                if parsed.path.startswith("/v1/agent/service/deregister/"):
                    if raw:
                        self.violate("deregister with non-empty body")
                    with outer.lock:
                        outer.deregistered.append(
                            parsed.path.rsplit("/", 1)[1])
                    return self._respond(200)
                self.violate("unexpected PUT endpoint")
                return self._respond(404, b"not found")

            def _register(self, raw):
                ctype = self.headers.get("Content-Type", "")
                if "application/json" not in ctype:
                    self.violate("register Content-Type %r" % ctype)
                try:
                    payload = json.loads(raw)
                except ValueError:
                    self.violate("register body is not JSON")
                    return self._respond(400, b"Request decode failed")
                for key in payload:
                    if key not in REGISTER_KEYS:
                        self.violate("unknown register key %r" % key)
                if not isinstance(payload.get("ID"), str) or \
                        not payload.get("ID"):
                    self.violate("register without string ID")
                if not isinstance(payload.get("Name"), str) or \
                        not payload.get("Name"):
                    self.violate("register without string Name")
                if not isinstance(payload.get("Port"), int):
                    self.violate("register Port must be int")
                if "Tags" in payload and not (
                        isinstance(payload["Tags"], list) and
                        all(isinstance(t, str) for t in payload["Tags"])):
                    self.violate("register Tags must be []string")
                if "EnableTagOverride" in payload and not isinstance(
                        payload["EnableTagOverride"], bool):
                    self.violate("EnableTagOverride must be bool")
                check = payload.get("Check")
                if not isinstance(check, dict):
                    self.violate("register without Check object")
                else:
                    for key in check:
                        if key not in CHECK_KEYS:
                            self.violate("unknown Check key %r" % key)
                    ttl = check.get("TTL", "")
                    if not re.fullmatch(r"\d+(\.\d+)?(ms|s|m|h)", ttl):
                        self.violate("Check.TTL %r not a duration" % ttl)
                    if "Status" in check and \
                            check["Status"] not in VALID_STATUSES:
                        self.violate("Check.Status %r invalid"
                                     % check["Status"])
                    if "DeregisterCriticalServiceAfter" in check and \
                            not re.fullmatch(
                                r"\d+(\.\d+)?(ms|s|m|h)",
                                check["DeregisterCriticalServiceAfter"]):
                        self.violate("DeregisterCriticalServiceAfter %r"
                                     % check[
                                         "DeregisterCriticalServiceAfter"])
                with outer.lock:
                    outer.services[payload.get("ID", "")] = payload
                return self._respond(200)

            def _ttl(self, path, raw):
                check_id = path[len("/v1/agent/check/update/"):]
                if not check_id.startswith("service:"):
                    self.violate("TTL update checkID %r (want service:<id>)"
                                 % check_id)
                try:
                    payload = json.loads(raw)
                except ValueError:
                    self.violate("TTL body is not JSON")
                    return self._respond(400, b"Request decode failed")
                for key in payload:
                    if key not in ("Status", "Output"):
                        self.violate("unknown TTL key %r" % key)
                if payload.get("Status") not in VALID_STATUSES:
                    self.violate("TTL Status %r invalid"
                                 % payload.get("Status"))
                with outer.lock:
                    outer.ttl_updates.append((check_id, payload))
                return self._respond(200)

        self.server = ThreadingHTTPServer(("127.0.0.1", 0), Handler)
        self.address = "127.0.0.1:%d" % self.server.server_address[1]
        self.thread = threading.Thread(target=self.server.serve_forever,
                                       daemon=True)

    def start(self):
        self.thread.start()
        return self

    def stop(self):
        self.server.shutdown()
        self.server.server_close()


def advertised_config(agent, socket_path_key=True, **overrides):
    cfg = {
        "consul": agent.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [{
            "name": "app", "exec": ["sleep", "60"], "port": 8900,
            "tags": ["prod", "blue"],
            "interfaces": ["static:127.0.0.1"],
            "health": {"exec": ["true"], "interval": 1, "ttl": 10},
            "consul": {"enableTagOverride": True,
                       "deregisterCriticalServiceAfter": "90m"},
        }],
    }
    cfg.update(overrides)
    return cfg


def test_register_ttl_deregister_transcript(daemon_factory):
    """The full advertised-job lifecycle produces only contract-clean
    requests: register (exact key set, duration-typed TTL, tags,
    EnableTagOverride, DeregisterCriticalServiceAfter), TTL pass
    updates against service:<id>, deregister on shutdown."""
    agent = StrictAgent().start()
    d = daemon_factory(advertised_config(agent)).start()
    try:
        d.wait_for_socket()
        assert wait_until(lambda: agent.services), d.log()[-2000:]
        (sid, reg), = agent.services.items()
        assert sid.startswith("app-")  # <name>-<hostname>
        assert reg["Name"] == "app"
        assert reg["Port"] == 8900
        assert reg["Address"] == "127.0.0.1"
        assert reg["Tags"] == ["prod", "blue"]
        assert reg["EnableTagOverride"] is True
        assert reg["Check"]["TTL"] == "10s"
        assert reg["Check"]["DeregisterCriticalServiceAfter"] == "90m"
        assert "set by containerpilot" in reg["Check"]["Notes"]
        assert wait_until(lambda: agent.ttl_updates)
        check_id, ttl = agent.ttl_updates[0]
        assert check_id == "service:" + sid
        assert ttl == {"Status": "passing", "Output": "ok"}
    finally:
        d.terminate()
        assert d.wait(timeout=30) == 0
    assert wait_until(lambda: sid in agent.deregistered)
    assert agent.violations == []
    agent.stop()


def test_watch_queries_transcript(daemon_factory):
    """Polled health queries carry passing=1 (+tag, +dc) and nothing
    else; tag/dc filters round-trip."""
    agent = StrictAgent().start()
    agent.health["db"] = [{"ID": "db1", "Address": "10.0.0.7",
                           "Port": 5432}]
    d = daemon_factory({
        "consul": agent.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [{"name": "main-app", "exec": ["sleep", "60"]}],
        "watches": [{"name": "db", "interval": 1, "tag": "primary",
                     "dc": "dc1"}],
    }).start()
    try:
        d.wait_for_socket()
        assert wait_until(lambda: len(agent.health_queries) >= 2)
        q = agent.health_queries[-1]
        assert q.get("passing") == "1"
        assert q.get("tag") == "primary"
        assert q.get("dc") == "dc1"
    finally:
        d.terminate()
        assert d.wait(timeout=30) == 0
    assert agent.violations == []
    agent.stop()


def test_acl_denied_registration_retries_then_succeeds(daemon_factory):
    """A 403 'Permission denied' (documented ACL failure) is survived:
    the daemon keeps running and registration succeeds once the ACL
    allows it — matching the reference's log-and-retry behavior."""
    agent = StrictAgent(acl_deny_first=3).start()
    cfg = advertised_config(agent)
    cfg["jobs"][0]["initial_status"] = "passing"
    d = daemon_factory(cfg).start()
    try:
        d.wait_for_socket()
        assert wait_until(lambda: agent.services, timeout=30), \
            d.log()[-2000:]
        assert agent.acl_deny_remaining == 0
        assert "registration failed" in d.log()
    finally:
        d.terminate()
        assert d.wait(timeout=30) == 0
    assert agent.violations == []
    agent.stop()


def test_blocking_query_index_echo_and_reset(daemon_factory):
    """Blocking watches implement the documented X-Consul-Index
    protocol: first query index=0, then echo the served index; when the
    agent's index goes BACKWARDS the client resets to 0 (consul's
    documented 'reset the index if it goes backwards' rule)."""
    agent = StrictAgent(index_sequence=[10, 12, 3, 7]).start()
    agent.health["db"] = [{"ID": "db1", "Address": "10.0.0.7",
                           "Port": 5432}]
    d = daemon_factory({
        "consul": agent.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [{"name": "main-app", "exec": ["sleep", "60"]}],
        "watches": [{"name": "db", "interval": 1, "blocking": True}],
    }).start()
    try:
        d.wait_for_socket()
        assert wait_until(lambda: len(agent.health_queries) >= 5,
                          timeout=30), agent.health_queries
        indexes = [int(q.get("index", "-1"))
                   for q in agent.health_queries[:5]]
        # first query: 0. served 10 -> echo 10; served 12 -> echo 12;
        # served 3 (backwards!) -> reset to 0; served 7 -> echo 7
        assert indexes[0] == 0, indexes
        assert indexes[1] == 10, indexes
        assert indexes[2] == 12, indexes
        assert indexes[3] == 0, indexes
        assert indexes[4] == 7, indexes
    finally:
        d.terminate()
        assert d.wait(timeout=30) == 0
    assert agent.violations == []
    agent.stop()


def test_chunked_health_response_parsed(daemon_factory, tmp_path):
    """Go's net/http can serve chunked bodies; the daemon's client must
    dechunk them. A chunked health response still produces a correct
    watch change event."""
    agent = StrictAgent(chunked_health=True).start()
    agent.health["db"] = [{"ID": "db1", "Address": "10.9.9.9",
                           "Port": 1234}]
    marker = tmp_path / "fired"
    d = daemon_factory({
        "consul": agent.address,
        "stopTimeout": 1,
        "logging": {"level": "DEBUG"},
        "jobs": [
            {"name": "main-app", "exec": ["sleep", "60"]},
            {"name": "onchange",
             "exec": ["sh", "-c", "echo fired >> %s" % marker],
             "when": {"source": "watch.db", "each": "changed"}},
        ],
        "watches": [{"name": "db", "interval": 1}],
    }).start()
    try:
        d.wait_for_socket()
        # change fires once the (chunked) health set is parsed
        assert wait_until(lambda: marker.exists(), timeout=20), \
            d.log()[-2000:]
    finally:
        d.terminate()
        assert d.wait(timeout=30) == 0
    assert agent.violations == []
    agent.stop()
