"""An in-process mock of the Consul agent HTTP API.

Implements exactly the endpoints the daemon uses (the same four the
reference calls through the official client — see SURVEY.md §2 #14-15):

  PUT /v1/agent/service/register
  PUT /v1/agent/check/update/<checkID>
  PUT /v1/agent/service/deregister/<serviceID>
  GET /v1/health/service/<name>?passing=1[&tag=..][&dc=..]

Tests drive upstream-change scenarios by mutating `health` and assert on
`services` / `ttl_updates` / request history. The reference runs a real
`consul agent -dev` in its API tests; no consul binary exists in this
environment so the wire protocol is faked at the HTTP layer instead.
"""

import json
import threading
import time
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer
from urllib.parse import urlparse, parse_qs


class MockConsul:
    def __init__(self, port=0, certfile=None, keyfile=None):
        self.lock = threading.Lock()
        self.services = {}      # id -> registration payload
        self.ttl_updates = []   # (checkID, payload)
        self.deregistered = []  # serviceIDs
        self.requests = []      # (method, path)
        self.health = {}        # service name -> list of dicts(ID,Address,Port)
        self.tokens = []        # X-Consul-Token header values seen
        self.fail_mode = False  # when True: every endpoint returns 500
        self.index = 1          # consul-style modify index (global)
        self.changed = threading.Condition(self.lock)

        outer = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, fmt, *args):
                pass

            def _respond(self, code, body=b"", ctype="application/json"):
                self.send_response(code)
                self.send_header("Content-Type", ctype)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_GET(self):
                parsed = urlparse(self.path)
                if outer.fail_mode:
                    return self._respond(500, b"boom")
                with outer.lock:
                    outer.requests.append(("GET", self.path))
                    tok = self.headers.get("X-Consul-Token")
                    if tok:
                        outer.tokens.append(tok)
                if parsed.path.startswith("/v1/health/service/"):
                    name = parsed.path[len("/v1/health/service/"):]
                    qs = parse_qs(parsed.query)
                    tag = qs.get("tag", [None])[0]
                    want_index = int(qs.get("index", ["0"])[0])
                    wait_s = 0
                    if "wait" in qs:
                        wait_s = int(qs["wait"][0].rstrip("s") or 0)
                    with outer.lock:
                        # consul blocking query: hold until the index
                        # advances past the caller's or the wait elapses
                        if want_index and wait_s:
                            deadline = time.time() + wait_s
                            while (outer.index <= want_index and
                                   time.time() < deadline):
                                outer.changed.wait(
                                    max(0.05, deadline - time.time()))
                        entries = list(outer.health.get(name, []))
                        current_index = outer.index
                    if tag is not None:
                        entries = [e for e in entries
                                   if tag in e.get("Tags", [])]
                    body = json.dumps([
                        {"Node": {"Node": "mock"},
                         "Service": {"ID": e["ID"],
                                     "Service": name,
                                     "Address": e.get("Address", ""),
                                     "Port": e.get("Port", 0),
                                     "Tags": e.get("Tags", [])},
                         "Checks": []}
                        for e in entries]).encode()
                    self.send_response(200)
                    self.send_header("Content-Type", "application/json")
                    self.send_header("X-Consul-Index", str(current_index))
                    self.send_header("Content-Length", str(len(body)))
                    self.end_headers()
                    self.wfile.write(body)
                    return
                if parsed.path == "/v1/agent/self":
                    return self._respond(200, b'{"Config":{}}')
                return self._respond(404, b"not found")

            def do_PUT(self):
                parsed = urlparse(self.path)
                if outer.fail_mode:
                    return self._respond(500, b"boom")
                length = int(self.headers.get("Content-Length", 0))
                raw = self.rfile.read(length) if length else b""
                payload = json.loads(raw) if raw else {}
                with outer.lock:
                    outer.requests.append(("PUT", parsed.path))
                    tok = self.headers.get("X-Consul-Token")
                    if tok:
                        outer.tokens.append(tok)
                    if parsed.path == "/v1/agent/service/register":
                        outer.services[payload.get("ID", "")] = payload
                        return self._respond(200)
                    if parsed.path.startswith("/v1/agent/check/update/"):
                        check_id = parsed.path[
                            len("/v1/agent/check/update/"):]
                        outer.ttl_updates.append((check_id, payload))
                        return self._respond(200)
                    if parsed.path.startswith(
                            "/v1/agent/service/deregister/"):
                        sid = parsed.path[
                            len("/v1/agent/service/deregister/"):]
                        outer.deregistered.append(sid)
                        outer.services.pop(sid, None)
                        return self._respond(200)
                return self._respond(404, b"not found")

            do_POST = do_PUT

        self.server = ThreadingHTTPServer(("127.0.0.1", port), Handler)
        if certfile:
            import ssl
            ctx = ssl.SSLContext(ssl.PROTOCOL_TLS_SERVER)
            ctx.load_cert_chain(certfile, keyfile)
            self.server.socket = ctx.wrap_socket(self.server.socket,
                                                 server_side=True)
        self.port = self.server.server_address[1]
        self.thread = threading.Thread(target=self.server.serve_forever,
                                       daemon=True)

    @property
    def address(self):
        return "127.0.0.1:%d" % self.port

    def start(self):
        self.thread.start()
        return self

    def set_health(self, service, entries):
        with self.lock:
            self.health[service] = entries
            self.index += 1
            self.changed.notify_all()

    def stop(self):
        self.server.shutdown()
        self.server.server_close()
