#!/usr/bin/env python3
"""Reloadable stand-in for nginx in the autopilot integration scenario
(reference: integration_tests/fixtures/nginx + test_discovery_consul).

Loads the upstream list from its config file ONCE at startup and again
on SIGHUP — never per request — so GET /upstreams proves the process
actually re-read its configuration after a reload signal.
"""
import json
import signal
import sys
from http.server import BaseHTTPRequestHandler, HTTPServer

PORT = int(sys.argv[1])
UPSTREAM_FILE = sys.argv[2]
STATE = {"upstreams": [], "reloads": 0}


def load(*_):
    try:
        with open(UPSTREAM_FILE) as f:
            STATE["upstreams"] = [ln.strip() for ln in f if ln.strip()]
    except OSError:
        STATE["upstreams"] = []
    STATE["reloads"] += 1


signal.signal(signal.SIGHUP, load)
load()


class Handler(BaseHTTPRequestHandler):
    def do_GET(self):
        body = json.dumps(STATE).encode()
        self.send_response(200)
        self.send_header("Content-Type", "application/json")
        self.send_header("Content-Length", str(len(body)))
        self.end_headers()
        self.wfile.write(body)

    def log_message(self, *args):
        pass


HTTPServer(("127.0.0.1", PORT), Handler).serve_forever()
