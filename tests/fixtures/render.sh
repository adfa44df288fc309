#!/bin/sh
# consul-template stand-in: query consul for the healthy backend set,
# render the proxy's upstream file, and SIGHUP the proxy to reload it
# (reference: fixtures/nginx/nginx.conf.ctmpl + `nginx -s reload`).
set -e
python3 - "$CONSUL_ADDR" "$UPSTREAM_FILE" <<'PY'
import json
import sys
import urllib.request

addr, out = sys.argv[1], sys.argv[2]
url = "http://%s/v1/health/service/backend?passing=1" % addr
with urllib.request.urlopen(url, timeout=5) as resp:
    doc = json.load(resp)
ups = sorted("%s:%s" % (e["Service"]["Address"], e["Service"]["Port"])
             for e in doc)
with open(out, "w") as f:
    f.write("\n".join(ups) + ("\n" if ups else ""))
PY
kill -HUP "$CONTAINERPILOT_PROXY_PID"
