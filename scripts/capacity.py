"""Single-daemon capacity probe: scale the stress shape up (more jobs =
more real fork/exec health checks per second) and measure sustained
rates + windowed p99 dispatch latency.

Usage: python3 scripts/capacity.py [jobs ...]   (default: 100 300 500)
Env: CPILOT_CAPACITY_WATCHES (default 0), CPILOT_CAPACITY_WINDOW (20s)
"""
import json
import os
import sys
import tempfile
import time

sys.path.insert(0, os.getcwd())
from containerpilot_amd import harness  # noqa: E402
from bench import stress_config, free_port, scrape, histogram_p99  # noqa: E402

job_counts = [int(a) for a in sys.argv[1:]] or [100, 300, 500]
watches = int(os.environ.get("CPILOT_CAPACITY_WATCHES", "0"))
window = int(os.environ.get("CPILOT_CAPACITY_WINDOW", "20"))

for jobs in job_counts:
    wd = tempfile.mkdtemp()
    port = free_port()
    cfg = stress_config("localhost:79", port, jobs, watches, 100,
                        os.path.join(wd, "cp.socket"))
    d = harness.Daemon(config_dict=cfg, workdir=wd)
    d.start()
    d.wait_for_socket(timeout=60)
    time.sleep(8)
    s0 = scrape(port)
    t0 = time.time()
    time.sleep(window)
    s1 = scrape(port)
    el = time.time() - t0
    pub = (s1["published"] - s0["published"]) / el
    dlv = (s1["delivered"] - s0["delivered"]) / el
    p99 = histogram_p99(s0["buckets"], s1["buckets"])
    print(json.dumps({
        "jobs": jobs,
        "checks_per_sec_target": jobs * 10,
        "published_per_sec": round(pub),
        "completion_pct": round(100.0 * pub / (2 * jobs * 10), 1),
        "delivered_per_sec": round(dlv),
        "p99_dispatch_ms": round(p99 * 1000, 3) if p99 else None,
    }), flush=True)
    d.cleanup()
