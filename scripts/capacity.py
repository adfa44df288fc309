"""Single-daemon capacity probe: scale the stress shape up (more jobs =
more real fork/exec health checks per second) and measure sustained
rates + windowed p99 dispatch latency."""
import sys, time, os, tempfile
sys.path.insert(0, os.getcwd())
from containerpilot_amd import harness
from bench import stress_config, free_port, scrape, histogram_p99

for jobs in (100, 300, 500):
    wd = tempfile.mkdtemp()
    port = free_port()
    cfg = stress_config("localhost:79", port, jobs, 0, 100,
                        os.path.join(wd, "cp.socket"))
    d = harness.Daemon(config_dict=cfg, workdir=wd)
    d.start(); d.wait_for_socket(timeout=60); time.sleep(8)
    s0 = scrape(port); t0 = time.time()
    time.sleep(20)
    s1 = scrape(port); el = time.time() - t0
    pub = (s1["published"] - s0["published"]) / el
    dlv = (s1["delivered"] - s0["delivered"]) / el
    p99 = histogram_p99(s0["buckets"], s1["buckets"])
    import json
    print(json.dumps({"jobs": jobs, "checks_per_sec_target": jobs * 10,
                      "published_per_sec": round(pub),
                      "delivered_per_sec": round(dlv),
                      "p99_dispatch_ms": round(p99 * 1000, 3)}))
    d.cleanup()
