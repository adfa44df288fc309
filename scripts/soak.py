"""Stress soak with reloads + maintenance toggles: RSS/CPU stability
evidence. Duration via argv[1] seconds (default 300)."""
import json, os, sys, tempfile, time
sys.path.insert(0, os.getcwd())
from containerpilot_amd import harness
from containerpilot_amd.mockconsul import MockConsul
from bench import stress_config, free_port

mc = MockConsul().start()
for i in range(50):
    mc.set_health("upstream-%02d" % i, [{"ID": "u-%d" % i, "Address": "10.0.0.1", "Port": 1000+i}])
wd = tempfile.mkdtemp(prefix="soak-")
port = free_port()
cfg = stress_config(mc.address, port, 100, 50, 100, os.path.join(wd, "cp.socket"))
d = harness.Daemon(config_dict=cfg, workdir=wd)
d.start(); d.wait_for_socket()

def sample(pid):
    with open(f"/proc/{pid}/status") as f:
        rss = [l for l in f if l.startswith("VmRSS")][0].split()[1]
    with open(f"/proc/{pid}/stat") as f:
        parts = f.read().split()
    cpu = (int(parts[13]) + int(parts[14])) / os.sysconf("SC_CLK_TCK")
    return int(rss), cpu

duration = int(sys.argv[1]) if len(sys.argv) > 1 else 300
samples = []
t0 = time.time()
last_cpu = sample(d.proc.pid)[1]
reloads = 0
maint = 0
while time.time() - t0 < duration:
    time.sleep(10)
    rss, cpu = sample(d.proc.pid)
    samples.append({"t": round(time.time()-t0), "rss_kb": rss,
                    "cpu_pct": round((cpu-last_cpu)/10*100, 1)})
    last_cpu = cpu
    n = len(samples)
    if n % 10 == 0:  # reload every ~100s
        d.control("POST", "/v3/reload")
        reloads += 1
        time.sleep(1); d.wait_for_socket(timeout=20)
    elif n % 10 == 4:  # maintenance window every ~100s
        d.control("POST", "/v3/maintenance/enable")
        maint += 1
    elif n % 10 == 6:
        d.control("POST", "/v3/maintenance/disable")

alive = d.proc.poll() is None
d.terminate(); rc = d.wait(timeout=60)
mc.stop()
out = {"duration_s": duration, "reloads": reloads,
       "maintenance_windows": maint, "alive_throughout": alive,
       "clean_exit_rc": rc,
       "rss_kb_first": samples[0]["rss_kb"], "rss_kb_last": samples[-1]["rss_kb"],
       "rss_kb_max": max(s["rss_kb"] for s in samples),
       "cpu_pct_mean": round(sum(s["cpu_pct"] for s in samples)/len(samples), 1),
       "samples": samples}
with open("gpurun_out/soak.json", "w") as f:
    json.dump(out, f, indent=1)
print(json.dumps({k: v for k, v in out.items() if k != "samples"}))
