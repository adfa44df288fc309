"""Long soak with fault windows: RSS/CPU stability + recovery evidence.

Faults injected while the stress shape runs (100 jobs x 100ms checks +
50 watches): periodic reloads, maintenance toggles, mock-consul
outage/recovery windows, and SIGUSR1 log rotation. Asserts flat RSS
(fitted slope), zero stuck jobs at the end (published-event rate in the
final window still at the structural rate), and a clean exit.

Usage: python3 scripts/soak.py [seconds] [outfile] [jobs]
"""
import json
import os
import signal
import sys
import tempfile
import time

sys.path.insert(0, os.getcwd())
from containerpilot_amd import harness  # noqa: E402
from containerpilot_amd.mockconsul import MockConsul  # noqa: E402
from bench import stress_config, free_port, scrape  # noqa: E402


def set_upstreams(mc):
    for i in range(50):
        mc.set_health("upstream-%02d" % i,
                      [{"ID": "u-%d" % i, "Address": "10.0.0.1",
                        "Port": 1000 + i}])


duration = int(sys.argv[1]) if len(sys.argv) > 1 else 300
outfile = sys.argv[2] if len(sys.argv) > 2 else "gpurun_out/soak.json"
n_jobs = int(sys.argv[3]) if len(sys.argv) > 3 else 100

mc = MockConsul().start()
consul_port = int(mc.address.split(":")[1])
set_upstreams(mc)
wd = tempfile.mkdtemp(prefix="soak-")
port = free_port()
cfg = stress_config(mc.address, port, n_jobs, 50, 100,
                    os.path.join(wd, "cp.socket"))
log_file = os.path.join(wd, "cp.log")
cfg["logging"] = {"level": "ERROR", "output": log_file}
d = harness.Daemon(config_dict=cfg, workdir=wd)
d.start()
d.wait_for_socket()


def sample(pid):
    with open("/proc/%d/status" % pid) as f:
        rss = [ln for ln in f if ln.startswith("VmRSS")][0].split()[1]
    with open("/proc/%d/stat" % pid) as f:
        parts = f.read().split()
    cpu = (int(parts[13]) + int(parts[14])) / os.sysconf("SC_CLK_TCK")
    return int(rss), cpu


samples = []
t0 = time.time()
last_cpu = sample(d.proc.pid)[1]
reloads = maint = outages = rotations = 0
outage_until = 0.0

while time.time() - t0 < duration:
    time.sleep(10)
    now = time.time()
    rss, cpu = sample(d.proc.pid)
    samples.append({"t": round(now - t0), "rss_kb": rss,
                    "cpu_pct": round((cpu - last_cpu) / 10 * 100, 1)})
    last_cpu = cpu
    n = len(samples)

    if outage_until and now >= outage_until:
        # recovery: bring the agent back on the same port
        mc = MockConsul(port=consul_port).start()
        set_upstreams(mc)
        outage_until = 0.0

    if n % 10 == 0:  # reload every ~100s
        d.control("POST", "/v3/reload")
        reloads += 1
        time.sleep(1)
        d.wait_for_socket(timeout=20)
    elif n % 10 == 4:  # maintenance window every ~100s
        d.control("POST", "/v3/maintenance/enable")
        maint += 1
    elif n % 10 == 6:
        d.control("POST", "/v3/maintenance/disable")
    elif n % 15 == 8 and not outage_until:  # consul outage ~20s
        mc.stop()
        outages += 1
        outage_until = now + 20
    elif n % 20 == 12:  # logrotate: rename + SIGUSR1 reopen
        try:
            os.rename(log_file, log_file + ".1")
        except OSError:
            pass
        d.signal(signal.SIGUSR1)
        rotations += 1

# end-of-run health: leave any open fault windows (maintenance mode,
# consul outage) before measuring the steady rate
d.control("POST", "/v3/maintenance/disable")
if outage_until:
    mc = MockConsul(port=consul_port).start()
    set_upstreams(mc)
    outage_until = 0.0
time.sleep(3)
# daemon must still publish at the structural rate: zero stuck jobs
s0 = scrape(port)
time.sleep(10)
s1 = scrape(port)
final_rate = (s1["published"] - s0["published"]) / 10.0
rotated_log_grows = os.path.exists(log_file)

alive = d.proc.poll() is None
d.terminate()
rc = d.wait(timeout=60)
mc.stop()

# RSS slope (KB/s) over the post-warmup samples, least squares
post = [s for s in samples if s["t"] > 60]
if len(post) >= 2:
    xs = [s["t"] for s in post]
    ys = [s["rss_kb"] for s in post]
    mx = sum(xs) / len(xs)
    my = sum(ys) / len(ys)
    denom = sum((x - mx) ** 2 for x in xs) or 1.0
    slope = sum((x - mx) * (y - my) for x, y in zip(xs, ys)) / denom
else:
    slope = 0.0

out = {"duration_s": duration, "reloads": reloads,
       "maintenance_windows": maint, "consul_outages": outages,
       "log_rotations": rotations,
       "alive_throughout": alive, "clean_exit_rc": rc,
       "final_published_per_sec": round(final_rate, 1),
       "expected_published_per_sec": n_jobs * 20,
       "log_reopened_after_rotation": rotated_log_grows,
       "rss_kb_first": samples[0]["rss_kb"],
       "rss_kb_last": samples[-1]["rss_kb"],
       "rss_kb_max": max(s["rss_kb"] for s in samples),
       "rss_slope_kb_per_s": round(slope, 3),
       "cpu_pct_mean": round(sum(s["cpu_pct"] for s in samples) /
                             len(samples), 1),
       "samples": samples}
os.makedirs(os.path.dirname(outfile) or ".", exist_ok=True)
with open(outfile, "w") as f:
    json.dump(out, f, indent=1)
print(json.dumps({k: v for k, v in out.items() if k != "samples"}))
