"""Run the stress shape with loop-stall + slow-dispatch probes enabled
and summarize where the reactor loses time (CPILOT_LOOP_DEBUG)."""
import collections
import json
import os
import re
import sys
import tempfile
import time

sys.path.insert(0, os.getcwd())
from containerpilot_amd import harness  # noqa: E402
from bench import stress_config, free_port, scrape, histogram_p99  # noqa: E402

jobs = int(sys.argv[1]) if len(sys.argv) > 1 else 300
wd = tempfile.mkdtemp()
port = free_port()
cfg = stress_config("localhost:79", port, jobs, 0, 100,
                    os.path.join(wd, "cp.socket"))
d = harness.Daemon(config_dict=cfg, workdir=wd,
                   env={"CPILOT_LOOP_DEBUG": "1", "CPILOT_SPAWN_DEBUG": "0"})

d.start()
d.wait_for_socket(timeout=60)
time.sleep(6)
marker = len(d.log())
s0 = scrape(port)
t0 = time.time()
time.sleep(15)
s1 = scrape(port)
el = time.time() - t0
pub = (s1["published"] - s0["published"]) / el
p99 = histogram_p99(s0["buckets"], s1["buckets"])
log = d.log()[marker:]
stalls = re.findall(r"loop stall: (\w+) fd=-?\d+ took ([0-9.]+) ms", log)
tstalls = re.findall(r"timer stall: id=\d+ interval_ms=(\d+) took ([0-9.]+) ms", log)
agg = collections.Counter()
mx = collections.defaultdict(float)
tot = collections.defaultdict(float)
for ph, ms in stalls:
    agg[ph] += 1
    mx[ph] = max(mx[ph], float(ms))
    tot[ph] += float(ms)
slow = re.findall(r"slow dispatch: \{(\w+) [^}]*\} waited ([0-9.]+) ms"
                  r" \(queue (\d+)\)", log)
rtts = [(float(a), float(b), float(c)) for a, b, c in re.findall(
    r"spawn rtt: \S+ total=([0-9.]+)ms spawn_phase=([0-9.]+)ms"
    r" exec_phase=([0-9.]+)ms", log)]
codes = collections.Counter(c for c, _, _ in slow)
worst = sorted((float(ms) for _, ms, _ in slow), reverse=True)[:10]
print(json.dumps({
    "jobs": jobs, "published_per_sec": round(pub),
    "p99_ms": round(p99 * 1000, 3) if p99 else None,
    "stall_counts": dict(agg), "stall_max_ms": dict(mx),
    "stall_total_ms": {k: round(v, 1) for k, v in tot.items()},
    "slow_dispatch_count": len(slow), "slow_codes": dict(codes),
    "slow_worst_ms": worst,
    "timer_stalls": len(tstalls),
    "timer_worst_ms": sorted((float(m) for _, m in tstalls))[-5:],
    "slow_rtts": len(rtts),
    "rtt_samples": sorted(rtts, reverse=True)[:8],
    "rtt_mean_spawn_phase": round(sum(r[1] for r in rtts) /
                                  max(1, len(rtts)), 1),
    "rtt_mean_exec_phase": round(sum(r[2] for r in rtts) /
                                 max(1, len(rtts)), 1),
    "loop_phases": re.findall(r"loop phases: .*", log)[-2:],
    "top_fds": re.findall(r"top fd .*", log)[-4:],
    "posted_items": re.findall(r"posted items: .*", log)[-2:],
    "item_tags": re.findall(r"item tag .*", log)[-3:],

}), flush=True)
d.cleanup()
