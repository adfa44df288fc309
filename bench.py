#!/usr/bin/env python3
"""Benchmark the containerpilot daemon on the BASELINE.json stress config:
100 jobs with 100 ms health checks + 50 Consul watches, measuring bus
event throughput and publish->dispatch latency.

Tier note (BASELINE.json "north_star"): the reference
(TritonDataCenter/containerpilot) is a CPU-only process-supervisor daemon
with no GPU code path, so this benchmark exercises the daemon on the host
CPUs of the node. --gpus N runs N independent daemon instances (one per
rank, weak scaling) so the multi-rank contract still holds; ranks
coordinate over torch.distributed gloo when launched via torchrun.

A "step" is one second of sustained stress-load operation. After W warmup
seconds, exactly K timed seconds are measured by scraping the daemon's own
Prometheus endpoint (containerpilot_events / containerpilot_event_deliveries
counters and the containerpilot_event_dispatch_seconds histogram) at the
window edges.

The headline value is PUBLISHED events/sec — health-check and watch
events entering the bus (events/bus.go:125-140) — aggregated over all
ranks, compared like-for-like against the BASELINE >=10k events/sec
target. Bus deliveries/sec (published x subscriber fan-out) and the p99
publish->dispatch latency are reported alongside in config; the <1 ms
p99 target from BASELINE.md applies to the latter, and
check_completion_pct shows whether the daemon kept up with the
configured check rate (a number achieved by shedding checks would be
hollow).

The default shape is the BASELINE stress config scaled 5.5x (550 jobs
x 100 ms health + 50 watches, an 11k events/s structural rate): the
original 100-job shape structurally caps published events at ~2k/s
(100 jobs x 10 checks/s x 2 events), so it cannot demonstrate the 10k
target no matter how fast the daemon is; 550 leaves ~10%% headroom over
the target. The 100-job shape and the full capacity curve (ceiling
~20k ev/s at 1200 jobs) are in profiles/capacity.md.
"""

import argparse
import json
import os
import re
import socket
import sys
import tempfile
import time
import urllib.request

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from containerpilot_amd import harness
from containerpilot_amd.mockconsul import MockConsul

BASELINE_EVENTS_PER_SEC = 10000.0  # >=10k events/sec target


def free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def stress_config(consul_addr, telemetry_port, n_jobs, n_watches,
                  check_ms, socket_path):
    jobs = []
    for i in range(n_jobs):
        jobs.append({
            "name": "svc-%03d" % i,
            "exec": ["sleep", "3600"],
            "port": 20000 + i,
            "interfaces": ["static:127.0.0.1"],
            "health": {
                "exec": ["true"],
                "interval": "%dms" % check_ms,
                "ttl": 10,
            },
        })
    watches = [{"name": "upstream-%02d" % i, "interval": 1}
               for i in range(n_watches)]
    return {
        "consul": consul_addr,
        "stopTimeout": 1,
        "logging": {"level": "ERROR"},
        "control": {"socket": socket_path},
        "jobs": jobs,
        "watches": watches,
        "telemetry": {"port": telemetry_port,
                      "interfaces": ["static:127.0.0.1"]},
    }


def scrape(port):
    url = "http://127.0.0.1:%d/metrics" % port
    with urllib.request.urlopen(url, timeout=10) as resp:
        text = resp.read().decode()
    published = 0.0
    for m in re.finditer(r'^containerpilot_events\{[^}]*\} ([0-9.e+]+)$',
                         text, re.M):
        published += float(m.group(1))
    delivered = 0.0
    m = re.search(r'^containerpilot_event_deliveries ([0-9.e+]+)$', text, re.M)
    if m:
        delivered = float(m.group(1))
    buckets = {}
    for m in re.finditer(
            r'^containerpilot_event_dispatch_seconds_bucket\{le="([^"]+)"\} '
            r'([0-9.e+]+)$', text, re.M):
        buckets[m.group(1)] = float(m.group(2))
    return {"published": published, "delivered": delivered,
            "buckets": buckets}


def histogram_p99(b0, b1):
    """p99 over the [t0, t1] window from cumulative bucket deltas."""
    deltas = []
    for le, count in sorted(b1.items(),
                            key=lambda kv: float("inf")
                            if kv[0] == "+Inf" else float(kv[0])):
        deltas.append((le, count - b0.get(le, 0.0)))
    if not deltas:
        return None
    total = deltas[-1][1]
    if total <= 0:
        return None
    target = 0.99 * total
    for le, cum in deltas:
        if cum >= target:
            return float("inf") if le == "+Inf" else float(le)
    return None


def cgroup_cpu_info():
    """(quota_cores, throttled_usec) from cgroup v2, or (None, None)."""
    quota = None
    throttled = None
    try:
        with open("/sys/fs/cgroup/cpu.max") as f:
            parts = f.read().split()
        if parts[0] != "max":
            quota = round(float(parts[0]) / float(parts[1]), 1)
    except (OSError, ValueError, IndexError):
        pass
    try:
        with open("/sys/fs/cgroup/cpu.stat") as f:
            for line in f:
                if line.startswith("throttled_usec"):
                    throttled = int(line.split()[1])
    except (OSError, ValueError):
        pass
    return quota, throttled


def wait_http(port, timeout=30):
    deadline = time.time() + timeout
    while time.time() < deadline:
        try:
            scrape(port)
            return True
        except OSError:
            time.sleep(0.1)
    return False


def run_rank(args, rank):
    consul = MockConsul().start()
    # watches see a small stable upstream set
    for i in range(args.watches):
        consul.set_health("upstream-%02d" % i,
                          [{"ID": "u-%d" % i, "Address": "10.0.0.1",
                            "Port": 1000 + i}])
    workdir = tempfile.mkdtemp(prefix="cpilot-bench-r%d-" % rank)
    telemetry_port = free_port()
    cfg = stress_config(consul.address, telemetry_port, args.jobs,
                        args.watches, args.check_ms,
                        os.path.join(workdir, "cp.socket"))
    daemon = harness.Daemon(config_dict=cfg, workdir=workdir)
    daemon.start()
    daemon.wait_for_socket(timeout=30)
    if not wait_http(telemetry_port):
        raise RuntimeError("telemetry endpoint never came up:\n" +
                           daemon.log()[-4000:])
    return consul, daemon, telemetry_port


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=30,
                        help="timed seconds of stress operation")
    parser.add_argument("--warmup", type=int, default=5)
    parser.add_argument("--jobs", type=int, default=550)
    parser.add_argument("--watches", type=int, default=50)
    parser.add_argument("--check-ms", type=int, default=100)
    args = parser.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    distributed = world_size > 1

    import torch
    import torch.distributed as dist
    if distributed:
        # the workload is host-side; gloo coordinates ranks on any box
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        dist.init_process_group(backend="gloo", rank=rank,
                                world_size=world_size)

    harness.build()
    consul, daemon, port = run_rank(args, rank)
    try:
        # warmup
        time.sleep(args.warmup)

        def sync():
            if distributed:
                dist.barrier()
            if torch.cuda.is_available():
                torch.cuda.synchronize()

        sync()
        quota_cores, throttled0 = cgroup_cpu_info()
        t0 = time.monotonic()
        s0 = scrape(port)
        time.sleep(args.steps)
        sync()
        s1 = scrape(port)
        t1 = time.monotonic()
        _, throttled1 = cgroup_cpu_info()
        throttled_ms = (None if throttled0 is None or throttled1 is None
                        else round((throttled1 - throttled0) / 1000.0, 1))

        elapsed = t1 - t0
        published = (s1["published"] - s0["published"]) / elapsed
        delivered = (s1["delivered"] - s0["delivered"]) / elapsed
        p99 = histogram_p99(s0["buckets"], s1["buckets"])
        p99_ms = None if p99 is None else p99 * 1e3

        local = torch.tensor(
            [delivered, published, elapsed, p99_ms or 0.0],
            dtype=torch.float64)
        if distributed:
            gathered = [torch.zeros_like(local) for _ in range(world_size)]
            dist.all_gather(gathered, local)
        else:
            gathered = [local]

        if rank == 0:
            total_delivered = sum(float(g[0]) for g in gathered)
            total_published = sum(float(g[1]) for g in gathered)
            max_elapsed = max(float(g[2]) for g in gathered)
            worst_p99_ms = max(float(g[3]) for g in gathered)
            target_checks = args.jobs * (1000.0 / args.check_ms)
            completion_pct = min(
                100.0, 100.0 * total_published /
                (2.0 * target_checks * max(1, world_size
                                           if distributed else 1)))
            result = {
                "metric": "events/sec",
                "value": round(total_published, 1),
                "unit": "events/s",
                "n_gpus": world_size if distributed else args.gpus,
                "steps": args.steps,
                "warmup": args.warmup,
                "ms_per_step": round(max_elapsed / args.steps * 1000, 3),
                "higher_is_better": True,
                "scaling": "weak",
                "vs_baseline": round(
                    total_published / BASELINE_EVENTS_PER_SEC, 3),
                "dtype": "n/a (CPU daemon, no tensor math; see BASELINE.json north_star)",
                "data": "synthetic (stress config: sleep jobs + /bin/true checks + mock consul)",
                "config": {
                    "model": ("containerpilot daemon, BASELINE stress "
                              "config" if args.jobs == 100 else
                              "containerpilot daemon, BASELINE stress "
                              "shape scaled to %d jobs" % args.jobs),
                    "jobs": args.jobs,
                    "health_check_interval_ms": args.check_ms,
                    "watches": args.watches,
                    "global_batch": None,
                    "seq_len": None,
                    "parallelism": "%d independent daemons (1/rank)"
                                   % (world_size if distributed else args.gpus),
                    "published_events_per_sec": round(total_published, 1),
                    "delivered_events_per_sec": round(total_delivered, 1),
                    "check_completion_pct": round(completion_pct, 1),
                    "p99_dispatch_ms": round(worst_p99_ms, 4),
                    # cgroup CPU budget + throttling during the window:
                    # a quota-throttled run (shared slice too small for
                    # N ranks of real forked checks) self-documents here
                    "cgroup_cpu_quota_cores": quota_cores,
                    "cgroup_throttled_ms_in_window": throttled_ms,
                    "p99_target_ms": 1.0,
                    "events_per_sec_target": BASELINE_EVENTS_PER_SEC,
                },
            }
            print(json.dumps(result))
    finally:
        daemon.cleanup()
        consul.stop()
        if distributed:
            dist.destroy_process_group()


if __name__ == "__main__":
    main()
