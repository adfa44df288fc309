"""Elimination matrix for the box RSS growth: which ingredient grows?"""
import sys, time, os, tempfile
sys.path.insert(0, os.getcwd())
from containerpilot_amd import harness
from containerpilot_amd.mockconsul import MockConsul
from bench import stress_config, free_port

def heap_kb(pid):
    # sum main-arena alloc via the daemon's own memdebug is indirect;
    # use VmRSS + VmData
    out = {}
    with open(f"/proc/{pid}/status") as f:
        for l in f:
            if l.startswith(("VmRSS", "VmData")):
                k, v = l.split()[0], int(l.split()[1])
                out[k.rstrip(':')] = v
    return out

def run(label, consul_addr, jobs=100, watches=50, health=True, secs=100):
    mc = None
    if consul_addr == "mock":
        mc = MockConsul().start()
        for i in range(watches):
            mc.set_health("upstream-%02d"%i, [{"ID":"u","Address":"1.2.3.4","Port":1}])
        consul_addr = mc.address
    wd = tempfile.mkdtemp()
    cfg = stress_config(consul_addr, free_port(), jobs, watches, 100, os.path.join(wd,"cp.socket"))
    if not health:
        for j in cfg["jobs"]:
            j.pop("health", None); j.pop("port", None)
    d = harness.Daemon(config_dict=cfg, workdir=wd)
    d.start(); d.wait_for_socket(); time.sleep(5)
    h0 = heap_kb(d.proc.pid); t0 = time.time()
    time.sleep(secs)
    h1 = heap_kb(d.proc.pid)
    dt = time.time() - t0
    print(f"{label}: rss {h0['VmRSS']}->{h1['VmRSS']} ({(h1['VmRSS']-h0['VmRSS'])/dt:.1f} KB/s) "
          f"data {h0['VmData']}->{h1['VmData']} ({(h1['VmData']-h0['VmData'])/dt:.1f} KB/s)")
    d.cleanup()
    if mc: mc.stop()

run("full(mock)", "mock")
run("closed-port", "localhost:79")
run("no-health", "localhost:79", health=False)
run("no-watch", "localhost:79", watches=0)
