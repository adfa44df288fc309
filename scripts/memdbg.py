import sys, time, os, tempfile
sys.path.insert(0, os.getcwd())
from containerpilot_amd import harness
from containerpilot_amd.mockconsul import MockConsul
from bench import stress_config, free_port
def rss(pid):
    with open(f"/proc/{pid}/status") as f:
        return int([l for l in f if l.startswith("VmRSS")][0].split()[1])
mc = MockConsul().start()
for i in range(50):
    mc.set_health("upstream-%02d"%i, [{"ID":"u","Address":"1.2.3.4","Port":1}])
wd = tempfile.mkdtemp()
cfg = stress_config(mc.address, free_port(), 100, 50, 100, os.path.join(wd,"cp.socket"))
cfg["logging"]["level"] = "WARN"
d = harness.Daemon(config_dict=cfg, workdir=wd, env={"CPILOT_MEMDEBUG":"1"})
d.start(); d.wait_for_socket()
for i in range(12):
    time.sleep(10)
    print(f"t={10*(i+1)} rss={rss(d.proc.pid)}")
log = d.log()
d.cleanup(); mc.stop()
for line in log.splitlines():
    if "memdebug" in line:
        print(line.split("memdebug: ")[1])
