import faulthandler, sys
import os, sys; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
faulthandler.enable()
print("step: import", flush=True)
import gpushare_amd._amdsmi as smi
print("step: available ->", smi.available(), smi.lib_path(), flush=True)
smi.init()
print("step: init ok", flush=True)
n = smi.device_count()
print("step: count", n, flush=True)
for i in range(n):
    info = smi.device_info(i)
    print("step: info", i, info, flush=True)
print("step: ecc", smi.ecc_count(0), flush=True)
from gpushare_amd.device import kfd_topology
topo = kfd_topology.resolve()
print("step: topo", {k: vars(v) for k, v in topo.items()}, flush=True)
import gpushare_amd._canary as canary
print("step: canary count", canary.device_count(), flush=True)
print("step: probe", canary.probe(0, 64, True), flush=True)
from gpushare_amd.device.amdsmi_source import AmdSmiSource
src = AmdSmiSource()
print("step: source", [vars(g) for g in src.devices()], flush=True)
print("DEBUG DONE", flush=True)
