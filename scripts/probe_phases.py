"""Per-step phase timing probe for large fleets (diagnoses step-time outliers)."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
from inferno_amd.core.system import System
from inferno_amd.utils.synthetic import make_fleet_spec, PoissonTrace
from inferno_amd.engine.fastpath import FastSweep

n_models = int(sys.argv[1]) if len(sys.argv) > 1 else 16384
spec = make_fleet_spec(n_models, seed=1234)
system, opt = System.from_spec(spec)
for a in system.accelerators.values(): a.calculate()
names = sorted(system.servers)
fs = FastSweep(system, names, backend="gpu", device="cuda")
trace = PoissonTrace(len(names), seed=1234 + 7)
n = len(names)
in_t = np.array([system.servers[x].load.avgInTokens for x in names], np.int32)
out_t = np.array([system.servers[x].load.avgOutTokens for x in names], np.int32)
pre = [trace.rates_at(s) for s in range(38)]
fs.load_override = (pre[0].astype(np.float32), in_t, out_t)
fs._reconcile_gpu()  # init device state + warm path
rows = []
for step in range(38):
    t0 = time.perf_counter()
    fs.load_override = (pre[step].astype(np.float32), in_t, out_t)
    arrs = fs._refresh_dynamic()
    t1 = time.perf_counter()
    fs._native_reconcile(arrs)
    torch.cuda.synchronize()
    t2 = time.perf_counter()
    of = fs._gpu["pin_out_f"].numpy(); oi = fs._gpu["pin_out_i"].numpy()
    w = (oi[0].copy(), of[0].copy())
    t3 = time.perf_counter()
    nz = int((pre[step] > 0).sum())
    rows.append((step, (t1-t0)*1e3, (t2-t1)*1e3, (t3-t2)*1e3, nz))
for r in rows:
    print(f"step {r[0]:2d} refresh {r[1]:7.2f} ms  gpu {r[2]:7.2f} ms  out {r[3]:5.2f} ms  nonzero {r[4]}")
