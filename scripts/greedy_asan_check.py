#!/usr/bin/env python3
"""AddressSanitizer pass over the native greedy solver (host C++).

Compiles ops/native/greedy.cpp standalone with g++ -fsanitize=address and
runs randomized differential workloads through it via ctypes. Run with ASAN
preloaded so the runtime intercepts allocations:

    LD_PRELOAD=$(g++ -print-file-name=libasan.so) \
        python3 scripts/greedy_asan_check.py

(`make asan-greedy` wraps exactly that.) Exit 0 = no ASAN reports and all
decisions match the Python golden.
"""
import ctypes
import os
import subprocess
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SRC = os.path.join(REPO, "inferno_amd", "ops", "native", "greedy.cpp")


def build_asan_lib() -> str:
    out = os.path.join(tempfile.mkdtemp(prefix="wva-asan-"), "libgreedy_asan.so")
    subprocess.run(
        ["g++", "-O1", "-g", "-std=c++17", "-shared", "-fPIC",
         "-fsanitize=address", "-fno-omit-frame-pointer", SRC, "-o", out],
        check=True,
    )
    return out


def main() -> int:
    if "asan" not in (os.environ.get("LD_PRELOAD") or ""):
        # re-exec with ASAN preloaded (needed because python isn't ASAN-built)
        libasan = subprocess.run(
            ["g++", "-print-file-name=libasan.so"], capture_output=True, text=True
        ).stdout.strip()
        env = dict(os.environ, LD_PRELOAD=libasan,
                   ASAN_OPTIONS="detect_leaks=0")  # python itself "leaks"
        return subprocess.run([sys.executable, __file__], env=env).returncode

    lib = ctypes.CDLL(build_asan_lib())
    lib.wva_greedy_solve.restype = ctypes.c_int

    from inferno_amd.config import SaturationPolicy
    from inferno_amd.core.system import System
    from inferno_amd.engine import SweepEngine
    from inferno_amd.solver.greedy import solve_greedy
    from tests.fixtures import make_spec
    from tests.test_native_greedy import cells_from_cpu_sweep

    def p(a):
        return a.ctypes.data_as(ctypes.c_void_p)

    mismatches = 0
    n_checked = 0
    for seed in range(12):
        rng = np.random.default_rng(seed)
        policy = ["None", "PriorityExhaustive", "PriorityRoundRobin",
                  "RoundRobin"][seed % 4]
        cap = {t: int(rng.integers(0, 40)) for t in
               ("AMD-MI300X-192GB", "AMD-MI325X-256GB", "AMD-MI355X-288GB")}
        kw = dict(n_servers=int(rng.integers(2, 24)), seed=3000 + seed,
                  unlimited=False, capacity=dict(cap), saturation_policy=policy,
                  delayed_best_effort=bool(seed % 2))
        a, opt = System.from_spec(make_spec(**kw))
        b, _ = System.from_spec(make_spec(**kw))
        names = sorted(a.servers)
        SweepEngine(backend="cpu").sweep(b)
        solve_greedy(b, delayed_best_effort=opt.delayedBestEffort,
                     saturation_policy=SaturationPolicy.parse(opt.saturationPolicy))

        fs, cells = cells_from_cpu_sweep(a, names)
        # build the SoA inputs the way dist.py does
        type_names = sorted({acc.type for acc in a.accelerators.values()}
                            | set(a.capacity))
        type_index = {t: i for i, t in enumerate(type_names)}
        feas = cells["feasible"].astype(bool)
        idx = np.nonzero(feas)[0]
        value = cells["value"]
        srv_of = cells["cell_server"]
        order = idx[np.lexsort((value[idx].astype(np.float64), srv_of[idx]))]
        n_srv = len(names)
        counts = np.bincount(srv_of[order], minlength=n_srv)
        seg = np.zeros(n_srv + 1, dtype=np.int32)
        np.cumsum(counts, out=seg[1:])
        units = np.empty(fs.n_cells, np.int32)
        tidx = np.empty(fs.n_cells, np.int32)
        for k in range(fs.n_cells):
            srv = fs._srv_objs[fs.cell_server[k]]
            acc = a.accelerators[fs.acc_names[fs.cell_acc_idx[k]]]
            units[k] = a.models[srv.model_name].get_num_instances(acc.name) * acc.multiplicity
            tidx[k] = type_index[acc.type]
        cand_value = np.ascontiguousarray(value[order], np.float32)
        cand_tidx = np.where(cells["zero_empty"][order].astype(bool), -1,
                             tidx[order]).astype(np.int32)
        cand_units = np.ascontiguousarray(units[order], np.int32)
        cand_reps = np.ascontiguousarray(cells["num_replicas"][order], np.int32)
        capacity = np.array([int(a.capacity.get(t, 0)) for t in type_names],
                            np.int32)
        prio = np.array([s.priority(a) for s in fs._srv_objs], np.int32)
        pol = {"None": 0, "PriorityExhaustive": 1, "PriorityRoundRobin": 2,
               "RoundRobin": 3}[policy]
        out_cand = np.full(n_srv, -1, np.int32)
        out_reps = np.zeros(n_srv, np.int32)
        rc = lib.wva_greedy_solve(
            ctypes.c_int(n_srv), ctypes.c_int(len(capacity)), p(cand_value),
            p(cand_tidx), p(cand_units), p(cand_reps), p(seg), p(prio),
            p(capacity), ctypes.c_int(1 if opt.delayedBestEffort else 0),
            ctypes.c_int(pol), p(out_cand), p(out_reps))
        assert rc == 0
        for s_i, name in enumerate(names):
            gb = b.servers[name].allocation
            n_checked += 1
            if out_cand[s_i] < 0:
                if gb is not None and gb.accelerator != "":
                    mismatches += 1
                continue
            cell = int(order[out_cand[s_i]])
            acc_key = fs.acc_names[int(fs.cell_acc_idx[cell])]
            if gb is None or gb.accelerator != acc_key or \
                    gb.num_replicas != int(out_reps[s_i]):
                mismatches += 1
    print(f"ASAN greedy check: {n_checked} decisions, {mismatches} mismatches")
    return 0 if mismatches == 0 else 1


if __name__ == "__main__":
    raise SystemExit(main())
