#!/usr/bin/env python3
"""GPU differential soak: many randomized fleets, CPU golden vs HIP sweep.

Broader than the pytest differential suite (which keeps seeds fixed and
small for CI time): sweeps fleet shapes, SLO regimes, load scales and
token distributions, and reports mismatch statistics. Run on a GPU box:

  python scripts/gpu_soak.py --fleets 40 --servers 32
"""
from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--fleets", type=int, default=40)
    p.add_argument("--servers", type=int, default=32)
    p.add_argument("--seed0", type=int, default=10_000)
    p.add_argument("--huge", action="store_true",
                   help="push every 4th server into the XL/GMEM N tiers "
                        "(uncapped batch states, round-2 spill paths)")
    args = p.parse_args()

    from inferno_amd.core import System
    from inferno_amd.engine import SweepEngine
    from tests.fixtures import make_spec

    total_cells = 0
    feas_mismatch = 0
    acc_mismatch = 0
    rep_off1 = 0
    rep_bad = 0
    val_bad = 0

    for f in range(args.fleets):
        seed = args.seed0 + f
        # vary regimes: arrival scale, replicas floor, accelerator count
        kw = dict(
            n_servers=args.servers,
            seed=seed,
            arrival_scale=[0.06, 6.0, 60.0, 600.0, 6000.0, 60000.0][f % 6],
            min_num_replicas=[0, 1][f % 2],
            n_accelerators=[1, 2, 3][f % 3],
        )
        analyzer = ["mm1k", "mm1k", "mg1"][f % 3]
        sa, sb = make_spec(**kw), make_spec(**kw)
        sa.optimizer.analyzer = analyzer
        sb.optimizer.analyzer = analyzer
        a, opt = System.from_spec(sa)
        b, _ = System.from_spec(sb)
        if args.huge:
            # force a slice of servers into the XL (8192<N<=32768) and GMEM
            # (N>32768) geometry tiers: tiny K with big maxBatchSize*atTokens
            import numpy as _np

            hrng = _np.random.default_rng(seed ^ 0xBEEF)
            for s in (a, b):
                srng = _np.random.default_rng(seed ^ 0xBEEF)  # same per system
                for i, srv_name in enumerate(sorted(s.servers)):
                    if i % 4 != 0:
                        continue
                    srv = s.servers[srv_name]
                    k_out = int(srng.integers(5, 40))
                    if srv.load is not None and srv.load.arrivalRate > 0:
                        srv.load.avgOutTokens = k_out
                    for perf in s.models[srv.model_name].perf_data.values():
                        perf.maxBatchSize = 256
                        perf.atTokens = 2048
        SweepEngine(backend="cpu").sweep(a)
        SweepEngine(backend="gpu").sweep(b)
        for name in a.servers:
            am = a.servers[name].all_allocations
            bm = b.servers[name].all_allocations
            if set(am) != set(bm):
                feas_mismatch += 1
                continue
            for acc in am:
                total_cells += 1
                x, y = am[acc], bm[acc]
                if x.accelerator != y.accelerator:
                    acc_mismatch += 1
                    continue
                dr = abs(x.num_replicas - y.num_replicas)
                # ceil-boundary allowance: lambda* may legally differ between
                # backends anywhere inside the bisection's 1e-6 y-tolerance
                # (amplified on flat metric plateaus), so replicas =
                # ceil(rate/rate*) can flip by ~1e-4 relative at huge counts
                rep_tol = max(1, int(1e-4 * max(x.num_replicas, y.num_replicas)))
                if 0 < dr <= rep_tol:
                    rep_off1 += 1
                elif dr > rep_tol:
                    rep_bad += 1
                    print(f"REPLICA MISMATCH {name}/{acc}: cpu={x.num_replicas} "
                          f"gpu={y.num_replicas} (seed={seed})")
                # a ceil-boundary replica flip legitimately moves value by
                # exactly the per-replica cost — allow that on top of the
                # relative tolerance
                per_rep = abs(x.cost) / max(x.num_replicas, 1)
                if x.value != 0 and abs(x.value - y.value) > max(
                    1e-3 * abs(x.value), 1e-2
                ) + dr * per_rep:
                    val_bad += 1
                    print(f"VALUE MISMATCH {name}/{acc}: cpu={x.value} gpu={y.value} "
                          f"(seed={seed})")
    print(
        f"soak: fleets={args.fleets} cells={total_cells} "
        f"feas_mismatch={feas_mismatch} acc_mismatch={acc_mismatch} "
        f"replicas_off_by_1={rep_off1} replicas_bad={rep_bad} value_bad={val_bad}"
    )
    ok = feas_mismatch == 0 and rep_bad == 0 and val_bad == 0 and (
        rep_off1 <= max(2, total_cells // 100)
    )
    print("SOAK", "PASS" if ok else "FAIL")
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
