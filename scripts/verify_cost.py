#!/usr/bin/env python3
"""Cost-parity verification: HIP sweep vs the CPU reference solver on the
SAME synthetic trace (BASELINE.md row "Total allocation cost vs reference
solver (same trace): must match/beat").

Two identical systems step through the same bursty Poisson trace; each is
solved by its own backend and applies its own decisions (exactly what each
controller would do in production). Reports per-step total allocation cost
for both, the relative cost delta, and the per-server decision agreement.

  python scripts/verify_cost.py --models 128 --steps 20
"""
from __future__ import annotations

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--models", type=int, default=128)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--seed", type=int, default=4242)
    p.add_argument("--analyzer", choices=["mm1k", "mg1"], default="mm1k")
    args = p.parse_args()

    import numpy as np

    from inferno_amd.config import ServerLoadSpec
    from inferno_amd.core import allocation_from_data
    from inferno_amd.core.system import System
    from inferno_amd.engine import SweepEngine
    from inferno_amd.parallel import ShardedSolver
    from inferno_amd.utils.synthetic import PoissonTrace, make_fleet_spec

    def build():
        spec = make_fleet_spec(args.models, seed=args.seed)
        spec.optimizer.analyzer = args.analyzer
        system, opt = System.from_spec(spec)
        for a in system.accelerators.values():
            a.calculate()
        return system, opt

    sys_gpu, opt = build()
    sys_cpu, _ = build()
    solver_gpu = ShardedSolver(SweepEngine(backend="gpu"))
    solver_cpu = ShardedSolver(SweepEngine(backend="cpu"))

    all_names = sorted(sys_gpu.servers)
    trace = PoissonTrace(len(all_names), seed=args.seed + 7)
    in_toks = {n: sys_gpu.servers[n].load.avgInTokens for n in all_names}
    out_toks = {n: sys_gpu.servers[n].load.avgOutTokens for n in all_names}

    cost_g_total = cost_c_total = 0.0
    max_rel_delta = 0.0
    agree = disagree = rep_off = 0
    for step in range(args.steps):
        rates = trace.rates_at(step)
        for system in (sys_gpu, sys_cpu):
            for i, name in enumerate(all_names):
                system.servers[name].load = ServerLoadSpec(
                    arrivalRate=float(rates[i]),
                    avgInTokens=in_toks[name],
                    avgOutTokens=out_toks[name],
                )
        rg = solver_gpu.solve(sys_gpu, opt)
        rc = solver_cpu.solve(sys_cpu, opt)
        cg = float(np.sum(rg.winners.cost[rg.winners.valid]))
        cc = float(np.sum(rc.winners.cost[rc.winners.valid]))
        cost_g_total += cg
        cost_c_total += cc
        if cc > 0:
            max_rel_delta = max(max_rel_delta, abs(cg - cc) / cc)
        for i in range(len(all_names)):
            ga, ca = int(rg.winners.acc_idx[i]), int(rc.winners.acc_idx[i])
            gr, cr = int(rg.winners.num_replicas[i]), int(rc.winners.num_replicas[i])
            if ga == ca and gr == cr:
                agree += 1
            elif ga == ca and abs(gr - cr) <= 1:
                rep_off += 1
            else:
                disagree += 1
        # each applies its own decisions
        for system, result in ((sys_gpu, rg), (sys_cpu, rc)):
            for name, data in result.solution.items():
                server = system.servers[name]
                server.spec.currentAlloc = data
                server.cur_allocation = allocation_from_data(data)

    n = len(all_names) * args.steps
    summary = {
        "models": args.models,
        "steps": args.steps,
        "analyzer": args.analyzer,
        "total_cost_gpu": round(cost_g_total, 2),
        "total_cost_cpu_reference": round(cost_c_total, 2),
        "cost_ratio_gpu_over_cpu": round(cost_g_total / cost_c_total, 8)
        if cost_c_total else None,
        "max_step_rel_cost_delta": round(max_rel_delta, 8),
        "decisions": n,
        "exact_agreement": agree,
        "replica_off_by_one": rep_off,
        "disagreement": disagree,
    }
    print(json.dumps(summary))
    ok = (
        disagree == 0
        and rep_off <= max(1, n // 200)
        and max_rel_delta < 1e-3
    )
    print("COST-PARITY", "PASS" if ok else "FAIL")
    sys.exit(0 if ok else 1)


if __name__ == "__main__":
    main()
