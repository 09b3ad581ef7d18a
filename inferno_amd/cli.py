"""Offline solver CLI: solve a SystemData JSON document (the reference's
pkg/config wire format) and print the allocation solution.

  python -m inferno_amd.cli solve system.json [--backend cpu|gpu] [--json]
  python -m inferno_amd.cli analyze system.json --server NAME:NS
"""
from __future__ import annotations

import argparse
import json
import sys


def _load_system(path: str):
    from .config import system_spec_from_json
    from .core.system import System

    with open(path) as f:
        doc = json.load(f)
    spec = system_spec_from_json(doc)
    system, opt = System.from_spec(spec)
    for a in system.accelerators.values():
        a.calculate()
    return system, opt


def cmd_solve(args) -> int:
    from .engine import SweepEngine
    from .parallel import ShardedSolver

    system, opt = _load_system(args.spec)
    solver = ShardedSolver(SweepEngine(backend=args.backend))
    result = solver.solve(system, opt)
    if args.json:
        out = {
            name: {
                "accelerator": d.accelerator,
                "numReplicas": d.numReplicas,
                "maxBatch": d.maxBatch,
                "cost": round(d.cost, 2),
                "itlAverage": round(d.itlAverage, 3),
                "ttftAverage": round(d.ttftAverage, 3),
            }
            for name, d in result.solution.items()
        }
        print(json.dumps({"allocations": out}, indent=2))
    else:
        total = 0.0
        print(f"{'server':<28} {'accelerator':<14} {'replicas':>8} {'cost':>10} "
              f"{'itl(ms)':>9} {'ttft(ms)':>9}")
        for name in sorted(result.solution):
            d = result.solution[name]
            total += d.cost
            print(f"{name:<28} {d.accelerator or '(none)':<14} {d.numReplicas:>8} "
                  f"{d.cost:>10.2f} {d.itlAverage:>9.2f} {d.ttftAverage:>9.2f}")
        by_type = result.allocation_by_type
        print("-" * 82)
        for t in sorted(by_type):
            a = by_type[t]
            lim = f"/{a.limit}" if a.limit else ""
            print(f"{t:<28} {'':<14} {a.count:>8}{lim} {a.cost:>10.2f}")
        print(f"{'TOTAL':<28} {'':<14} {'':>8} {total:>10.2f}")
    return 0


def cmd_analyze(args) -> int:
    from .api import v1alpha1 as api
    from .controller.modelanalyzer import ModelAnalyzer
    from .engine import SweepEngine

    system, _ = _load_system(args.spec)
    name, _, ns = args.server.partition(":")
    va = api.VariantAutoscaling(name=name, namespace=ns or "default")
    resp = ModelAnalyzer(system, SweepEngine(backend=args.backend)).analyze_model(va)
    if not resp.allocations:
        print(f"no feasible allocations for {args.server}", file=sys.stderr)
        return 1
    print(f"{'accelerator':<16} {'replicas':>8} {'batch':>6} {'cost':>10} "
          f"{'value':>10} {'itl(ms)':>9} {'ttft(ms)':>9} {'rho':>6}")
    for acc in sorted(resp.allocations):
        a = resp.allocations[acc].allocation
        print(f"{acc:<16} {a.num_replicas:>8} {a.batch_size:>6} {a.cost:>10.2f} "
              f"{a.value:>10.2f} {a.itl:>9.2f} {a.ttft:>9.2f} {a.rho:>6.3f}")
    return 0


def main() -> None:
    p = argparse.ArgumentParser(prog="inferno_amd.cli")
    sub = p.add_subparsers(dest="cmd", required=True)
    ps = sub.add_parser("solve", help="global cost/SLO solve of a SystemData JSON")
    ps.add_argument("spec")
    ps.add_argument("--backend", choices=["auto", "gpu", "cpu"], default="auto")
    ps.add_argument("--json", action="store_true")
    ps.set_defaults(fn=cmd_solve)
    pa = sub.add_parser("analyze", help="per-variant candidate allocations")
    pa.add_argument("spec")
    pa.add_argument("--server", required=True, help="name:namespace")
    pa.add_argument("--backend", choices=["auto", "gpu", "cpu"], default="auto")
    pa.set_defaults(fn=cmd_analyze)
    args = p.parse_args()
    sys.exit(args.fn(args))


if __name__ == "__main__":
    main()
