#!/usr/bin/env python3
"""Deep CPU fuzz: log-space chain vs the Go-style recurrence oracle over
thousands of random configurations (broader than the seeded pytest
differential). Exercises the expm1-stable tail across all regimes including
r -> 1 and tiny/huge rates.

  python scripts/cpu_fuzz.py --configs 2000
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from inferno_amd.analyzer.queue import (
    EPSILON, Configuration, DecodeParms, PrefillParms, QueueAnalyzer,
    RequestSize, ServiceParms, StateDependentChain, build_service_rates,
)
from tests.oracle import chain_stats_recurrence


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--configs", type=int, default=2000)
    p.add_argument("--seed", type=int, default=999)
    args = p.parse_args()
    rng = np.random.default_rng(args.seed)
    worst = 0.0
    bad = 0
    checked = 0
    for i in range(args.configs):
        N = int(rng.integers(1, 512))
        K = 11 * N
        cfg = Configuration(N, 10 * N, ServiceParms(
            prefill=PrefillParms(gamma=float(rng.uniform(0.01, 50)),
                                 delta=float(rng.uniform(1e-4, 1.0))),
            decode=DecodeParms(alpha=float(rng.uniform(0.1, 100)),
                               beta=float(rng.uniform(1e-3, 5))),
        ))
        req = RequestSize(int(rng.integers(0, 4000)), int(rng.integers(1, 2000)))
        serv = build_service_rates(cfg, req)
        if not np.all(np.isfinite(serv)) or np.any(serv <= 0):
            continue
        chain = StateDependentChain(K, serv)
        lo = float(serv[0]) * EPSILON
        hi = float(serv[-1]) * (1 - EPSILON)
        # include near-r=1 probes beyond the admissible cap on purpose
        rates = [lo, lo + 0.5 * (hi - lo), hi,
                 float(serv[-1]) * (1 - 1e-8), float(serv[-1])]
        for lam in rates:
            got = chain.solve(lam)
            want = chain_stats_recurrence(K, serv, lam)
            for k in ("p0", "pK", "throughput", "avg_num_in_system",
                      "avg_num_in_servers"):
                g = getattr(got, k)
                w = want[k]
                denom = max(abs(w), 1e-300)
                rel = abs(g - w) / denom
                if w > 1e-280:  # below that the recurrence itself underflows
                    worst = max(worst, rel)
                    if rel > 1e-6:
                        bad += 1
                        if bad <= 5:
                            print(f"MISMATCH cfg={i} lam={lam} {k}: {g} vs {w} rel={rel:.2e}")
                checked += 1
    print(f"configs={args.configs} checks={checked} worst_rel={worst:.3e} mismatches={bad}")
    print("FUZZ PASS" if bad == 0 else "FUZZ FAIL")
    return 0 if bad == 0 else 1


if __name__ == "__main__":
    raise SystemExit(main())
