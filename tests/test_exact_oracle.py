"""Exact-arithmetic oracle for the state-dependent M/M/1/K chain.

The strongest numerics check available: solve the stationary distribution
with python Fractions (NO floating point anywhere — exact rationals), using
the exact values of the same float32 service rates the production solver
consumes, and require the log-space float64 solver to match to ~1e-12.
This is independent of both the reference's recurrence (tests/oracle.py
covers that) and of this repo's own log-space algebra."""
from fractions import Fraction

import numpy as np
import pytest

from inferno_amd.analyzer.queue import StateDependentChain


def exact_chain(lam: Fraction, serv: list[Fraction], K: int):
    """Stationary distribution of the birth-death chain with arrival lam and
    state-dependent service serv[n-1] for n=1..N, constant serv[N-1] for
    n=N+1..K. Pure Fraction arithmetic."""
    N = len(serv)
    p = [Fraction(1)]
    for n in range(1, K + 1):
        s = serv[min(n, N) - 1]
        p.append(p[-1] * lam / s)
    Z = sum(p)
    p = [x / Z for x in p]
    throughput = lam * (1 - p[K])
    avg_n = sum(Fraction(n) * p[n] for n in range(K + 1))
    avg_in_serv = sum(Fraction(min(n, N)) * p[n] for n in range(K + 1))
    resp = avg_n / throughput
    serv_t = avg_in_serv / throughput
    wait = resp - serv_t
    return {
        "throughput": throughput,
        "avg_num_in_system": avg_n,
        "avg_num_in_servers": avg_in_serv,
        "avg_resp_time": resp,
        "avg_serv_time": serv_t,
        "avg_wait_time": wait,
        "p0": p[0],
        "pK": p[K],
    }


FIELDS = ("throughput", "avg_num_in_system", "avg_num_in_servers",
          "avg_resp_time", "avg_serv_time", "avg_wait_time", "p0", "pK")


class TestExactRationalOracle:
    @pytest.mark.parametrize("seed", range(6))
    def test_matches_exact_arithmetic(self, seed):
        rng = np.random.default_rng(seed)
        N = int(rng.integers(4, 48))
        K = 11 * N
        # realistic service-rate shape: s(n) = n/(prefill + decode tail)
        alpha = rng.uniform(5, 30)
        beta = rng.uniform(0.05, 0.6)
        gamma = rng.uniform(1, 10)
        delta = rng.uniform(1e-4, 1e-2)
        in_tok, out_tok = int(rng.integers(8, 512)), int(rng.integers(2, 256))
        n = np.arange(1, N + 1, dtype=np.float32)
        serv32 = (n / (np.float32(gamma) + np.float32(delta) * np.float32(in_tok) * n
                       + np.float32(out_tok - 1)
                       * (np.float32(alpha) + np.float32(beta) * n))).astype(np.float32)

        chain = StateDependentChain(K, serv32)
        lam = float(serv32[-1]) * float(rng.uniform(0.05, 0.97))
        got = chain.solve(lam)

        # the oracle consumes the EXACT values of the same float32 rates and
        # the exact float64 lambda — Fraction(float) is exact
        serv_exact = [Fraction(float(s)) for s in serv32]
        want = exact_chain(Fraction(lam), serv_exact, K)

        resp = float(want["avg_resp_time"])
        for f in FIELDS:
            g = getattr(got, f)
            w = float(want[f])
            # wait = resp - serv cancels catastrophically when utilization is
            # tiny (exact wait ~1e-16 of a ~1e3 response): judge it at the
            # float64 cancellation floor of the subtraction, like the
            # reference's own clamp-to-zero does
            abs_tol = 1e-11 * resp if f == "avg_wait_time" else 1e-300
            assert g == pytest.approx(w, rel=5e-12, abs=abs_tol), (
                f"{f}: solver={g!r} exact={w!r} (seed={seed}, N={N}, lam={lam})"
            )

    def test_deep_saturation_tail(self):
        # lam just under s(N): the geometric tail is nearly flat — the
        # expm1-stable closed forms must agree with exact rationals
        serv32 = np.linspace(0.01, 0.05, 16).astype(np.float32)
        K = 11 * 16
        chain = StateDependentChain(K, serv32)
        for frac in (0.999, 0.9999, 1.0 - 1e-7):
            lam = float(serv32[-1]) * frac
            got = chain.solve(lam)
            want = exact_chain(Fraction(lam), [Fraction(float(s)) for s in serv32], K)
            for f in FIELDS:
                assert getattr(got, f) == pytest.approx(float(want[f]), rel=1e-9), (
                    f"{f} at frac={frac}"
                )

    def test_tiny_lambda(self):
        serv32 = np.linspace(0.02, 0.3, 8).astype(np.float32)
        chain = StateDependentChain(88, serv32)
        lam = float(serv32[0]) * 1e-3
        got = chain.solve(lam)
        want = exact_chain(Fraction(lam), [Fraction(float(s)) for s in serv32], 88)
        for f in FIELDS:
            assert getattr(got, f) == pytest.approx(float(want[f]), rel=5e-12)
