"""Perf-parameter estimation from benchmark measurements.

The reference's fitting methodology (docs/tutorials/parameter-estimation.md:
80-195): run a *synchronous* benchmark (effective batch b=1) and a
*throughput* benchmark (effective batch b=B), read average ITL and TTFT from
each, and solve the WVA perf equations

    ITL(b)  = alpha + beta * b
    TTFT(b) = gamma + delta * inTokens * b

for (alpha, beta) and (gamma, delta). Used to validate the MI355X-derived
profiles (perfmodel.mi355x.derive_profile) against the emulator: configure
the emulator with a derived profile, benchmark it, fit, and compare.
"""
from __future__ import annotations

from dataclasses import dataclass


@dataclass
class BenchPoint:
    """One benchmark run's averages (ms / counts)."""

    batch: float  # effective concurrency during the run
    itl_ms: float  # avg inter-token latency
    ttft_ms: float  # avg time to first token
    avg_input_tokens: float


@dataclass
class FittedParms:
    alpha: float
    beta: float
    gamma: float
    delta: float


def fit_from_benchmarks(sync: BenchPoint, tput: BenchPoint) -> FittedParms:
    """Two-point fit of the perf equations (exact for two points)."""
    db = tput.batch - sync.batch
    if db == 0:
        raise ValueError("benchmark points need distinct batch sizes")
    beta = (tput.itl_ms - sync.itl_ms) / db
    alpha = sync.itl_ms - beta * sync.batch
    dtb = tput.avg_input_tokens * tput.batch - sync.avg_input_tokens * sync.batch
    if dtb == 0:
        raise ValueError("benchmark points need distinct token*batch products")
    delta = (tput.ttft_ms - sync.ttft_ms) / dtb
    gamma = sync.ttft_ms - delta * sync.avg_input_tokens * sync.batch
    return FittedParms(alpha=alpha, beta=beta, gamma=gamma, delta=delta)


def benchmark_sim(sim, concurrency: int, input_tokens: int,
                  output_tokens: int) -> BenchPoint:
    """Cohort benchmark on the emulator's virtual clock: submit exactly
    ``concurrency`` identical requests at t=0 and run the batch to
    completion, then read ITL/TTFT from the same stats the Prometheus series
    expose. The cohort prefills together and decodes at a constant batch, so
    the measurement is deterministic.

    concurrency=1 is the synchronous benchmark (guidellm --rate-type
    synchronous); a large cohort is the throughput benchmark, matching
    parameter-estimation.md:80-195.
    """
    t0 = (sim.ttft_sum_s, sim.ttft_count, sim.tpot_sum_s, sim.tpot_count,
          len(sim.finished))
    for _ in range(concurrency):
        sim.submit(input_tokens, output_tokens)
    while len(sim.finished) - t0[4] < concurrency:
        sim.step()

    ttft_s = sim.ttft_sum_s - t0[0]
    ttft_n = sim.ttft_count - t0[1]
    tpot_s = sim.tpot_sum_s - t0[2]
    tpot_n = sim.tpot_count - t0[3]
    return BenchPoint(
        batch=float(concurrency),
        itl_ms=(tpot_s / tpot_n) * 1000.0 if tpot_n else 0.0,
        ttft_ms=(ttft_s / ttft_n) * 1000.0 if ttft_n else 0.0,
        avg_input_tokens=float(input_tokens),
    )
