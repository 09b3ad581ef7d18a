"""Offline emulator experiments (the reference's tools/vllm-emulator/
experiment.py analogue, text output instead of matplotlib): sweep the
discrete-event vLLM model over arrival-rate grids and report TTFT/ITL/memory
statistics — the tool used to sanity-check fitted perf parameters.

  python -m inferno_amd.emulator.experiment --rates 30,60,120,240 \
      --requests 200 --decode-ms 50 --prefill-ms 100
"""
from __future__ import annotations

import argparse
import json

from .sim import VLLMSim


def run_one(rpm: float, n_requests: int, *, decode_ms: float, prefill_ms: float,
            mem_mb: float, kv_mb: float, max_batch: int, in_tokens: int,
            out_tokens: int) -> dict:
    sim = VLLMSim(
        decode_time_ms=decode_ms,
        prefill_time_ms=prefill_ms,
        mem_size_mb=mem_mb,
        kv_mb_per_token=kv_mb,
        max_batch_size=max_batch,
    )
    gap_s = 60.0 / rpm if rpm > 0 else 0.0
    t = 0.0
    submitted = 0
    peak_mem = 0.0
    peak_running = 0
    while submitted < n_requests or sim.waiting or sim.running:
        while submitted < n_requests and t <= sim.clock:
            sim.submit(input_tokens=in_tokens, output_tokens=out_tokens)
            submitted += 1
            t += gap_s
        if sim.waiting or sim.running:
            sim.step()
            peak_mem = max(peak_mem, sim.device.used_mb)
            peak_running = max(peak_running, sim.num_requests_running)
        else:
            sim.clock = t
    effective_rate = sim.success_total / sim.clock if sim.clock > 0 else 0.0
    return {
        "rpm": rpm,
        "completed": sim.success_total,
        "avg_ttft_ms": round(sim.avg_ttft_s * 1000, 2),
        "avg_itl_ms": round(sim.avg_tpot_s * 1000, 2),
        "throughput_rps": round(effective_rate, 3),
        "peak_batch": peak_running,
        "peak_mem_mb": round(peak_mem, 1),
        "preemptions": sim.preemptions,
    }


def main() -> None:
    p = argparse.ArgumentParser()
    p.add_argument("--rates", default="30,60,120,240,480",
                   help="comma-separated arrival rates (req/min)")
    p.add_argument("--requests", type=int, default=200)
    p.add_argument("--decode-ms", type=float, default=50.0)
    p.add_argument("--prefill-ms", type=float, default=100.0)
    p.add_argument("--mem-mb", type=float, default=80000.0)
    p.add_argument("--kv-mb", type=float, default=4.0)
    p.add_argument("--max-batch", type=int, default=256)
    p.add_argument("--input-tokens", type=int, default=64)
    p.add_argument("--output-tokens", type=int, default=64)
    args = p.parse_args()
    for rpm in (float(r) for r in args.rates.split(",")):
        res = run_one(
            rpm,
            args.requests,
            decode_ms=args.decode_ms,
            prefill_ms=args.prefill_ms,
            mem_mb=args.mem_mb,
            kv_mb=args.kv_mb,
            max_batch=args.max_batch,
            in_tokens=args.input_tokens,
            out_tokens=args.output_tokens,
        )
        print(json.dumps(res))


if __name__ == "__main__":
    main()
