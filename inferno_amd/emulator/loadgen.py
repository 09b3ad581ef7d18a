"""Load generator for the emulator/e2e benchmarks.

Ref tools/vllm-emulator/loadgen.py:10-130: Poisson (expovariate) or
deterministic arrivals with piecewise rate schedules [[duration_s, rpm],
...], seeded, firing OpenAI chat completions at a target URL.
"""
from __future__ import annotations

import argparse
import asyncio
import json
import random
import time
from dataclasses import dataclass


@dataclass
class Stats:
    sent: int = 0
    completed: int = 0
    errors: int = 0
    latencies: list = None

    def __post_init__(self):
        if self.latencies is None:
            self.latencies = []


def parse_schedule(s: str) -> list[tuple[float, float]]:
    """'[[60, 30], [120, 90]]' -> [(duration_s, requests_per_minute), ...]"""
    raw = json.loads(s)
    return [(float(d), float(rpm)) for d, rpm in raw]


async def _fire(client, url: str, model: str, in_tokens: int, max_tokens: int,
                stats: Stats) -> None:
    body = {
        "model": model,
        "messages": [{"role": "user", "content": "tok " * in_tokens}],
        "max_tokens": max_tokens,
    }
    t0 = time.perf_counter()
    try:
        r = await client.post(url, json=body, timeout=600.0)
        r.raise_for_status()
        stats.completed += 1
        stats.latencies.append(time.perf_counter() - t0)
    except Exception:
        stats.errors += 1


async def run_load(
    base_url: str,
    schedule: list[tuple[float, float]],
    model: str = "default/default",
    deterministic: bool = False,
    seed: int = 0,
    avg_input_tokens: int = 32,
    max_tokens: int = 32,
) -> Stats:
    import httpx

    rng = random.Random(seed)
    stats = Stats()
    url = "/v1/chat/completions"
    async with httpx.AsyncClient(base_url=base_url) as client:
        tasks = []
        for duration, rpm in schedule:
            phase_end = time.perf_counter() + duration
            rate_per_s = rpm / 60.0
            while time.perf_counter() < phase_end:
                if rate_per_s <= 0:
                    await asyncio.sleep(min(duration, 1.0))
                    continue
                gap = (1.0 / rate_per_s) if deterministic else rng.expovariate(rate_per_s)
                await asyncio.sleep(min(gap, max(phase_end - time.perf_counter(), 0)))
                if time.perf_counter() >= phase_end:
                    break
                stats.sent += 1
                tasks.append(
                    asyncio.create_task(
                        _fire(client, url, model, avg_input_tokens, max_tokens, stats)
                    )
                )
        if tasks:
            await asyncio.gather(*tasks, return_exceptions=True)
    return stats


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--url", default="http://127.0.0.1:8000")
    p.add_argument("--schedule", default="[[60, 30]]",
                   help="[[duration_s, requests_per_minute], ...]")
    p.add_argument("--model", default="default/default")
    p.add_argument("--deterministic", action="store_true")
    p.add_argument("--seed", type=int, default=0)
    p.add_argument("--input-tokens", type=int, default=32)
    p.add_argument("--max-tokens", type=int, default=32)
    args = p.parse_args()
    stats = asyncio.run(
        run_load(
            args.url,
            parse_schedule(args.schedule),
            model=args.model,
            deterministic=args.deterministic,
            seed=args.seed,
            avg_input_tokens=args.input_tokens,
            max_tokens=args.max_tokens,
        )
    )
    lat = sorted(stats.latencies)
    p50 = lat[len(lat) // 2] if lat else 0.0
    print(
        json.dumps(
            {
                "sent": stats.sent,
                "completed": stats.completed,
                "errors": stats.errors,
                "latency_p50_s": round(p50, 4),
            }
        )
    )


if __name__ == "__main__":
    main()
