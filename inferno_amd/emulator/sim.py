"""Discrete-event vLLM emulator core.

Re-implementation of the reference's emulator model
(tools/vllm-emulator/vllm_model.py:46-467): a stepping clock at decode-time
granularity, KV-cache memory accounting against a device size, continuous
batching with admission by free KV memory and eviction (preemption) of the
most-recently admitted request under memory pressure, and per-request
token-time tracking feeding the exact ``vllm:*`` Prometheus series the
collector scrapes.

The core is synchronous (``step()`` advances one decode tick) so tests run
on a virtual clock; the FastAPI server wraps it with a real-time loop.
"""
from __future__ import annotations

import itertools
from dataclasses import dataclass
from typing import Callable, Optional

# defaults mirror tools/vllm-emulator/server.py:22-33
DEFAULT_DECODE_TIME_MS = 50.0
DEFAULT_PREFILL_TIME_MS = 100.0
DEFAULT_MEM_SIZE_MB = 80000.0
DEFAULT_KV_MB_PER_TOKEN = 4.0
DEFAULT_USABLE_MEM_RATIO = 0.8
DEFAULT_MAX_BATCH_SIZE = 256

# MI355X-parameterized profile: 288 GB HBM3E per GPU
MI355X_MEM_SIZE_MB = 288000.0


@dataclass
class Request:
    rid: int
    input_tokens: int
    output_tokens: int
    arrival_time: float = 0.0
    first_token_time: Optional[float] = None
    finish_time: Optional[float] = None
    generated: int = 0
    prefilled: bool = False
    on_finish: Optional[Callable[["Request"], None]] = None

    @property
    def kv_tokens(self) -> int:
        """Tokens resident in KV cache: prompt + generated so far."""
        return self.input_tokens + self.generated


class Device:
    """KV-cache memory accounting (ref vllm_model.py:79-144)."""

    def __init__(self, mem_size_mb: float = DEFAULT_MEM_SIZE_MB,
                 kv_mb_per_token: float = DEFAULT_KV_MB_PER_TOKEN,
                 usable_ratio: float = DEFAULT_USABLE_MEM_RATIO):
        self.capacity_mb = mem_size_mb * usable_ratio
        self.kv_mb_per_token = kv_mb_per_token
        self.used_mb = 0.0

    def fits(self, tokens: int) -> bool:
        return self.used_mb + tokens * self.kv_mb_per_token <= self.capacity_mb

    def allocate(self, tokens: int) -> bool:
        need = tokens * self.kv_mb_per_token
        if self.used_mb + need > self.capacity_mb:
            return False
        self.used_mb += need
        return True

    def free(self, tokens: int) -> None:
        self.used_mb = max(0.0, self.used_mb - tokens * self.kv_mb_per_token)

    @property
    def utilization(self) -> float:
        return self.used_mb / self.capacity_mb if self.capacity_mb > 0 else 0.0


class VLLMSim:
    """Continuous-batching inference-server emulator."""

    def __init__(
        self,
        decode_time_ms: float = DEFAULT_DECODE_TIME_MS,
        prefill_time_ms: float = DEFAULT_PREFILL_TIME_MS,
        mem_size_mb: float = DEFAULT_MEM_SIZE_MB,
        kv_mb_per_token: float = DEFAULT_KV_MB_PER_TOKEN,
        usable_ratio: float = DEFAULT_USABLE_MEM_RATIO,
        max_batch_size: int = DEFAULT_MAX_BATCH_SIZE,
        decode_parms: Optional[tuple[float, float]] = None,
        prefill_parms: Optional[tuple[float, float]] = None,
    ):
        self.decode_time_s = decode_time_ms / 1000.0
        self.prefill_time_s = prefill_time_ms / 1000.0
        # optional batch-dependent service model (the WVA perf equations:
        # decode(b) = alpha + beta*b, prefill(b) = gamma + delta*inTok*b, ms)
        # — lets the emulator act as an oracle for parameter estimation
        # (ref docs/tutorials/parameter-estimation.md:80-195)
        self.decode_parms = decode_parms  # (alpha_ms, beta_ms)
        self.prefill_parms = prefill_parms  # (gamma_ms, delta_ms)
        self.device = Device(mem_size_mb, kv_mb_per_token, usable_ratio)
        self.max_batch_size = max_batch_size
        self.clock = 0.0
        self.waiting: list[Request] = []
        self.running: list[Request] = []
        self.finished: list[Request] = []
        self._ids = itertools.count(1)
        # cumulative stats for the metrics endpoint
        self.success_total = 0
        self.prompt_tokens_sum = 0.0
        self.prompt_tokens_count = 0
        self.generation_tokens_sum = 0.0
        self.generation_tokens_count = 0
        self.ttft_sum_s = 0.0
        self.ttft_count = 0
        self.tpot_sum_s = 0.0
        self.tpot_count = 0
        self.preemptions = 0

    # ------------------------------------------------------------------
    def submit(self, input_tokens: int, output_tokens: int,
               on_finish: Optional[Callable[[Request], None]] = None) -> Request:
        req = Request(
            rid=next(self._ids),
            input_tokens=max(int(input_tokens), 0),
            output_tokens=max(int(output_tokens), 1),
            arrival_time=self.clock,
            on_finish=on_finish,
        )
        self.waiting.append(req)
        return req

    def _admit(self) -> None:
        """Admit waiting requests while KV memory and batch slots allow
        (ref vllm_model.py:371-400)."""
        while (
            self.waiting
            and len(self.running) < self.max_batch_size
            and self.device.fits(self.waiting[0].input_tokens + 1)
        ):
            req = self.waiting.pop(0)
            self.device.allocate(req.input_tokens + 1)
            req.prefilled = False
            self.running.append(req)

    def _evict_one(self) -> bool:
        """Preempt the most recently admitted request back to the queue head
        (ref vllm_model.py:402-413 eviction under memory pressure); its KV is
        freed and it restarts from prefill on re-admission (TTFT keeps the
        first observed value)."""
        if not self.running:
            return False
        req = self.running.pop()  # last admitted
        self.device.free(req.kv_tokens + 1)
        req.generated = 0
        req.prefilled = False
        self.waiting.insert(0, req)
        self.preemptions += 1
        return True

    def step(self) -> None:
        """One scheduler iteration (ref vllm_model.py:432-456 one_iteration):
        admit, prefill the newly admitted batch, then one decode step for all
        running requests. Before decoding, one KV token per running request
        is reserved, evicting from the tail (most recently admitted) under
        memory pressure. Advances the clock by the iteration time."""
        self._admit()

        b = len(self.running)
        if self.decode_parms is not None and b > 0:
            alpha, beta = self.decode_parms
            step_time = (alpha + beta * b) / 1000.0
        else:
            step_time = self.decode_time_s
        new = [r for r in self.running if not r.prefilled]
        if new:
            if self.prefill_parms is not None:
                gamma, delta = self.prefill_parms
                avg_in = sum(r.input_tokens for r in new) / len(new)
                step_time += (gamma + delta * avg_in * len(new)) / 1000.0
            else:
                step_time += self.prefill_time_s
            for r in new:
                r.prefilled = True

        self.clock += step_time

        # reserve one KV token per running request, evicting under pressure
        while self.running and not self.device.fits(len(self.running)):
            self._evict_one()
        for r in self.running:
            self.device.allocate(1)

        still_running: list[Request] = []
        for r in self.running:
            r.generated += 1
            if r.first_token_time is None:
                r.first_token_time = self.clock
                self.ttft_sum_s += self.clock - r.arrival_time
                self.ttft_count += 1
            else:
                self.tpot_sum_s += step_time
                self.tpot_count += 1
            if r.generated >= r.output_tokens:
                r.finish_time = self.clock
                self.device.free(r.kv_tokens + 1)
                self._record_finish(r)
                if r.on_finish is not None:
                    r.on_finish(r)
            else:
                still_running.append(r)
        self.running = still_running

    def _record_finish(self, r: Request) -> None:
        self.finished.append(r)
        self.success_total += 1
        self.prompt_tokens_sum += r.input_tokens
        self.prompt_tokens_count += 1
        self.generation_tokens_sum += r.generated
        self.generation_tokens_count += 1

    def run_until_idle(self, max_steps: int = 100000) -> None:
        steps = 0
        while (self.waiting or self.running) and steps < max_steps:
            self.step()
            steps += 1

    # -- derived metrics ------------------------------------------------
    @property
    def num_requests_running(self) -> int:
        return len(self.running)

    @property
    def num_requests_waiting(self) -> int:
        return len(self.waiting)

    @property
    def avg_ttft_s(self) -> float:
        return self.ttft_sum_s / self.ttft_count if self.ttft_count else 0.0

    @property
    def avg_tpot_s(self) -> float:
        return self.tpot_sum_s / self.tpot_count if self.tpot_count else 0.0
