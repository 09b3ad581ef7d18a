"""M/G/1/K evaluator — the cheap closed-form alternative to the
state-dependent chain (BASELINE config 4's "M/G/1 model").

Model: the batch server is an M/G/1/K queue with a single service rate
mu = s(N) (the marginal per-request rate at max batch) and K = 11*N states;
waiting time carries the Pollaczek-Khinchine variability factor (1+cv^2)/2
(cv^2 = 1 recovers M/M/1/K). The effective concurrency is utilization-based
(eff = rho*N — the average number being served in a batch server at
utilization rho), which keeps EvalTTFT/EvalITL monotone in lambda so the
same bisection sizing applies. No per-state chain is needed: every
evaluation is O(1), which is why this is the cheap path on both CPU and GPU.
"""
from __future__ import annotations

from dataclasses import dataclass

from .mm1k import MG1K
from .queue import (
    EPSILON,
    STABILITY_SAFETY_FRACTION,
    AnalysisMetrics,
    AnalyzerError,
    Configuration,
    RequestSize,
    TargetPerf,
    TargetRate,
    binary_search,
    build_service_rates,
)


@dataclass
class _Point:
    throughput: float  # req/msec
    wait: float  # msec
    eff: float  # effective concurrency
    rho: float


class MG1QueueEvaluator:
    """Same sizing API surface as QueueAnalyzer, closed-form internals."""

    def __init__(self, cfg: Configuration, req: RequestSize, cv2: float = 1.0):
        cfg.check()
        req.check()
        self.max_batch_size = cfg.max_batch_size
        self.max_queue_size = cfg.max_queue_size
        self.service_parms = cfg.service_parms
        self.request_size = req
        self.cv2 = float(cv2)
        import numpy as np

        serv = build_service_rates(cfg, req)
        if not np.all(np.isfinite(serv)) or np.any(serv <= 0):
            raise AnalyzerError(f"invalid service rates for configuration {cfg}")
        self.mu = float(serv[-1])  # s(N), req/msec
        self.rate_min = float(serv[0]) * EPSILON * 1000.0
        self.rate_max = self.mu * (1.0 - EPSILON) * 1000.0
        self.K = cfg.max_queue_size + cfg.max_batch_size
        self._model = MG1K(self.K, cv2=self.cv2)

    # -- internals ---------------------------------------------------------
    def _point(self, lam: float) -> _Point:
        st = self._model.solve(lam, self.mu)
        if not st.is_valid:
            raise AnalyzerError(f"invalid M/G/1/K at lam={lam}, mu={self.mu}")
        rho = min(max(lam / self.mu, 0.0), 1.0)
        return _Point(
            throughput=st.throughput, wait=st.avg_wait_time, eff=rho * self.max_batch_size,
            rho=rho,
        )

    def _eval_ttft(self, lam: float) -> float:
        p = self._point(lam)
        return p.wait + self.service_parms.prefill.prefill_time(
            self.request_size.avg_input_tokens, p.eff
        )

    def _eval_itl(self, lam: float) -> float:
        p = self._point(lam)
        return self.service_parms.decode.decode_time(p.eff)

    # -- public API (QueueAnalyzer-compatible) ------------------------------
    def analyze(self, request_rate: float) -> AnalysisMetrics:
        if request_rate <= 0:
            raise AnalyzerError(f"invalid request rate {request_rate}")
        if request_rate > self.rate_max:
            raise AnalyzerError(f"rate={request_rate}, max allowed rate={self.rate_max}")
        p = self._point(request_rate / 1000.0)
        prefill_time = self.service_parms.prefill.prefill_time(
            self.request_size.avg_input_tokens, p.eff
        )
        token_time = self.service_parms.decode.decode_time(p.eff)
        return AnalysisMetrics(
            throughput=p.throughput * 1000.0,
            avg_resp_time=p.wait + 1.0 / self.mu,
            avg_wait_time=p.wait,
            avg_num_in_serv=p.eff,
            avg_prefill_time=prefill_time,
            avg_token_time=token_time,
            max_rate=self.rate_max,
            rho=p.rho,
        )

    def size(self, targets: TargetPerf):
        targets.check()
        lam_min = self.rate_min / 1000.0
        lam_max = self.rate_max / 1000.0
        lam_ttft = lam_max
        if targets.target_ttft > 0:
            lam_ttft, ind = binary_search(lam_min, lam_max, targets.target_ttft,
                                          self._eval_ttft)
            if ind < 0:
                raise AnalyzerError("targetTTFT below the bounded region")
        lam_itl = lam_max
        if targets.target_itl > 0:
            lam_itl, ind = binary_search(lam_min, lam_max, targets.target_itl,
                                         self._eval_itl)
            if ind < 0:
                raise AnalyzerError("targetITL below the bounded region")
        lam_tps = lam_max
        if targets.target_tps > 0:
            lam_tps = lam_max * (1.0 - STABILITY_SAFETY_FRACTION)
        lam = min(lam_ttft, lam_itl, lam_tps)
        metrics = self.analyze(lam * 1000.0)
        target_rate = TargetRate(
            rate_target_ttft=lam_ttft * 1000.0,
            rate_target_itl=lam_itl * 1000.0,
            rate_target_tps=lam_tps * 1000.0,
        )
        achieved = TargetPerf(
            target_ttft=metrics.avg_wait_time + metrics.avg_prefill_time,
            target_itl=metrics.avg_token_time,
            target_tps=metrics.throughput * float(self.request_size.avg_output_tokens),
        )
        return target_rate, metrics, achieved
