"""Queueing analyzer — CPU golden reference for the HIP sweep kernels.

Re-derivation of the reference's state-dependent M/M/1/K evaluator
(``pkg/analyzer/queueanalyzer.go``, ``pkg/analyzer/mm1modelstatedependent.go``)
in **log-space closed form** instead of the forward probability recurrence:

The birth-death chain with constant arrival rate ``lam`` and state-dependent
service rate ``s(n)`` (``s(n) = s(N)`` for ``n > N``) has stationary
probabilities ``p[n] ∝ lam^n / prod_{i=1..n} s(i)``, i.e.

    log p[n] = n*log(lam) - S(min(n, N)) - max(n - N, 0)*log(s(N))

where ``S(m) = sum_{i=1..m} log s(i)`` is a prefix sum over only the N batch
states. The 10*N queue states (n > N) form a geometric tail with ratio
``r = lam/s(N) < 1`` inside the admissible rate range, so every statistic the
sizing needs (normalization, E[n], p[K], head sums) is an O(N) reduction plus
an O(1) analytic tail — this is what makes the GPU sweep O(N) per bisection
step instead of the reference's O(K)=O(11N) sequential recurrence with
overflow rescaling (mm1modelstatedependent.go:70-116). Numerically the two
formulations agree to ~1e-12 relative in float64.

Semantics matched to the reference:
  * service rates are computed in float32 then widened (queueanalyzer.go:99-131
    mixes float32 inputs with float64 probabilities);
  * ``Analyze`` / ``Size`` / ``EffectiveConcurrency`` follow
    queueanalyzer.go:134-302 including unit conventions (rates are req/msec
    internally, req/sec at the API surface) and clamping;
  * the bisection follows ``pkg/analyzer/utils.go:26-70`` (relative tolerance
    1e-6, <=100 iterations, below/within/above indicator).
"""
from __future__ import annotations

import math
from dataclasses import dataclass

import numpy as np

# small disturbance around a value (ref queueanalyzer.go:8)
EPSILON = 1e-3
# fraction below max throughput kept for stability (ref queueanalyzer.go:11)
STABILITY_SAFETY_FRACTION = 0.1

# bisection constants (ref pkg/analyzer/utils.go:8-9)
SEARCH_TOLERANCE = 1e-6
MAX_SEARCH_ITERATIONS = 100


class AnalyzerError(ValueError):
    """Raised where the reference returns an error (infeasible / invalid input)."""


@dataclass
class PrefillParms:
    gamma: float = 0.0
    delta: float = 0.0

    def prefill_time(self, avg_input_tokens: int, batch_size: float) -> float:
        """prefill time (msec); 0 when there are no input tokens.

        Ref: queueanalyzer.go:257-262.
        """
        if avg_input_tokens == 0:
            return 0.0
        return float(
            np.float32(self.gamma)
            + np.float32(self.delta) * np.float32(avg_input_tokens) * np.float32(batch_size)
        )


@dataclass
class DecodeParms:
    alpha: float = 0.0
    beta: float = 0.0

    def decode_time(self, batch_size: float) -> float:
        """decode time per token (msec). Ref: queueanalyzer.go:264-266."""
        return float(np.float32(self.alpha) + np.float32(self.beta) * np.float32(batch_size))


@dataclass
class ServiceParms:
    prefill: PrefillParms
    decode: DecodeParms


@dataclass
class RequestSize:
    avg_input_tokens: int
    avg_output_tokens: int

    def check(self) -> None:
        if self.avg_input_tokens < 0 or self.avg_output_tokens < 1:
            raise AnalyzerError(f"invalid request size {self}")


@dataclass
class Configuration:
    max_batch_size: int
    max_queue_size: int
    service_parms: ServiceParms

    def check(self) -> None:
        if (
            self.max_batch_size <= 0
            or self.max_queue_size < 0
            or self.service_parms is None
            or self.service_parms.prefill is None
            or self.service_parms.decode is None
        ):
            raise AnalyzerError(f"invalid configuration {self}")


@dataclass
class AnalysisMetrics:
    """Solution metrics. Ref: queueanalyzer.go:59-69."""

    throughput: float  # requests/sec
    avg_resp_time: float  # msec
    avg_wait_time: float  # msec
    avg_num_in_serv: float
    avg_prefill_time: float  # msec
    avg_token_time: float  # msec
    max_rate: float  # requests/sec
    rho: float


@dataclass
class TargetPerf:
    target_ttft: float = 0.0  # msec
    target_itl: float = 0.0  # msec
    target_tps: float = 0.0  # tokens/sec

    def check(self) -> None:
        if self.target_itl < 0 or self.target_ttft < 0 or self.target_tps < 0:
            raise AnalyzerError(f"invalid target data values {self}")


@dataclass
class TargetRate:
    rate_target_ttft: float
    rate_target_itl: float
    rate_target_tps: float


@dataclass
class ChainStats:
    """Raw statistics of one chain solve (internal units: req/msec, msec)."""

    throughput: float  # req/msec, lam * (1 - p[K])
    avg_num_in_system: float
    avg_num_in_servers: float
    avg_resp_time: float  # msec
    avg_serv_time: float  # msec
    avg_wait_time: float  # msec
    p0: float
    pK: float


def within_tolerance(x: float, value: float, tolerance: float) -> bool:
    """Ref: pkg/analyzer/utils.go:12-23."""
    if x == value:
        return True
    if value == 0 or tolerance < 0:
        return False
    return abs((x - value) / value) <= tolerance


def binary_search(x_min, x_max, y_target, eval_fn):
    """Find x* with f(x*)=y_target for monotone f over [x_min, x_max].

    Returns (x_star, indicator): indicator is -1 when the target lies below
    the bounded region, +1 above, 0 within. Matches the semantics of the
    reference's BinarySearch (pkg/analyzer/utils.go:26-70): boundary values
    within relative tolerance return immediately; otherwise <=100 bisection
    iterations with early exit on relative tolerance.
    Raises AnalyzerError when eval_fn fails.
    """
    if x_min > x_max:
        raise AnalyzerError(f"invalid range [{x_min}, {x_max}]")
    y_lo = eval_fn(x_min)
    if within_tolerance(y_lo, y_target, SEARCH_TOLERANCE):
        return x_min, 0
    y_hi = eval_fn(x_max)
    if within_tolerance(y_hi, y_target, SEARCH_TOLERANCE):
        return x_max, 0

    increasing = y_lo < y_hi
    if (increasing and y_target < y_lo) or (not increasing and y_target > y_lo):
        return x_min, -1  # below the bounded region
    if (increasing and y_target > y_hi) or (not increasing and y_target < y_hi):
        return x_max, +1  # above the bounded region

    x_star = 0.5 * (x_min + x_max)
    for _ in range(MAX_SEARCH_ITERATIONS):
        x_star = 0.5 * (x_min + x_max)
        y_star = eval_fn(x_star)
        if within_tolerance(y_star, y_target, SEARCH_TOLERANCE):
            break
        # fixed point: interval collapsed to <=1 ulp — every remaining
        # iteration re-evaluates this same midpoint until the cap, so the
        # result is already final (identical x_star, fewer evaluations)
        if x_star == x_min or x_star == x_max:
            break
        if (increasing and y_target < y_star) or (not increasing and y_target > y_star):
            x_max = x_star
        else:
            x_min = x_star
    return x_star, 0


def build_service_rates(cfg: Configuration, req: RequestSize) -> np.ndarray:
    """State-dependent service rates s(n), n=1..N, in req/msec (float32).

    s(n) = n / (prefill(inTok, n) + numDecode * decode(n)) with
    numDecode = outTok - 1, except the decode-only single-token special case.
    Ref: queueanalyzer.go:99-114.
    """
    N = cfg.max_batch_size
    parms = cfg.service_parms
    n = np.arange(1, N + 1, dtype=np.float32)
    if req.avg_input_tokens == 0:
        prefill = np.zeros(N, dtype=np.float32)
    else:
        prefill = (
            np.float32(parms.prefill.gamma)
            + np.float32(parms.prefill.delta) * np.float32(req.avg_input_tokens) * n
        ).astype(np.float32)
    num_decode = req.avg_output_tokens - 1
    if req.avg_input_tokens == 0 and req.avg_output_tokens == 1:
        num_decode = 1
    decode = (np.float32(parms.decode.alpha) + np.float32(parms.decode.beta) * n).astype(np.float32)
    # degenerate perf parms can yield a zero/negative service time; the
    # resulting inf/nan rate is rejected by the callers' validity checks
    # (AnalyzerError / infeasible cell) — silence only the expected warning
    with np.errstate(divide="ignore", invalid="ignore"):
        serv = (n / (prefill + np.float32(num_decode) * decode)).astype(np.float32)
    return serv


class StateDependentChain:
    """Log-space solver of the state-dependent M/M/1/K birth-death chain.

    Equivalent (to float64 rounding) to the reference's rescaled forward
    recurrence (mm1modelstatedependent.go:70-116) but O(N) instead of O(K):
    only the N batch states are materialized; the 10*N saturated queue states
    are a geometric tail with closed-form sums.
    """

    def __init__(self, K: int, serv_rate: np.ndarray):
        self.K = int(K)
        self.serv_rate = np.asarray(serv_rate, dtype=np.float32)
        self.N = len(self.serv_rate)
        # prefix sums of log service rates (float64), S[m] = sum_{i=1..m} log s(i)
        self._log_s = np.log(self.serv_rate.astype(np.float64))
        self._S = np.concatenate(([0.0], np.cumsum(self._log_s)))

    def solve(self, lam: float) -> ChainStats:
        """Solve at arrival rate lam (req/msec) and return chain statistics."""
        if lam < 0:
            raise AnalyzerError(f"invalid model lambda={lam}")
        if lam == 0:
            return ChainStats(0.0, 0.0, 0.0, math.nan, math.nan, math.nan, 1.0, 0.0)
        K, N = self.K, self.N
        log_lam = math.log(lam)
        n_head = np.arange(0, N + 1, dtype=np.float64)
        # unnormalized log p[n] for the head states n=0..N
        t = n_head * log_lam - self._S
        m = float(np.max(t))
        w = np.exp(t - m)  # scaled head probabilities

        head_sum = float(np.sum(w))
        head_n_sum = float(np.sum(n_head * w))

        # geometric tail n = N+1..K with ratio r = lam/s(N)
        log_r = log_lam - float(self._log_s[-1])
        r = math.exp(log_r)
        wN = float(w[N])
        Q = K - N  # number of tail states
        if Q > 0 and wN > 0.0:
            rQ = math.exp(Q * log_r)
            if Q * abs(log_r) < 1e-6:
                # r ~ 1: flat tail (relative error < Q|log r|/2 < 5e-7); the
                # closed forms below cancel catastrophically in this regime
                tail_sum = wN * Q
                tail_n_sum = wN * (Q * N + Q * (Q + 1) / 2.0)
                wK = wN * rQ
            else:
                # expm1-stable forms: 1-r and 1-r^Q computed without
                # cancellation, and the arithmetico-geometric numerator
                # rewritten as (1-r^Q) - Q r^Q (1-r) so its error is
                # O(eps / (Q|log r|)) <= ~4e-10 at the 1e-6 threshold
                one_m_r = -math.expm1(log_r)
                one_m_rQ = -math.expm1(Q * log_r)
                g = r * one_m_rQ / one_m_r  # sum_{j=1..Q} r^j
                # sum_{j=1..Q} j r^j (arithmetico-geometric)
                jg = r * (one_m_rQ - Q * rQ * one_m_r) / (one_m_r * one_m_r)
                tail_sum = wN * g
                tail_n_sum = wN * (N * g + jg)
                wK = wN * rQ
        else:
            tail_sum = 0.0
            tail_n_sum = 0.0
            wK = wN if Q == 0 else 0.0

        Z = head_sum + tail_sum
        p0 = float(w[0]) / Z
        pK = wK / Z
        avg_n_sys = (head_n_sum + tail_n_sum) / Z
        # avg in servers: sum_{i=1..N} i p[i] + (1 - sum_{i=0..N} p[i]) * N
        # (ref mm1modelstatedependent.go:47-57)
        avg_n_serv = head_n_sum / Z + (1.0 - head_sum / Z) * N

        throughput = lam * (1.0 - pK)
        avg_resp = avg_n_sys / throughput if throughput > 0 else math.nan
        avg_serv = avg_n_serv / throughput if throughput > 0 else math.nan
        avg_wait = max(avg_resp - avg_serv, 0.0)
        return ChainStats(
            throughput=throughput,
            avg_num_in_system=avg_n_sys,
            avg_num_in_servers=avg_n_serv,
            avg_resp_time=avg_resp,
            avg_serv_time=avg_serv,
            avg_wait_time=avg_wait,
            p0=p0,
            pK=pK,
        )


def effective_concurrency(
    avg_service_time: float,
    parms: ServiceParms,
    req: RequestSize,
    max_batch_size: int,
) -> float:
    """Invert avg service time to an effective batch level n in [0, N].

    n satisfies prefill(n) + (outTok-1)*decode(n) = avgServiceTime.
    Ref: queueanalyzer.go:288-302.
    """
    tokens = float(req.avg_output_tokens - 1)
    numerator = avg_service_time - (parms.prefill.gamma + parms.decode.alpha * tokens)
    denominator = parms.prefill.delta * float(req.avg_input_tokens) + parms.decode.beta * tokens
    if denominator == 0:
        n = math.inf if numerator > 0 else 0.0
    else:
        n = numerator / denominator
    return min(max(n, 0.0), float(max_batch_size))


class QueueAnalyzer:
    """Analyzer of an inference-server queue. Ref: queueanalyzer.go:14-21,87-131."""

    def __init__(self, cfg: Configuration, req: RequestSize):
        cfg.check()
        req.check()
        self.max_batch_size = cfg.max_batch_size
        self.max_queue_size = cfg.max_queue_size
        self.service_parms = cfg.service_parms
        self.request_size = req
        self.serv_rate = build_service_rates(cfg, req)
        # degenerate perf parameters (e.g. all-zero alpha/beta/gamma/delta
        # from a malformed CR) produce non-positive or non-finite rates; the
        # cell is infeasible rather than propagating inf/nan into the chain
        if not np.all(np.isfinite(self.serv_rate)) or np.any(self.serv_rate <= 0):
            raise AnalyzerError(f"invalid service rates for configuration {cfg}")
        lambda_min = float(self.serv_rate[0]) * EPSILON
        lambda_max = float(self.serv_rate[-1]) * (1.0 - EPSILON)
        self.rate_min = lambda_min * 1000.0  # req/sec
        self.rate_max = lambda_max * 1000.0  # req/sec
        K = cfg.max_queue_size + cfg.max_batch_size
        self.chain = StateDependentChain(K, self.serv_rate)

    # -- internal evaluators (x is lambda in req/msec) --------------------

    def _eval_ttft(self, lam: float) -> float:
        """queueing wait + prefill at effective concurrency. Ref: queueanalyzer.go:270-279."""
        st = self.chain.solve(lam)
        eff = effective_concurrency(
            st.avg_serv_time, self.service_parms, self.request_size, self.max_batch_size
        )
        return st.avg_wait_time + self.service_parms.prefill.prefill_time(
            self.request_size.avg_input_tokens, eff
        )

    def _eval_itl(self, lam: float) -> float:
        """decode time at effective concurrency. Ref: queueanalyzer.go:283-286."""
        st = self.chain.solve(lam)
        eff = effective_concurrency(
            st.avg_serv_time, self.service_parms, self.request_size, self.max_batch_size
        )
        return self.service_parms.decode.decode_time(eff)

    def eval_serv_time(self, lam: float) -> float:
        """Avg service time at lam (req/msec). Ref: pkg/analyzer/utils.go:75-81."""
        return self.chain.solve(lam).avg_serv_time

    def eval_waiting_time(self, lam: float) -> float:
        """Avg waiting time at lam (req/msec). Ref: pkg/analyzer/utils.go:84-90."""
        return self.chain.solve(lam).avg_wait_time

    # -- public API --------------------------------------------------------

    def analyze(self, request_rate: float) -> AnalysisMetrics:
        """Evaluate metrics at a given request rate (req/sec).

        Ref: queueanalyzer.go:134-174.
        """
        if request_rate <= 0:
            raise AnalyzerError(f"invalid request rate {request_rate}")
        if request_rate > self.rate_max:
            raise AnalyzerError(f"rate={request_rate}, max allowed rate={self.rate_max}")
        st = self.chain.solve(request_rate / 1000.0)
        eff = effective_concurrency(
            st.avg_serv_time, self.service_parms, self.request_size, self.max_batch_size
        )
        prefill_time = self.service_parms.prefill.prefill_time(
            self.request_size.avg_input_tokens, eff
        )
        token_time = self.service_parms.decode.decode_time(eff)
        rho = min(max(st.avg_num_in_servers / float(self.max_batch_size), 0.0), 1.0)
        return AnalysisMetrics(
            throughput=st.throughput * 1000.0,
            avg_resp_time=st.avg_resp_time,
            avg_wait_time=st.avg_wait_time,
            avg_num_in_serv=st.avg_num_in_servers,
            avg_prefill_time=prefill_time,
            avg_token_time=token_time,
            max_rate=self.rate_max,
            rho=rho,
        )

    def size(self, targets: TargetPerf):
        """Max request rates achieving the SLO targets.

        Returns (TargetRate, AnalysisMetrics at min rate, achieved TargetPerf).
        Raises AnalyzerError when a target is below the feasible region
        (indicator < 0), matching Size (queueanalyzer.go:185-255).
        """
        targets.check()
        lambda_min = self.rate_min / 1000.0
        lambda_max = self.rate_max / 1000.0

        lam_ttft = lambda_max
        if targets.target_ttft > 0:
            lam_ttft, ind = binary_search(
                lambda_min, lambda_max, targets.target_ttft, self._eval_ttft
            )
            if ind < 0:
                raise AnalyzerError(
                    f"failed to calculate lambdaStarTTFT, targetTTFT={targets.target_ttft}: "
                    "target is below the bounded region"
                )

        lam_itl = lambda_max
        if targets.target_itl > 0:
            lam_itl, ind = binary_search(
                lambda_min, lambda_max, targets.target_itl, self._eval_itl
            )
            if ind < 0:
                raise AnalyzerError(
                    f"failed to calculate lambdaStarITL, targetITL={targets.target_itl}: "
                    "target is below the bounded region"
                )

        lam_tps = lambda_max
        if targets.target_tps > 0:
            lam_tps = lambda_max * (1.0 - STABILITY_SAFETY_FRACTION)

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
