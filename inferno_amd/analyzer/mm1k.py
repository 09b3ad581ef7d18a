"""Closed-form finite-capacity queue models.

``MM1K`` mirrors the reference's classic geometric M/M/1/K
(``pkg/analyzer/mm1kmodel.go:28-108``). ``MG1K`` is the M/G/1-style
evaluator referenced by the sweep configs (BASELINE.json config 4): a
finite-capacity approximation that corrects waiting time by the squared
coefficient of variation of service time via the Pollaczek-Khinchine
factor — the reference carries the hook for it but only ships M/M/1/K;
here both are closed-form device functions (no chain needed) used as the
cheap path when the state-dependent chain is not required.
"""
from __future__ import annotations

import math
from dataclasses import dataclass


@dataclass
class QueueStats:
    rho: float
    throughput: float  # effective departure rate (same units as lambda)
    avg_num_in_system: float
    avg_resp_time: float
    avg_wait_time: float
    avg_serv_time: float
    avg_queue_length: float
    p0: float
    pK: float
    is_valid: bool


class MM1K:
    """M/M/1/K with geometric state probabilities. Ref: mm1kmodel.go."""

    def __init__(self, K: int):
        if K < 1:
            raise ValueError(f"invalid K={K}")
        self.K = int(K)

    def solve(self, lam: float, mu: float) -> QueueStats:
        K = self.K
        rho = 1.0 if lam == mu else (lam / mu if mu > 0 else math.inf)
        # validity per QueueModel.Solve (queuemodel.go:27-37): rho in [0, rhoMax=K)
        if rho < 0 or rho >= K or lam < 0 or mu <= 0:
            return QueueStats(rho, 0, 0, 0, 0, 0, 0, 1.0, 0.0, False)
        if rho == 1.0:
            p0 = 1.0 / (K + 1)
            pK = p0
            avg_n = K / 2.0
        else:
            p0 = (1.0 - rho) / (1.0 - rho ** (K + 1))
            pK = p0 * rho**K
            # E[n] = rho/(1-rho) - (K+1) rho^(K+1) / (1 - rho^(K+1))
            avg_n = rho / (1.0 - rho) - (K + 1) * rho ** (K + 1) / (1.0 - rho ** (K + 1))
        throughput = lam * (1.0 - pK)
        avg_resp = avg_n / throughput if throughput > 0 else math.nan
        avg_serv = 1.0 / mu
        avg_wait = max(avg_resp - avg_serv, 0.0)
        return QueueStats(
            rho=rho,
            throughput=throughput,
            avg_num_in_system=avg_n,
            avg_resp_time=avg_resp,
            avg_wait_time=avg_wait,
            avg_serv_time=avg_serv,
            avg_queue_length=throughput * avg_wait,
            p0=p0,
            pK=pK,
            is_valid=True,
        )


class MG1K:
    """M/G/1(/K) approximation with general service-time variability.

    Waiting time follows the Pollaczek-Khinchine mean-value formula scaled by
    (1 + cv^2)/2; blocking probability is approximated by the M/M/1/K pK at
    the same utilization (exact for cv=1, asymptotically correct for small
    and large rho). cv2 is the squared coefficient of variation of service
    time (cv2=1 recovers M/M/1/K behavior for the waiting time).
    """

    def __init__(self, K: int, cv2: float = 1.0):
        if K < 1:
            raise ValueError(f"invalid K={K}")
        if cv2 < 0:
            raise ValueError(f"invalid cv2={cv2}")
        self.K = int(K)
        self.cv2 = float(cv2)
        self._mm1k = MM1K(K)

    def solve(self, lam: float, mu: float) -> QueueStats:
        base = self._mm1k.solve(lam, mu)
        if not base.is_valid:
            return base
        factor = (1.0 + self.cv2) / 2.0
        avg_wait = base.avg_wait_time * factor
        avg_resp = avg_wait + base.avg_serv_time
        avg_n = base.throughput * avg_resp  # Little's law
        return QueueStats(
            rho=base.rho,
            throughput=base.throughput,
            avg_num_in_system=avg_n,
            avg_resp_time=avg_resp,
            avg_wait_time=avg_wait,
            avg_serv_time=base.avg_serv_time,
            avg_queue_length=base.throughput * avg_wait,
            p0=base.p0,
            pK=base.pK,
            is_valid=True,
        )
