"""Go-style forward-recurrence oracle for the state-dependent M/M/1/K chain.

Independent re-implementation of the reference algorithm
(mm1modelstatedependent.go:38-116): float64 forward recurrence
p[n+1] = p[n]*lambda/s(n) with overflow rescaling, then normalization and
the avgNumInServers partial-sum statistics. Used ONLY by tests, as the
differential oracle for inferno_amd.analyzer.StateDependentChain's log-space
closed form (the production path).
"""
from __future__ import annotations

import math

import numpy as np


def chain_stats_recurrence(K: int, serv_rate: np.ndarray, lam: float):
    serv_rate = np.asarray(serv_rate, dtype=np.float32)
    num = len(serv_rate)
    p = np.zeros(K + 1, dtype=np.float64)
    p[0] = 1.0
    scale = np.finfo(np.float64).max / K
    for n in range(K):
        s = float(serv_rate[min(n, num - 1)])
        p[n + 1] = p[n] * lam / s
        while p[n + 1] < 0 or math.isinf(p[n + 1]) or math.isnan(p[n + 1]):
            p[: n + 1] /= scale
            p[n + 1] = p[n] * lam / s
    total = float(np.sum(p))
    p /= total

    avg_in_system = 0.0
    avg_in_servers = 0.0
    sum_p = p[0]
    for i in range(1, K + 1):
        avg_in_system += i * p[i]
        sum_p += p[i]
        if i == num:
            avg_in_servers = avg_in_system + (1.0 - sum_p) * num
    throughput = lam * (1.0 - p[K])
    avg_resp = avg_in_system / throughput if throughput > 0 else math.nan
    avg_serv = avg_in_servers / throughput if throughput > 0 else math.nan
    avg_wait = max(avg_resp - avg_serv, 0.0)
    return {
        "p0": p[0],
        "pK": p[K],
        "throughput": throughput,
        "avg_num_in_system": avg_in_system,
        "avg_num_in_servers": avg_in_servers,
        "avg_resp_time": avg_resp,
        "avg_serv_time": avg_serv,
        "avg_wait_time": avg_wait,
    }
