"""Exponential-backoff retry helpers (wait.Backoff equivalents).

Ref internal/utils/utils.go:31-104: Standard (100ms x2 5 steps), Reconcile
(500ms x2 5 steps) and Prometheus (5s x2 6 steps) profiles.
"""
from __future__ import annotations

import random
import time
from dataclasses import dataclass
from typing import Callable, TypeVar

T = TypeVar("T")


@dataclass
class Backoff:
    duration: float  # initial sleep seconds
    factor: float = 2.0
    jitter: float = 0.0
    steps: int = 5


STANDARD_BACKOFF = Backoff(duration=0.1, factor=2.0, jitter=0.1, steps=5)
RECONCILE_BACKOFF = Backoff(duration=0.5, factor=2.0, steps=5)
PROMETHEUS_BACKOFF = Backoff(duration=5.0, factor=2.0, jitter=0.1, steps=6)


def retry_with_backoff(fn: Callable[[], T], backoff: Backoff = STANDARD_BACKOFF,
                       retryable: Callable[[Exception], bool] = lambda e: True) -> T:
    """Run fn with exponential backoff; raises the last error when steps are
    exhausted or the error is non-retryable."""
    delay = backoff.duration
    last: Exception | None = None
    for step in range(backoff.steps):
        try:
            return fn()
        except Exception as e:  # noqa: BLE001 - retry layer
            last = e
            if not retryable(e) or step == backoff.steps - 1:
                raise
        sleep = delay
        if backoff.jitter:
            sleep += delay * backoff.jitter * random.random()
        time.sleep(sleep)
        delay *= backoff.factor
    raise last if last else RuntimeError("unreachable")
