"""Minimal Prometheus stand-in: real scrape loop + the query API subset the
collector uses.

Scrapes the configured targets' /metrics (Prometheus exposition format) on an
interval, keeps a sliding window of samples per (metric, labelset), and
answers ``GET /api/v1/query`` for exactly the PromQL shapes the controller
issues (internal/collector/collector.go:170-209 — byte-identical strings
built by inferno_amd.controller.collector):

    sum(rate(metric{label="v",...}[1m]))
    sum(rate(A{..}[1m]))/sum(rate(B{..}[1m]))
    up

Run: python -m inferno_amd.testing.promstub --port 0 --target http://host:p \
        [--interval 1.0]
Prints "promstub listening on <port>" when ready.
"""
from __future__ import annotations

import argparse
import re
import threading
import time
from collections import deque
from typing import Optional

_RATE_RE = re.compile(
    r"^sum\(rate\(([a-zA-Z_:][a-zA-Z0-9_:]*)\{([^}]*)\}\[(\d+)([smh])\]\)\)$"
)
_SELECTOR_RE = re.compile(r"^([a-zA-Z_:][a-zA-Z0-9_:]*)(?:\{([^}]*)\})?$")
_LABEL_RE = re.compile(r'([a-zA-Z_][a-zA-Z0-9_]*)="([^"]*)"')


class SeriesDB:
    """Sliding-window sample store per (metric, frozenset(labels))."""

    def __init__(self, window_s: float = 300.0):
        self.window_s = window_s
        self._lock = threading.Lock()
        self._series: dict[tuple[str, frozenset], deque] = {}

    def add(self, metric: str, labels: dict[str, str], value: float, ts: float) -> None:
        key = (metric, frozenset(labels.items()))
        with self._lock:
            dq = self._series.setdefault(key, deque())
            dq.append((ts, value))
            cutoff = ts - self.window_s
            while dq and dq[0][0] < cutoff:
                dq.popleft()

    def sum_rate(self, metric: str, matchers: dict[str, str], range_s: float,
                 now: Optional[float] = None) -> Optional[float]:
        """sum(rate(metric{matchers}[range])) — per-series simple rate
        (last-first)/(t_last-t_first) over the range window, summed."""
        now = now if now is not None else time.time()
        total = None
        with self._lock:
            for (m, lset), dq in self._series.items():
                if m != metric:
                    continue
                labels = dict(lset)
                if any(labels.get(k) != v for k, v in matchers.items()):
                    continue
                pts = [(t, v) for t, v in dq if t >= now - range_s]
                if len(pts) < 2:
                    continue
                t0, v0 = pts[0]
                t1, v1 = pts[-1]
                if t1 <= t0:
                    continue
                # counter-reset guard (prometheus rate semantics)
                delta = v1 - v0 if v1 >= v0 else v1
                total = (total or 0.0) + delta / (t1 - t0)
        return total

    def latest(self, metric: str, matchers: dict[str, str]) -> Optional[float]:
        """Instant-vector selector: sum of the most recent sample per
        matching series (the collector's availability probe uses this,
        collector.go:87-156)."""
        total = None
        with self._lock:
            for (m, lset), dq in self._series.items():
                if m != metric or not dq:
                    continue
                labels = dict(lset)
                if any(labels.get(k) != v for k, v in matchers.items()):
                    continue
                total = (total or 0.0) + dq[-1][1]
        return total


def scrape_once(db: SeriesDB, target: str, client) -> None:
    from prometheus_client.parser import text_string_to_metric_families

    r = client.get(f"{target}/metrics")
    r.raise_for_status()
    ts = time.time()
    for family in text_string_to_metric_families(r.text):
        for sample in family.samples:
            db.add(sample.name, dict(sample.labels), float(sample.value), ts)


def evaluate(db: SeriesDB, promql: str) -> Optional[float]:
    promql = promql.strip()
    if promql == "up":
        return 1.0
    # ratio split: only a "/" at paren depth 0 outside quotes is an operator
    # (label values like model_name="default/default" contain slashes)
    depth = 0
    in_q = False
    for i, ch in enumerate(promql):
        if ch == '"':
            in_q = not in_q
        elif not in_q and ch == "(":
            depth += 1
        elif not in_q and ch == ")":
            depth -= 1
        elif not in_q and ch == "/" and depth == 0:
            a = evaluate(db, promql[:i])
            b = evaluate(db, promql[i + 1:])
            if a is None or b is None or b == 0:
                return None
            return a / b
    m = _RATE_RE.match(promql)
    if m is not None:
        metric, labels_s, num, unit = m.groups()
        matchers = _matchers(labels_s)
        range_s = float(num) * {"s": 1, "m": 60, "h": 3600}[unit]
        return db.sum_rate(metric, matchers, range_s)
    m = _SELECTOR_RE.match(promql)
    if m is not None:
        metric, labels_s = m.groups()
        return db.latest(metric, _matchers(labels_s or ""))
    return None


def _matchers(labels_s: str) -> dict[str, str]:
    # drop empty-string matchers (the collector emits namespace="" when the
    # emulator exports no namespace label)
    return {k: v for k, v in _LABEL_RE.findall(labels_s) if v != ""}


def build_app(db: SeriesDB):
    # module-level so FastAPI can resolve the deferred "Request" annotations
    global FastAPI, Request, JSONResponse
    from fastapi import FastAPI, Request
    from fastapi.responses import JSONResponse

    app = FastAPI(title="promstub")

    @app.get("/api/v1/query")
    async def query(request: Request):
        q = request.query_params.get("query", "")
        val = evaluate(db, q)
        results = []
        if val is not None:
            results.append({"metric": {}, "value": [time.time(), f"{val:.10g}"]})
        return JSONResponse({
            "status": "success",
            "data": {"resultType": "vector", "result": results},
        })

    @app.get("/-/ready")
    async def ready():
        return JSONResponse({"ok": True})

    return app


def main() -> None:
    import socket

    import httpx
    import uvicorn

    p = argparse.ArgumentParser()
    p.add_argument("--port", type=int, default=0)
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--target", action="append", default=[],
                   help="base URL(s) to scrape /metrics from")
    p.add_argument("--interval", type=float, default=1.0)
    p.add_argument("--tls-cert", default="", help="serve the query API over TLS")
    p.add_argument("--tls-key", default="")
    args = p.parse_args()

    db = SeriesDB()
    client = httpx.Client(timeout=5.0)
    stop = threading.Event()

    def scrape_loop():
        while not stop.is_set():
            for t in args.target:
                try:
                    scrape_once(db, t, client)
                except Exception:
                    pass
            stop.wait(args.interval)

    threading.Thread(target=scrape_loop, daemon=True).start()

    app = build_app(db)
    sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
    sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
    sock.bind((args.host, args.port))
    port = sock.getsockname()[1]
    print(f"promstub listening on {port}", flush=True)
    cfg = uvicorn.Config(
        app, log_level="warning",
        ssl_certfile=args.tls_cert or None, ssl_keyfile=args.tls_key or None,
    )
    uvicorn.Server(cfg).run(sockets=[sock])


if __name__ == "__main__":
    main()
