"""Prometheus metrics for every service (exceeds reference parity: the
reference has no metrics endpoint — SURVEY.md 5.5 'No Prometheus').

``instrument(app, service)`` adds an HTTP middleware recording request
counts and latency histograms per (service, path, method, status) and a
``GET /metrics`` endpoint in the Prometheus text format. GFKB-specific
gauges (store rows, failure count) are registered by the GFKB service.
"""

from __future__ import annotations

import time
from typing import Callable, Optional

from fastapi import FastAPI, Request, Response

try:
    from prometheus_client import (
        CONTENT_TYPE_LATEST,
        Counter,
        Gauge,
        Histogram,
        generate_latest,
    )

    _AVAILABLE = True
except ImportError:  # pragma: no cover
    _AVAILABLE = False

if _AVAILABLE:
    REQUESTS = Counter(
        "kakveda_http_requests_total",
        "HTTP requests",
        ["service", "method", "path", "status"],
    )
    LATENCY = Histogram(
        "kakveda_http_request_seconds",
        "HTTP request latency",
        ["service", "path"],
        buckets=(0.001, 0.005, 0.01, 0.025, 0.05, 0.1, 0.25, 0.5, 1.0, 2.5, 5.0),
    )
    GFKB_ROWS = Gauge("kakveda_gfkb_rows", "Live fingerprint rows in the GFKB store")
    GFKB_FAILURE_RECORDS = Gauge(
        "kakveda_gfkb_failure_records", "Append-only failure records (all versions)"
    )
    WARN_DECISIONS = Counter(
        "kakveda_warn_decisions_total", "Pre-flight warning decisions", ["action"]
    )
    WARN_SHED = Counter(
        "kakveda_warn_shed_total",
        "Pre-flight requests shed by admission control (503)",
    )


def _route_template(request: Request) -> str:
    route = request.scope.get("route")
    return getattr(route, "path", request.url.path)


def instrument(app: FastAPI, service: str) -> bool:
    """Attach metrics middleware + /metrics; no-op if the client is absent."""
    if not _AVAILABLE:
        return False

    @app.middleware("http")
    async def _metrics_mw(request: Request, call_next: Callable):
        t0 = time.perf_counter()
        response: Response = await call_next(request)
        path = _route_template(request)
        if path != "/metrics":
            REQUESTS.labels(service, request.method, path, response.status_code).inc()
            LATENCY.labels(service, path).observe(time.perf_counter() - t0)
        return response

    @app.get("/metrics")
    async def metrics():
        return Response(generate_latest(), media_type=CONTENT_TYPE_LATEST)

    return True


def observe_gfkb(rows: int, failure_records: int) -> None:
    if _AVAILABLE:
        GFKB_ROWS.set(rows)
        GFKB_FAILURE_RECORDS.set(failure_records)


def observe_warn(action: Optional[str]) -> None:
    if _AVAILABLE and action:
        WARN_DECISIONS.labels(action).inc()


def observe_shed() -> None:
    if _AVAILABLE:
        WARN_SHED.inc()
