"""Timing + percentile helpers shared by the dashboard, evals and benches."""

from __future__ import annotations

import time
from typing import Dict, List, Optional


def percentiles(vals: List[float]) -> Dict[str, float]:
    """p50/p95 of a sample (empty-safe; reference app.py:2449-2464 math)."""
    if not vals:
        return {"p50": 0.0, "p95": 0.0}
    vs = sorted(vals)
    return {
        "p50": vs[len(vs) // 2],
        "p95": vs[min(len(vs) - 1, int(len(vs) * 0.95))],
    }


class Stopwatch:
    """Accumulates named spans: ``with sw.span("warn"): ...``."""

    def __init__(self):
        self.spans: List[Dict[str, float]] = []

    class _Span:
        def __init__(self, sw: "Stopwatch", name: str):
            self.sw = sw
            self.name = name
            self.t0: Optional[float] = None

        def __enter__(self):
            self.t0 = time.perf_counter()
            return self

        def __exit__(self, *exc):
            self.sw.spans.append(
                {
                    "name": self.name,
                    "duration_ms": (time.perf_counter() - self.t0) * 1000.0,
                }
            )
            return False

    def span(self, name: str) -> "Stopwatch._Span":
        return Stopwatch._Span(self, name)

    def total_ms(self) -> float:
        return sum(s["duration_ms"] for s in self.spans)
