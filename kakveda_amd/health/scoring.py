"""Health-score timeline engine.

Formula parity with /root/reference/services/health_scoring/app.py:58-107:
rolling window of the last 50 failures per app;
``score = max(0, base - 5*sum(weights) - 2.5*sum(recurrences-1))``;
``failure_rate = min(1, n/10)``; recovery placeholder ``30 + 10*penalty``.

Implemented as an incremental engine: the window's weighted sum and the
per-type counts are maintained as running aggregates (O(1) per event, not
the reference's O(window) rescan), which is what lets the 50k-QPS ingest
path (BASELINE config 5) batch thousands of events per tick.
"""

from __future__ import annotations

import threading
from collections import Counter, deque
from typing import Any, Deque, Dict, Iterable, Optional

from kakveda_amd.core.schemas import HealthPoint, utcnow
from kakveda_amd.core.store import JsonlLog

_DEFAULT_WEIGHTS = {"low": 1.0, "medium": 3.0, "high": 7.0}


class _AppWindow:
    __slots__ = ("events", "weighted", "type_counts")

    def __init__(self, size: int):
        self.events: Deque[Dict[str, Any]] = deque(maxlen=size)
        self.weighted = 0.0
        self.type_counts: Counter = Counter()

    def push(self, failure_type: str, severity: str, weight: float, ts: Any) -> None:
        if len(self.events) == self.events.maxlen:
            old = self.events[0]
            self.weighted -= old["weight"]
            self.type_counts[old["failure_type"]] -= 1
            if self.type_counts[old["failure_type"]] <= 0:
                del self.type_counts[old["failure_type"]]
        self.events.append(
            {"failure_type": failure_type, "severity": severity, "weight": weight, "ts": ts}
        )
        self.weighted += weight
        self.type_counts[failure_type] += 1

    @property
    def recurrent_penalty(self) -> float:
        return 2.5 * sum(max(0, c - 1) for c in self.type_counts.values())


class HealthScorer:
    def __init__(
        self,
        log: Optional[JsonlLog] = None,
        base_score: float = 100.0,
        weights: Optional[Dict[str, float]] = None,
        window_size: int = 50,
    ):
        self.log = log
        self.base = base_score
        self.weights = dict(weights or _DEFAULT_WEIGHTS)
        self.window_size = window_size
        self._windows: Dict[str, _AppWindow] = {}
        self._lock = threading.Lock()

    def observe(self, event: Dict[str, Any]) -> HealthPoint:
        """Fold one failure.detected event and emit a HealthPoint."""
        app_id = str(event.get("app_id", "unknown"))
        sev = str(event.get("severity", "low"))
        ftype = str(event.get("failure_type"))
        w = float(self.weights.get(sev, 1.0))
        with self._lock:
            win = self._windows.get(app_id)
            if win is None:
                win = self._windows[app_id] = _AppWindow(self.window_size)
            win.push(ftype, sev, w, event.get("ts"))

            n = len(win.events)
            penalty = win.recurrent_penalty
            score = max(0.0, self.base - win.weighted * 5.0 - penalty)
            top = win.type_counts.most_common(1)
            last = win.events[-1]
            point = HealthPoint(
                ts=utcnow(),
                app_id=app_id,
                score=score,
                failure_rate=min(1.0, n / 10.0),
                recurrent_penalty=penalty,
                avg_recovery_time_sec=30.0 + 10.0 * penalty,
                notes={
                    "window_failures": n,
                    "weighted": win.weighted,
                    "top_failure": top[0][0] if top else None,
                    "last_failure": last["failure_type"],
                    "last_severity": last["severity"],
                },
            )
        if self.log is not None:
            self.log.append(point.model_dump(mode="json"))
        return point

    def observe_batch(self, events: Iterable[Dict[str, Any]]) -> list[HealthPoint]:
        return [self.observe(e) for e in events]

    def timeline(self, app_id: str, limit: int = 100) -> list[Dict[str, Any]]:
        if self.log is None:
            return []
        return self.log.tail(limit, where=lambda r: r.get("app_id") == app_id)
