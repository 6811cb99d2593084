"""GFKB HTTP service fronting the GfkbEngine.

Endpoint parity with /root/reference/services/gfkb/app.py:74-198:
GET /failures, POST /failures/match, POST /failures/upsert,
GET /patterns, POST /patterns/upsert. The engine behind them is the
HBM-resident embedding store + fused cosine-topk kernel (gfkb/engine.py)
instead of a per-request TF-IDF refit.
"""

from __future__ import annotations

import os
import queue
import threading
from typing import Any, Dict, List, Optional

from fastapi import FastAPI
from pydantic import BaseModel, Field

from kakveda_amd.core.schemas import (
    FailureMatchRequest,
    FailureMatchResponse,
    Severity,
)
from kakveda_amd.gfkb.engine import GfkbEngine


class MatchBatcher:
    """Coalesces concurrent /failures/match requests into one fused-kernel
    launch (serving concurrency).

    Adaptive batching with zero added idle latency: the collector thread
    blocks for the first request, then drains whatever else is queued at
    that instant (requests naturally accumulate while the previous
    batch's kernel is in flight) — a single client sees one thread-hop,
    N concurrent clients share one engine.match_batch launch instead of
    serialising N kernel launches behind the engine lock.
    """

    def __init__(self, engine: GfkbEngine, max_batch: int = 256):
        self.engine = engine
        self.max_batch = max_batch
        self._q: "queue.SimpleQueue" = queue.SimpleQueue()
        self._thread: Optional[threading.Thread] = None
        self._start_lock = threading.Lock()
        self.batches = 0  # observability: launches issued
        self.requests = 0  # requests served through the batcher

    def _ensure_thread(self) -> None:
        if self._thread is None or not self._thread.is_alive():
            with self._start_lock:
                if self._thread is None or not self._thread.is_alive():
                    self._thread = threading.Thread(
                        target=self._loop, name="gfkb-match-batcher", daemon=True
                    )
                    self._thread.start()

    def match(self, signature_text: str, failure_type: Optional[str] = None):
        self._ensure_thread()
        item: Dict[str, Any] = {
            "text": signature_text,
            "ftype": failure_type,
            "ev": threading.Event(),
            "res": None,
            "err": None,
        }
        self._q.put(item)
        if not item["ev"].wait(timeout=120.0):
            raise TimeoutError("match batcher timed out")
        if item["err"] is not None:
            raise item["err"]
        return item["res"]

    def _loop(self) -> None:
        while True:
            batch = [self._q.get()]
            while len(batch) < self.max_batch:
                try:
                    batch.append(self._q.get_nowait())
                except queue.Empty:
                    break
            try:
                results = self.engine.match_batch(
                    [b["text"] for b in batch], [b["ftype"] for b in batch]
                )
                for b, r in zip(batch, results):
                    b["res"] = r
            except Exception as exc:  # surfaced to every waiting request
                for b in batch:
                    b["err"] = exc
            self.batches += 1
            self.requests += len(batch)
            for b in batch:
                b["ev"].set()


class UpsertFailureRequest(BaseModel):
    failure_type: str
    root_cause: Optional[str] = None
    context_signature: Dict[str, Any] = Field(default_factory=dict)
    impact_severity: Severity = Severity.medium
    resolution: Optional[str] = None
    signature_text: str
    app_id: str


class UpsertPatternRequest(BaseModel):
    name: str
    failure_ids: List[str]
    affected_apps: List[str]
    description: Optional[str] = None


def create_app(
    data_dir: Optional[str] = None,
    device: Optional[str] = None,
    engine: Optional[GfkbEngine] = None,
) -> FastAPI:
    app = FastAPI(title="Kakveda-AMD GFKB")
    if engine is None:
        import torch

        dev = device or ("cuda" if torch.cuda.is_available() else "cpu")
        engine = GfkbEngine(
            data_dir=data_dir or os.environ.get("DATA_DIR", "/app/data"), device=dev
        )
    app.state.engine = engine
    # micro-batcher on by default; KAKVEDA_MATCH_BATCHER=0 opts out
    batcher = (
        MatchBatcher(engine)
        if os.environ.get("KAKVEDA_MATCH_BATCHER", "1") == "1"
        else None
    )
    app.state.batcher = batcher

    # handlers are plain `def` so FastAPI runs them in its threadpool: the
    # engine is synchronous (GPU kernels, file appends) and must not block
    # the event loop — a slow match would stall every concurrent request
    # including /healthz (matches the reference's sync-handler behaviour)

    # anyio's default 40-token thread limiter caps how many requests can
    # be parked inside `match` waiting on the batcher, which caps the
    # micro-batch size and hence service throughput. Raise it so the
    # batcher, not the threadpool, sets the batch size. (The front-door
    # warn app registers the same hook: ASGITransport composition never
    # runs THIS app's lifespan.)
    @app.on_event("startup")
    async def _raise_threadpool():
        from kakveda_amd.services.wiring import raise_thread_limiter

        raise_thread_limiter()

    @app.get("/failures")
    def list_failures():
        return {"failures": engine.list_failures()}

    @app.post("/failures/match", response_model=FailureMatchResponse)
    def match(req: FailureMatchRequest):
        from kakveda_amd.core.metrics import observe_gfkb

        if batcher is not None:
            matches = batcher.match(req.signature_text, failure_type=req.failure_type)
        else:
            matches = engine.match(req.signature_text, failure_type=req.failure_type)
        observe_gfkb(engine.store.count, len(engine.failures))
        return FailureMatchResponse(matches=matches)

    @app.post("/failures/upsert")
    def upsert(req: UpsertFailureRequest):
        rec, created = engine.upsert_failure(
            failure_type=req.failure_type,
            signature_text=req.signature_text,
            context_signature=req.context_signature,
            impact_severity=req.impact_severity.value,
            root_cause=req.root_cause,
            resolution=req.resolution,
            app_id=req.app_id,
        )
        return {"ok": True, "created": created, "failure": rec}

    @app.get("/patterns")
    def list_patterns():
        return {"patterns": engine.list_patterns()}

    @app.post("/patterns/upsert")
    def upsert_pattern(req: UpsertPatternRequest):
        rec, created = engine.upsert_pattern(
            name=req.name,
            failure_ids=req.failure_ids,
            affected_apps=req.affected_apps,
            description=req.description,
        )
        return {"ok": True, "created": created, "pattern": rec}

    @app.get("/healthz")
    def healthz():
        from kakveda_amd.core.metrics import observe_gfkb

        observe_gfkb(engine.store.count, len(engine.failures))
        return {"ok": True, "failures": len(engine.failures), "rows": engine.store.count}

    return app
