"""Health-scoring service: failure.detected -> health timeline.

Endpoint parity with /root/reference/services/health_scoring/app.py:39-130
(``POST /events/failure``, ``GET /health/{app_id}?limit=``, ``/healthz``),
backed by the incremental HealthScorer engine (kakveda_amd.health).
"""

from __future__ import annotations

import os
from typing import Optional

from fastapi import FastAPI

from kakveda_amd.core.config import ConfigStore
from kakveda_amd.core.store import JsonlLog
from kakveda_amd.health.scoring import HealthScorer
from kakveda_amd.services import TOPIC_FAILURE_DETECTED
from kakveda_amd.services.wiring import Transport


def create_app(
    event_bus_url: Optional[str] = None,
    data_dir: Optional[str] = None,
    self_url: Optional[str] = None,
    transport: Optional[Transport] = None,
    config: Optional[ConfigStore] = None,
) -> FastAPI:
    app = FastAPI(title="Kakveda-AMD Health Scoring")
    bus = event_bus_url or os.environ.get("EVENT_BUS_URL", "http://event-bus:8100")
    me = self_url or os.environ.get("SELF_URL", "http://health-scoring:8106")
    ddir = data_dir or os.environ.get("DATA_DIR", "/app/data")
    tx = transport or Transport()
    cfg = config or ConfigStore()

    os.makedirs(ddir, exist_ok=True)
    scorer = HealthScorer(
        log=JsonlLog(os.path.join(ddir, "health.jsonl")),
        base_score=float(cfg.get("health_score.base_score", 100)),
        weights={
            str(k): float(v)
            for k, v in (cfg.get("health_score.severity_weights") or {}).items()
        }
        or None,
    )
    app.state.scorer = scorer
    app.state.transport = tx

    async def subscribe() -> None:
        await tx.post(
            f"{bus}/subscribe",
            json={"topic": TOPIC_FAILURE_DETECTED, "callback_url": f"{me}/events/failure"},
        )

    app.state.subscribe = subscribe

    @app.on_event("startup")
    async def _startup():
        # the bus may come up after us (all services start concurrently
        # under `kakveda up`): retry in the background until subscribed;
        # /subscribe dedups, so LocalCluster's direct call stays safe
        import asyncio as _aio

        async def _retry():
            for _ in range(30):
                try:
                    await subscribe()
                    return
                except Exception:
                    await _aio.sleep(1.0)

        _aio.get_event_loop().create_task(_retry())

    @app.post("/events/failure")
    async def on_failure(event: dict):
        point = scorer.observe(event)
        return {"ok": True, "health": point.model_dump(mode="json")}

    @app.get("/health/{app_id}")
    async def health(app_id: str, limit: int = 100):
        return {"app_id": app_id, "points": scorer.timeline(app_id, limit=limit)}

    @app.get("/healthz")
    async def healthz():
        return {"ok": True}

    return app
