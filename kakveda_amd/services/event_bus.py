"""Event bus: HTTP pub/sub hub.

Endpoint parity with /root/reference/services/event_bus/app.py:28-59:
``POST /subscribe`` (dedup callback urls), ``POST /publish`` (concurrent
best-effort fan-out, 3 s timeout, drop-on-error), ``GET /topics``.
Subscriptions are in-memory (non-durable), as in the reference.
"""

from __future__ import annotations

import asyncio
from typing import Any, Dict, List, Optional

from fastapi import FastAPI
from pydantic import BaseModel

from kakveda_amd.core.runtime import setup_logging
from kakveda_amd.services.wiring import Transport


class SubscribeRequest(BaseModel):
    topic: str
    callback_url: str


class PublishRequest(BaseModel):
    topic: str
    payload: Dict[str, Any]


def create_app(transport: Optional[Transport] = None) -> FastAPI:
    app = FastAPI(title="Kakveda-AMD Event Bus")
    log = setup_logging("event-bus")
    topics: Dict[str, List[str]] = {}
    tx = transport or Transport()
    app.state.topics = topics
    app.state.transport = tx

    @app.post("/subscribe")
    async def subscribe(req: SubscribeRequest):
        urls = topics.setdefault(req.topic, [])
        if req.callback_url not in urls:
            urls.append(req.callback_url)
        return {"ok": True, "topic": req.topic, "subscribers": len(urls)}

    @app.post("/publish")
    async def publish(req: PublishRequest):
        urls = list(topics.get(req.topic, []))

        async def _deliver(url: str) -> bool:
            try:
                await tx.post(url, json=req.payload, timeout=3.0)
                return True
            except Exception as exc:  # best-effort: drop on error
                log.warning("drop %s -> %s: %s", req.topic, url, exc)
                return False

        results = await asyncio.gather(*(_deliver(u) for u in urls))
        return {"ok": True, "delivered": sum(results), "subscribers": len(urls)}

    @app.get("/topics")
    async def list_topics():
        return {"topics": {t: list(u) for t, u in topics.items()}}

    @app.get("/healthz")
    async def healthz():
        return {"ok": True}

    return app


app = create_app()
