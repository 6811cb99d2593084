"""Event bus: HTTP pub/sub hub.

Endpoint parity with /root/reference/services/event_bus/app.py:28-59:
``POST /subscribe`` (dedup callback urls), ``POST /publish`` (concurrent
best-effort fan-out, 3 s timeout, drop-on-error), ``GET /topics``.

Subscriptions are in-memory by default, as in the reference (its release
notes list this as a known limitation). Setting ``KAKVEDA_BUS_DURABLE``
to a directory makes them durable: every new subscription is appended to
``subscriptions.jsonl`` there and reloaded on startup, so a bus restart
no longer drops the fan-out graph (subscribers also re-subscribe with
retry on their own restarts; ``/subscribe`` dedups either way).
"""

from __future__ import annotations

import asyncio
import json
import os
from pathlib import Path
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


def create_app(
    transport: Optional[Transport] = None, durable_dir: Optional[str] = None
) -> FastAPI:
    app = FastAPI(title="Kakveda-AMD Event Bus")
    log = setup_logging("event-bus")
    topics: Dict[str, List[str]] = {}
    tx = transport or Transport()
    app.state.topics = topics
    app.state.transport = tx

    # optional durability (exceeds reference parity, opt-in)
    durable = durable_dir or os.environ.get("KAKVEDA_BUS_DURABLE") or ""
    sub_log: Optional[Path] = None
    if durable:
        sub_log = Path(durable) / "subscriptions.jsonl"
        sub_log.parent.mkdir(parents=True, exist_ok=True)
        if sub_log.exists():
            for line in sub_log.read_text().splitlines():
                try:
                    rec = json.loads(line)
                    urls = topics.setdefault(rec["topic"], [])
                    if rec["callback_url"] not in urls:
                        urls.append(rec["callback_url"])
                except (ValueError, KeyError):
                    continue
    app.state.sub_log = sub_log

    @app.post("/subscribe")
    async def subscribe(req: SubscribeRequest):
        urls = topics.setdefault(req.topic, [])
        if req.callback_url not in urls:
            urls.append(req.callback_url)
            if sub_log is not None:
                with open(sub_log, "a") as f:
                    f.write(json.dumps(
                        {"topic": req.topic, "callback_url": req.callback_url}
                    ) + "\n")
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
