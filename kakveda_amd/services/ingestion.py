"""Ingestion edge: validate a trace and republish it as trace.ingested.

Parity with /root/reference/services/ingestion/app.py:15-21.
"""

from __future__ import annotations

import os
from typing import Optional

from fastapi import FastAPI

from kakveda_amd.core.schemas import IngestRequest
from kakveda_amd.services import TOPIC_TRACE_INGESTED
from kakveda_amd.services.wiring import Transport


def create_app(
    event_bus_url: Optional[str] = None, transport: Optional[Transport] = None
) -> FastAPI:
    app = FastAPI(title="Kakveda-AMD Ingestion")
    bus = event_bus_url or os.environ.get("EVENT_BUS_URL", "http://event-bus:8100")
    tx = transport or Transport()
    app.state.transport = tx

    @app.post("/ingest")
    async def ingest(req: IngestRequest):
        await tx.post(
            f"{bus}/publish",
            json={"topic": TOPIC_TRACE_INGESTED, "payload": req.trace.model_dump(mode="json")},
        )
        return {"ok": True, "trace_id": req.trace.trace_id}

    @app.get("/healthz")
    async def healthz():
        return {"ok": True}

    return app


app = create_app()
