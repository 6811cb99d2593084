"""Reference demo agent implementing the agent contract.

Parity with /root/reference/services/agent_echo/app.py:13-47 and the
contract in docs/agents.md: GET /health, GET /capabilities, POST /invoke.
"""

from __future__ import annotations

from typing import Any, Dict, List

from fastapi import FastAPI
from pydantic import BaseModel, Field


class InvokeRequest(BaseModel):
    event: Dict[str, Any] = Field(default_factory=dict)


def create_app() -> FastAPI:
    app = FastAPI(title="Kakveda-AMD Agent Echo")

    @app.get("/health")
    async def health():
        return {"ok": True, "agent": "agent-echo"}

    @app.get("/capabilities")
    async def capabilities():
        return {
            "name": "agent-echo",
            "capabilities": ["echo"],
            "events_in": ["*"],
            "events_out": ["echo.reply"],
        }

    @app.post("/invoke")
    async def invoke(req: InvokeRequest):
        events_out: List[Dict[str, Any]] = [
            {"topic": "echo.reply", "payload": req.event}
        ]
        return {"ok": True, "events_out": events_out}

    return app


app = create_app()
