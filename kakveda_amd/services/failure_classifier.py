"""Failure classifier: trace.ingested -> FailureSignal -> GFKB + event.

Rule parity with /root/reference/services/failure_classifier/app.py:31-89:
a prompt that *wants citations* combined with a response that *contains
citation markers* (with no sources provided) is a HALLUCINATION_CITATION
of medium severity with the canned root-cause/mitigation strings. The rule
path stays CPU-deterministic (BASELINE config 1); embedding happens inside
the GFKB engine at upsert time.
"""

from __future__ import annotations

import os
from typing import Optional

from fastapi import FastAPI

from kakveda_amd.core.schemas import FailureSignal, Severity, TracePayload
from kakveda_amd.core.signature import (
    detect_citation_markers,
    normalize_prompt,
    signature_text,
)
from kakveda_amd.services import TOPIC_FAILURE_DETECTED, TOPIC_TRACE_INGESTED
from kakveda_amd.services.wiring import Transport

_WANTS_CITATION_WORDS = (
    "citation",
    "citations",
    "reference",
    "references",
    "sources",
    "bibliography",
)

ROOT_CAUSE = "Model produced citations without provided sources"
MITIGATION = "Ask model to explicitly say 'no sources available' when none are provided"


def classify_trace(trace: TracePayload) -> Optional[FailureSignal]:
    """Return a FailureSignal when the citation-hallucination rule fires."""
    prompt_norm = normalize_prompt(trace.prompt)
    wants_citations = any(w in prompt_norm for w in _WANTS_CITATION_WORDS)
    has_markers = detect_citation_markers(trace.response).has_citation_markers
    if not (wants_citations and has_markers):
        return None
    return FailureSignal(
        trace_id=trace.trace_id,
        ts=trace.ts,
        app_id=trace.app_id,
        failure_type="HALLUCINATION_CITATION",
        severity=Severity.medium,
        root_cause=ROOT_CAUSE,
        mitigation=MITIGATION,
        context_signature={
            "prompt_shape": trace.prompt[:120],
            "model": trace.model,
            "tools": trace.tools,
            "env": trace.env,
        },
    )


def create_app(
    event_bus_url: Optional[str] = None,
    gfkb_url: Optional[str] = None,
    self_url: Optional[str] = None,
    transport: Optional[Transport] = None,
) -> FastAPI:
    app = FastAPI(title="Kakveda-AMD Failure Classifier")
    bus = event_bus_url or os.environ.get("EVENT_BUS_URL", "http://event-bus:8100")
    gfkb = gfkb_url or os.environ.get("GFKB_URL", "http://gfkb:8101")
    me = self_url or os.environ.get("SELF_URL", "http://failure-classifier:8103")
    tx = transport or Transport()
    app.state.transport = tx

    async def subscribe() -> None:
        await tx.post(
            f"{bus}/subscribe",
            json={"topic": TOPIC_TRACE_INGESTED, "callback_url": f"{me}/events/trace"},
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

        _aio.get_event_loop().create_task(_retry())  # bus may not be up yet; LocalCluster calls subscribe() itself

    @app.post("/events/trace")
    async def on_trace(event: dict):
        trace = TracePayload.model_validate(event)
        signal = classify_trace(trace)
        if signal is None:
            return {"ok": True, "failure": False}

        sig_text = signature_text(trace.prompt, trace.tools, trace.env)
        await tx.post(
            f"{gfkb}/failures/upsert",
            json={
                "failure_type": signal.failure_type,
                "root_cause": signal.root_cause,
                "context_signature": signal.context_signature,
                "impact_severity": signal.severity.value,
                "resolution": signal.mitigation,
                "signature_text": sig_text,
                "app_id": trace.app_id,
            },
        )
        await tx.post(
            f"{bus}/publish",
            json={
                "topic": TOPIC_FAILURE_DETECTED,
                "payload": signal.model_dump(mode="json"),
            },
        )
        return {"ok": True, "failure": True, "failure_type": signal.failure_type}

    @app.get("/healthz")
    async def healthz():
        return {"ok": True}

    return app
