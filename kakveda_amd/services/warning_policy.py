"""Warning policy: the pre-flight check (the north-star hot path).

Parity with /root/reference/services/warning_policy/app.py:19-72: build
the signature text, match against the GFKB, attach a pattern id via the
demo name heuristic, and warn/block when the best score clears the
config-driven similarity threshold (default 0.8). Response message format
matches the reference so dashboards/clients render identically.
"""

from __future__ import annotations

import os
from typing import Optional

from fastapi import FastAPI, HTTPException

from kakveda_amd.core.config import ConfigStore
from kakveda_amd.core.schemas import WarningRequest, WarningResponse
from kakveda_amd.core.signature import signature_text
from kakveda_amd.services.wiring import Transport

_DEMO_PATTERN_NAME = "Citation hallucination without sources"


def create_app(
    gfkb_url: Optional[str] = None,
    transport: Optional[Transport] = None,
    config: Optional[ConfigStore] = None,
) -> FastAPI:
    app = FastAPI(title="Kakveda-AMD Warning Policy")
    gfkb = gfkb_url or os.environ.get("GFKB_URL", "http://gfkb:8101")
    tx = transport or Transport()
    cfg = config or ConfigStore()
    app.state.transport = tx

    # front door: raise the sync-handler thread limiter for this event
    # loop (covers the locally-composed gfkb app too — its own lifespan
    # never runs under ASGITransport). See wiring.raise_thread_limiter.
    @app.on_event("startup")
    async def _raise_threadpool():
        from kakveda_amd.services.wiring import raise_thread_limiter

        raise_thread_limiter()

    # admission control: past ~512 outstanding requests per worker the
    # service degrades non-linearly (measured overload knee —
    # profiles/serving_http.md), so shed load with a fast 503 +
    # Retry-After instead of letting every request queue into the knee.
    # KAKVEDA_MAX_INFLIGHT=0 disables.
    max_inflight = int(os.environ.get("KAKVEDA_MAX_INFLIGHT", "512"))
    inflight = {"n": 0}
    app.state.inflight = inflight

    @app.post("/warn", response_model=WarningResponse)
    async def warn(req: WarningRequest):
        if max_inflight > 0 and inflight["n"] >= max_inflight:
            from kakveda_amd.core.metrics import observe_shed

            observe_shed()
            raise HTTPException(
                status_code=503,
                detail="warning-policy at capacity; retry",
                headers={"Retry-After": "1"},
            )
        inflight["n"] += 1
        try:
            return await _warn(req)
        finally:
            inflight["n"] -= 1

    async def _warn(req: WarningRequest) -> WarningResponse:
        threshold = float(cfg.get("failure_matching.similarity_threshold", 0.8))
        action_default = str(cfg.get("warning_policy.default_action", "warn"))

        sig_txt = signature_text(req.prompt, req.tools, req.env)
        resp = await tx.post(
            f"{gfkb}/failures/match", json={"signature_text": sig_txt}, timeout=3.5
        )
        matches = resp.json().get("matches", [])
        best = matches[0] if matches else None
        score = float(best.get("score", 0.0)) if best else 0.0

        pattern_id = None
        try:
            presp = await tx.get(f"{gfkb}/patterns")
            patterns = presp.json().get("patterns", [])
            if best and best.get("failure_type") == "HALLUCINATION_CITATION":
                for p in reversed(patterns):
                    if p.get("name") == _DEMO_PATTERN_NAME:
                        pattern_id = p.get("pattern_id")
                        break
        except Exception:
            pattern_id = None

        from kakveda_amd.core.metrics import observe_warn

        if best and score >= threshold:
            observe_warn(action_default)
            msg = (
                f"This execution matches past failure type {best.get('failure_type')} "
                f"(failure_id={best.get('failure_id')}, similarity={score:.2f}). "
                f"Suggested mitigation: {best.get('suggested_mitigation') or 'n/a'}"
            )
            return WarningResponse(
                action=action_default,
                confidence=score,
                pattern_id=pattern_id,
                references=[best],
                message=msg,
            )
        observe_warn("silent" if action_default == "silent" else "warn")
        return WarningResponse(
            action="silent" if action_default == "silent" else "warn",
            confidence=score,
            pattern_id=pattern_id,
            references=[],
            message="No high-similarity match found in GFKB.",
        )

    @app.get("/healthz")
    async def healthz():
        return {"ok": True}

    return app
