"""Pattern detector: failure.detected -> recurring-pattern entities.

Parity with /root/reference/services/pattern_detector/app.py:19-58 — on a
HALLUCINATION_CITATION event, group GFKB failures by type and upsert the
named demo pattern once >= 2 apps are affected. The generalised clustering
path (streaming k-means over fingerprint embeddings) lives
in kakveda_amd.patterns and can be driven via POST /cluster/run.
"""

from __future__ import annotations

import os
from typing import Optional

from fastapi import FastAPI

from kakveda_amd.services import TOPIC_FAILURE_DETECTED
from kakveda_amd.services.wiring import Transport

_DEMO_PATTERN_NAME = "Citation hallucination without sources"
_DEMO_PATTERN_DESC = "Same prompt pattern causes hallucinated citations across apps"


def create_app(
    event_bus_url: Optional[str] = None,
    gfkb_url: Optional[str] = None,
    self_url: Optional[str] = None,
    transport: Optional[Transport] = None,
    engine=None,
) -> FastAPI:
    app = FastAPI(title="Kakveda-AMD Pattern Detector")
    bus = event_bus_url or os.environ.get("EVENT_BUS_URL", "http://event-bus:8100")
    gfkb = gfkb_url or os.environ.get("GFKB_URL", "http://gfkb:8101")
    me = self_url or os.environ.get("SELF_URL", "http://pattern-detector:8104")
    tx = transport or Transport()
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
        if event.get("failure_type") != "HALLUCINATION_CITATION":
            return {"ok": True, "pattern": False}

        resp = await tx.get(f"{gfkb}/failures")
        failures = resp.json().get("failures", [])
        by_type: dict[str, dict] = {}
        for f in failures:
            ft = f.get("failure_type", "")
            slot = by_type.setdefault(ft, {"ids": set(), "apps": set()})
            slot["ids"].add(f.get("failure_id"))
            slot["apps"].update(f.get("affected_apps", []))

        slot = by_type.get("HALLUCINATION_CITATION")
        if not slot or len(slot["apps"]) < 2:
            return {"ok": True, "pattern": False}

        await tx.post(
            f"{gfkb}/patterns/upsert",
            json={
                "name": _DEMO_PATTERN_NAME,
                "failure_ids": sorted(x for x in slot["ids"] if x),
                "affected_apps": sorted(slot["apps"]),
                "description": _DEMO_PATTERN_DESC,
            },
        )
        return {"ok": True, "pattern": True}

    @app.post("/cluster/run")
    async def cluster_run(body: Optional[dict] = None):
        """Generalised pattern mining: streaming k-means over the GFKB
        fingerprint embeddings (GPU kernels when the engine is on GPU;
        RCCL all-reduce of centroid partials when sharded)."""
        if engine is None:
            return {"ok": False, "error": "no local engine attached"}
        from kakveda_amd.patterns.miner import PatternMiner

        body = body or {}
        miner = PatternMiner(
            engine,
            n_clusters=int(body.get("n_clusters", 16)),
            min_apps=int(body.get("min_apps", 2)),
        )
        patterns = miner.mine(iters=int(body.get("iters", 8)))
        return {"ok": True, "patterns": patterns}

    @app.get("/healthz")
    async def healthz():
        return {"ok": True}

    return app
