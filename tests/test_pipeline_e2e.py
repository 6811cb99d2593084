"""Deterministic CPU end-to-end pipeline (BASELINE config 1).

Mirrors the reference's scripts/demo_client.py flow:
warn -> generate (deterministic citation stub) -> ingest -> classify ->
GFKB upsert -> pattern detection -> health scoring, all in one process.
"""

from datetime import datetime, timezone

from kakveda_amd.services.cluster import LocalCluster

PROMPT = "Summarize this article and include references even if none are provided."


def _stub_generate(prompt: str) -> str:
    """The deterministic model stub (reference dashboard/app.py:1193-1199):
    always emits citation markers so the failure pipeline is reproducible."""
    return (
        "Here is a summary with sources. [1] Example citation. [2] Another. "
        "References: [1] A. Author 2020. [2] B. Author 2021."
    )


def _trace(app_id: str, i: int) -> dict:
    return {
        "trace_id": f"t-{app_id}-{i}",
        "ts": datetime.now(timezone.utc).isoformat(),
        "app_id": app_id,
        "prompt": PROMPT,
        "response": _stub_generate(PROMPT),
        "model": "e2e-stub",
        "tools": [],
        "env": {"e2e": True},
    }


async def test_full_pipeline(tmp_path):
    cluster = LocalCluster(data_dir=str(tmp_path), device="cpu")
    await cluster.start()

    # 1. cold warn: no matches yet
    w0 = await cluster.warn("app-A", PROMPT)
    assert w0["references"] == []
    assert w0["action"] in ("warn", "silent")

    # 2. ingest a failing trace from app-A -> classifier fires -> GFKB row
    r = await cluster.ingest(_trace("app-A", 1))
    assert r["ok"]
    failures = cluster.engine.list_failures()
    assert len(failures) == 1
    assert failures[0]["failure_type"] == "HALLUCINATION_CITATION"

    # 3. warm warn on the same prompt: high-confidence match
    w1 = await cluster.warn("app-A", PROMPT)
    assert w1["confidence"] >= 0.8
    assert w1["references"], w1
    assert w1["references"][0]["failure_id"] == "F-0001"
    assert "HALLUCINATION_CITATION" in w1["message"]

    # 4. second app -> pattern appears (>=2 affected apps)
    await cluster.ingest(_trace("app-B", 1))
    patterns = cluster.engine.list_patterns()
    assert len(patterns) == 1
    assert patterns[0]["name"] == "Citation hallucination without sources"
    assert set(patterns[0]["affected_apps"]) == {"app-A", "app-B"}

    # 5. warn now carries the pattern id
    w2 = await cluster.warn("app-B", PROMPT)
    assert w2["pattern_id"] == patterns[0]["pattern_id"]

    # 6. health degraded for app-A after repeated failures
    for i in range(2, 6):
        await cluster.ingest(_trace("app-A", i))
    scorer = cluster.health_scoring.state.scorer
    points = scorer.timeline("app-A", limit=10)
    assert points
    assert points[-1]["score"] < 100.0
    assert points[-1]["recurrent_penalty"] > 0

    # 7. unrelated prompt stays below threshold
    w3 = await cluster.warn("app-A", "What's the capital of France?")
    assert w3["confidence"] < 0.8
    assert w3["references"] == []

    await cluster.aclose()


async def test_event_bus_topics(tmp_path):
    cluster = LocalCluster(data_dir=str(tmp_path), device="cpu")
    await cluster.start()
    topics = cluster.event_bus.state.topics
    assert "trace.ingested" in topics
    assert "failure.detected" in topics
    await cluster.aclose()


async def test_event_bus_durable_subscriptions(tmp_path, monkeypatch):
    """KAKVEDA_BUS_DURABLE persists subscriptions across a bus restart
    (the reference's in-memory bus loses them — a known limitation its
    release notes call out; this is opt-in extra durability)."""
    import httpx

    from kakveda_amd.services.event_bus import create_app as bus_app

    monkeypatch.setenv("KAKVEDA_BUS_DURABLE", str(tmp_path))
    bus1 = bus_app()
    async with httpx.AsyncClient(
        transport=httpx.ASGITransport(app=bus1), base_url="http://bus"
    ) as client:
        r = await client.post(
            "/subscribe",
            json={"topic": "trace.ingested", "callback_url": "http://x/events"},
        )
        assert r.json()["subscribers"] == 1
        # dedup does not double-append
        await client.post(
            "/subscribe",
            json={"topic": "trace.ingested", "callback_url": "http://x/events"},
        )
    assert (tmp_path / "subscriptions.jsonl").read_text().count("http://x/events") == 1

    # "restart": a fresh app instance reloads the subscription graph
    bus2 = bus_app()
    async with httpx.AsyncClient(
        transport=httpx.ASGITransport(app=bus2), base_url="http://bus"
    ) as client:
        topics = (await client.get("/topics")).json()["topics"]
        assert topics == {"trace.ingested": ["http://x/events"]}
