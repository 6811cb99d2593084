"""Whole-platform end-to-end on GPU: the LocalCluster pipeline with the
GFKB engine (encoder + store + fused kernel) running on cuda:0."""

from datetime import datetime, timezone

import pytest

pytestmark = pytest.mark.gpu

PROMPT = "Summarize this report and include references even if none are provided."


def _trace(app_id: str, i: int) -> dict:
    return {
        "trace_id": f"g-{app_id}-{i}",
        "ts": datetime.now(timezone.utc).isoformat(),
        "app_id": app_id,
        "prompt": PROMPT,
        "response": "Findings. [1] Source A. [2] Source B. References included.",
        "model": "e2e-stub",
        "tools": [],
        "env": {"e2e": True},
    }


async def test_pipeline_on_gpu(tmp_path):
    from kakveda_amd.services.cluster import LocalCluster

    cluster = LocalCluster(data_dir=str(tmp_path), device="cuda")
    await cluster.start()

    await cluster.ingest(_trace("app-A", 1))
    w = await cluster.warn("app-A", PROMPT)
    assert w["confidence"] >= 0.8, w
    assert w["references"][0]["failure_id"] == "F-0001"

    # pad the store so the fused kernel scans a non-trivial prefix
    for i in range(100):
        cluster.engine.upsert_failure(
            "T", f"intent_tags: | prompt_hint:noise {i} | tools: | env_keys:", {},
            app_id="x",
        )
    w2 = await cluster.warn("app-B", PROMPT)
    assert w2["confidence"] >= 0.8
    assert w2["references"][0]["failure_id"] == "F-0001"

    # unrelated prompt stays silent
    w3 = await cluster.warn("app-A", "what's for lunch today?")
    assert w3["confidence"] < 0.8

    # pattern mining on GPU through the service endpoint
    resp = await cluster.tx.post(
        f"{cluster.urls['pattern_detector']}/cluster/run",
        json={"n_clusters": 4, "min_apps": 1},
    )
    body = resp.json()
    assert body["ok"] and body["patterns"], body
    await cluster.aclose()
