"""GFKB service HTTP contract tests (endpoint shapes of gfkb/app.py)."""

import httpx

from kakveda_amd.gfkb.engine import GfkbEngine
from kakveda_amd.services.gfkb_service import create_app

SIG_A = (
    "intent_tags:intent:citations_required | prompt_hint:summarize with "
    "references | tools: | env_keys:a"
)
SIG_B = "intent_tags: | prompt_hint:translate to french | tools: | env_keys:b"


def _client(tmp_path):
    engine = GfkbEngine(data_dir=str(tmp_path), device="cpu", dim=128, hash_dim=2048)
    app = create_app(engine=engine)
    return httpx.AsyncClient(
        transport=httpx.ASGITransport(app=app), base_url="http://gfkb:8101"
    ), engine


async def test_upsert_and_list(tmp_path):
    client, _ = _client(tmp_path)
    r1 = await client.post(
        "/failures/upsert",
        json={
            "failure_type": "HALLUCINATION_CITATION",
            "signature_text": SIG_A,
            "context_signature": {"model": "stub"},
            "impact_severity": "medium",
            "app_id": "app-A",
        },
    )
    body = r1.json()
    assert body["created"] and body["failure"]["failure_id"] == "F-0001"

    r2 = await client.post(
        "/failures/upsert",
        json={
            "failure_type": "HALLUCINATION_CITATION",
            "signature_text": SIG_A,
            "context_signature": {},
            "app_id": "app-B",
        },
    )
    assert r2.json()["created"] is False
    assert r2.json()["failure"]["version"] == 2

    failures = (await client.get("/failures")).json()["failures"]
    assert len(failures) == 2  # append-only version rows
    await client.aclose()


async def test_match_endpoint_and_type_filter(tmp_path):
    client, _ = _client(tmp_path)
    for sig, ftype, app_id in ((SIG_A, "HALLUCINATION_CITATION", "a"), (SIG_B, "OTHER", "b")):
        await client.post(
            "/failures/upsert",
            json={
                "failure_type": ftype,
                "signature_text": sig,
                "context_signature": {},
                "app_id": app_id,
            },
        )
    m = (await client.post("/failures/match", json={"signature_text": SIG_A})).json()
    assert m["matches"][0]["failure_type"] == "HALLUCINATION_CITATION"
    assert m["matches"][0]["score"] > 0.99

    # type filter applied after the top-k cut (reference gfkb/app.py:89-92)
    m2 = (
        await client.post(
            "/failures/match",
            json={"signature_text": SIG_A, "failure_type": "OTHER"},
        )
    ).json()
    assert all(x["failure_type"] == "OTHER" for x in m2["matches"])
    await client.aclose()


async def test_patterns_endpoints(tmp_path):
    client, _ = _client(tmp_path)
    p1 = (
        await client.post(
            "/patterns/upsert",
            json={"name": "P", "failure_ids": ["F-0001"], "affected_apps": ["a"]},
        )
    ).json()
    assert p1["created"] and p1["pattern"]["pattern_id"] == "FP-0001"
    p2 = (
        await client.post(
            "/patterns/upsert",
            json={"name": "P", "failure_ids": ["F-0002"], "affected_apps": ["b"]},
        )
    ).json()
    assert not p2["created"]
    pats = (await client.get("/patterns")).json()["patterns"]
    assert len(pats) == 1 and sorted(pats[0]["failure_ids"]) == ["F-0001", "F-0002"]

    hz = (await client.get("/healthz")).json()
    assert hz["ok"] and hz["rows"] == 0  # patterns only: no embedding rows
    await client.aclose()
