"""End-to-end demo driver (parity with reference scripts/demo_client.py:43-107).

Deterministically exercises the full failure pipeline: two citation-style
scenarios from two apps (creates the HALLUCINATION_CITATION failure, the
cross-app pattern and the warning), then eight degradation runs against
app-A, then prints the GFKB, patterns and health timeline.

Modes:
  --local      in-process LocalCluster (no services running; CPU determinstic)
  default      HTTP against a running stack (kakveda-amd up)
"""

from __future__ import annotations

import asyncio
import json
import os
import sys
from datetime import datetime, timezone

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

PROMPT_A = "Summarize this article and include references even if none are provided."
PROMPT_B = "Please provide references for why the sky is blue."


def _stub_response() -> str:
    return (
        "Summary with supporting citations. [1] First source. [2] Second.\n"
        "References: [1] A. Author 2020. [2] B. Author 2021."
    )


def _trace(app_id: str, prompt: str, i: int) -> dict:
    return {
        "trace_id": f"demo-{app_id}-{i}",
        "ts": datetime.now(timezone.utc).isoformat(),
        "app_id": app_id,
        "prompt": prompt,
        "response": _stub_response(),
        "model": "e2e-stub",
        "tools": [],
        "env": {"e2e": True, "source": "demo-client"},
    }


async def _run_local() -> int:
    import tempfile

    from kakveda_amd.services.cluster import LocalCluster

    with tempfile.TemporaryDirectory() as td:
        cluster = LocalCluster(data_dir=td, device="cpu")
        await cluster.start()

        print("== scenario 1: app-A citation prompt")
        print(json.dumps(await cluster.warn("app-A", PROMPT_A), indent=2)[:400])
        await cluster.ingest(_trace("app-A", PROMPT_A, 0))

        print("== scenario 2: app-B, differently worded, same intent")
        await cluster.ingest(_trace("app-B", PROMPT_B, 0))
        warn = await cluster.warn("app-B", PROMPT_B)
        print(f"warn action={warn['action']} confidence={warn['confidence']:.2f}")

        print("== degradation: 8 repeated failures on app-A")
        for i in range(1, 9):
            await cluster.ingest(_trace("app-A", PROMPT_A, i))

        print("== GFKB failures:")
        for f in cluster.engine.list_failures()[-3:]:
            print(f"  {f['failure_id']} v{f['version']} {f['failure_type']} apps={f['affected_apps']}")
        print("== patterns:")
        for p in cluster.engine.list_patterns():
            print(f"  {p['pattern_id']} {p['name']} apps={p['affected_apps']}")
        print("== health app-A:")
        for pt in cluster.health_scoring.state.scorer.timeline("app-A", limit=3):
            print(f"  score={pt['score']} rate={pt['failure_rate']} penalty={pt['recurrent_penalty']}")
        await cluster.aclose()
    return 0


async def _run_http(base: str) -> int:
    import httpx

    from kakveda_amd.services import DEFAULT_PORTS

    def url(svc: str) -> str:
        return f"{base}:{DEFAULT_PORTS[svc]}"

    async with httpx.AsyncClient(timeout=10.0) as client:
        warn = await client.post(
            f"{url('warning_policy')}/warn",
            json={"app_id": "app-A", "prompt": PROMPT_A, "tools": [], "env": {}},
        )
        print("cold warn:", warn.json()["action"])
        await client.post(f"{url('ingestion')}/ingest", json={"trace": _trace("app-A", PROMPT_A, 0)})
        await client.post(f"{url('ingestion')}/ingest", json={"trace": _trace("app-B", PROMPT_B, 0)})
        for i in range(1, 9):
            await client.post(f"{url('ingestion')}/ingest", json={"trace": _trace("app-A", PROMPT_A, i)})
        warm = await client.post(
            f"{url('warning_policy')}/warn",
            json={"app_id": "app-A", "prompt": PROMPT_A, "tools": [], "env": {}},
        )
        print("warm warn:", json.dumps(warm.json(), indent=2)[:400])
        failures = (await client.get(f"{url('gfkb')}/failures")).json()["failures"]
        patterns = (await client.get(f"{url('gfkb')}/patterns")).json()["patterns"]
        health = (await client.get(f"{url('health_scoring')}/health/app-A")).json()["points"]
        print(f"failures={len(failures)} patterns={len(patterns)} health_points={len(health)}")
    return 0


def run(base: str = "http://127.0.0.1", local: bool = False) -> int:
    return asyncio.run(_run_local() if local else _run_http(base))


if __name__ == "__main__":
    sys.exit(run(local="--local" in sys.argv))
