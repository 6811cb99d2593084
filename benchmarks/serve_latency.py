"""Request-level serving latency: single-query POST /warn round trips
through the real service path (ASGI in-process transport -> warning_policy
-> gfkb_service -> GPU engine with an adopted N-entry corpus).

This measures what a caller of the pre-flight API sees per request —
signature build + encode (B=1) + fused kernel over the whole corpus +
policy — as opposed to bench.py's batched throughput.

Run: python benchmarks/serve_latency.py [--entries 10000000] [--seconds 15]
"""

from __future__ import annotations

import argparse
import asyncio
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


async def run(args) -> dict:
    from kakveda_amd import ops
    from kakveda_amd.gfkb.engine import EmbeddingStore, GfkbEngine
    from kakveda_amd.services.gfkb_service import create_app as gfkb_app
    from kakveda_amd.services.warning_policy import create_app as warn_app
    from kakveda_amd.services.wiring import Transport

    device = "cuda" if torch.cuda.is_available() else "cpu"

    import tempfile

    with tempfile.TemporaryDirectory() as td:
        engine = GfkbEngine(data_dir=td, device=device, dim=args.dim)
        # seed one real failure so some requests match above threshold
        engine.upsert_failure(
            "HALLUCINATION_CITATION",
            "intent_tags:intent:citations_required | prompt_hint:please provide "
            "references for why the sky is blue. | tools: | env_keys:e2e,source",
            {"m": 1},
            app_id="app-A",
        )
        # swap in a large store: identity rows first (attach_store
        # re-encodes them at the front), then a big anonymous corpus so
        # every lookup scans all N rows (worst case)
        n = args.entries
        store = EmbeddingStore(args.dim, device=device, capacity=n + 1024)
        engine.attach_store(store)
        gen = torch.Generator(device=device).manual_seed(9)
        fill = 1 << 21
        for s0 in range(0, n, fill):
            e0 = min(s0 + fill, n)
            rows = torch.randn(e0 - s0, args.dim, generator=gen, device=device)
            rows = rows / rows.norm(dim=-1, keepdim=True)
            rows = rows.to(store.dtype)
            if device == "cuda":
                ops.l2normalize_(rows)
            store.append(rows)
            del rows

        tx = Transport()
        tx.register_local("http://gfkb:8101", gfkb_app(engine=engine))
        tx.register_local(
            "http://warning-policy:8104",
            warn_app(gfkb_url="http://gfkb:8101", transport=tx),
        )

        body = {
            "app_id": "app-A",
            "prompt": "please provide references for why the sky is blue.",
            "tools": [],
            "env": {"e2e": "1", "source": "none"},
        }
        # warmup
        for _ in range(5):
            await tx.post("http://warning-policy:8104/warn", json=body)

        lat = []
        matched = 0
        stop_at = time.perf_counter() + args.seconds

        async def client_loop():
            nonlocal matched
            while time.perf_counter() < stop_at:
                ts = time.perf_counter()
                r = await tx.post("http://warning-policy:8104/warn", json=body)
                lat.append(time.perf_counter() - ts)
                if r.json().get("references"):
                    matched += 1

        t0 = time.perf_counter()
        await asyncio.gather(*(client_loop() for _ in range(args.clients)))
        wall = time.perf_counter() - t0
        lat_ms = sorted(x * 1000 for x in lat)
        gfkb = tx.local_app("http://gfkb:8101")
        batcher = getattr(gfkb.state, "batcher", None)
        return {
            "metric": "serve_warn_latency",
            "unit": "ms",
            "clients": args.clients,
            "requests": len(lat),
            "rps": round(len(lat) / wall, 1),
            "p50_ms": round(lat_ms[len(lat_ms) // 2], 2),
            "p99_ms": round(lat_ms[max(0, int(len(lat_ms) * 0.99) - 1)], 2),
            "matched": matched,
            "entries": n,
            "device": device,
            "data": "synthetic",
            "batcher": (
                {
                    "batches": batcher.batches,
                    "requests": batcher.requests,
                    "avg_batch": round(batcher.requests / max(1, batcher.batches), 2),
                }
                if batcher is not None
                else None
            ),
        }


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--entries", type=int, default=10_000_000)
    ap.add_argument("--seconds", type=float, default=15.0)
    ap.add_argument("--dim", type=int, default=768)
    ap.add_argument("--clients", type=int, default=1,
                    help="concurrent client loops (micro-batcher coalesces them)")
    args = ap.parse_args()
    out = asyncio.run(run(args))
    print(json.dumps(out))
    return 0


if __name__ == "__main__":
    sys.exit(main())
