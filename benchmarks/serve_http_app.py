"""uvicorn app factory for the multi-process HTTP serving benchmark.

Builds the single-container serving composition (warning_policy ->
gfkb_service -> engine, wired in-process exactly like deploy/Dockerfile's
entrypoint) with a synthetic N-entry corpus, and returns the warn ASGI
app for uvicorn to serve over real TCP.  Configured via env so uvicorn
workers (separate processes) can each build their own engine:

  KAKVEDA_BENCH_ENTRIES  corpus rows (default 50k; GPU runs use 10M)
  KAKVEDA_BENCH_DIM      embedding dim (default 768)

Launch (done by serve_http_bench.py):
  python -m uvicorn serve_http_app:create_app --factory \
      --app-dir benchmarks --host 127.0.0.1 --port 8104 --workers 2
"""

from __future__ import annotations

import os
import tempfile

import torch


def create_app():
    from kakveda_amd import ops
    from kakveda_amd.gfkb.engine import EmbeddingStore, GfkbEngine
    from kakveda_amd.services.gfkb_service import create_app as gfkb_app
    from kakveda_amd.services.warning_policy import create_app as warn_app
    from kakveda_amd.services.wiring import Transport

    entries = int(os.environ.get("KAKVEDA_BENCH_ENTRIES", "50000"))
    dim = int(os.environ.get("KAKVEDA_BENCH_DIM", "768"))
    device = "cuda" if torch.cuda.is_available() else "cpu"

    td = tempfile.mkdtemp(prefix="kakveda_httpbench_")
    engine = GfkbEngine(data_dir=td, device=device, dim=dim)
    # one real failure so requests with the demo prompt match >= threshold
    engine.upsert_failure(
        "HALLUCINATION_CITATION",
        "intent_tags:intent:citations_required | prompt_hint:please provide "
        "references for why the sky is blue. | tools: | env_keys:e2e,source",
        {"m": 1},
        app_id="app-A",
    )
    store = EmbeddingStore(dim, device=device, capacity=entries + 1024)
    engine.attach_store(store)
    gen = torch.Generator(device=device).manual_seed(9)
    fill = 1 << 21
    for s0 in range(0, entries, fill):
        e0 = min(s0 + fill, entries)
        rows = torch.randn(e0 - s0, dim, generator=gen, device=device)
        rows = rows / rows.norm(dim=-1, keepdim=True)
        rows = rows.to(store.dtype)
        if device == "cuda":
            ops.l2normalize_(rows)
        store.append(rows)
        del rows

    tx = Transport()
    tx.register_local("http://gfkb:8101", gfkb_app(engine=engine))
    return warn_app(gfkb_url="http://gfkb:8101", transport=tx)
