import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))
"""Layer-by-layer serving cost: engine.match_batch vs MatchBatcher vs HTTP."""
import tempfile, threading, time
import torch
from kakveda_amd import ops
from kakveda_amd.gfkb.engine import EmbeddingStore, GfkbEngine
from kakveda_amd.services.gfkb_service import MatchBatcher

dev = "cuda"
with tempfile.TemporaryDirectory() as td:
    engine = GfkbEngine(data_dir=td, device=dev, dim=768)
    engine.upsert_failure(
        "HALLUCINATION_CITATION",
        "intent_tags:intent:citations_required | prompt_hint:please provide "
        "references for why the sky is blue. | tools: | env_keys:e2e,source",
        {"m": 1}, app_id="app-A")
    n = 10_000_000
    store = EmbeddingStore(768, device=dev, capacity=n + 1024)
    engine.attach_store(store)
    gen = torch.Generator(device=dev).manual_seed(9)
    for s0 in range(0, n, 1 << 21):
        e0 = min(s0 + (1 << 21), n)
        rows = torch.randn(e0 - s0, 768, generator=gen, device=dev)
        rows = (rows / rows.norm(dim=-1, keepdim=True)).to(store.dtype)
        store.append(rows)

    sig = ("intent_tags:intent:citations_required | prompt_hint:please provide "
           "references for why the sky is blue. | tools: | env_keys:e2e,source")

    # layer 1: engine.match_batch at B=19 (the observed avg batch)
    texts = [sig] * 19
    for _ in range(3):
        engine.match_batch(texts)
    t0 = time.perf_counter()
    for _ in range(10):
        engine.match_batch(texts)
    t_engine = (time.perf_counter() - t0) / 10
    print(f"engine.match_batch(19) @10M: {t_engine*1000:.2f} ms")

    # layer 1b: encode only
    t0 = time.perf_counter()
    for _ in range(10):
        engine.encoder.encode_texts(texts)
    torch.cuda.synchronize()
    print(f"encode_texts(19): {(time.perf_counter()-t0)/10*1000:.2f} ms")

    # layer 2: MatchBatcher with 64 threads hammering
    batcher = MatchBatcher(engine)
    stop = time.perf_counter() + 8
    counts = [0] * 64
    def worker(i):
        while time.perf_counter() < stop:
            batcher.match(sig)
            counts[i] += 1
    ths = [threading.Thread(target=worker, args=(i,)) for i in range(64)]
    t0 = time.perf_counter()
    for t in ths: t.start()
    for t in ths: t.join()
    wall = time.perf_counter() - t0
    tot = sum(counts)
    print(f"batcher 64 threads: {tot/wall:.0f} req/s, batches={batcher.batches}, "
          f"avg_batch={batcher.requests/max(1,batcher.batches):.1f}, "
          f"cycle={wall/max(1,batcher.batches)*1000:.2f} ms")
