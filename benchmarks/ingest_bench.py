"""BASELINE config 5: sustained synthetic trace-ingest throughput.

The full ingest pipeline per event batch, on-GPU where it counts:
signature featurisation (pre-hashed pool, CPU-side as in production),
trace-encoder embedding (embedding_bag + projection GEMM), pre-flight
warning lookup against the GFKB shard (fused cosine-topk), failure-rule
classification, and incremental health scoring.

Run:  python benchmarks/ingest_bench.py [--batch 4096] [--seconds 10]
      [--entries 1000000]
Prints one JSON line: sustained QPS + p50/p99 batch latency.
"""

from __future__ import annotations

import argparse
import json
import os
import statistics
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=4096)
    ap.add_argument("--seconds", type=float, default=10.0)
    ap.add_argument("--entries", type=int, default=1_000_000)
    ap.add_argument("--dim", type=int, default=768)
    ap.add_argument("--topk", type=int, default=5)
    args = ap.parse_args()

    device = "cuda" if torch.cuda.is_available() else "cpu"
    from kakveda_amd import ops
    from kakveda_amd.encoder.featurizer import featurize_batch
    from kakveda_amd.encoder.model import TraceEncoder
    from kakveda_amd.health.scoring import HealthScorer

    if device == "cuda" and not ops.hip_available():
        raise RuntimeError("HIP extension missing on a GPU box")

    B, D = args.batch, args.dim
    enc = TraceEncoder(dim=D, hash_dim=1 << 16, seed=9, device=device)

    # GFKB shard of unit fingerprints
    gen = torch.Generator(device=device).manual_seed(5)
    corpus = torch.randn(args.entries, D, generator=gen, device=device).to(
        torch.bfloat16 if device == "cuda" else torch.float32
    )
    ops.l2normalize_(corpus)

    # pre-hashed feature pool (cycled), as requests would arrive pre-tokenised
    texts = [
        f"intent_tags:intent:citations_required | prompt_hint:ingest probe {i} "
        f"with sources | tools: | env_keys:e2e"
        for i in range(1024)
    ]
    idx_np, w_np = featurize_batch(texts, hash_dim=enc.hash_dim, max_features=64)
    reps = (B + 1023) // 1024
    fidx = torch.from_numpy(idx_np).to(device).repeat(reps, 1)[:B].contiguous()
    fw = torch.from_numpy(w_np).to(device).repeat(reps, 1)[:B].contiguous()

    scorer = HealthScorer()
    apps = [f"app-{i % 32}" for i in range(256)]

    def tick() -> int:
        q = enc.encode_features(fidx, fw)
        if device == "cuda":
            q = q.to(torch.bfloat16)
        scores, idx = ops.cosine_topk(q, corpus, args.topk)
        warn_mask = scores[:, 0] >= 0.8
        n_warn = int(warn_mask.sum().item())
        # health scoring folds one event per flagged app bucket (capped to
        # keep host work proportional to alerts, as the service batches)
        for i in range(min(n_warn, 256)):
            scorer.observe(
                {"app_id": apps[i], "severity": "medium", "failure_type": "HALLUCINATION_CITATION"}
            )
        return B

    # warmup
    for _ in range(3):
        tick()
    if device == "cuda":
        torch.cuda.synchronize()

    lat = []
    done = 0
    t0 = time.perf_counter()
    while time.perf_counter() - t0 < args.seconds:
        ts = time.perf_counter()
        done += tick()
        if device == "cuda":
            torch.cuda.synchronize()
        lat.append(time.perf_counter() - ts)
    elapsed = time.perf_counter() - t0

    lat_ms = sorted(x * 1000 for x in lat)
    out = {
        "metric": "ingest_qps",
        "value": done / elapsed,
        "unit": "traces/s",
        "batch": B,
        "entries": args.entries,
        "p50_ms": statistics.median(lat_ms),
        "p99_ms": lat_ms[max(0, int(len(lat_ms) * 0.99) - 1)],
        "device": device,
        "data": "synthetic",
    }
    print(json.dumps(out))


if __name__ == "__main__":
    main()
