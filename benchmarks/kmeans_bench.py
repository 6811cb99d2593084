"""BASELINE config 4: streaming k-means throughput over fingerprints.

One iteration = assignment (fused cosine kernel, points as queries vs
centroids) + segmented centroid update (+ RCCL all-reduce when
distributed). Reports points/sec per Lloyd iteration.

Run: python benchmarks/kmeans_bench.py [--points 10000000] [--clusters 64]
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--points", type=int, default=10_000_000)
    ap.add_argument("--clusters", type=int, default=64)
    ap.add_argument("--dim", type=int, default=768)
    ap.add_argument("--iters", type=int, default=5)
    ap.add_argument("--batch", type=int, default=1_000_000)
    args = ap.parse_args()

    device = "cuda" if torch.cuda.is_available() else "cpu"
    from kakveda_amd import ops
    from kakveda_amd.patterns.kmeans import StreamingKMeans

    if device == "cuda" and not ops.hip_available():
        raise RuntimeError("HIP extension missing on a GPU box")

    N, D, C = args.points, args.dim, args.clusters
    gen = torch.Generator(device=device).manual_seed(11)
    dtype = torch.bfloat16 if device == "cuda" else torch.float32
    pts = torch.empty(N, D, dtype=dtype, device=device)
    for s in range(0, N, 1 << 20):
        e = min(s + (1 << 20), N)
        pts[s:e] = torch.randn(e - s, D, generator=gen, device=device, dtype=torch.float32).to(dtype)
    ops.l2normalize_(pts) if device == "cuda" else None

    km = StreamingKMeans(C, D, device=device, seed=3, decay=0.0)

    def iteration():
        # mini-batched full pass (keeps the top-1 workspace bounded)
        for s in range(0, N, args.batch):
            km.step(pts[s : min(s + args.batch, N)])

    iteration()  # warmup
    if device == "cuda":
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.iters):
        iteration()
    if device == "cuda":
        torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / args.iters

    print(
        json.dumps(
            {
                "metric": "kmeans_points_per_sec",
                "value": N / dt,
                "unit": "points/s/iteration",
                "points": N,
                "clusters": C,
                "dim": D,
                "sec_per_iteration": dt,
                "device": device,
                "data": "synthetic",
            }
        )
    )


if __name__ == "__main__":
    main()
