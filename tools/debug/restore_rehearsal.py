"""Sidecar restore rehearsal at scale (ROUND3.md candidate 4).

Writes an N-row packed bf16 embedding sidecar to box-local disk, then
times the exact restart path GfkbEngine._restore_rows takes: mmap +
chunked H2D upload into a fresh EmbeddingStore, with a self-match
exactness spot check. Usage:

  python tools/debug/restore_rehearsal.py [N]     (default 20M = 30.7 GB)
"""
import shutil
import sys
import tempfile
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from kakveda_amd.gfkb.engine import EmbeddingSidecar, EmbeddingStore  # noqa: E402

n = int(float(sys.argv[1])) if len(sys.argv) > 1 else 20_000_000
dim = 768
dev = "cuda" if torch.cuda.is_available() else "cpu"
dt = torch.bfloat16 if dev == "cuda" else torch.float32

td = tempfile.mkdtemp(prefix="restore_rehearsal_")
try:
    sc = EmbeddingSidecar(Path(td) / "embeddings.bin", dim, dt)
    g = torch.Generator(device=dev).manual_seed(5)
    gb = n * dim * (2 if dt == torch.bfloat16 else 4) / 1e9
    t0 = time.perf_counter()
    fill = 1 << 20
    for s in range(0, n, fill):
        e = min(s + fill, n)
        t = torch.randn(e - s, dim, generator=g, device=dev,
                        dtype=torch.float32)
        t = (t / t.norm(dim=-1, keepdim=True)).to(dt)
        sc.append(t)
    if dev == "cuda":
        torch.cuda.synchronize()
    t1 = time.perf_counter()
    print(f"sidecar write: {n} rows ({gb:.1f} GB) in {t1 - t0:.1f}s "
          f"= {gb / (t1 - t0):.2f} GB/s", flush=True)

    store = EmbeddingStore(dim, device=dev, capacity=n + 1024)
    t2 = time.perf_counter()
    rows = 0
    for chunk in sc.load_chunks():
        store.append(chunk)
        rows += chunk.shape[0]
    if dev == "cuda":
        torch.cuda.synchronize()
    t3 = time.perf_counter()
    assert rows == n, (rows, n)
    print(f"restore: {rows} rows in {t3 - t2:.1f}s = {gb / (t3 - t2):.2f} "
          f"GB/s  (100M extrapolation ~{(t3 - t2) * 1e8 / n:.0f}s)",
          flush=True)

    # exactness spot check: a restored row must self-match at score ~1
    probe = store.search(
        torch.randn(1, dim, generator=g, device=dev).to(dt), 1)
    q = None
    for chunk in sc.load_chunks(chunk_rows=16384):
        q = chunk[12345 % chunk.shape[0]:12345 % chunk.shape[0] + 1]
        break
    s_, i_ = store.search(q.to(dev).to(dt), 1)
    print(f"self-match: idx={int(i_[0, 0])} score={float(s_[0, 0]):.4f}",
          flush=True)
    assert float(s_[0, 0]) > 0.99
finally:
    shutil.rmtree(td, ignore_errors=True)
