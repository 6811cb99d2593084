"""Multi-process sharded GFKB tests on CPU (gloo, world_size 2).

Verifies the distributed path is correct by construction: a sharded search
over 2 ranks must equal a single-store search on the same data
(all-gather merge equals single-shard result).
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

WORLD = 2


def _worker(rank: int, world: int, port: int, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    import torch.distributed as td

    td.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from kakveda_amd.parallel.sharded import ShardedStore

        torch.manual_seed(7)  # same data on all ranks (SPMD insert contract)
        dim, n, k = 64, 500, 5
        data = torch.randn(n, dim)
        data = data / data.norm(dim=-1, keepdim=True)
        queries = torch.randn(8, dim)
        queries = queries / queries.norm(dim=-1, keepdim=True)

        store = ShardedStore(dim, device="cpu", capacity=128)
        first = store.append(data[:300])
        assert first == 0
        store.append(data[300:])
        assert store.total == n
        # round-robin shard balance
        assert store.local.count == n // world

        scores, idx = store.search(queries, k)

        # reference: single store over the same data
        from kakveda_amd.gfkb.engine import EmbeddingStore

        ref = EmbeddingStore(dim, device="cpu", capacity=1024)
        ref.append(data)
        ref_scores, ref_idx = ref.search(queries, k)

        assert torch.allclose(scores, ref_scores, atol=1e-5), (
            scores - ref_scores
        ).abs().max()
        # indices equal where scores are distinct
        gathered = (queries.float() @ data.float().t()).gather(1, idx)
        assert torch.allclose(gathered, scores, atol=1e-5)

        if rank == 0:
            q.put(("ok", scores[:2].tolist()))
    except Exception as e:  # surface failures to the parent
        q.put(("err", f"rank{rank}: {type(e).__name__}: {e}"))
        raise
    finally:
        td.destroy_process_group()


def test_sharded_store_matches_single(tmp_path):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    port = 29781
    procs = [
        ctx.Process(target=_worker, args=(r, WORLD, port, q)) for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    for p in procs:
        assert p.exitcode == 0, f"worker exited {p.exitcode}"
    status, payload = q.get()
    assert status == "ok", payload


def test_sharded_single_process_fallback():
    """world=1 path needs no process group."""
    from kakveda_amd.parallel.sharded import ShardedStore

    torch.manual_seed(3)
    store = ShardedStore(32, device="cpu")
    data = torch.randn(50, 32)
    data = data / data.norm(dim=-1, keepdim=True)
    store.append(data)
    s, i = store.search(data[:4], 3)
    assert i[:, 0].tolist() == [0, 1, 2, 3]
    assert (s[:, 0] > 0.99).all()


def test_force_collectives_world1(monkeypatch):
    """KAKVEDA_FORCE_COLLECTIVES=1 runs the all-gather merge even at
    world=1 (the single-GPU RCCL execution test depends on this path)."""
    import torch.distributed as td

    monkeypatch.setenv("MASTER_ADDR", "127.0.0.1")
    monkeypatch.setenv("MASTER_PORT", "29823")
    monkeypatch.setenv("KAKVEDA_FORCE_COLLECTIVES", "1")
    td.init_process_group("gloo", rank=0, world_size=1)
    try:
        from kakveda_amd.gfkb.engine import EmbeddingStore
        from kakveda_amd.parallel.sharded import ShardedStore

        torch.manual_seed(3)
        data = torch.randn(200, 64)
        data = data / data.norm(dim=-1, keepdim=True)
        q = torch.randn(4, 64)
        q = q / q.norm(dim=-1, keepdim=True)
        store = ShardedStore(64, device="cpu", capacity=256)
        assert store.force_collectives
        store.append(data)
        scores, idx = store.search(q, 5)
        ref = EmbeddingStore(64, device="cpu", capacity=256)
        ref.append(data)
        ref_scores, ref_idx = ref.search(q, 5)
        assert torch.allclose(scores, ref_scores, atol=1e-5)
        assert torch.equal(idx, ref_idx)
    finally:
        td.destroy_process_group()
