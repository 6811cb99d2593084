"""Distributed GFKB service test (CPU, gloo, world=2): rank 0 coordinates
inserts/searches over broadcast commands; results must equal a single
store over the same data."""

import os

import torch
import torch.multiprocessing as mp

WORLD = 2


def _worker(rank: int, world: int, port: int, q):
    os.environ.update(
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT=str(port),
        RANK=str(rank),
        LOCAL_RANK=str(rank),
        WORLD_SIZE=str(world),
    )
    import torch.distributed as td

    td.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from kakveda_amd.gfkb.dist_server import DistGfkbCoordinator, worker_loop

        dim = 64
        if rank != 0:
            worker_loop(coord_dim=dim, capacity=2048)
            return

        coord = DistGfkbCoordinator(dim=dim, capacity=2048)
        torch.manual_seed(123)
        data = torch.randn(400, dim)
        data = data / data.norm(dim=-1, keepdim=True)
        queries = data[:8] .clone()

        coord.append(data[:250])
        coord.append(data[250:])
        assert coord.total() == 400

        scores, idx = coord.search(queries, 5)
        # exact self-match must come back first
        assert idx[:, 0].tolist() == list(range(8)), idx[:, 0]
        assert bool((scores[:, 0] > 0.99).all())

        # reference: single local store
        from kakveda_amd.gfkb.engine import EmbeddingStore

        ref = EmbeddingStore(dim, device="cpu", capacity=1024)
        ref.append(data)
        rs, ri = ref.search(queries, 5)
        assert torch.allclose(scores, rs, atol=1e-5)

        coord.stop()
        q.put(("ok", None))
    except Exception as e:
        q.put(("err", f"rank{rank}: {type(e).__name__}: {e}"))
        raise
    finally:
        td.destroy_process_group()


def test_dist_gfkb_service_roundtrip():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [
        ctx.Process(target=_worker, args=(r, WORLD, 29783, q)) for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    for p in procs:
        assert p.exitcode == 0, f"worker exited {p.exitcode}"
    status, payload = q.get()
    assert status == "ok", payload


def _engine_worker(rank: int, world: int, port: int, q):
    """The REAL dist-server wiring: rank 0 runs a GfkbEngine whose store
    is the coordinator facade (attach_store migration included); rank 1
    sits in worker_loop. Upsert + match must round-trip through the
    broadcast command plane."""
    import tempfile

    os.environ.update(
        MASTER_ADDR="127.0.0.1",
        MASTER_PORT=str(port),
        RANK=str(rank),
        LOCAL_RANK=str(rank),
        WORLD_SIZE=str(world),
    )
    import torch.distributed as td

    td.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from kakveda_amd.gfkb.dist_server import (
            DistGfkbCoordinator,
            _CoordinatorStore,
            worker_loop,
        )

        dim = 128
        if rank != 0:
            worker_loop(coord_dim=dim, capacity=2048)
            return

        from kakveda_amd.gfkb.engine import GfkbEngine

        with tempfile.TemporaryDirectory() as td_dir:
            engine = GfkbEngine(data_dir=td_dir, device="cpu", dim=dim)
            sig = (
                "intent_tags:intent:citations_required | prompt_hint:dist "
                "round trip | tools: | env_keys:k"
            )
            engine.upsert_failure("T", sig, {}, app_id="a")  # pre-swap row
            coord = DistGfkbCoordinator(dim=dim, capacity=2048)
            engine.attach_store(_CoordinatorStore(coord))
            # the JSONL-restored identity must survive the swap
            m = engine.match(sig)
            assert m and m[0].failure_id == "F-0001" and m[0].score > 0.99, m
            # new upserts go through the coordinator (broadcast append)
            rec, created = engine.upsert_failure(
                "T2",
                "intent_tags: | prompt_hint:other dist | tools: | env_keys:",
                {},
                app_id="b",
            )
            assert created and rec["failure_id"] == "F-0002"
            m2 = engine.match(
                "intent_tags: | prompt_hint:other dist | tools: | env_keys:"
            )
            assert m2 and m2[0].failure_id == "F-0002" and m2[0].score > 0.99
            coord.stop()
        q.put(("ok", None))
    except Exception as e:
        q.put(("err", f"rank{rank}: {type(e).__name__}: {e}"))
        raise
    finally:
        td.destroy_process_group()


def test_dist_gfkb_engine_over_coordinator():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [
        ctx.Process(target=_engine_worker, args=(r, WORLD, 29787, q))
        for r in range(WORLD)
    ]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=120)
    for p in procs:
        assert p.exitcode == 0, f"worker exited {p.exitcode}"
    status, payload = q.get()
    assert status == "ok", payload
