"""RCCL-backend distributed tests on a single GPU (2 ranks on cuda:0).

Round-1 gap: every distributed test ran gloo-on-CPU, so
the nccl(=RCCL) branch of parallel/dist.py and the all-gather merge in
parallel/sharded.py had never executed on the real backend. These tests
initialise a 2-rank nccl process group with BOTH ranks on the one leased
MI355X and drive the same SPMD contract the 8-GPU bench uses:

- ShardedStore.append/search with the fused HIP kernel per shard and the
  RCCL all-gather (score, global-id) merge,
- DistGfkbCoordinator broadcast command plane (broadcast_object_list +
  tensor broadcast over RCCL).

A 2-rank-on-one-device world exercises every RCCL call site with real
device buffers; only the xGMI link layer (driver-measured in SCALE_rNN)
differs from the 8-GPU case.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

WORLD = 2


def _init(rank: int, world: int, port: int):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    import torch.distributed as td

    torch.cuda.set_device(0)
    td.init_process_group("nccl", rank=rank, world_size=world)
    return td


def _sharded_worker(rank: int, world: int, port: int, q):
    td = _init(rank, world, port)
    try:
        from kakveda_amd import ops
        from kakveda_amd.gfkb.engine import EmbeddingStore
        from kakveda_amd.parallel.sharded import ShardedStore

        assert ops.hip_available(), "HIP extension must be built"
        dev = "cuda:0"
        torch.manual_seed(11)  # same data on all ranks (SPMD contract)
        dim, n, k, nq = 768, 20000, 5, 64
        data = torch.randn(n, dim)
        data = (data / data.norm(dim=-1, keepdim=True)).to(dev, torch.bfloat16)
        queries = torch.randn(nq, dim)
        queries = (queries / queries.norm(dim=-1, keepdim=True)).to(dev, torch.bfloat16)

        store = ShardedStore(dim, device=dev, capacity=4096)
        store.append(data[: n // 2])
        store.append(data[n // 2 :])
        assert store.total == n
        assert store.local.count == n // world

        scores, idx = store.search(queries, k)
        torch.cuda.synchronize()

        ref = EmbeddingStore(dim, device=dev, capacity=n)
        ref.append(data)
        ref_scores, ref_idx = ref.search(queries, k)
        torch.cuda.synchronize()

        assert torch.allclose(scores, ref_scores, atol=1e-3), (
            (scores - ref_scores).abs().max().item()
        )
        # sharded result must score-match a direct gather at the merged ids
        gathered = (queries.float() @ data.float().t()).gather(1, idx)
        assert torch.allclose(gathered, scores, atol=1e-3)
        if rank == 0:
            q.put(("ok", float(scores.sum())))
    except Exception as e:
        q.put(("err", f"rank{rank}: {type(e).__name__}: {e}"))
        raise
    finally:
        td.destroy_process_group()


def _coord_worker(rank: int, world: int, port: int, q):
    td = _init(rank, world, port)
    try:
        from kakveda_amd.gfkb.dist_server import DistGfkbCoordinator, worker_loop

        if rank != 0:
            worker_loop(coord_dim=768, capacity=8192)
            return
        coord = DistGfkbCoordinator(dim=768, capacity=8192)
        torch.manual_seed(23)
        data = torch.randn(4096, 768)
        data = (data / data.norm(dim=-1, keepdim=True)).to("cuda:0", torch.bfloat16)
        first = coord.append(data)
        assert first == 0 and coord.total() == 4096
        qs = data[:16].clone()
        scores, idx = coord.search(qs, 5)
        torch.cuda.synchronize()
        # self-queries must find themselves first with score ~1
        assert (idx[:, 0].cpu() == torch.arange(16)).all(), idx[:, 0]
        assert (scores[:, 0] > 0.99).all()
        coord.stop()
        q.put(("ok", float(scores[:, 0].mean())))
    except Exception as e:
        q.put(("err", f"rank{rank}: {type(e).__name__}: {e}"))
        raise
    finally:
        td.destroy_process_group()


def _run(target, port):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=target, args=(r, WORLD, port, q)) for r in range(WORLD)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
    alive = [p for p in procs if p.is_alive()]
    for p in alive:
        p.terminate()
    assert not alive, "worker hung"
    for p in procs:
        assert p.exitcode == 0, f"worker exited {p.exitcode}"
    status, payload = q.get()
    assert status == "ok", payload


def test_rccl_sharded_store_matches_single():
    _run(_sharded_worker, 29815)


def test_rccl_coordinator_roundtrip():
    _run(_coord_worker, 29817)
