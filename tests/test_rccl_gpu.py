"""RCCL-backend (nccl-on-ROCm) execution tests on one GPU.

Round-1 gap: the nccl branch of parallel/dist.py and the
all-gather merge in parallel/sharded.py had only ever run on gloo/CPU.
RCCL refuses two ranks on one device ("Duplicate GPU detected", verified
on an MI355X box — see profiles/rccl_notes.md), so a single leased GPU
cannot host a multi-rank nccl world. These tests do the strongest
single-box thing instead: a REAL RCCL communicator (world size 1) with
every collective call site executed on device buffers —

- ``ShardedStore.search`` with KAKVEDA_FORCE_COLLECTIVES=1 runs the
  actual ``all_gather`` merge over RCCL,
- ``StreamingKMeans.step`` all-reduces centroid sums/counts over RCCL,
- the dist_server command plane broadcasts over the initialised group.

Multi-rank semantics are covered by the gloo world=2 tests
(test_sharded_cpu.py, test_dist_gfkb.py) and by the driver's 8-GPU SCALE
run of bench.py.
"""

import os

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu


def _worker(port: int, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["KAKVEDA_FORCE_COLLECTIVES"] = "1"
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    import torch.distributed as td

    torch.cuda.set_device(0)
    td.init_process_group("nccl", rank=0, world_size=1)
    try:
        from kakveda_amd import ops
        from kakveda_amd.gfkb.engine import EmbeddingStore
        from kakveda_amd.parallel.sharded import ShardedStore

        assert ops.hip_available(), "HIP extension must be built"
        dev = "cuda:0"
        torch.manual_seed(11)
        dim, n, k, nq = 768, 20000, 5, 64
        data = torch.randn(n, dim)
        data = (data / data.norm(dim=-1, keepdim=True)).to(dev, torch.bfloat16)
        queries = torch.randn(nq, dim)
        queries = (queries / queries.norm(dim=-1, keepdim=True)).to(dev, torch.bfloat16)

        store = ShardedStore(dim, device=dev, capacity=4096)
        assert store.force_collectives and store.world == 1
        store.append(data)
        scores, idx = store.search(queries, k)  # all_gather over RCCL
        torch.cuda.synchronize()

        ref = EmbeddingStore(dim, device=dev, capacity=n)
        ref.append(data)
        ref_scores, _ = ref.search(queries, k)
        torch.cuda.synchronize()
        assert torch.allclose(scores, ref_scores, atol=1e-3), (
            (scores - ref_scores).abs().max().item()
        )
        gathered = (queries.float() @ data.float().t()).gather(1, idx)
        assert torch.allclose(gathered, scores, atol=1e-3)

        # k-means centroid all-reduce over RCCL (kakveda_amd/patterns/kmeans.py)
        from kakveda_amd.patterns.kmeans import StreamingKMeans

        km = StreamingKMeans(8, dim, device=dev, seed=3)
        assign, mean_cos = km.step(data[:4096].float())
        torch.cuda.synchronize()
        assert assign.shape[0] == 4096 and -1.0 <= mean_cos <= 1.0

        # dist_server command plane broadcast over the initialised group
        obj = ["probe", (1, 2)]
        td.broadcast_object_list(obj, src=0)
        t = torch.ones(64, device=dev)
        td.broadcast(t, src=0)
        td.all_reduce(t)
        torch.cuda.synchronize()
        assert float(t.sum()) == 64.0

        q.put(("ok", float(scores.sum())))
    except Exception as e:
        q.put(("err", f"{type(e).__name__}: {e}"))
        raise
    finally:
        td.destroy_process_group()


def test_rccl_collective_paths_world1():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    p = ctx.Process(target=_worker, args=(29819, q))
    p.start()
    p.join(timeout=300)
    if p.is_alive():
        p.terminate()
        raise AssertionError("worker hung")
    assert p.exitcode == 0, f"worker exited {p.exitcode}"
    status, payload = q.get()
    assert status == "ok", payload
