"""Distributed GFKB serving: one process per GPU, rank 0 fronts HTTP.

The bench proves the sharded data plane; this module makes it a *service*
(SURVEY.md 2.5 'sharded GFKB'): rank 0 runs the FastAPI app and broadcasts
every store-mutating or collective operation as a command over
torch.distributed (RCCL on GPUs, gloo on CPU); ranks 1..N-1 sit in a
worker loop executing the same ShardedStore calls SPMD-style, so the
all-gather merge inside ``ShardedStore.search`` lines up across ranks.

Launch (8 GPUs):
  python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \\
      --master-addr 127.0.0.1 -m kakveda_amd.gfkb.dist_server --port 8101

Commands are one fixed-size int64[8] header tensor broadcast; payloads
(embedding/query batches) travel as tensor broadcasts — RCCL end to end
on GPUs, no pickling.
"""

from __future__ import annotations

import argparse
from typing import Optional

import torch
import torch.distributed as td

from kakveda_amd.parallel.dist import init_from_env
from kakveda_amd.parallel.sharded import ShardedStore

CMD_APPEND = 1
CMD_SEARCH = 2
CMD_STOP = 3

#: fixed dtype codes for the command header (no pickling on the wire)
_DTYPES = {0: torch.float32, 1: torch.bfloat16, 2: torch.float16}
_DTYPE_IDS = {v: k for k, v in _DTYPES.items()}


class DistGfkbCoordinator:
    """Rank-0 handle: broadcasts a command header, then performs the same
    local collective call every worker performs.

    The command plane is ONE fixed-size int64[8] tensor broadcast
    (cmd, n, d, k, dtype_id) followed by the payload broadcast — no
    broadcast_object_list (which costs two broadcasts plus host-side
    pickling per request). On GPUs the header
    tensor rides RCCL like the payload.
    """

    def __init__(self, dim: int = 768, capacity: int = 1 << 20):
        self.rank, self.world, self.device = init_from_env()
        self.dim = dim
        self.store = ShardedStore(dim, device=str(self.device), capacity=capacity)
        self._hdr = torch.zeros(8, dtype=torch.int64, device=self.device)

    # -- plumbing ----------------------------------------------------------

    def _bcast_cmd(self, cmd: int, n: int = 0, d: int = 0, k: int = 0,
                   dtype: torch.dtype = torch.float32) -> None:
        if self.world > 1:
            self._hdr[0] = cmd
            self._hdr[1] = n
            self._hdr[2] = d
            self._hdr[3] = k
            self._hdr[4] = _DTYPE_IDS.get(dtype, 0)
            td.broadcast(self._hdr, src=0)

    def _bcast_tensor(self, t: torch.Tensor) -> torch.Tensor:
        if self.world > 1:
            td.broadcast(t, src=0)
        return t

    # -- operations (call on rank 0 only) ----------------------------------

    def append(self, rows: torch.Tensor) -> int:
        rows = rows.to(self.device)
        self._bcast_cmd(CMD_APPEND, rows.shape[0], rows.shape[1], dtype=rows.dtype)
        self._bcast_tensor(rows.contiguous())
        return self.store.append(rows)

    def search(self, queries: torch.Tensor, k: int):
        q = queries.to(self.device)
        self._bcast_cmd(CMD_SEARCH, q.shape[0], q.shape[1], int(k), dtype=q.dtype)
        self._bcast_tensor(q.contiguous())
        return self.store.search(q, k)

    def total(self) -> int:
        return self.store.total

    def stop(self) -> None:
        self._bcast_cmd(CMD_STOP)


def worker_loop(coord_dim: int = 768, capacity: int = 1 << 20) -> None:
    """Ranks 1..N-1: execute broadcast commands until CMD_STOP."""
    rank, world, device = init_from_env()
    store = ShardedStore(coord_dim, device=str(device), capacity=capacity)
    hdr = torch.zeros(8, dtype=torch.int64, device=device)

    def recv_tensor(n: int, d: int, dtype: torch.dtype) -> torch.Tensor:
        t = torch.empty(n, d, dtype=dtype, device=device)
        td.broadcast(t, src=0)
        return t

    while True:
        td.broadcast(hdr, src=0)
        cmd, n, d, k, dtid = (int(x) for x in hdr[:5].tolist())
        if cmd == CMD_STOP:
            break
        dtype = _DTYPES.get(dtid, torch.float32)
        if cmd == CMD_APPEND:
            store.append(recv_tensor(n, d, dtype))
        elif cmd == CMD_SEARCH:
            store.search(recv_tensor(n, d, dtype), k)


def main(argv: Optional[list] = None) -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--port", type=int, default=8101)
    ap.add_argument("--dim", type=int, default=768)
    ap.add_argument("--data-dir", default="./data")
    args = ap.parse_args(argv)

    rank, world, device = init_from_env()
    if rank != 0:
        worker_loop(args.dim)
        return

    # rank 0: the HTTP service on top of a distributed engine. The engine's
    # EmbeddingStore is replaced by the coordinator-backed sharded store.
    import uvicorn

    from kakveda_amd.gfkb.engine import GfkbEngine
    from kakveda_amd.services.gfkb_service import create_app

    coord = DistGfkbCoordinator(dim=args.dim)
    engine = GfkbEngine(data_dir=args.data_dir, device=str(device), dim=args.dim)
    # migrate JSONL-restored rows into the coordinator-backed store (the
    # engine's __init__ rebuilt them into the throwaway local store)
    engine.attach_store(_CoordinatorStore(coord))  # type: ignore[arg-type]
    app = create_app(engine=engine)
    try:
        uvicorn.run(app, host="0.0.0.0", port=args.port)
    finally:
        coord.stop()


class _CoordinatorStore:
    """EmbeddingStore facade over the coordinator (duck-typed)."""

    def __init__(self, coord: DistGfkbCoordinator):
        self.coord = coord
        self.dim = coord.dim
        self.device = coord.device
        self.dtype = coord.store.local.dtype

    @property
    def count(self) -> int:
        return self.coord.total()

    def append(self, rows: torch.Tensor) -> int:
        return self.coord.append(rows.to(self.dtype))

    def search(self, queries: torch.Tensor, k: int, valid_n=None):
        # valid_n snapshots are a single-store liveness optimisation; the
        # coordinator path is SPMD-collective and searches its live count
        return self.coord.search(queries.to(self.dtype), k)

    @property
    def data(self) -> torch.Tensor:
        return self.coord.store.local.data


if __name__ == "__main__":
    main()
