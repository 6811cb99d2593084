"""Sharded GFKB store: each rank owns a slice of the fingerprint corpus in
its own HBM; searches run the fused cosine-topk kernel per shard and merge
candidates with one all-gather over xGMI (SURVEY.md sections 2.5 and 5.7).

Sharding is round-robin by insertion order (global row r lives on rank
r % world at local row r // world), which keeps shards balanced under
continuous inserts without rebalancing. Candidate payloads are tiny
((score, global-id) pairs, k per shard per query), so the merge is
latency-bound; compute-heavy local search dominates.
"""

from __future__ import annotations

from typing import Tuple

import torch
import torch.distributed as td

from kakveda_amd.gfkb.engine import EmbeddingStore
from kakveda_amd.parallel.dist import get_world, is_distributed


class ShardedStore:
    """A distributed EmbeddingStore with all-gather top-k merge.

    All ranks must call ``append`` / ``search`` collectively with the same
    arguments (SPMD): inserts carry the full batch to every rank, each rank
    keeps its round-robin share; searches evaluate the same query batch on
    every shard.
    """

    def __init__(self, dim: int, device: str = "cpu", capacity: int = 1024):
        import os

        self.rank, self.world = get_world()
        self.dim = dim
        self.local = EmbeddingStore(dim, device=device, capacity=capacity)
        self.total = 0  # global row count
        # test hook: run the all-gather merge even at world=1 so a 1-GPU
        # box executes the real RCCL collective path (RCCL refuses two
        # ranks on one device — "Duplicate GPU detected" — so this is the
        # strongest single-box exercise of the nccl branch; the multi-rank
        # path itself is covered by gloo world=2 tests and the driver's
        # 8-GPU SCALE run)
        self.force_collectives = os.environ.get("KAKVEDA_FORCE_COLLECTIVES") == "1"

    @property
    def device(self) -> torch.device:
        return self.local.device

    def append(self, rows: torch.Tensor) -> int:
        """Collectively append [n, D] rows; returns the first global id."""
        n = rows.shape[0]
        first_global = self.total
        # rows whose global id r satisfies r % world == rank
        ids = torch.arange(first_global, first_global + n, device=rows.device)
        mine = (ids % self.world) == self.rank
        if bool(mine.any()):
            self.local.append(rows[mine])
        self.total += n
        return first_global

    def load_shard(self, rows: torch.Tensor, total: int) -> None:
        """Benchmark/restore fast path: install this rank's shard directly.
        ``rows`` are the rank's round-robin rows in local order (global id
        of local row l is l*world + rank); ``total`` is the global count."""
        if rows.dtype == self.local.dtype:
            self.local.adopt(rows.contiguous())
        else:
            self.local.append(rows)
        self.total = total

    def _local_to_global(self, local_idx: torch.Tensor) -> torch.Tensor:
        """Round-robin inverse: local row l on rank r is global l*world + r."""
        out = local_idx * self.world + self.rank
        return torch.where(local_idx < 0, local_idx, out)

    def search(self, queries: torch.Tensor, k: int) -> Tuple[torch.Tensor, torch.Tensor]:
        """Global top-k: local fused search + all-gather + merge.

        Returns (scores f32 [B, k], global ids i64 [B, k]) identical on all
        ranks, equal to a single-store search over the same data.
        """
        B = queries.shape[0]
        scores, lidx = self.local.search(queries, k)
        gidx = self._local_to_global(lidx)

        if not is_distributed() or (self.world == 1 and not self.force_collectives):
            return scores, gidx

        # scores and int64 ids gathered separately: f32 cannot carry ids
        # beyond 2^24 exactly, and exactness matters at 100M+ rows.
        buf_s = [torch.empty_like(scores) for _ in range(self.world)]
        buf_i = [torch.empty_like(gidx) for _ in range(self.world)]
        td.all_gather(buf_s, scores.contiguous())
        td.all_gather(buf_i, gidx.contiguous())
        all_scores = torch.cat(buf_s, dim=1)  # [B, world*k]
        all_idx = torch.cat(buf_i, dim=1)

        top_s, sel = torch.topk(all_scores, k, dim=1)
        top_i = all_idx.gather(1, sel)
        return top_s, top_i

    def barrier(self) -> None:
        if is_distributed():
            td.barrier()
