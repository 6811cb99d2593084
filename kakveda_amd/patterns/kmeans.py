"""Streaming spherical k-means over fingerprint embeddings.

The GPU generalisation of the reference's single hard-coded pattern rule
(reference: services/pattern_detector/app.py:29-58; mapping in SURVEY.md
section 2.5): assignment is the same MFMA cosine kernel as the GFKB search
(argmax = top-1), the centroid update is a segmented reduction, and
multi-GPU runs all-reduce the centroid sums/counts over RCCL/xGMI before
renormalising (BASELINE config 4).
"""

from __future__ import annotations

from typing import Tuple

import torch
import torch.distributed as td

from kakveda_amd import ops
from kakveda_amd.parallel.dist import is_distributed


class StreamingKMeans:
    def __init__(
        self,
        n_clusters: int,
        dim: int,
        device: str = "cpu",
        seed: int = 0,
        decay: float = 0.9,
    ):
        self.k = n_clusters
        self.dim = dim
        self.device = torch.device(device)
        self.decay = decay  # EMA weight for streaming (mini-batch) updates
        gen = torch.Generator(device="cpu").manual_seed(seed)
        init = torch.randn(n_clusters, dim, generator=gen, dtype=torch.float32)
        init = init / init.norm(dim=-1, keepdim=True)
        self.centroids = init.to(self.device)
        self._counts = torch.zeros(n_clusters, dtype=torch.float32, device=self.device)

    def _centroids_matcher(self) -> torch.Tensor:
        if self.device.type == "cuda":
            return self.centroids.to(torch.bfloat16)
        return self.centroids

    def assign(self, points: torch.Tensor) -> torch.Tensor:
        """Nearest-centroid (max cosine) id per point -> int64 [N]."""
        pts = points.to(self.device)
        if self.device.type == "cuda":
            pts = pts.to(torch.bfloat16)
        return ops.kmeans_assign(pts, self._centroids_matcher())

    def step(self, points: torch.Tensor) -> Tuple[torch.Tensor, float]:
        """One (mini-)batch update. Returns (assignments, mean cosine).

        Distributed: every rank passes its own points; centroid partials
        are all-reduced so all ranks hold identical centroids after.
        The quality metric reuses the assignment's top-1 scores (free)
        instead of re-gathering centroids per point.
        """
        pts = points.to(self.device)
        if self.device.type == "cuda":
            scores, assign = ops.kmeans_assign_scored(
                pts.to(torch.bfloat16), self._centroids_matcher()
            )
            mean_cos = scores.mean()
        else:
            sims = pts.float() @ self.centroids.t()
            top = sims.max(dim=1)
            assign = top.indices
            mean_cos = top.values.mean()
        sums, counts = ops.kmeans_update(
            pts.float() if self.device.type == "cpu" else pts.to(torch.bfloat16),
            assign,
            self.k,
        )
        if is_distributed():
            td.all_reduce(sums)
            td.all_reduce(counts)

        # EMA merge, then renormalise to the unit sphere (spherical k-means)
        upd = counts > 0
        mean = torch.where(
            upd.unsqueeze(1), sums / counts.clamp_min(1.0).unsqueeze(1), self.centroids
        )
        new = self.decay * self.centroids + (1.0 - self.decay) * mean
        new = torch.where(upd.unsqueeze(1), new, self.centroids)
        norm = new.norm(dim=-1, keepdim=True).clamp_min(1e-12)
        self.centroids = new / norm
        self._counts = self.decay * self._counts + counts

        return assign, float(mean_cos.item())

    def fit(self, points: torch.Tensor, iters: int = 10) -> torch.Tensor:
        """Full-batch Lloyd iterations (decay ignored: hard reassignment)."""
        old_decay, self.decay = self.decay, 0.0
        try:
            assign = self.assign(points)
            for _ in range(iters):
                assign, _ = self.step(points)
        finally:
            self.decay = old_decay
        return assign

    def cluster_sizes(self) -> torch.Tensor:
        return self._counts.clone()
