"""GPU ops: hand-written CDNA4 (gfx950) HIP kernels + CPU references.

Kernels (kakveda_amd/ops/hip/):
- cosine_topk  — fused MFMA-tiled score GEMM (queries x corpus^T) with an
  LDS-resident per-row top-k epilogue per corpus chunk + a merge kernel.
  Replaces the reference's per-request TF-IDF refit + full-corpus cosine
  (/root/reference/services/shared/similarity.py:14-20,
  /root/reference/services/gfkb/app.py:79-102).
- l2normalize_ — in-place row L2-normalisation (bf16, vectorised loads).
- embedding_bag — weighted gather-sum over the hashed-feature table
  (the trace-encoder front end).
- kmeans_assign / kmeans_update — streaming k-means for the pattern
  detector (assignment reuses the score GEMM; update is a segmented
  reduction).

Policy: on a CUDA(ROCm) device the HIP extension is REQUIRED — ops raise
if it is missing rather than silently falling back to eager PyTorch. The
torch fallbacks exist for CPU (tests, BASELINE config 1) only.
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

import torch

_EXT = None
_EXT_ERR: Optional[str] = None


def _load_extension():
    global _EXT, _EXT_ERR
    if _EXT is not None or _EXT_ERR is not None:
        return _EXT
    try:
        import importlib

        _EXT = importlib.import_module("kakveda_amd.ops._kakveda_hip")
    except ImportError as exc:  # pragma: no cover - exercised on GPU boxes
        _EXT_ERR = str(exc)
        _EXT = None
    return _EXT


def hip_available() -> bool:
    return _load_extension() is not None


def _require_ext():
    ext = _load_extension()
    if ext is None:
        raise RuntimeError(
            "kakveda_amd HIP extension (_kakveda_hip) is not built but a GPU "
            "op was requested on a CUDA device. Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950). "
            f"Import error: {_EXT_ERR}"
        )
    return ext


# ---------------------------------------------------------------------------
# cosine top-k
# ---------------------------------------------------------------------------

#: per-row candidate capacity of the fused HIP kernel (kakveda_kernels.hip)
KMAX = 8

def cosine_topk(
    queries: torch.Tensor, corpus: torch.Tensor, k: int, valid_n: Optional[int] = None
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Top-k inner products of each query row against every corpus row.

    queries: [B, D], corpus: [N, D]; rows are expected pre-normalised so the
    inner product is the cosine. Returns (scores f32 [B, k], idx i64 [B, k])
    sorted descending per row. ``valid_n`` limits the search to the first
    valid_n corpus rows (the live prefix of a preallocated store).

    The HIP kernel keeps at most KMAX=8 candidates per row; a request with
    k > 8 on GPU returns the true top-8 padded with (-inf, -1) so CPU and
    GPU agree on the leading 8 columns instead of the GPU path raising.
    """
    n = int(valid_n) if valid_n is not None else corpus.shape[0]
    B = queries.shape[0]
    if n <= 0:
        return (
            torch.full((B, k), float("-inf"), dtype=torch.float32, device=queries.device),
            torch.full((B, k), -1, dtype=torch.int64, device=queries.device),
        )
    if queries.device.type == "cuda":
        ext = _require_ext()
        kk = min(int(k), KMAX)
        scores, idx = ext.cosine_topk(queries, corpus, kk, n)
        if kk < k:
            pad_s = torch.full((B, k - kk), float("-inf"), dtype=scores.dtype, device=scores.device)
            pad_i = torch.full((B, k - kk), -1, dtype=idx.dtype, device=idx.device)
            scores = torch.cat([scores, pad_s], dim=1)
            idx = torch.cat([idx, pad_i], dim=1)
        return scores, idx
    return cosine_topk_ref(queries, corpus, k, n)


def cosine_topk_ref(
    queries: torch.Tensor, corpus: torch.Tensor, k: int, valid_n: Optional[int] = None
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Plain-PyTorch reference (fp32 accumulate), used on CPU and in tests."""
    n = int(valid_n) if valid_n is not None else corpus.shape[0]
    kk = min(k, n)
    sims = queries.float() @ corpus[:n].float().t()
    scores, idx = torch.topk(sims, kk, dim=1)
    if kk < k:
        pad_s = torch.full((queries.shape[0], k - kk), float("-inf"), device=queries.device)
        pad_i = torch.full((queries.shape[0], k - kk), -1, dtype=idx.dtype, device=queries.device)
        scores = torch.cat([scores, pad_s], dim=1)
        idx = torch.cat([idx, pad_i], dim=1)
    return scores.float(), idx.long()


# ---------------------------------------------------------------------------
# row L2 normalisation
# ---------------------------------------------------------------------------

def l2normalize_(t: torch.Tensor, start_row: int = 0, end_row: Optional[int] = None) -> torch.Tensor:
    """In-place L2-normalise rows [start_row, end_row) of a [N, D] tensor."""
    end = int(end_row) if end_row is not None else t.shape[0]
    if end <= start_row:
        return t
    if t.device.type == "cuda":
        ext = _require_ext()
        ext.l2normalize_(t, int(start_row), end)
        return t
    seg = t[start_row:end]
    norm = seg.float().norm(dim=-1, keepdim=True).clamp_min(1e-12)
    seg.copy_((seg.float() / norm).to(t.dtype))
    return t


# ---------------------------------------------------------------------------
# embedding bag (hashed-feature gather-sum)
# ---------------------------------------------------------------------------

def embedding_bag(table: torch.Tensor, idx: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """out[b] = sum_l w[b,l] * table[idx[b,l]]  -> [B, D] float32."""
    if table.device.type == "cuda":
        ext = _require_ext()
        return ext.embedding_bag(table, idx.to(torch.int32), w.float())
    flat = table[idx.reshape(-1).long()].reshape(*idx.shape, table.shape[1]).float()
    return (flat * w.unsqueeze(-1).float()).sum(dim=1)


# ---------------------------------------------------------------------------
# k-means (pattern detector)
# ---------------------------------------------------------------------------

def kmeans_assign_scored(
    points: torch.Tensor, centroids: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor]:
    """(top-1 cosine f32 [N], nearest centroid id i64 [N]) per point.

    GPU, C<=64: the dedicated LDS-resident-centroid kernel (point stream
    straight from HBM to MFMA, no per-window barriers — built for the
    pattern detector's N-huge/C-tiny shape). GPU, C>64: the general fused
    cosine_topk with k=1. CPU: eager matmul+argmax.
    """
    if points.device.type == "cuda":
        ext = _require_ext()
        if centroids.shape[0] <= 64:
            scores, idx = ext.kmeans_assign(points, centroids)
            return scores, idx.long()
        scores, idx = ext.cosine_topk(points, centroids, 1, centroids.shape[0])
        return scores[:, 0], idx[:, 0]
    sims = points.float() @ centroids.float().t()
    top = sims.max(dim=1)
    return top.values, top.indices


def kmeans_assign(points: torch.Tensor, centroids: torch.Tensor) -> torch.Tensor:
    """Nearest (max-cosine) centroid id per point -> int64 [N]."""
    return kmeans_assign_scored(points, centroids)[1]


def kmeans_update(
    points: torch.Tensor, assign: torch.Tensor, n_clusters: int
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Segmented reduction: per-cluster sum + count -> ([C, D] f32, [C] f32).

    GPU: LDS-partial segmented-reduction HIP kernel (points read once per
    64-dim tile). CPU fallback: index_add_.
    """
    if points.device.type == "cuda" and n_clusters <= 512:
        ext = _require_ext()
        return ext.kmeans_update(
            points.to(torch.bfloat16), assign.to(torch.int32), int(n_clusters)
        )
    D = points.shape[1]
    sums = torch.zeros(n_clusters, D, dtype=torch.float32, device=points.device)
    sums.index_add_(0, assign.long(), points.float())
    counts = torch.zeros(n_clusters, dtype=torch.float32, device=points.device)
    counts.index_add_(0, assign.long(), torch.ones_like(assign, dtype=torch.float32))
    return sums, counts
