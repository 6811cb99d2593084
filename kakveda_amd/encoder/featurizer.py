"""Deterministic hashed n-gram featurisation of signature texts.

The reference scores similarity with a TF-IDF vectoriser over 1-2 word
grams refit on every request (similarity.py:14-20). Here the feature space
is fixed up front: word uni+bigrams hashed into ``hash_dim`` buckets with
sub-linear (sqrt) term-frequency weights. Identical texts map to identical
sparse vectors (cosine 1.0), texts sharing intent tags / prompt hints share
mass — which preserves the reference's behavioural contract (repeated
citation-style prompts match >= 0.8, unrelated prompts do not) without any
per-query fitting, so the hot path becomes a pure GPU kernel.
"""

from __future__ import annotations

import math
import re
from typing import Dict, List, Sequence, Tuple

import numpy as np

_TOKEN_RE = re.compile(r"[a-z0-9_:.\-]+")

_FNV_OFFSET = 0xCBF29CE484222325
_FNV_PRIME = 0x100000001B3
_MASK64 = (1 << 64) - 1


def _fnv1a(s: str, seed: int = 0) -> int:
    """64-bit FNV-1a — stable across processes (unlike Python's hash())."""
    h = (_FNV_OFFSET ^ (seed * 0x9E3779B97F4A7C15)) & _MASK64
    for b in s.encode("utf-8"):
        h ^= b
        h = (h * _FNV_PRIME) & _MASK64
    return h


def tokenize(text: str) -> List[str]:
    return _TOKEN_RE.findall(text.lower())


def featurize(text: str, hash_dim: int = 1 << 16, seed: int = 0) -> Tuple[np.ndarray, np.ndarray]:
    """Hash a text into (indices, weights) of its sparse feature vector.

    Weights are sqrt(term count), L2-normalised, matching TF-IDF's
    sublinear-tf flavour closely enough for the threshold contract.
    Returns int32 indices (sorted, unique) and float32 weights.
    """
    toks = tokenize(text)
    grams: Dict[int, float] = {}
    for i, t in enumerate(toks):
        grams[_fnv1a(t, seed)] = grams.get(_fnv1a(t, seed), 0.0) + 1.0
        if i + 1 < len(toks):
            big = _fnv1a(toks[i] + " " + toks[i + 1], seed)
            grams[big] = grams.get(big, 0.0) + 1.0
    if not grams:
        return np.zeros(0, dtype=np.int32), np.zeros(0, dtype=np.float32)

    buckets: Dict[int, float] = {}
    for h, cnt in grams.items():
        idx = h % hash_dim
        sign = 1.0 if (h >> 62) & 1 else -1.0  # hashing-trick sign to cancel collisions
        buckets[idx] = buckets.get(idx, 0.0) + sign * math.sqrt(cnt)

    idxs = np.fromiter(buckets.keys(), dtype=np.int32, count=len(buckets))
    ws = np.fromiter(buckets.values(), dtype=np.float32, count=len(buckets))
    order = np.argsort(idxs)
    idxs, ws = idxs[order], ws[order]
    norm = float(np.linalg.norm(ws))
    if norm > 0:
        ws = ws / norm
    return idxs, ws


def featurize_batch(
    texts: Sequence[str], hash_dim: int = 1 << 16, seed: int = 0, max_features: int = 64
) -> Tuple[np.ndarray, np.ndarray]:
    """Featurise a batch into padded [B, L] index/weight arrays.

    Padding uses index 0 with weight 0 (the weight masks the contribution).
    Features beyond ``max_features`` keep the largest-|weight| ones.
    """
    B = len(texts)
    idx_out = np.zeros((B, max_features), dtype=np.int32)
    w_out = np.zeros((B, max_features), dtype=np.float32)
    for b, text in enumerate(texts):
        idxs, ws = featurize(text, hash_dim=hash_dim, seed=seed)
        if len(idxs) > max_features:
            keep = np.argsort(-np.abs(ws))[:max_features]
            idxs, ws = idxs[keep], ws[keep]
            norm = float(np.linalg.norm(ws))
            if norm > 0:
                ws = ws / norm
        idx_out[b, : len(idxs)] = idxs
        w_out[b, : len(ws)] = ws
    return idx_out, w_out
