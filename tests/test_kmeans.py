"""Streaming k-means correctness (CPU; GPU path covered in test_gpu_ops)."""

import torch

from kakveda_amd.patterns.kmeans import StreamingKMeans
from kakveda_amd.patterns.miner import PatternMiner


def _blobs(k=4, per=50, dim=64, seed=0):
    g = torch.Generator().manual_seed(seed)
    centers = torch.randn(k, dim, generator=g)
    centers = centers / centers.norm(dim=-1, keepdim=True)
    pts = centers.repeat_interleave(per, 0) + 0.05 * torch.randn(
        k * per, dim, generator=g
    )
    pts = pts / pts.norm(dim=-1, keepdim=True)
    labels = torch.arange(k).repeat_interleave(per)
    return pts, labels, centers


def test_kmeans_recovers_blobs():
    pts, labels, _ = _blobs()
    km = StreamingKMeans(4, 64, device="cpu", seed=1)
    assign = km.fit(pts, iters=15)
    # each true blob should map to a single cluster (purity ~1)
    for b in range(4):
        got = assign[labels == b]
        purity = (got == got.mode().values).float().mean().item()
        assert purity > 0.95, purity


def test_kmeans_step_matches_reference():
    """One Lloyd step equals the plain-torch reference update."""
    pts, _, _ = _blobs(k=3, per=30, dim=32, seed=2)
    km = StreamingKMeans(3, 32, device="cpu", seed=3, decay=0.0)
    c0 = km.centroids.clone()
    assign, _ = km.step(pts)

    sims = pts @ c0.t()
    ref_assign = sims.argmax(dim=1)
    assert torch.equal(assign, ref_assign)
    ref = torch.zeros_like(c0)
    cnt = torch.zeros(3)
    ref.index_add_(0, ref_assign, pts)
    cnt.index_add_(0, ref_assign, torch.ones(len(pts)))
    mask = cnt > 0
    ref[mask] = ref[mask] / cnt[mask].unsqueeze(1)
    ref[mask] = ref[mask] / ref[mask].norm(dim=-1, keepdim=True)
    assert torch.allclose(km.centroids[mask], ref[mask], atol=1e-5)


def test_pattern_miner(tmp_path):
    from kakveda_amd.gfkb.engine import GfkbEngine

    eng = GfkbEngine(data_dir=str(tmp_path), device="cpu", dim=128, hash_dim=2048)
    # two semantic groups of failures across two apps each
    for i in range(6):
        eng.upsert_failure(
            "HALLUCINATION_CITATION",
            f"intent_tags:intent:citations_required | prompt_hint:cite sources v{i} | tools: | env_keys:",
            {},
            app_id=f"app-{i % 2}",
        )
    for i in range(6):
        eng.upsert_failure(
            "TIMEOUT",
            f"intent_tags: | prompt_hint:slow request totally different {i} | tools: | env_keys:",
            {},
            app_id=f"app-{i % 3}",
        )
    miner = PatternMiner(eng, n_clusters=2, min_apps=2)
    patterns = miner.mine(iters=12)
    assert patterns, "expected at least one mined pattern"
    listed = eng.list_patterns()
    assert len(listed) == len(patterns)
    # every pattern spans >= 2 apps (reference gating semantics)
    for p in patterns:
        assert len(p["affected_apps"]) >= 2
