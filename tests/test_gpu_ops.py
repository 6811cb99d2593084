"""GPU numerics tests: HIP kernels vs plain-PyTorch fp32 references."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _dev():
    return torch.device("cuda:0")


def _rand_unit(n, d, seed=0, device=None):
    g = torch.Generator(device="cpu").manual_seed(seed)
    x = torch.randn(n, d, generator=g, dtype=torch.float32)
    x = x / x.norm(dim=-1, keepdim=True)
    return x.to(device=device or _dev(), dtype=torch.bfloat16)


def test_extension_loaded():
    from kakveda_amd import ops

    assert ops.hip_available(), "HIP extension must be built on GPU boxes"


@pytest.mark.parametrize(
    "B,N,k",
    [
        (1, 100, 5),
        (100, 1000, 5),
        (128, 4096, 8),
        (300, 70000, 5),
        (1024, 300000, 8),
    ],
)
def test_cosine_topk_vs_torch(B, N, k):
    from kakveda_amd import ops

    D = 768
    q = _rand_unit(B, D, seed=1)
    c = _rand_unit(N, D, seed=2)

    scores, idx = ops.cosine_topk(q, c, k)
    torch.cuda.synchronize()

    # fp32 reference on the SAME bf16-rounded inputs
    sims = q.float() @ c.float().t()
    ref_scores, ref_idx = torch.topk(sims, k, dim=1)

    assert scores.shape == (B, k) and idx.shape == (B, k)
    # scores sorted descending
    assert (scores[:, :-1] >= scores[:, 1:] - 1e-6).all()
    # kernel scores match reference top-k scores (bf16 accumulation tolerance)
    assert torch.allclose(scores, ref_scores, atol=2e-2, rtol=1e-2), (
        (scores - ref_scores).abs().max().item()
    )
    # each claimed index really achieves its claimed score
    gathered = sims.gather(1, idx.clamp_min(0))
    assert torch.allclose(gathered, scores, atol=1e-4), (
        (gathered - scores).abs().max().item()
    )


def test_cosine_topk_valid_n():
    from kakveda_amd import ops

    q = _rand_unit(16, 768, seed=3)
    c = _rand_unit(1000, 768, seed=4)
    # plant a huge match OUTSIDE the valid prefix: must not be returned
    c[900] = q[0].clone()
    scores, idx = ops.cosine_topk(q, c, 5, valid_n=800)
    torch.cuda.synchronize()
    assert (idx < 800).all()
    sims = q.float() @ c[:800].float().t()
    ref_scores, _ = torch.topk(sims, 5, dim=1)
    assert torch.allclose(scores, ref_scores, atol=2e-2, rtol=1e-2)


def test_cosine_topk_exact_match_first():
    from kakveda_amd import ops

    c = _rand_unit(5000, 768, seed=5)
    q = c[1234:1235].clone()
    scores, idx = ops.cosine_topk(q, c, 5)
    torch.cuda.synchronize()
    assert idx[0, 0].item() == 1234
    assert scores[0, 0].item() > 0.99


def test_l2normalize():
    from kakveda_amd import ops

    g = torch.Generator(device="cpu").manual_seed(9)
    x = (torch.randn(500, 768, generator=g) * 3.0).to(_dev(), torch.bfloat16)
    ref = x.float() / x.float().norm(dim=-1, keepdim=True).clamp_min(1e-12)
    ops.l2normalize_(x)
    torch.cuda.synchronize()
    assert torch.allclose(x.float(), ref, atol=1e-2), (x.float() - ref).abs().max()
    norms = x.float().norm(dim=-1)
    assert torch.allclose(norms, torch.ones_like(norms), atol=2e-2)


def test_l2normalize_partial_rows():
    from kakveda_amd import ops

    x = torch.ones(10, 768, device=_dev(), dtype=torch.bfloat16) * 2.0
    ops.l2normalize_(x, start_row=4, end_row=8)
    torch.cuda.synchronize()
    assert x[0, 0].item() == 2.0 and x[9, 0].item() == 2.0
    assert abs(x[5].float().norm().item() - 1.0) < 2e-2


def test_embedding_bag_vs_torch():
    from kakveda_amd import ops

    g = torch.Generator(device="cpu").manual_seed(11)
    V, D, B, L = 5000, 768, 64, 48
    table = torch.randn(V, D, generator=g).to(_dev(), torch.bfloat16)
    idx = torch.randint(0, V, (B, L), generator=g, dtype=torch.int32).to(_dev())
    w = torch.randn(B, L, generator=g, dtype=torch.float32).to(_dev())
    w[:, 40:] = 0.0  # padding

    out = ops.embedding_bag(table, idx, w)
    torch.cuda.synchronize()
    flat = table.float()[idx.reshape(-1).long()].reshape(B, L, D)
    ref = (flat * w.unsqueeze(-1)).sum(dim=1)
    assert torch.allclose(out, ref, atol=5e-2, rtol=1e-2), (out - ref).abs().max()


def test_encoder_gpu_matches_cpu():
    from kakveda_amd.encoder import TraceEncoder

    texts = [
        "intent_tags:intent:citations_required | prompt_hint:please cite sources | tools: | env_keys:a",
        "intent_tags: | prompt_hint:what is the weather | tools: | env_keys:b",
    ]
    cpu = TraceEncoder(dim=768, hash_dim=8192, seed=3, device="cpu")
    gpu = TraceEncoder(dim=768, hash_dim=8192, seed=3, device="cuda")
    e_cpu = cpu.encode_texts(texts)
    e_gpu = gpu.encode_texts(texts).cpu()
    torch.cuda.synchronize()
    # bf16 table/GEMM on GPU vs fp32 CPU: cosine of the two encodings ~ 1
    cos = (e_cpu * e_gpu).sum(dim=-1)
    assert (cos > 0.99).all(), cos


def test_gfkb_engine_on_gpu(tmp_path):
    from kakveda_amd.gfkb.engine import GfkbEngine

    eng = GfkbEngine(data_dir=str(tmp_path), device="cuda", dim=768, hash_dim=8192)
    sig = (
        "intent_tags:intent:citations_required | prompt_hint:explain with sources"
        " | tools: | env_keys:x"
    )
    eng.upsert_failure("HALLUCINATION_CITATION", sig, {}, app_id="a")
    for i in range(50):
        eng.upsert_failure(
            "T", f"intent_tags: | prompt_hint:filler {i} | tools: | env_keys:", {},
            app_id="a",
        )
    matches = eng.match(sig)
    assert matches and matches[0].failure_id == "F-0001"
    assert matches[0].score > 0.98


def test_kmeans_gpu_matches_cpu():
    from kakveda_amd.patterns.kmeans import StreamingKMeans

    g = torch.Generator().manual_seed(21)
    centers = torch.randn(4, 768, generator=g)
    centers = centers / centers.norm(dim=-1, keepdim=True)
    pts = centers.repeat_interleave(64, 0) + 0.05 * torch.randn(256, 768, generator=g)
    pts = pts / pts.norm(dim=-1, keepdim=True)

    km_cpu = StreamingKMeans(4, 768, device="cpu", seed=5)
    km_gpu = StreamingKMeans(4, 768, device="cuda", seed=5)
    a_cpu = km_cpu.fit(pts, iters=10)
    a_gpu = km_gpu.fit(pts.to("cuda"), iters=10).cpu()
    torch.cuda.synchronize()
    # same partition (bf16 rounding can only flip points on blob borders)
    agree = (a_cpu == a_gpu).float().mean().item()
    assert agree > 0.98, agree
    cos = (km_cpu.centroids * km_gpu.centroids.cpu()).sum(-1)
    assert (cos > 0.99).all(), cos


def test_kmeans_update_kernel_vs_torch():
    from kakveda_amd import ops

    g = torch.Generator().manual_seed(31)
    N, D, C = 20000, 768, 64
    pts = torch.randn(N, D, generator=g).to(_dev(), torch.bfloat16)
    assign = torch.randint(0, C, (N,), generator=g, dtype=torch.int32).to(_dev())

    sums, counts = ops.kmeans_update(pts, assign, C)
    torch.cuda.synchronize()

    ref_s = torch.zeros(C, D, device=_dev())
    ref_s.index_add_(0, assign.long(), pts.float())
    ref_c = torch.zeros(C, device=_dev())
    ref_c.index_add_(0, assign.long(), torch.ones(N, device=_dev()))

    assert torch.equal(counts, ref_c)
    assert torch.allclose(sums, ref_s, atol=0.5, rtol=1e-2), (
        (sums - ref_s).abs().max().item()
    )


def test_cosine_topk_argument_validation():
    from kakveda_amd import ops

    q = _rand_unit(8, 768, seed=40)
    c = _rand_unit(100, 768, seed=41)
    # k > KMAX is clamped to the kernel's top-8 and padded with (-inf, -1)
    # so the GPU path agrees with CPU on the leading KMAX columns instead
    # of raising (CPU/GPU divergence)
    s9, i9 = ops.cosine_topk(q, c, 9)
    s8, i8 = ops.cosine_topk(q, c, 8)
    torch.cuda.synchronize()
    assert s9.shape == (8, 9) and i9.shape == (8, 9)
    assert torch.equal(s9[:, :8], s8) and torch.equal(i9[:, :8], i8)
    assert torch.isinf(s9[:, 8]).all() and (i9[:, 8] == -1).all()
    with pytest.raises(RuntimeError):
        ops.cosine_topk(q.float(), c, 5)  # wrong dtype
    q100 = torch.randn(8, 100, device=_dev(), dtype=torch.bfloat16)
    c100 = torch.randn(50, 100, device=_dev(), dtype=torch.bfloat16)
    with pytest.raises(RuntimeError):
        ops.cosine_topk(q100, c100, 5)  # D not a multiple of 64


def test_k1_argmax_fast_path():
    """k=1 takes the dedicated argmax epilogue; results must match torch."""
    from kakveda_amd import ops

    q = _rand_unit(512, 768, seed=42)
    c = _rand_unit(50000, 768, seed=43)
    scores, idx = ops.cosine_topk(q, c, 1)
    torch.cuda.synchronize()
    sims = q.float() @ c.float().t()
    ref_s, ref_i = sims.max(dim=1)
    assert torch.allclose(scores[:, 0], ref_s, atol=2e-2, rtol=1e-2)
    gathered = sims.gather(1, idx)
    assert torch.allclose(gathered[:, 0], scores[:, 0], atol=1e-4)


@pytest.mark.gpu
@pytest.mark.parametrize("variant", ["eager", "fast", "rege", "8pbl", "8pe", "8pe2"])
def test_kernel_variants_match_default(variant):
    """Every KAKVEDA_KNN_KERNEL variant must produce the same top-k
    scores as the default ballot kernel (env is read once per process,
    so variants run in a subprocess). The emission variant (8pe) only
    engages at N >= 64k (it needs the prepass floors), so it gets a
    bigger corpus."""
    import os
    import subprocess
    import sys

    n = 131072 if variant in ("8pe", "8pe2") else 8192
    code = (
        "import torch\n"
        "from kakveda_amd import ops\n"
        "q = torch.randn(256, 768, generator=torch.Generator(device='cuda')"
        ".manual_seed(3), device='cuda')\n"
        "q = (q / q.norm(dim=-1, keepdim=True)).to(torch.bfloat16)\n"
        f"c = torch.randn({n}, 768, generator=torch.Generator(device='cuda')"
        ".manual_seed(4), device='cuda')\n"
        "c = (c / c.norm(dim=-1, keepdim=True)).to(torch.bfloat16)\n"
        "s, i = ops.cosine_topk(q, c, 5)\n"
        "torch.cuda.synchronize()\n"
        "print('CSUM', float(s.double().sum()))\n"
    )
    outs = {}
    for ksel in (None, variant):
        env = dict(os.environ)
        env.pop("KAKVEDA_KNN_KERNEL", None)
        if ksel:
            env["KAKVEDA_KNN_KERNEL"] = ksel
        r = subprocess.run(
            [sys.executable, "-c", code], env=env, capture_output=True,
            text=True, timeout=300,
        )
        assert r.returncode == 0, r.stderr[-1500:]
        outs[ksel] = float(r.stdout.split("CSUM")[1].strip())
    assert abs(outs[None] - outs[variant]) < 1e-2, outs


def test_segmented_store_gpu_matches_ref():
    """Multi-segment HBM store: per-segment fused kernels + merge equal a
    single-kernel search over the same rows."""
    from kakveda_amd import ops
    from kakveda_amd.gfkb.engine import EmbeddingStore

    torch.manual_seed(9)
    dim, n = 768, 50000
    data = _rand_unit(n, dim, seed=60)
    store = EmbeddingStore(dim, device="cuda:0", capacity=1024, segment_rows=16384)
    for s in range(0, n, 7000):
        store.append(data[s : min(s + 7000, n)])
    assert store.count == n and store.n_segments > 1
    q = _rand_unit(64, dim, seed=61)
    scores, idx = store.search(q, 5)
    ref_s, ref_i = ops.cosine_topk(q, data, 5)
    torch.cuda.synchronize()
    assert torch.allclose(scores, ref_s, atol=1e-3), (scores - ref_s).abs().max()
    gathered = (q.float() @ data.float().t()).gather(1, idx)
    assert torch.allclose(gathered, scores, atol=1e-3)


def test_sidecar_gpu_restore(tmp_path):
    """GPU engine restart restores bf16 rows bit-exactly from the packed
    sidecar without re-encoding."""
    from kakveda_amd.gfkb.engine import GfkbEngine

    eng = GfkbEngine(data_dir=str(tmp_path), device="cuda", dim=768, hash_dim=16384)
    sigs = [
        f"intent_tags:intent:citations_required | prompt_hint:gpu restore {i} | "
        "tools: | env_keys:e2e"
        for i in range(32)
    ]
    for s in sigs:
        eng.upsert_failure("HALLUCINATION_CITATION", s, {}, app_id="a")
    rows = eng.store.row_range(0, 32).clone()
    assert eng.sidecar.count() == 32

    from kakveda_amd.encoder.model import TraceEncoder

    calls = []
    orig = TraceEncoder.encode_texts
    TraceEncoder.encode_texts = lambda self, t: (calls.append(len(t)), orig(self, t))[1]
    try:
        eng2 = GfkbEngine(data_dir=str(tmp_path), device="cuda", dim=768, hash_dim=16384)
    finally:
        TraceEncoder.encode_texts = orig
    assert eng2.store.count == 32 and not calls
    assert torch.equal(eng2.store.row_range(0, 32), rows)
    m = eng2.match(sigs[7])
    assert m and m[0].score > 0.98


@pytest.mark.parametrize("C,N", [(64, 10007), (5, 300), (64, 256), (33, 70000)])
def test_kmeans_assign_kernel_vs_torch(C, N):
    """Dedicated LDS-resident-centroid assignment kernel (C<=64) matches
    the fp32 torch argmax on the same bf16 inputs."""
    from kakveda_amd import ops

    D = 768
    pts = _rand_unit(N, D, seed=70 + C)
    cents = _rand_unit(C, D, seed=71 + C)
    scores, idx = ops.kmeans_assign_scored(pts, cents)
    torch.cuda.synchronize()
    assert scores.shape == (N,) and idx.shape == (N,)
    sims = pts.float() @ cents.float().t()
    ref_v, ref_i = sims.max(dim=1)
    # scores match the fp32 reference within bf16-accumulation tolerance
    assert torch.allclose(scores, ref_v, atol=2e-2, rtol=1e-2), (
        (scores - ref_v).abs().max().item()
    )
    # each claimed centroid really achieves the claimed score
    gathered = sims.gather(1, idx.unsqueeze(1))[:, 0]
    assert torch.allclose(gathered, scores, atol=1e-4)
    # and the claimed centroid is a true argmax (scores equal at the max)
    assert torch.allclose(gathered, ref_v, atol=1e-4)


def test_kmeans_assign_tie_breaks_to_first():
    """Duplicate centroids: ties resolve to the lowest centroid id,
    matching torch.argmax/CPU-reference first-occurrence semantics."""
    from kakveda_amd import ops

    D = 768
    cents = _rand_unit(8, D, seed=80)
    cents = torch.cat([cents, cents[:4]])  # ids 8..11 duplicate 0..3
    pts = cents[torch.randint(0, 12, (500,), generator=torch.Generator().manual_seed(81))]
    _, idx = ops.kmeans_assign_scored(pts.contiguous(), cents.contiguous())
    torch.cuda.synchronize()
    assert (idx < 8).all(), idx.max()


def test_emission_kernel_exact_vs_torch():
    """The 8pe threshold-emission path (prepass floors + GEMM-core
    emission + emit_merge_topk) is exact vs fp32 torch.topk, including a
    valid_n prefix and k=8."""
    import os
    import subprocess
    import sys

    code = (
        "import torch\n"
        "from kakveda_amd import ops\n"
        "g = torch.Generator(device='cuda').manual_seed(9)\n"
        "q = torch.randn(300, 768, generator=g, device='cuda')\n"
        "q = (q / q.norm(dim=-1, keepdim=True)).to(torch.bfloat16)\n"
        "c = torch.randn(200001, 768, generator=g, device='cuda')\n"
        "c = (c / c.norm(dim=-1, keepdim=True)).to(torch.bfloat16)\n"
        "for k, vn in ((5, None), (8, None), (5, 150001), (1, None)):\n"
        "    s, i = ops.cosine_topk(q, c, k, valid_n=vn)\n"
        "    torch.cuda.synchronize()\n"
        "    n = vn or c.shape[0]\n"
        "    sims = q.float() @ c[:n].float().t()\n"
        "    rs, ri = torch.topk(sims, k, dim=1)\n"
        "    assert (i >= 0).all() and (i < n).all(), (k, vn)\n"
        "    assert torch.allclose(s, rs, atol=2e-2, rtol=1e-2), (k, vn)\n"
        "    gathered = sims.gather(1, i)\n"
        "    assert torch.allclose(gathered, s, atol=1e-4), (k, vn)\n"
        "    assert torch.allclose(gathered, rs, atol=1e-4), (k, vn)\n"
        "# B=1 single-query serving shape through the emission path\n"
        "s1, i1 = ops.cosine_topk(q[:1].contiguous(), c, 5)\n"
        "torch.cuda.synchronize()\n"
        "sims1 = q[:1].float() @ c.float().t()\n"
        "rs1, _ = torch.topk(sims1, 5, dim=1)\n"
        "assert torch.allclose(sims1.gather(1, i1), rs1, atol=1e-4)\n"
        "print('OK8PE')\n"
    )
    env = dict(os.environ)
    env["KAKVEDA_KNN_KERNEL"] = "8pe"
    r = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    assert "OK8PE" in r.stdout


def test_emission_correlated_queries_exact():
    """Correlated (template-generated) query batches make qualifying
    groups BURST on hot corpus tiles — the emission stash must
    drain-and-refill, not bail (a fixed-capacity poison path fired
    ~600x/row-tile on the bench and silently doubled step time via the
    overflow fallback). Exactness on encoder-style correlated queries at
    an emission-regime corpus size is the regression guard."""
    from kakveda_amd import ops
    from kakveda_amd.encoder.featurizer import featurize_batch
    from kakveda_amd.encoder.model import TraceEncoder

    dev = "cuda"
    D, B, N = 768, 1024, 131072
    enc = TraceEncoder(dim=D, hash_dim=1 << 16, seed=1234, device=dev)
    texts = [
        f"intent_tags:intent:citations_required | prompt_hint:synthetic probe "
        f"{i} explain with sources | tools:t{i % 7} | env_keys:e2e,k{i % 5}"
        for i in range(512)
    ]
    idx_np, w_np = featurize_batch(texts, hash_dim=enc.hash_dim, max_features=64)
    fi = torch.from_numpy(idx_np).to(dev).repeat(2, 1)[:B].contiguous()
    fw = torch.from_numpy(w_np).to(dev).repeat(2, 1)[:B].contiguous()
    q = enc.encode_features(fi, fw).to(torch.bfloat16)
    c = _rand_unit(N, D, seed=90)
    scores, idx = ops.cosine_topk(q, c, 5)
    torch.cuda.synchronize()
    sims = q.float() @ c.float().t()
    ref_s, _ = torch.topk(sims, 5, dim=1)
    assert torch.allclose(scores, ref_s, atol=2e-2, rtol=1e-2)
    gathered = sims.gather(1, idx)
    assert torch.allclose(gathered, scores, atol=1e-4)
    assert torch.allclose(gathered, ref_s, atol=1e-4)


@pytest.mark.parametrize("D", [256, 1536])
def test_emission_non_default_dims(D):
    """The emission path at non-768 dims (D % 64 == 0): window counts,
    prepass and floors must all generalise."""
    from kakveda_amd import ops

    q = _rand_unit(128, D, seed=95)
    c = _rand_unit(131072, D, seed=96)
    scores, idx = ops.cosine_topk(q, c, 5)
    torch.cuda.synchronize()
    sims = q.float() @ c.float().t()
    ref_s, _ = torch.topk(sims, 5, dim=1)
    assert torch.allclose(scores, ref_s, atol=2e-2, rtol=1e-2)
    assert torch.allclose(sims.gather(1, idx), ref_s, atol=1e-4)


@pytest.mark.parametrize("B", [1, 3, 8])
def test_smallb_streaming_search_exact(B):
    """B <= 8 requests route to the streaming small-batch emission kernel
    (no MFMA tile padding); must stay exact vs fp32 torch."""
    from kakveda_amd import ops

    q = _rand_unit(B, 768, seed=100 + B)
    c = _rand_unit(300000, 768, seed=101)
    scores, idx = ops.cosine_topk(q, c, 5)
    torch.cuda.synchronize()
    sims = q.float() @ c.float().t()
    ref_s, _ = torch.topk(sims, 5, dim=1)
    assert torch.allclose(scores, ref_s, atol=2e-2, rtol=1e-2)
    assert torch.allclose(sims.gather(1, idx), ref_s, atol=1e-4)
    # valid_n prefix respected on the streaming path too
    scores2, idx2 = ops.cosine_topk(q, c, 5, valid_n=123457)
    torch.cuda.synchronize()
    assert (idx2 < 123457).all()
    ref2, _ = torch.topk(sims[:, :123457], 5, dim=1)
    assert torch.allclose(scores2, ref2, atol=2e-2, rtol=1e-2)


@pytest.mark.gpu
def test_smallb_v4_matches_default():
    """The default v4 (8-lanes-per-row remap) and the legacy v2
    per-lane-row kernel (KAKVEDA_SMALLB=2) must produce the same top-k
    (env read once per process, so the variant runs in a subprocess)."""
    import os
    import subprocess
    import sys

    code = (
        "import torch\n"
        "from kakveda_amd import ops\n"
        "q = torch.randn(3, 768, generator=torch.Generator(device='cuda')"
        ".manual_seed(7), device='cuda')\n"
        "q = (q / q.norm(dim=-1, keepdim=True)).to(torch.bfloat16)\n"
        "c = torch.randn(300000, 768, generator=torch.Generator(device='cuda')"
        ".manual_seed(8), device='cuda')\n"
        "c = (c / c.norm(dim=-1, keepdim=True)).to(torch.bfloat16)\n"
        "s, i = ops.cosine_topk(q, c, 5)\n"
        "s2, i2 = ops.cosine_topk(q, c, 5, valid_n=123457)\n"
        "torch.cuda.synchronize()\n"
        "assert (i2 < 123457).all()\n"
        "print('CSUM', float(s.double().sum()) + float(s2.double().sum()))\n"
    )
    outs = {}
    for sel in (None, "2"):
        env = dict(os.environ)
        env.pop("KAKVEDA_SMALLB", None)
        if sel:
            env["KAKVEDA_SMALLB"] = sel
        r = subprocess.run(
            [sys.executable, "-c", code], env=env, capture_output=True,
            text=True, timeout=300,
        )
        assert r.returncode == 0, r.stderr[-1500:]
        outs[sel] = float(r.stdout.split("CSUM")[1].strip())
    assert abs(outs[None] - outs["2"]) < 1e-2, outs


@pytest.mark.gpu
def test_smallb_dim_tail_segments():
    """smallb v4 streams rows in 4-segment (1 KiB) batches; D % 256 != 0
    leaves tail segments (e.g. D=832 -> 13 segments = 3 batches + 1
    tail). Must stay exact vs fp32 torch on that path too."""
    from kakveda_amd import ops

    q = _rand_unit(2, 832, seed=210)
    c = _rand_unit(70000, 832, seed=211)
    scores, idx = ops.cosine_topk(q, c, 5)
    torch.cuda.synchronize()
    sims = q.float() @ c.float().t()
    ref_s, _ = torch.topk(sims, 5, dim=1)
    assert torch.allclose(scores, ref_s, atol=2e-2, rtol=1e-2)
    assert torch.allclose(sims.gather(1, idx), ref_s, atol=1e-4)
