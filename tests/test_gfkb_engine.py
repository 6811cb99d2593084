"""GFKB engine semantics: versioned upserts, matching, patterns, rebuild."""

from kakveda_amd.gfkb.engine import GfkbEngine


def _engine(tmp_path):
    return GfkbEngine(data_dir=str(tmp_path), device="cpu", dim=256, hash_dim=4096)


SIG = (
    "intent_tags:intent:citations_required | prompt_hint:please provide references "
    "for why the sky is blue. | tools: | env_keys:e2e,source"
)


def test_upsert_creates_then_versions(tmp_path):
    eng = _engine(tmp_path)
    rec1, created1 = eng.upsert_failure(
        "HALLUCINATION_CITATION", SIG, {"model": "stub"}, app_id="app-A"
    )
    assert created1 and rec1["failure_id"] == "F-0001" and rec1["version"] == 1

    rec2, created2 = eng.upsert_failure(
        "HALLUCINATION_CITATION", SIG, {"model": "stub"}, app_id="app-B"
    )
    assert not created2
    assert rec2["failure_id"] == "F-0001"
    assert rec2["version"] == 2
    assert rec2["occurrences"] == 2
    assert rec2["affected_apps"] == ["app-A", "app-B"]
    # one embedding row per identity, two JSONL version rows
    assert eng.store.count == 1
    assert len(eng.failures) == 2


def test_match_exact_signature_scores_high(tmp_path):
    eng = _engine(tmp_path)
    eng.upsert_failure("HALLUCINATION_CITATION", SIG, {}, app_id="app-A")
    eng.upsert_failure(
        "OTHER_TYPE",
        "intent_tags: | prompt_hint:completely different thing | tools: | env_keys:",
        {},
        app_id="app-B",
    )
    matches = eng.match(SIG)
    assert matches
    assert matches[0].failure_id == "F-0001"
    assert matches[0].score >= 0.99
    assert matches[0].failure_type == "HALLUCINATION_CITATION"


def test_match_type_filter(tmp_path):
    eng = _engine(tmp_path)
    eng.upsert_failure("TYPE_A", SIG, {}, app_id="a")
    matches = eng.match(SIG, failure_type="TYPE_B")
    assert matches == []


def test_rebuild_from_log(tmp_path):
    eng = _engine(tmp_path)
    eng.upsert_failure("HALLUCINATION_CITATION", SIG, {}, app_id="app-A")
    eng.upsert_failure("HALLUCINATION_CITATION", SIG, {}, app_id="app-B")

    # fresh engine over the same dir rebuilds the HBM mirror from JSONL
    eng2 = _engine(tmp_path)
    assert eng2.store.count == 1
    matches = eng2.match(SIG)
    assert matches and matches[0].version == 2


def test_pattern_upsert_identity_by_name(tmp_path):
    eng = _engine(tmp_path)
    p1, created1 = eng.upsert_pattern("P", ["F-0001"], ["a"], "desc")
    assert created1 and p1["pattern_id"] == "FP-0001"
    p2, created2 = eng.upsert_pattern("P", ["F-0002"], ["b"], None)
    assert not created2
    assert p2["failure_ids"] == ["F-0001", "F-0002"]
    assert p2["affected_apps"] == ["a", "b"]
    assert p2["description"] == "desc"
    # list dedups to latest per pattern_id
    pats = eng.list_patterns()
    assert len(pats) == 1 and pats[0]["failure_ids"] == ["F-0001", "F-0002"]


def test_store_growth(tmp_path):
    eng = _engine(tmp_path)
    for i in range(40):
        eng.upsert_failure(
            "T",
            f"intent_tags: | prompt_hint:unique prompt number {i} | tools: | env_keys:",
            {},
            app_id="a",
        )
    assert eng.store.count == 40
    m = eng.match(
        "intent_tags: | prompt_hint:unique prompt number 17 | tools: | env_keys:"
    )
    assert m[0].score >= 0.99


def test_attach_store_migrates_rows(tmp_path):
    """Swapping the embedding store re-encodes known identities (the
    distributed server path: JSONL-restored rows must survive)."""
    from kakveda_amd.gfkb.engine import EmbeddingStore

    eng = _engine(tmp_path)
    eng.upsert_failure("HALLUCINATION_CITATION", SIG, {}, app_id="a")
    eng.upsert_failure(
        "T", "intent_tags: | prompt_hint:other | tools: | env_keys:", {}, app_id="b"
    )
    fresh = EmbeddingStore(256, device="cpu", capacity=64)
    eng.attach_store(fresh)
    assert eng.store is fresh and eng.store.count == 2
    m = eng.match(SIG)
    assert m and m[0].failure_id == "F-0001" and m[0].score >= 0.99


def test_segmented_store_growth_and_search():
    """Growth past segment_rows allocates fixed segments (no 2x copy);
    multi-segment search merges exactly to the single-store result."""
    import torch

    from kakveda_amd.gfkb.engine import EmbeddingStore
    from kakveda_amd.ops import cosine_topk_ref

    torch.manual_seed(5)
    dim = 64
    store = EmbeddingStore(dim, device="cpu", capacity=8, segment_rows=32)
    data = torch.randn(100, dim)
    data = data / data.norm(dim=-1, keepdim=True)
    first = store.append(data[:20])
    assert first == 0 and store.n_segments == 1  # doubled in place while small
    store.append(data[20:70])
    store.append(data[70:])
    assert store.count == 100
    assert store.n_segments > 1  # grew by fixed segments past segment_rows
    assert all(int(s.shape[0]) <= 32 for s in store._segments)

    q = torch.randn(7, dim)
    q = q / q.norm(dim=-1, keepdim=True)
    scores, idx = store.search(q, 5)
    ref_s, ref_i = cosine_topk_ref(q, data, 5)
    assert torch.allclose(scores, ref_s, atol=1e-5)
    gathered = (q.float() @ data.float().t()).gather(1, idx)
    assert torch.allclose(gathered, scores, atol=1e-5)

    # row_range: within one segment (view) and across a boundary (copy)
    assert torch.equal(store.row_range(0, 10), data[:10])
    assert torch.equal(store.row_range(25, 45), data[25:45])
    assert torch.equal(store.row_range(0, 100), data)


def test_segmented_store_adopt_then_grow():
    """The bench/restore adopt path stays zero-copy; later inserts grow
    segment-wise without touching the adopted tensor."""
    import torch

    from kakveda_amd.gfkb.engine import EmbeddingStore

    torch.manual_seed(6)
    dim = 32
    big = torch.randn(50, dim)
    store = EmbeddingStore(dim, device="cpu", capacity=4, segment_rows=16)
    store.adopt(big)
    assert store.count == 50 and store._segments[0] is big  # zero copy
    more = torch.randn(10, dim)
    store.append(more)
    assert store.count == 60 and store.n_segments == 2
    assert store._segments[0] is big  # adopted tensor untouched
    assert torch.equal(store.row_range(45, 60), torch.cat([big[45:], more]))


def test_sidecar_restore_skips_reencode(tmp_path, monkeypatch):
    """Restart restores embeddings from the packed sidecar (mmap+upload),
    not by re-encoding every identity; a stale sidecar falls back to
    re-encode and is rewritten."""
    import torch

    eng = _engine(tmp_path)
    eng.upsert_failure("HALLUCINATION_CITATION", SIG, {}, app_id="a")
    for i in range(5):
        eng.upsert_failure(
            "T", f"intent_tags: | prompt_hint:noise {i} | tools: | env_keys:", {}, app_id="b"
        )
    assert eng.sidecar.count() == 6
    rows_before = eng.store.row_range(0, 6).clone()

    # restart: sidecar matches -> encode_texts must NOT be called in rebuild
    from kakveda_amd.encoder.model import TraceEncoder

    calls = []
    orig = TraceEncoder.encode_texts

    def counting(self, texts):
        calls.append(len(texts))
        return orig(self, texts)

    monkeypatch.setattr(TraceEncoder, "encode_texts", counting)
    eng2 = GfkbEngine(data_dir=str(tmp_path), device="cpu", dim=256, hash_dim=4096)
    assert eng2.store.count == 6
    assert not calls, "rebuild re-encoded despite a valid sidecar"
    assert torch.equal(eng2.store.row_range(0, 6), rows_before)
    m = eng2.match(SIG)
    assert m and m[0].failure_id == "F-0001" and m[0].score >= 0.99

    # corrupt the sidecar (truncate one row): rebuild re-encodes + rewrites
    bin_path = tmp_path / "embeddings.bin"
    data = bin_path.read_bytes()
    bin_path.write_bytes(data[: len(data) - eng.sidecar.row_bytes])
    eng3 = GfkbEngine(data_dir=str(tmp_path), device="cpu", dim=256, hash_dim=4096)
    assert calls, "stale sidecar must trigger re-encode"
    assert eng3.store.count == 6 and eng3.sidecar.count() == 6
    m = eng3.match(SIG)
    assert m and m[0].failure_id == "F-0001" and m[0].score >= 0.99
