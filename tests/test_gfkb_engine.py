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
