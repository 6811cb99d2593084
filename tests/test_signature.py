"""Signature/fingerprint parity tests (reference fingerprint.py semantics)."""

from kakveda_amd.core.signature import (
    detect_citation_markers,
    fingerprint,
    normalize_prompt,
    prompt_intent_tags,
    signature_text,
)


def test_normalize_prompt():
    assert normalize_prompt("  Hello   World \n ") == "hello world"


def test_intent_tags_citations():
    tags = prompt_intent_tags("Please summarize this and include references [importantly]")
    assert "intent:citations_required" in tags
    assert "task:summarization" in tags
    assert "instruction:include_references" in tags
    assert tags == sorted(tags)


def test_signature_text_layout():
    sig = signature_text(
        "Explain why the sky is blue with citations",
        tools=["web", "calc", "web"],
        env={"b": 1, "a": 2},
    )
    parts = sig.split(" | ")
    assert parts[0].startswith("intent_tags:")
    assert "intent:citations_required" in parts[0]
    assert parts[1] == "prompt_hint:explain why the sky is blue with citations"
    assert parts[2] == "tools:calc,web"
    assert parts[3] == "env_keys:a,b"


def test_signature_matches_reference_sample():
    # Exact string from /root/reference/data/failures.jsonl (wire contract).
    sig = signature_text(
        "Please provide references for why the sky is blue.",
        tools=[],
        env={"source": "kids-agent", "e2e": True},
    )
    assert sig == (
        "intent_tags:intent:citations_required | "
        "prompt_hint:please provide references for why the sky is blue. | "
        "tools: | env_keys:e2e,source"
    )


def test_fingerprint_stable():
    fp1 = fingerprint("explain x with sources", ["t"], {"k": 1})
    fp2 = fingerprint("explain   X with sources ", ["t"], {"k": 2})  # same env KEYS
    assert fp1 == fp2
    assert len(fp1) == 16


def test_citation_markers():
    assert detect_citation_markers("as shown in [1] and [2]").has_citation_markers
    assert detect_citation_markers("(Smith, 2020) argued").has_citation_markers
    assert detect_citation_markers("see doi: 10.1234/abc").has_citation_markers
    assert detect_citation_markers("See the References section").has_citation_markers
    assert not detect_citation_markers("no markers here").has_citation_markers
