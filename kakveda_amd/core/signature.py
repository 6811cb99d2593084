"""Prompt signatures: normalisation, intent tags, signature text, fingerprints.

Behaviourally equivalent to the reference's
/root/reference/services/shared/fingerprint.py:16-87 — the *output strings*
must match exactly because signature_text is persisted in failures.jsonl and
is the identity key for GFKB upserts. The tag vocabulary and the
``"intent_tags:... | prompt_hint:... | tools:... | env_keys:..."`` layout are
therefore part of the wire contract.
"""

from __future__ import annotations

import hashlib
import re
from typing import Any, Dict, List, NamedTuple

_WS_RE = re.compile(r"\s+")

# Markers that indicate a response contains citations (reference
# fingerprint.py:9-13): numeric brackets, (Author, year), DOIs.
_CITATION_RES = (
    re.compile(r"\[[0-9]+\]"),
    re.compile(r"\([A-Za-z]+,\s*\d{4}\)"),
    re.compile(r"doi:\s*\S+"),
)

_CITE_WORDS = ("citation", "citations", "reference", "references", "sources", "bibliography")
_SUMM_WORDS = ("summarize", "summary", "tl;dr")
_EXPL_WORDS = ("explain", "explanation", "describe")


def normalize_prompt(prompt: str) -> str:
    """Lowercase, trim, and collapse all whitespace runs to single spaces."""
    return _WS_RE.sub(" ", prompt.strip().lower())


def prompt_intent_tags(prompt: str) -> List[str]:
    """Coarse app-agnostic intent tags for a prompt.

    Prompts that carry the same failure risk share tags even when worded
    differently; this is the stable component of the similarity signal.
    """
    p = normalize_prompt(prompt)
    tags: set[str] = set()

    wants_citations = any(w in p for w in _CITE_WORDS)
    if wants_citations:
        tags.add("intent:citations_required")
    if any(w in p for w in _SUMM_WORDS):
        tags.add("task:summarization")
    if any(w in p for w in _EXPL_WORDS):
        tags.add("task:explanation")
    if "even if not provided" in p or "even if none" in p:
        tags.add("constraint:no_sources_provided")
    if wants_citations and "include" in p:
        tags.add("instruction:include_references")

    return sorted(tags)


def signature_text(prompt: str, tools: List[str], env: Dict[str, Any]) -> str:
    """Build the canonical signature string for a request context.

    App-agnostic by design (no app_id / trace_id). Layout is wire-pinned:
    ``intent_tags:<t1,t2> | prompt_hint:<first 80 normalised chars> |
    tools:<sorted unique> | env_keys:<sorted>``.
    """
    tags = prompt_intent_tags(prompt)
    hint = normalize_prompt(prompt)[:80]
    return " | ".join(
        (
            "intent_tags:" + ",".join(tags),
            "prompt_hint:" + hint,
            "tools:" + ",".join(sorted(set(tools))),
            "env_keys:" + ",".join(sorted(env.keys())),
        )
    )


def fingerprint(prompt: str, tools: List[str], env: Dict[str, Any]) -> str:
    """16-hex-char sha256 fingerprint of the signature text."""
    sig = signature_text(prompt, tools, env)
    return hashlib.sha256(sig.encode("utf-8")).hexdigest()[:16]


class CitationCheck(NamedTuple):
    has_citation_markers: bool


def detect_citation_markers(text: str) -> CitationCheck:
    """True when a response contains citation-like markers."""
    t = text or ""
    if any(r.search(t) for r in _CITATION_RES):
        return CitationCheck(True)
    low = t.lower()
    return CitationCheck("references" in low or "bibliography" in low)
