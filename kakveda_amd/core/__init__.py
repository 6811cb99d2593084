"""Core shared runtime: wire schemas, signatures, config, stores, logging.

Behavioural parity with the reference's ``services/shared/`` package
(models.py, fingerprint.py, config.py, runtime.py), re-implemented fresh.
"""

from kakveda_amd.core.schemas import (  # noqa: F401
    CanonicalFailureRecord,
    FailureMatch,
    FailureMatchRequest,
    FailureMatchResponse,
    FailureSignal,
    HealthPoint,
    IngestRequest,
    PatternEntity,
    Severity,
    TracePayload,
    WarningRequest,
    WarningResponse,
)
from kakveda_amd.core.signature import (  # noqa: F401
    detect_citation_markers,
    fingerprint,
    normalize_prompt,
    prompt_intent_tags,
    signature_text,
)
