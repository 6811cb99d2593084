"""Wire schemas for every cross-service payload.

Field-compatible with the reference's pydantic models
(/root/reference/services/shared/models.py:10-121) so that clients, the
JSONL data files (data/failures.jsonl etc.) and the HTTP endpoints keep
their exact shapes. The implementation is original; only the field names,
types and defaults are pinned by the compatibility contract.
"""

from __future__ import annotations

from datetime import datetime, timezone
from enum import Enum
from typing import Any, Dict, List, Optional

from pydantic import BaseModel, Field


def utcnow() -> datetime:
    """Timezone-aware UTC now (the reference serialises Z-suffixed UTC)."""
    return datetime.now(timezone.utc)


class Severity(str, Enum):
    low = "low"
    medium = "medium"
    high = "high"


#: health-score weights per severity (reference config/config.yaml:7-10)
SEVERITY_WEIGHTS = {Severity.low: 1.0, Severity.medium: 3.0, Severity.high: 7.0}


class TracePayload(BaseModel):
    """One observed LLM interaction (prompt/response + context)."""

    trace_id: str
    ts: datetime
    app_id: str
    agent_id: Optional[str] = None

    prompt: str
    response: str

    model: Optional[str] = None
    temperature: Optional[float] = None

    tools: List[str] = Field(default_factory=list)
    env: Dict[str, Any] = Field(default_factory=dict)


class IngestRequest(BaseModel):
    trace: TracePayload


class FailureSignal(BaseModel):
    """Emitted by the failure classifier when a trace exhibits a failure."""

    trace_id: str
    ts: datetime
    app_id: str

    failure_type: str
    severity: Severity

    root_cause: Optional[str] = None
    mitigation: Optional[str] = None

    context_signature: Dict[str, Any]


class CanonicalFailureRecord(BaseModel):
    """A versioned GFKB entry (JSONL row shape of data/failures.jsonl)."""

    failure_id: str
    version: int
    created_at: datetime
    updated_at: datetime

    failure_type: str
    root_cause: Optional[str] = None
    context_signature: Dict[str, Any]

    impact_severity: Severity
    resolution: Optional[str] = None

    occurrences: int = 0
    affected_apps: List[str] = Field(default_factory=list)

    signature_text: str


class FailureMatchRequest(BaseModel):
    signature_text: str
    failure_type: Optional[str] = None


class FailureMatch(BaseModel):
    failure_id: str
    version: int
    score: float
    failure_type: str
    suggested_mitigation: Optional[str] = None


class FailureMatchResponse(BaseModel):
    matches: List[FailureMatch]


class PatternEntity(BaseModel):
    """A recurring failure pattern (JSONL row shape of data/patterns.jsonl)."""

    pattern_id: str
    name: str
    created_at: datetime
    failure_ids: List[str]
    affected_apps: List[str]
    description: Optional[str] = None


class WarningRequest(BaseModel):
    app_id: str
    agent_id: Optional[str] = None
    prompt: str
    tools: List[str] = Field(default_factory=list)
    env: Dict[str, Any] = Field(default_factory=dict)


class WarningResponse(BaseModel):
    action: str  # block | warn | silent
    confidence: float
    pattern_id: Optional[str] = None
    references: List[FailureMatch] = Field(default_factory=list)
    message: str


class HealthPoint(BaseModel):
    """One health-timeline sample (JSONL row shape of data/health.jsonl)."""

    ts: datetime
    app_id: str
    score: float
    failure_rate: float
    recurrent_penalty: float
    avg_recovery_time_sec: float
    notes: Dict[str, Any] = Field(default_factory=dict)
