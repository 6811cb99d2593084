"""The kakveda-amd microservice constellation.

Wire-compatible with the reference's 9 FastAPI services (SURVEY.md section
1: same endpoints, ports, event topics and JSONL shapes), re-implemented as
app factories over a pluggable transport so the whole pipeline can run
either as real HTTP containers or mounted in one process (tests,
BASELINE config 1's deterministic CPU end-to-end run).

Default ports (reference docker-compose.yml:10-163): event-bus 8100,
gfkb 8101, ingestion 8102, failure-classifier 8103, pattern-detector 8104,
warning-policy 8105, health-scoring 8106, dashboard 8110, agent-echo 8120.
"""

DEFAULT_PORTS = {
    "event_bus": 8100,
    "gfkb": 8101,
    "ingestion": 8102,
    "failure_classifier": 8103,
    "pattern_detector": 8104,
    "warning_policy": 8105,
    "health_scoring": 8106,
    "dashboard": 8110,
    "agent_echo": 8120,
}

TOPIC_TRACE_INGESTED = "trace.ingested"
TOPIC_FAILURE_DETECTED = "failure.detected"
TOPIC_CHILD_SAFETY = "child_safety_alert"
