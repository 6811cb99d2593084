"""Env-driven runtime settings + structured JSON logging + request ids.

Parity with /root/reference/services/shared/runtime.py:54-142 — the JSON
log record fields (service, request_id, path, status, duration) and the
``X-Request-Id`` echo are part of the operational contract.
"""

from __future__ import annotations

import json
import logging
import os
import sys
import uuid
from dataclasses import dataclass, field
from typing import Any, Dict, Optional


@dataclass(frozen=True)
class RuntimeConfig:
    env: str = "dev"
    log_level: str = "INFO"
    log_format: str = "json"  # json | text
    request_id_header: str = "X-Request-Id"
    jwt_secret: str = "kakveda-dev-secret"
    jwt_issuer: str = "kakveda"
    redis_url: Optional[str] = None
    otel_enabled: bool = False
    extra: Dict[str, str] = field(default_factory=dict)


def get_runtime_config() -> RuntimeConfig:
    e = os.environ
    return RuntimeConfig(
        env=e.get("KAKVEDA_ENV", "dev"),
        log_level=e.get("KAKVEDA_LOG_LEVEL", "INFO").upper(),
        log_format=e.get("KAKVEDA_LOG_FORMAT", "json"),
        request_id_header=e.get("KAKVEDA_REQUEST_ID_HEADER", "X-Request-Id"),
        jwt_secret=e.get("KAKVEDA_JWT_SECRET", "kakveda-dev-secret"),
        jwt_issuer=e.get("KAKVEDA_JWT_ISSUER", "kakveda"),
        redis_url=e.get("KAKVEDA_REDIS_URL") or None,
        otel_enabled=e.get("KAKVEDA_OTEL_ENABLED", "0") in ("1", "true", "yes"),
    )


class _JsonFormatter(logging.Formatter):
    def __init__(self, service: str):
        super().__init__()
        self.service = service

    def format(self, record: logging.LogRecord) -> str:
        payload: Dict[str, Any] = {
            "ts": self.formatTime(record, "%Y-%m-%dT%H:%M:%S"),
            "level": record.levelname,
            "service": self.service,
            "msg": record.getMessage(),
        }
        for key in ("request_id", "path", "status", "duration_ms", "method"):
            val = getattr(record, key, None)
            if val is not None:
                payload[key] = val
        if record.exc_info:
            payload["exc"] = self.formatException(record.exc_info)
        return json.dumps(payload, ensure_ascii=False)


def setup_logging(service: str, level: Optional[str] = None, fmt: Optional[str] = None) -> logging.Logger:
    """Install a JSON (or plain-text) stdout handler for this service."""
    rc = get_runtime_config()
    logger = logging.getLogger(service)
    logger.setLevel(level or rc.log_level)
    logger.propagate = False
    if not logger.handlers:
        handler = logging.StreamHandler(sys.stdout)
        if (fmt or rc.log_format) == "json":
            handler.setFormatter(_JsonFormatter(service))
        else:
            handler.setFormatter(
                logging.Formatter("%(asctime)s %(levelname)s [" + service + "] %(message)s")
            )
        logger.addHandler(handler)
    return logger


def ensure_request_id(existing: Optional[str]) -> str:
    """Return the propagated request id, or mint a fresh one."""
    return existing if existing else uuid.uuid4().hex
