"""Optional OpenTelemetry wiring (parity with reference
services/shared/otel.py:6-59): OTLP HTTP exporter + BatchSpanProcessor,
gated by KAKVEDA_OTEL_ENABLED and import-guarded so the platform runs
without the opentelemetry packages installed."""

from __future__ import annotations

import os
from typing import Any, Optional


def setup_otel(service_name: str) -> Optional[Any]:
    """Install a tracer provider when enabled + importable; else None."""
    if os.environ.get("KAKVEDA_OTEL_ENABLED", "0") not in ("1", "true", "yes"):
        return None
    try:
        from opentelemetry import trace
        from opentelemetry.exporter.otlp.proto.http.trace_exporter import (
            OTLPSpanExporter,
        )
        from opentelemetry.sdk.resources import Resource
        from opentelemetry.sdk.trace import TracerProvider
        from opentelemetry.sdk.trace.export import BatchSpanProcessor
    except ImportError:
        return None

    provider = TracerProvider(
        resource=Resource.create({"service.name": service_name})
    )
    provider.add_span_processor(BatchSpanProcessor(OTLPSpanExporter()))
    trace.set_tracer_provider(provider)
    return provider


def instrument_fastapi(app: Any) -> bool:
    """Auto-instrument FastAPI routes when the instrumentation package is
    importable; returns whether instrumentation was applied."""
    if os.environ.get("KAKVEDA_OTEL_ENABLED", "0") not in ("1", "true", "yes"):
        return False
    try:
        from opentelemetry.instrumentation.fastapi import FastAPIInstrumentor
    except ImportError:
        return False
    FastAPIInstrumentor.instrument_app(app)
    return True
