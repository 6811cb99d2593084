"""Service launcher: ``python -m kakveda_amd.serve <service> [--port N]``.

Builds the requested service app from environment configuration (URLs via
*_URL env vars, DATA_DIR, CONFIG_PATH) and serves it with uvicorn. This is
what the CLI's ``up`` and the Docker images run.
"""

from __future__ import annotations

import argparse
import os


def build_app(name: str):
    from kakveda_amd.services import DEFAULT_PORTS

    if name not in DEFAULT_PORTS:
        raise SystemExit(f"unknown service '{name}'; one of {sorted(DEFAULT_PORTS)}")
    if name == "event_bus":
        from kakveda_amd.services.event_bus import create_app

        return create_app()
    if name == "ingestion":
        from kakveda_amd.services.ingestion import create_app

        return create_app()
    if name == "gfkb":
        from kakveda_amd.services.gfkb_service import create_app

        return create_app(data_dir=os.environ.get("DATA_DIR", "./data"))
    if name == "failure_classifier":
        from kakveda_amd.services.failure_classifier import create_app

        return create_app()
    if name == "pattern_detector":
        from kakveda_amd.services.pattern_detector import create_app

        return create_app()
    if name == "warning_policy":
        from kakveda_amd.services.warning_policy import create_app

        return create_app()
    if name == "health_scoring":
        from kakveda_amd.services.health_scoring import create_app

        return create_app(data_dir=os.environ.get("DATA_DIR", "./data"))
    if name == "agent_echo":
        from kakveda_amd.services.agent_echo import create_app

        return create_app()
    if name == "dashboard":
        from kakveda_amd.services.dashboard import create_app

        return create_app(data_dir=os.environ.get("DATA_DIR", "./data"))
    raise SystemExit(f"unhandled service {name}")


def main() -> None:
    from kakveda_amd.services import DEFAULT_PORTS

    ap = argparse.ArgumentParser(prog="python -m kakveda_amd.serve")
    ap.add_argument("service", choices=sorted(DEFAULT_PORTS))
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=None)
    args = ap.parse_args()

    import uvicorn

    from kakveda_amd.core.metrics import instrument as instrument_metrics
    from kakveda_amd.core.otel import instrument_fastapi, setup_otel

    app = build_app(args.service)
    setup_otel(f"kakveda-{args.service.replace('_', '-')}")
    instrument_fastapi(app)
    instrument_metrics(app, args.service)
    uvicorn.run(
        app,
        host=args.host,
        port=args.port or DEFAULT_PORTS[args.service],
        log_level="info",
    )


if __name__ == "__main__":
    main()
