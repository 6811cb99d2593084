"""LocalCluster: the full service constellation mounted in one process.

Runs the reference's end-to-end pipeline (SURVEY.md section 3.1/3.3:
warn -> generate(stub) -> ingest -> classify -> gfkb upsert -> pattern ->
health) deterministically on CPU with no network — the analogue of the
reference's docker-compose stack + scripts/demo_client.py, used by tests
and BASELINE config 1.
"""

from __future__ import annotations

from typing import Optional

from kakveda_amd.core.config import ConfigStore
from kakveda_amd.gfkb.engine import GfkbEngine
from kakveda_amd.services import DEFAULT_PORTS
from kakveda_amd.services import (
    event_bus as event_bus_mod,
    failure_classifier as fc_mod,
    gfkb_service as gfkb_mod,
    health_scoring as hs_mod,
    ingestion as ing_mod,
    pattern_detector as pd_mod,
    warning_policy as wp_mod,
)
from kakveda_amd.services.wiring import Transport


class LocalCluster:
    def __init__(
        self,
        data_dir: str,
        device: str = "cpu",
        config: Optional[ConfigStore] = None,
        engine: Optional[GfkbEngine] = None,
        with_dashboard: bool = False,
    ):
        self.tx = Transport()
        cfg = config or ConfigStore()
        self.urls = {
            name: f"http://{name.replace('_', '-')}:{port}"
            for name, port in DEFAULT_PORTS.items()
        }
        u = self.urls

        self.engine = engine or GfkbEngine(data_dir=data_dir, device=device)
        self.event_bus = event_bus_mod.create_app(transport=self.tx)
        self.gfkb = gfkb_mod.create_app(engine=self.engine)
        self.ingestion = ing_mod.create_app(event_bus_url=u["event_bus"], transport=self.tx)
        self.failure_classifier = fc_mod.create_app(
            event_bus_url=u["event_bus"],
            gfkb_url=u["gfkb"],
            self_url=u["failure_classifier"],
            transport=self.tx,
        )
        self.warning_policy = wp_mod.create_app(
            gfkb_url=u["gfkb"], transport=self.tx, config=cfg
        )
        self.pattern_detector = pd_mod.create_app(
            event_bus_url=u["event_bus"],
            gfkb_url=u["gfkb"],
            self_url=u["pattern_detector"],
            transport=self.tx,
            engine=self.engine,
        )
        self.health_scoring = hs_mod.create_app(
            event_bus_url=u["event_bus"],
            data_dir=data_dir,
            self_url=u["health_scoring"],
            transport=self.tx,
            config=cfg,
        )

        self.dashboard = None
        if with_dashboard:
            from kakveda_amd.services.dashboard import create_app as dash_create

            self.dashboard = dash_create(
                data_dir=data_dir,
                transport=self.tx,
                urls=u,
                self_url=u["dashboard"],
            )
            self.tx.register_local(u["dashboard"], self.dashboard)

        for name, asgi_app in (
            ("event_bus", self.event_bus),
            ("gfkb", self.gfkb),
            ("ingestion", self.ingestion),
            ("failure_classifier", self.failure_classifier),
            ("warning_policy", self.warning_policy),
            ("pattern_detector", self.pattern_detector),
            ("health_scoring", self.health_scoring),
        ):
            self.tx.register_local(u[name], asgi_app)

    async def start(self) -> None:
        """Perform the startup subscriptions (what container startup does)."""
        await self.failure_classifier.state.subscribe()
        await self.pattern_detector.state.subscribe()
        await self.health_scoring.state.subscribe()
        if self.dashboard is not None:
            await self.dashboard.state.subscribe()

    async def warn(self, app_id: str, prompt: str, tools=None, env=None) -> dict:
        resp = await self.tx.post(
            f"{self.urls['warning_policy']}/warn",
            json={
                "app_id": app_id,
                "prompt": prompt,
                "tools": tools or [],
                "env": env or {},
            },
        )
        return resp.json()

    async def ingest(self, trace: dict) -> dict:
        resp = await self.tx.post(
            f"{self.urls['ingestion']}/ingest", json={"trace": trace}
        )
        return resp.json()

    async def aclose(self) -> None:
        await self.tx.aclose()
