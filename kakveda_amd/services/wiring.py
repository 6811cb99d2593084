"""Service-to-service transport with in-process ASGI routing.

The reference wires services with hard-coded container URLs + httpx posts
(e.g. failure_classifier/app.py:14-15). Here outbound calls go through a
``Transport`` that resolves a URL either to a real HTTP client or — when
the target app is registered locally — to an httpx ASGITransport client,
so the full event pipeline runs in one process with identical semantics
(used by tests and the deterministic CPU e2e config).
"""

from __future__ import annotations

import asyncio
from typing import Any, Dict, Optional

import httpx


def raise_thread_limiter(default: int = 256) -> int:
    """Raise anyio's default 40-token sync-handler thread limiter for the
    CURRENT event loop (call from an app startup hook).

    FastAPI runs plain-`def` handlers through this limiter; 40 tokens cap
    how many requests can be parked inside the gfkb `match` handler
    waiting on the micro-batcher, which caps the batch size and hence
    service throughput (measured: the HTTP peak sat at exactly
    40/cycle/worker — profiles/serving_http.md). KAKVEDA_THREADPOOL
    overrides the new cap."""
    import os

    tokens = int(os.environ.get("KAKVEDA_THREADPOOL", str(default)))
    try:
        import anyio.to_thread

        anyio.to_thread.current_default_thread_limiter().total_tokens = tokens
    except Exception:  # pragma: no cover - anyio internals moved
        return 0
    return tokens


class Transport:
    def __init__(self, timeout: float = 3.0):
        self.timeout = timeout
        self._local: Dict[str, Any] = {}  # base_url -> ASGI app
        self._clients: Dict[str, httpx.AsyncClient] = {}

    def register_local(self, base_url: str, app: Any) -> None:
        self._local[base_url.rstrip("/")] = app
        self._clients.pop(base_url.rstrip("/"), None)

    def local_app(self, base_url: str) -> Optional[Any]:
        """The locally-registered ASGI app for a base URL, if any."""
        return self._local.get(base_url.rstrip("/"))

    def _client(self, url: str) -> tuple[httpx.AsyncClient, str]:
        for base, app in self._local.items():
            if url.startswith(base):
                cli = self._clients.get(base)
                if cli is None:
                    cli = httpx.AsyncClient(
                        transport=httpx.ASGITransport(app=app),
                        base_url=base,
                        timeout=self.timeout,
                    )
                    self._clients[base] = cli
                return cli, url[len(base) :] or "/"
        cli = self._clients.get("__net__")
        if cli is None:
            cli = httpx.AsyncClient(timeout=self.timeout)
            self._clients["__net__"] = cli
        return cli, url

    async def post(self, url: str, json: Any = None, timeout: Optional[float] = None) -> httpx.Response:
        cli, path = self._client(url)
        return await cli.post(path, json=json, timeout=timeout or self.timeout)

    async def get(self, url: str, params: Any = None, timeout: Optional[float] = None) -> httpx.Response:
        cli, path = self._client(url)
        return await cli.get(path, params=params, timeout=timeout or self.timeout)

    async def aclose(self) -> None:
        await asyncio.gather(*(c.aclose() for c in self._clients.values()), return_exceptions=True)
        self._clients.clear()
