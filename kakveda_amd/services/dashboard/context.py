"""Shared dashboard context: DB session factory, transport, settings."""

from __future__ import annotations

import hashlib
import json
import os
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import httpx
from fastapi import Request

from kakveda_amd.core.runtime import get_runtime_config, setup_logging
from kakveda_amd.services.dashboard.auth import (
    RateLimiter,
    RevocationStore,
    decode_jwt,
)
from kakveda_amd.services.wiring import Transport

COOKIE_NAME = "kv_token"
IMPERSONATE_COOKIE = "kv_view_as"
DEMO_USERS = [
    ("admin@kakveda.local", "admin123", "Admin", ["admin"]),
    ("operator@kakveda.local", "operator123", "Operator", ["operator"]),
    ("viewer@kakveda.local", "viewer123", "Viewer", ["viewer"]),
    ("demo@kakveda.local", "demo123", "Demo", ["viewer"]),
]

#: deterministic citation-bearing stub (reference dashboard/app.py:1193-1199):
#: produced whenever the model backend is unreachable, which makes the whole
#: failure pipeline reproducible with zero dependencies.
STUB_RESPONSE = (
    "Here is a concise answer with supporting citations. Key finding one. [1] "
    "Key finding two. [2]\n\nReferences:\n[1] A. Example, 2020.\n[2] B. Sample, 2021."
)


@dataclass
class DashboardContext:
    Session: Any
    tx: Transport
    urls: Dict[str, str]
    data_dir: str
    jwt_secret: str = ""
    revocation: Optional[RevocationStore] = None
    auth_limiter: Optional[RateLimiter] = None
    log: Any = None

    def __post_init__(self):
        rc = get_runtime_config()
        if not self.jwt_secret:
            self.jwt_secret = rc.jwt_secret
        # Redis-backed revocation/rate-limit when KAKVEDA_REDIS_URL is set
        # and reachable; in-memory fallback otherwise (reference
        # shared/redis_helpers.py:26-84 semantics)
        if self.revocation is None:
            self.revocation = RevocationStore(redis_url=rc.redis_url)
        if self.auth_limiter is None:
            self.auth_limiter = RateLimiter(20, 60, redis_url=rc.redis_url)
        if rc.env == "prod" and self.jwt_secret == "kakveda-dev-secret":
            # production guardrail (reference dashboard/app.py:1266-1269)
            raise RuntimeError(
                "refusing to start in prod with the default JWT secret; set "
                "KAKVEDA_JWT_SECRET"
            )
        if self.log is None:
            self.log = setup_logging("dashboard")

    # -- auth helpers --------------------------------------------------------

    def current_user(self, request: Request) -> Optional[Dict[str, Any]]:
        token = request.cookies.get(COOKIE_NAME) or ""
        if not token:
            auth = request.headers.get("Authorization", "")
            if auth.startswith("Bearer "):
                token = auth[7:]
        payload = decode_jwt(token, self.jwt_secret) if token else None
        if payload and self.revocation.is_revoked(payload.get("jti", "")):
            return None
        if payload:
            # admin "view as role" impersonation (reference app.py:2730-2763)
            view_as = request.cookies.get(IMPERSONATE_COOKIE)
            if view_as and "admin" in (payload.get("roles") or []):
                payload = dict(payload)
                payload["roles"] = [view_as]
                payload["impersonating"] = True
        return payload

    # -- model provider ------------------------------------------------------

    async def generate(
        self, prompt: str, model: Optional[str] = None, timeout: float = 60.0
    ) -> Dict[str, Any]:
        """Generate via Ollama-compatible backend; deterministic stub on
        failure. Returns {text, provider, model, latency_ms}."""
        base = os.environ.get("OLLAMA_URL", "http://ollama:11434")
        mdl = model or os.environ.get("KAKVEDA_MODEL", "llama3.2")
        t0 = time.perf_counter()
        try:
            async with httpx.AsyncClient(timeout=timeout) as client:
                resp = await client.post(
                    f"{base}/api/generate",
                    json={"model": mdl, "prompt": prompt, "stream": False},
                )
                resp.raise_for_status()
                text = resp.json().get("response", "")
                provider = "ollama"
        except Exception:
            text = STUB_RESPONSE
            provider = "stub"
            mdl = "deterministic-stub"
        return {
            "text": text,
            "provider": provider,
            "model": mdl,
            "latency_ms": (time.perf_counter() - t0) * 1000.0,
        }

    async def list_models(self) -> List[str]:
        base = os.environ.get("OLLAMA_URL", "http://ollama:11434")
        try:
            async with httpx.AsyncClient(timeout=3.0) as client:
                resp = await client.get(f"{base}/api/tags")
                return [m.get("name", "") for m in resp.json().get("models", [])]
        except Exception:
            return ["deterministic-stub"]


def estimate_tokens(text: str) -> int:
    """len/4 heuristic (reference app.py:139-147)."""
    return max(1, len(text) // 4)


def estimate_cost_usd_micro(tokens_in: int, tokens_out: int, model: str = "") -> int:
    """Micro-USD cost model (reference app.py:150-170 keeps integer micro
    dollars; rates are illustrative demo constants)."""
    rate_in, rate_out = 5, 15  # micro-USD per 1k tokens
    return (tokens_in * rate_in + tokens_out * rate_out) // 1000


def sha256_hex(s: str) -> str:
    return hashlib.sha256(s.encode()).hexdigest()


async def read_payload(request: Request) -> Dict[str, Any]:
    """Accept JSON or urlencoded form bodies (no python-multipart dep)."""
    ctype = request.headers.get("content-type", "")
    body = await request.body()
    if "application/json" in ctype:
        try:
            return json.loads(body or b"{}")
        except json.JSONDecodeError:
            return {}
    from urllib.parse import parse_qs

    parsed = parse_qs(body.decode("utf-8", "replace"))
    return {k: v[0] if len(v) == 1 else v for k, v in parsed.items()}
