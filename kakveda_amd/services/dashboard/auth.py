"""Auth primitives: pbkdf2 password hashing + stdlib HS256 JWT + RBAC.

Parity with reference services/dashboard/auth.py:15-62 and rbac.py:6-18.
No external crypto deps: pbkdf2 via hashlib, JWT via hmac/base64 (HS256
only, constant-time compare), which matches the reference's security
properties with fewer moving parts.
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import json
import os
import secrets
import time
from typing import Any, Dict, List, Optional  # noqa: F401

PBKDF2_ITERS = int(os.environ.get("KAKVEDA_PBKDF2_ITERS", "260000"))
JWT_TTL_MIN = 720

ROLES = ("admin", "operator", "viewer")


# -- passwords ---------------------------------------------------------------

def hash_password(password: str) -> str:
    salt = secrets.token_bytes(16)
    dk = hashlib.pbkdf2_hmac("sha256", password.encode(), salt, PBKDF2_ITERS)
    return f"pbkdf2_sha256${PBKDF2_ITERS}${salt.hex()}${dk.hex()}"


def verify_password(password: str, stored: str) -> bool:
    try:
        algo, iters, salt_hex, dk_hex = stored.split("$")
        if algo != "pbkdf2_sha256":
            return False
        dk = hashlib.pbkdf2_hmac(
            "sha256", password.encode(), bytes.fromhex(salt_hex), int(iters)
        )
        return hmac.compare_digest(dk.hex(), dk_hex)
    except Exception:
        return False


# -- JWT (HS256) -------------------------------------------------------------

def _b64url(data: bytes) -> str:
    return base64.urlsafe_b64encode(data).rstrip(b"=").decode()


def _b64url_dec(s: str) -> bytes:
    return base64.urlsafe_b64decode(s + "=" * (-len(s) % 4))


def make_jwt(
    sub: str,
    roles: List[str],
    secret: str,
    issuer: str = "kakveda",
    ttl_min: int = JWT_TTL_MIN,
) -> str:
    now = int(time.time())
    header = {"alg": "HS256", "typ": "JWT"}
    payload = {
        "iss": issuer,
        "sub": sub,
        "roles": roles,
        "iat": now,
        "jti": secrets.token_hex(8),
        "exp": now + ttl_min * 60,
    }
    signing = _b64url(json.dumps(header, separators=(",", ":")).encode()) + "." + _b64url(
        json.dumps(payload, separators=(",", ":")).encode()
    )
    sig = hmac.new(secret.encode(), signing.encode(), hashlib.sha256).digest()
    return signing + "." + _b64url(sig)


def decode_jwt(token: str, secret: str) -> Optional[Dict[str, Any]]:
    """Return the payload, or None on any failure (bad sig, expired)."""
    try:
        h, p, s = token.split(".")
        sig = hmac.new(secret.encode(), f"{h}.{p}".encode(), hashlib.sha256).digest()
        if not hmac.compare_digest(_b64url(sig), s):
            return None
        payload = json.loads(_b64url_dec(p))
        if int(payload.get("exp", 0)) < time.time():
            return None
        return payload
    except Exception:
        return None


def new_reset_token() -> str:
    return secrets.token_urlsafe(32)


# -- RBAC --------------------------------------------------------------------

def has_role(payload: Optional[Dict[str, Any]], role: str) -> bool:
    return bool(payload) and role in (payload.get("roles") or [])


def require_any(payload: Optional[Dict[str, Any]], roles: List[str]) -> bool:
    return bool(payload) and any(r in (payload.get("roles") or []) for r in roles)


# -- revocation + rate limiting (Redis-backed with in-memory fallback;
#    parity with the reference's shared/redis_helpers.py:26-84:
#    SET-with-TTL revocation, fixed-window INCR+EXPIRE rate limiting) ------


def _redis_client(redis_url: Optional[str], injected: Any = None) -> Any:
    """A live Redis client, or None (missing package / unreachable server
    -> silent in-memory fallback, like the reference)."""
    if injected is not None:
        return injected
    if not redis_url:
        return None
    try:
        import redis  # optional dependency

        client = redis.Redis.from_url(redis_url)
        client.ping()
        return client
    except Exception:
        return None


class RevocationStore:
    """Revoked JWT jti set. Redis: SET key with TTL (shared across
    replicas); fallback: per-process dict with expiry."""

    def __init__(self, redis_url: Optional[str] = None, prefix: str = "kv:revoked:",
                 client: Any = None):
        self.prefix = prefix
        self._redis = _redis_client(redis_url, client)
        self._revoked: Dict[str, float] = {}

    @property
    def backend(self) -> str:
        return "redis" if self._redis is not None else "memory"

    def revoke(self, jti: str, ttl_sec: float = JWT_TTL_MIN * 60):
        if self._redis is not None:
            try:
                self._redis.set(self.prefix + jti, "1", ex=max(1, int(ttl_sec)))
                return
            except Exception:
                self._redis = None  # degrade to memory mid-flight
        self._revoked[jti] = time.time() + ttl_sec

    def is_revoked(self, jti: str) -> bool:
        if self._redis is not None:
            try:
                return self._redis.get(self.prefix + jti) is not None
            except Exception:
                self._redis = None
        exp = self._revoked.get(jti)
        if exp is None:
            return False
        if exp < time.time():
            del self._revoked[jti]
            return False
        return True


class RateLimiter:
    """Fixed-window counter per key. Redis: INCR + EXPIRE on a
    per-window key (shared across replicas); fallback: per-process."""

    def __init__(self, limit: int = 30, window_sec: float = 60.0,
                 redis_url: Optional[str] = None, prefix: str = "kv:rl:",
                 client: Any = None):
        self.limit = limit
        self.window = window_sec
        self.prefix = prefix
        self._redis = _redis_client(redis_url, client)
        self._counts: Dict[str, tuple[int, float]] = {}

    @property
    def backend(self) -> str:
        return "redis" if self._redis is not None else "memory"

    def allow(self, key: str) -> bool:
        now = time.time()
        if self._redis is not None:
            try:
                wkey = f"{self.prefix}{key}:{int(now // self.window)}"
                n = int(self._redis.incr(wkey))
                if n == 1:
                    self._redis.expire(wkey, int(self.window) + 1)
                return n <= self.limit
            except Exception:
                self._redis = None
        count, start = self._counts.get(key, (0, now))
        if now - start > self.window:
            count, start = 0, now
        count += 1
        self._counts[key] = (count, start)
        return count <= self.limit
