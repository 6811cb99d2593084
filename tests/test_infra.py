"""Infra parity (VERDICT round 1 missing #2/#3/#4): Redis-backed
revocation/rate-limiting, DATABASE_URL engine selection, idempotent
startup migrations."""

import time


class FakeRedis:
    """Minimal Redis semantics (SET-with-TTL, GET, INCR, EXPIRE, PING) —
    lets the Redis code path run without a server/package."""

    def __init__(self):
        self.kv = {}  # key -> (value, expires_at|None)

    def _live(self, key):
        v = self.kv.get(key)
        if v is None:
            return None
        if v[1] is not None and v[1] < time.time():
            del self.kv[key]
            return None
        return v

    def ping(self):
        return True

    def set(self, key, value, ex=None):
        self.kv[key] = (value, time.time() + ex if ex else None)

    def get(self, key):
        v = self._live(key)
        return None if v is None else v[0]

    def incr(self, key):
        v = self._live(key)
        n = (int(v[0]) if v else 0) + 1
        self.kv[key] = (n, v[1] if v else None)
        return n

    def expire(self, key, ttl):
        v = self._live(key)
        if v:
            self.kv[key] = (v[0], time.time() + ttl)


def test_revocation_store_redis_backend():
    from kakveda_amd.services.dashboard.auth import RevocationStore

    r = FakeRedis()
    store = RevocationStore(client=r)
    assert store.backend == "redis"
    assert not store.is_revoked("jti-1")
    store.revoke("jti-1", ttl_sec=60)
    assert store.is_revoked("jti-1")
    assert r.get("kv:revoked:jti-1") is not None  # really went to redis
    # TTL expiry
    store.revoke("jti-2", ttl_sec=1)
    r.kv["kv:revoked:jti-2"] = (r.kv["kv:revoked:jti-2"][0], time.time() - 1)
    assert not store.is_revoked("jti-2")


def test_revocation_store_degrades_to_memory():
    from kakveda_amd.services.dashboard.auth import RevocationStore

    class DyingRedis(FakeRedis):
        def set(self, *a, **k):
            raise ConnectionError("gone")

    store = RevocationStore(client=DyingRedis())
    store.revoke("jti-x")  # redis dies mid-flight -> memory fallback
    assert store.backend == "memory"
    assert store.is_revoked("jti-x")


def test_rate_limiter_redis_fixed_window():
    from kakveda_amd.services.dashboard.auth import RateLimiter

    r = FakeRedis()
    rl = RateLimiter(limit=3, window_sec=60, client=r)
    assert rl.backend == "redis"
    assert [rl.allow("ip1") for _ in range(5)] == [True, True, True, False, False]
    assert rl.allow("ip2")  # separate key untouched
    # the window key carries a TTL (reference INCR+EXPIRE pattern)
    wkey = [k for k in r.kv if k.startswith("kv:rl:ip1")][0]
    assert r.kv[wkey][1] is not None


def test_unreachable_redis_url_falls_back():
    from kakveda_amd.services.dashboard.auth import RateLimiter, RevocationStore

    store = RevocationStore(redis_url="redis://127.0.0.1:1/0")
    rl = RateLimiter(redis_url="redis://127.0.0.1:1/0")
    assert store.backend == "memory" and rl.backend == "memory"
    store.revoke("a")
    assert store.is_revoked("a") and rl.allow("k")


def test_database_url_env_selects_engine(tmp_path, monkeypatch):
    from kakveda_amd.services.dashboard import db as dbm

    target = tmp_path / "via_url.db"
    monkeypatch.setenv("DATABASE_URL", f"sqlite:///{target}")
    Session = dbm.init_db(str(tmp_path / "ignored.db"))
    with Session() as s:
        s.add(dbm.Role(name="probe"))
        s.commit()
    assert target.exists()  # wrote through DATABASE_URL, not db_path
    assert not (tmp_path / "ignored.db").exists()


def test_migrations_add_missing_columns(tmp_path):
    """A database created by an older schema gains the model's new
    columns on startup (reference db.py:368-644 behaviour)."""
    from sqlalchemy import text

    from kakveda_amd.services.dashboard import db as dbm

    db_path = tmp_path / "old.db"
    eng = dbm.make_engine(str(db_path))
    with eng.begin() as conn:
        conn.execute(
            text(
                "CREATE TABLE warning_events ("
                "id INTEGER PRIMARY KEY, ts DATETIME, app_id VARCHAR(255), "
                "action VARCHAR(32), confidence FLOAT)"
            )
        )
        conn.execute(
            text(
                "INSERT INTO warning_events (app_id, action, confidence) "
                "VALUES ('app-A', 'warn', 0.9)"
            )
        )
    applied = dbm.migrate_db(eng)
    assert any("warning_events ADD COLUMN pattern_id" in d for d in applied)
    assert any("warning_events ADD COLUMN est_cost_usd_micro" in d for d in applied)
    # idempotent: second run applies nothing
    assert dbm.migrate_db(eng) == []
    # old row readable through the full model, new columns defaulted
    from sqlalchemy.orm import sessionmaker

    Session = sessionmaker(bind=eng)
    with Session() as s:
        w = s.query(dbm.WarningEvent).first()
        assert w.app_id == "app-A" and w.est_cost_usd_micro == 0
        # and inserts through the current model work
        s.add(dbm.WarningEvent(app_id="app-B", action="block", confidence=1.0))
        s.commit()


def test_migrations_create_missing_tables(tmp_path):
    from sqlalchemy import inspect

    from kakveda_amd.services.dashboard import db as dbm

    eng = dbm.make_engine(str(tmp_path / "fresh.db"))
    dbm.migrate_db(eng)
    assert "users" in inspect(eng).get_table_names()
