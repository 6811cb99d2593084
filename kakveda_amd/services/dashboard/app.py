"""Dashboard FastAPI app: middleware, auth, pipeline, warnings, runs.

Route parity with reference services/dashboard/app.py.
API-first: every page has a JSON ``/api`` counterpart; HTML is rendered
from the in-package Jinja environment (templates.py).
"""

from __future__ import annotations

import datetime as dt
import json
import os
import time
import uuid
from typing import Any, Dict, List, Optional

from fastapi import FastAPI, Request
from fastapi.responses import HTMLResponse, JSONResponse, RedirectResponse, Response

from kakveda_amd.core.schemas import utcnow
from kakveda_amd.services import (
    TOPIC_CHILD_SAFETY,
    TOPIC_FAILURE_DETECTED,
    TOPIC_TRACE_INGESTED,
)
from kakveda_amd.services.dashboard import db as dbm
from kakveda_amd.services.dashboard.auth import (
    ROLES,
    has_role,
    hash_password,
    make_jwt,
    new_reset_token,
    require_any,
    verify_password,
)
from kakveda_amd.services.dashboard.context import (
    COOKIE_NAME,
    DEMO_USERS,
    IMPERSONATE_COOKIE,
    DashboardContext,
    estimate_cost_usd_micro,
    estimate_tokens,
    read_payload,
    sha256_hex,
)
from kakveda_amd.services.dashboard.templates import render
from kakveda_amd.services.wiring import Transport


def audit(ctx: DashboardContext, actor: str, action: str, target: str = "", **meta):
    with ctx.Session() as s:
        s.add(
            dbm.AuditEvent(
                actor=actor, action=action, target=target, meta_json=json.dumps(meta)
            )
        )
        s.commit()


def bootstrap(ctx: DashboardContext) -> None:
    """Idempotent role + demo-user bootstrap with self-repair
    (reference app.py:1271-1329)."""
    with ctx.Session() as s:
        roles = {r.name: r for r in s.query(dbm.Role).all()}
        for name in ROLES:
            if name not in roles:
                role = dbm.Role(name=name)
                s.add(role)
                s.flush()
                roles[name] = role
        for email, password, display, user_roles in DEMO_USERS:
            user = s.query(dbm.User).filter_by(email=email).first()
            if user is None:
                user = dbm.User(
                    email=email,
                    password_hash=hash_password(password),
                    display_name=display,
                )
                s.add(user)
                s.flush()
            elif not verify_password(password, user.password_hash) and os.environ.get(
                "DASHBOARD_BOOTSTRAP_FORCE_PASSWORDS", "0"
            ) == "1":
                # password self-repair is opt-in (matching the reference's
                # DASHBOARD_BOOTSTRAP_FORCE_PASSWORDS gate) so a rotated
                # admin password is never silently reverted on restart
                user.password_hash = hash_password(password)
            have = {
                ur.role_id
                for ur in s.query(dbm.UserRole).filter_by(user_id=user.id).all()
            }
            for rname in user_roles:
                if roles[rname].id not in have:
                    s.add(dbm.UserRole(user_id=user.id, role_id=roles[rname].id))
        s.commit()


def user_roles(ctx: DashboardContext, user_id: int) -> List[str]:
    with ctx.Session() as s:
        rows = (
            s.query(dbm.Role.name)
            .join(dbm.UserRole, dbm.UserRole.role_id == dbm.Role.id)
            .filter(dbm.UserRole.user_id == user_id)
            .all()
        )
        return [r[0] for r in rows]


# ---------------------------------------------------------------------------


def create_app(
    data_dir: str = "/app/data",
    db_path: Optional[str] = None,
    transport: Optional[Transport] = None,
    urls: Optional[Dict[str, str]] = None,
    self_url: Optional[str] = None,
) -> FastAPI:
    import os

    from kakveda_amd.services import DEFAULT_PORTS

    app = FastAPI(title="Kakveda-AMD Dashboard")
    u = urls or {
        name: os.environ.get(
            f"{name.upper()}_URL", f"http://{name.replace('_', '-')}:{port}"
        )
        for name, port in DEFAULT_PORTS.items()
    }
    ctx = DashboardContext(
        Session=dbm.init_db(db_path or os.path.join(data_dir, "dashboard.db")),
        tx=transport or Transport(),
        urls=u,
        data_dir=data_dir,
    )
    app.state.ctx = ctx
    me = self_url or u.get("dashboard", "http://dashboard:8110")
    bootstrap(ctx)

    # -- middleware: request id, duration log, security headers, CSRF cookie --

    # JSON API routes require a logged-in user (reference guards these with
    # Depends(require_login)); the exceptions carry their own guard:
    # agent self-registration/heartbeat and external ingest are project
    # API-key-authenticated, /api/me returns its own 401 contract, and
    # /events/* are internal event-bus callbacks (see SECURITY.md).
    PUBLIC_API_PATHS = {
        "/api/me",
        "/api/agents/register",
        "/api/agents/heartbeat",
        "/api/ingest/run",
    }

    @app.middleware("http")
    async def _middleware(request: Request, call_next):
        rid = request.headers.get("X-Request-Id") or uuid.uuid4().hex
        t0 = time.perf_counter()
        path = request.url.path
        if (
            path.startswith("/api/") or path == "/eval/run"
        ) and path not in PUBLIC_API_PATHS and not ctx.current_user(request):
            response: Response = JSONResponse(
                {"ok": False, "error": "auth_required"}, status_code=401
            )
        else:
            response = await call_next(request)
        dur = (time.perf_counter() - t0) * 1000.0
        response.headers["X-Request-Id"] = rid
        response.headers["X-Content-Type-Options"] = "nosniff"
        response.headers["X-Frame-Options"] = "DENY"
        response.headers["Referrer-Policy"] = "same-origin"
        response.headers["Content-Security-Policy"] = (
            "default-src 'self'; style-src 'self' 'unsafe-inline'"
        )
        response.headers["Strict-Transport-Security"] = "max-age=63072000"
        if "kv_csrf" not in request.cookies:
            # cookie set, enforcement deliberately matching the reference
            # (dashboard/app.py:655-663) — documented divergence candidate
            response.set_cookie("kv_csrf", uuid.uuid4().hex, samesite="lax")
        ctx.log.info(
            "request",
            extra={
                "request_id": rid,
                "path": request.url.path,
                "method": request.method,
                "status": response.status_code,
                "duration_ms": round(dur, 2),
            },
        )
        return response

    # -- static assets (theme css, logo, progressive-enhancement js) ---------

    from kakveda_amd.services.dashboard.static import APP_JS, LOGO_SVG, STYLE_CSS

    @app.get("/static/style.css")
    async def style_css():
        return Response(STYLE_CSS, media_type="text/css",
                        headers={"Cache-Control": "public, max-age=3600"})

    @app.get("/static/logo.svg")
    async def logo_svg():
        return Response(LOGO_SVG, media_type="image/svg+xml",
                        headers={"Cache-Control": "public, max-age=3600"})

    @app.get("/static/app.js")
    async def app_js():
        return Response(APP_JS, media_type="application/javascript",
                        headers={"Cache-Control": "public, max-age=3600"})

    # -- auth ----------------------------------------------------------------

    @app.get("/login", response_class=HTMLResponse)
    async def login_page():
        return render("login.html", {})

    @app.get("/register", response_class=HTMLResponse)
    async def register_page():
        return render("register.html", {})

    @app.get("/forgot", response_class=HTMLResponse)
    async def forgot_page():
        return render("forgot.html", {})

    @app.get("/reset", response_class=HTMLResponse)
    async def reset_page(token: str = ""):
        return render("reset.html", {"token": token})

    @app.post("/login")
    async def login(request: Request):
        if not ctx.auth_limiter.allow(request.client.host if request.client else "x"):
            return JSONResponse({"ok": False, "error": "rate_limited"}, status_code=429)
        body = await read_payload(request)
        email = str(body.get("email", "")).strip().lower()
        password = str(body.get("password", ""))
        with ctx.Session() as s:
            user = s.query(dbm.User).filter_by(email=email, is_active=True).first()
        if not user or not verify_password(password, user.password_hash):
            audit(ctx, email, "login.failed")
            return JSONResponse({"ok": False, "error": "invalid_credentials"}, status_code=401)
        token = make_jwt(email, user_roles(ctx, user.id), ctx.jwt_secret)
        audit(ctx, email, "login.ok")
        accept = request.headers.get("accept", "")
        resp: Response
        if "text/html" in accept:
            resp = RedirectResponse("/", status_code=303)
        else:
            resp = JSONResponse({"ok": True, "token": token})
        resp.set_cookie(COOKIE_NAME, token, httponly=True, samesite="lax")
        return resp

    @app.get("/logout")
    async def logout(request: Request):
        payload = ctx.current_user(request)
        if payload:
            ctx.revocation.revoke(payload.get("jti", ""))
            audit(ctx, payload.get("sub", ""), "logout")
        resp = RedirectResponse("/login", status_code=303)
        resp.delete_cookie(COOKIE_NAME)
        resp.delete_cookie(IMPERSONATE_COOKIE)
        return resp

    @app.post("/register")
    async def register(request: Request):
        if not ctx.auth_limiter.allow("register:" + (request.client.host if request.client else "x")):
            return JSONResponse({"ok": False, "error": "rate_limited"}, status_code=429)
        body = await read_payload(request)
        email = str(body.get("email", "")).strip().lower()
        password = str(body.get("password", ""))
        if "@" not in email or len(password) < 8:
            return JSONResponse({"ok": False, "error": "invalid_input"}, status_code=400)
        with ctx.Session() as s:
            if s.query(dbm.User).filter_by(email=email).first():
                return JSONResponse({"ok": False, "error": "exists"}, status_code=409)
            user = dbm.User(email=email, password_hash=hash_password(password))
            s.add(user)
            s.flush()
            viewer = s.query(dbm.Role).filter_by(name="viewer").first()
            s.add(dbm.UserRole(user_id=user.id, role_id=viewer.id))
            s.commit()
        audit(ctx, email, "register")
        return {"ok": True}

    @app.post("/forgot")
    async def forgot(request: Request):
        body = await read_payload(request)
        email = str(body.get("email", "")).strip().lower()
        with ctx.Session() as s:
            user = s.query(dbm.User).filter_by(email=email).first()
            if user:
                token = new_reset_token()
                s.add(
                    dbm.PasswordResetToken(
                        user_id=user.id,
                        token=token,
                        expires_at=dt.datetime.utcnow() + dt.timedelta(hours=2),
                    )
                )
                s.commit()
                audit(ctx, email, "password.reset_requested")
                # dev-mode: token returned (prod would email it)
                return {"ok": True, "reset_token": token}
        return {"ok": True}

    @app.post("/reset")
    async def reset(request: Request):
        body = await read_payload(request)
        token = str(body.get("token", ""))
        password = str(body.get("password", ""))
        if len(password) < 8:
            return JSONResponse({"ok": False, "error": "weak_password"}, status_code=400)
        with ctx.Session() as s:
            row = s.query(dbm.PasswordResetToken).filter_by(token=token, used=False).first()
            if not row or row.expires_at < dt.datetime.utcnow():
                return JSONResponse({"ok": False, "error": "invalid_token"}, status_code=400)
            user = s.query(dbm.User).get(row.user_id)
            user.password_hash = hash_password(password)
            row.used = True
            s.commit()
        return {"ok": True}

    @app.get("/api/me")
    async def me_route(request: Request):
        payload = ctx.current_user(request)
        if not payload:
            return JSONResponse({"ok": False}, status_code=401)
        return {"ok": True, "user": payload.get("sub"), "roles": payload.get("roles")}

    @app.post("/admin/impersonate")
    async def impersonate(request: Request):
        payload = ctx.current_user(request)
        if not payload or "admin" not in (payload.get("roles") or []):
            return JSONResponse({"ok": False}, status_code=403)
        body = await read_payload(request)
        role = str(body.get("role", ""))
        resp = JSONResponse({"ok": True, "view_as": role or None})
        if role in ROLES:
            resp.set_cookie(IMPERSONATE_COOKIE, role, samesite="lax")
        else:
            resp.delete_cookie(IMPERSONATE_COOKIE)
        return resp

    # -- probes ----------------------------------------------------------------

    @app.get("/healthz")
    async def healthz():
        return {"ok": True}

    @app.get("/readyz")
    async def readyz():
        try:
            with ctx.Session() as s:
                s.query(dbm.User).first()
            return {"ok": True}
        except Exception as exc:
            return JSONResponse({"ok": False, "error": str(exc)}, status_code=503)

    # -- event-bus callbacks ---------------------------------------------------

    async def subscribe() -> None:
        for topic, path in (
            (TOPIC_TRACE_INGESTED, "/events/trace"),
            (TOPIC_CHILD_SAFETY, "/events/child-safety"),
        ):
            await ctx.tx.post(
                f"{u['event_bus']}/subscribe",
                json={"topic": topic, "callback_url": f"{me}{path}"},
            )

    app.state.subscribe = subscribe

    @app.on_event("startup")
    async def _startup():
        # bus may come up after us under `kakveda up` (all services start
        # concurrently): retry in the background; /subscribe dedups
        import asyncio as _aio

        async def _retry():
            for _ in range(30):
                try:
                    await subscribe()
                    return
                except Exception:
                    await _aio.sleep(1.0)

        _aio.get_event_loop().create_task(_retry())

    @app.post("/events/trace")
    async def on_trace(event: dict):
        with ctx.Session() as s:
            if s.query(dbm.TraceRun).filter_by(trace_id=event.get("trace_id", "")).first():
                return {"ok": True, "dedup": True}
            prompt = event.get("prompt", "")
            response = event.get("response", "")
            ti, to = estimate_tokens(prompt), estimate_tokens(response)
            s.add(
                dbm.TraceRun(
                    trace_id=event.get("trace_id", uuid.uuid4().hex),
                    app_id=event.get("app_id", "unknown"),
                    model=event.get("model") or "",
                    provider="event",
                    prompt=prompt,
                    response=response,
                    tokens_in=ti,
                    tokens_out=to,
                    cost_usd_micro=estimate_cost_usd_micro(ti, to),
                    source="event",
                )
            )
            s.commit()
        return {"ok": True}

    @app.post("/events/child-safety")
    async def on_child_safety(event: dict):
        with ctx.Session() as s:
            s.add(
                dbm.WarningEvent(
                    app_id=event.get("app_id", "unknown"),
                    action="block",
                    confidence=1.0,
                    message=f"child_safety_alert: {event.get('reason', '')}",
                )
            )
            s.commit()
        return {"ok": True}

    @app.post("/health/test")
    async def health_test(request: Request):
        """Synthetic failure injector (reference app.py:1762-1819)."""
        body = await read_payload(request)
        payload = {
            "trace_id": f"test-{uuid.uuid4().hex[:8]}",
            "ts": utcnow().isoformat(),
            "app_id": str(body.get("app_id", "app-A")),
            "failure_type": str(body.get("failure_type", "HALLUCINATION_CITATION")),
            "severity": str(body.get("severity", "medium")),
            "context_signature": {"injected": True},
        }
        await ctx.tx.post(
            f"{u['event_bus']}/publish",
            json={"topic": TOPIC_FAILURE_DETECTED, "payload": payload},
        )
        return {"ok": True, "injected": payload}

    # -- scenario runner (the primary end-to-end path) ---------

    @app.post("/scenarios/run")
    async def run_scenario(request: Request):
        payload = ctx.current_user(request)
        body = await read_payload(request)
        app_id = str(body.get("app_id", "app-A"))
        prompt = str(body.get("prompt", ""))
        if not prompt:
            return JSONResponse({"ok": False, "error": "prompt required"}, status_code=400)

        t_start = utcnow()
        spans: List[Dict[str, Any]] = []

        async def timed(name: str, coro):
            t0 = time.perf_counter()
            result = await coro
            spans.append(
                {"name": name, "duration_ms": (time.perf_counter() - t0) * 1000.0}
            )
            return result

        warn_resp = await timed(
            "warn_policy.call",
            ctx.tx.post(
                f"{u['warning_policy']}/warn",
                json={"app_id": app_id, "prompt": prompt, "tools": [], "env": {"scenario": True}},
                timeout=5.0,
            ),
        )
        warn = warn_resp.json()

        gen = await timed("model.generate", ctx.generate(prompt))
        trace_id = f"scn-{uuid.uuid4().hex[:12]}"
        trace = {
            "trace_id": trace_id,
            "ts": utcnow().isoformat(),
            "app_id": app_id,
            "prompt": prompt,
            "response": gen["text"],
            "model": gen["model"],
            "tools": [],
            "env": {"scenario": True},
        }
        await timed(
            "ingestion.ingest",
            ctx.tx.post(f"{u['ingestion']}/ingest", json={"trace": trace}, timeout=5.0),
        )

        ti, to = estimate_tokens(prompt), estimate_tokens(gen["text"])
        with ctx.Session() as s:
            # the ingest fan-out may already have persisted this trace via
            # the /events/trace callback: enrich that row instead
            run = s.query(dbm.TraceRun).filter_by(trace_id=trace_id).first()
            if run is None:
                run = dbm.TraceRun(trace_id=trace_id, app_id=app_id)
                s.add(run)
            run.provider = gen["provider"]
            run.model = gen["model"]
            run.prompt = prompt
            run.response = gen["text"]
            run.latency_ms = gen["latency_ms"]
            run.tokens_in = ti
            run.tokens_out = to
            run.cost_usd_micro = estimate_cost_usd_micro(ti, to)
            run.source = "scenario"
            s.flush()
            parent = dbm.TraceSpan(
                trace_run_id=run.id,
                name="scenario.run",
                start=t_start,
                end=utcnow(),
                duration_ms=sum(sp["duration_ms"] for sp in spans),
            )
            s.add(parent)
            s.flush()
            for sp in spans:
                s.add(
                    dbm.TraceSpan(
                        trace_run_id=run.id,
                        parent_id=parent.id,
                        name=sp["name"],
                        duration_ms=sp["duration_ms"],
                    )
                )
            s.add(
                dbm.ScenarioRun(
                    app_id=app_id,
                    prompt=prompt,
                    response=gen["text"],
                    warn_action=warn.get("action", ""),
                    warn_confidence=float(warn.get("confidence", 0.0)),
                    trace_id=trace_id,
                )
            )
            best = (warn.get("references") or [{}])[0]
            we = dbm.WarningEvent(
                app_id=app_id,
                action=warn.get("action", ""),
                confidence=float(warn.get("confidence", 0.0)),
                pattern_id=warn.get("pattern_id") or "",
                failure_id=best.get("failure_id", ""),
                message=warn.get("message", ""),
                prompt=prompt,
                est_cost_usd_micro=estimate_cost_usd_micro(ti, to),
            )
            s.add(we)
            s.commit()
            warning_id = we.id
        audit(ctx, (payload or {}).get("sub", "anonymous"), "scenario.run", app_id)
        if "text/html" in request.headers.get("accept", ""):
            return RedirectResponse(f"/warnings#w-{warning_id}", status_code=303)
        return {
            "ok": True,
            "trace_id": trace_id,
            "warning": warn,
            "response": gen["text"],
            "warning_event_id": warning_id,
        }

    # -- warnings + analytics --------------------------------------------------

    @app.get("/api/warnings")
    async def api_warnings(days: int = 90, limit: int = 500):
        cutoff = dt.datetime.utcnow() - dt.timedelta(days=days)
        with ctx.Session() as s:
            rows = (
                s.query(dbm.WarningEvent)
                .filter(dbm.WarningEvent.ts >= cutoff)
                .order_by(dbm.WarningEvent.ts.desc())
                .limit(limit)
                .all()
            )
            return {
                "warnings": [
                    {
                        "id": w.id,
                        "ts": w.ts.isoformat(),
                        "app_id": w.app_id,
                        "action": w.action,
                        "confidence": w.confidence,
                        "pattern_id": w.pattern_id,
                        "failure_id": w.failure_id,
                        "message": w.message,
                    }
                    for w in rows
                ]
            }

    @app.get("/api/warnings/analytics")
    async def warnings_analytics(days: int = 30):
        """Daily counts, per-app, per-pattern, cost impact
        (reference app.py:1926-2027)."""
        cutoff = dt.datetime.utcnow() - dt.timedelta(days=days)
        with ctx.Session() as s:
            rows = s.query(dbm.WarningEvent).filter(dbm.WarningEvent.ts >= cutoff).all()
        daily: Dict[str, int] = {}
        per_app: Dict[str, int] = {}
        per_pattern: Dict[str, int] = {}
        cost = 0
        for w in rows:
            day = w.ts.date().isoformat()
            daily[day] = daily.get(day, 0) + 1
            per_app[w.app_id] = per_app.get(w.app_id, 0) + 1
            if w.pattern_id:
                per_pattern[w.pattern_id] = per_pattern.get(w.pattern_id, 0) + 1
            cost += w.est_cost_usd_micro
        return {
            "days": days,
            "total": len(rows),
            "daily": dict(sorted(daily.items())),
            "per_app": per_app,
            "per_pattern": per_pattern,
            "est_cost_impact_usd_micro": cost,
        }

    # -- home + html pages -----------------------------------------------------

    @app.get("/", response_class=HTMLResponse)
    async def home(request: Request):
        payload = ctx.current_user(request)
        if not payload:
            return RedirectResponse("/login", status_code=303)
        failures: List[Dict] = []
        patterns: List[Dict] = []
        try:
            fr = await ctx.tx.get(f"{u['gfkb']}/failures")
            failures = fr.json().get("failures", [])[-10:]
            pr = await ctx.tx.get(f"{u['gfkb']}/patterns")
            patterns = pr.json().get("patterns", [])
        except Exception:
            pass
        with ctx.Session() as s:
            warnings = (
                s.query(dbm.WarningEvent).order_by(dbm.WarningEvent.ts.desc()).limit(10).all()
            )
            runs = s.query(dbm.TraceRun).order_by(dbm.TraceRun.ts.desc()).limit(10).all()
        return render(
            "home.html",
            {
                "user": payload.get("sub"),
                "roles": payload.get("roles"),
                "failures": failures,
                "patterns": patterns,
                "warnings": warnings,
                "runs": runs,
            },
        )

    @app.get("/health", response_class=HTMLResponse)
    async def health_page(request: Request):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        apps: Dict[str, List[Dict]] = {}
        with ctx.Session() as s:
            app_ids = [r[0] for r in s.query(dbm.TraceRun.app_id).distinct().limit(20)]
        for app_id in app_ids:
            try:
                resp = await ctx.tx.get(f"{u['health_scoring']}/health/{app_id}", params={"limit": 5})
                pts = resp.json().get("points", [])
                if pts:
                    apps[app_id] = pts
            except Exception:
                continue
        return render("health.html", {"apps": apps})

    @app.get("/failure/{fid}", response_class=HTMLResponse)
    async def failure_detail(request: Request, fid: str):
        """Failure detail with version addressing: /failure/F-0001 shows the
        latest version, /failure/F-0001v3 a specific one (reference
        dashboard/app.py:1822-1909)."""
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        base_id, _, ver = fid.partition("v")
        try:
            resp = await ctx.tx.get(f"{u['gfkb']}/failures")
            records = [
                r for r in resp.json().get("failures", [])
                if r.get("failure_id") == base_id
            ]
        except Exception:
            records = []
        if not records:
            return HTMLResponse("failure not found", status_code=404)
        if ver:
            chosen = [r for r in records if str(r.get("version")) == ver]
            record = chosen[0] if chosen else records[-1]
        else:
            record = records[-1]
        return render(
            "failure_detail.html",
            {"record": record, "versions": records, "fid": base_id},
        )

    @app.get("/warnings", response_class=HTMLResponse)
    async def warnings_page(request: Request):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        analytics = await warnings_analytics()
        listing = await api_warnings()
        return render(
            "warnings.html", {"analytics": analytics, "warnings": listing["warnings"]}
        )

    from kakveda_amd.services.dashboard.features import register_features

    register_features(app, ctx, u)
    return app
