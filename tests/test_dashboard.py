"""Dashboard service tests: auth/RBAC, scenario pipeline, runs, playground,
datasets/evals, prompts/experiments, agents, projects/keys, admin."""

import httpx
import pytest

from kakveda_amd.services.cluster import LocalCluster

PROMPT = "Summarize this and include references even if none are provided."


async def _cluster(tmp_path):
    cluster = LocalCluster(data_dir=str(tmp_path), device="cpu", with_dashboard=True)
    await cluster.start()
    return cluster


def _client(cluster):
    return httpx.AsyncClient(
        transport=httpx.ASGITransport(app=cluster.dashboard),
        base_url="http://dashboard:8110",
    )


async def _login(client, email="admin@kakveda.local", password="admin123"):
    resp = await client.post("/login", json={"email": email, "password": password})
    assert resp.status_code == 200, resp.text
    token = resp.json()["token"]
    client.cookies.set("kv_token", token)
    return token


async def test_auth_flow(tmp_path):
    cluster = await _cluster(tmp_path)
    async with _client(cluster) as client:
        # unauthenticated home redirects to login
        resp = await client.get("/", follow_redirects=False)
        assert resp.status_code == 303

        # bad credentials
        resp = await client.post("/login", json={"email": "admin@kakveda.local", "password": "nope"})
        assert resp.status_code == 401

        await _login(client)
        me = await client.get("/api/me")
        assert me.json()["roles"] == ["admin"]

        # register + login as the new user
        resp = await client.post(
            "/register", json={"email": "new@x.com", "password": "longenough1"}
        )
        assert resp.json()["ok"]

        # forgot/reset round trip
        resp = await client.post("/forgot", json={"email": "new@x.com"})
        token = resp.json()["reset_token"]
        resp = await client.post("/reset", json={"token": token, "password": "evenlonger2"})
        assert resp.json()["ok"]

        # security headers + request id present
        resp = await client.get("/api/me")
        assert "X-Request-Id" in resp.headers
        assert "Content-Security-Policy" in resp.headers
    await cluster.aclose()


async def test_scenario_pipeline_and_runs(tmp_path):
    cluster = await _cluster(tmp_path)
    async with _client(cluster) as client:
        await _login(client)
        r1 = await client.post("/scenarios/run", json={"app_id": "app-A", "prompt": PROMPT})
        assert r1.json()["ok"], r1.text
        # second run matches the first failure -> warn with references
        r2 = await client.post("/scenarios/run", json={"app_id": "app-B", "prompt": PROMPT})
        warn = r2.json()["warning"]
        assert warn["confidence"] >= 0.8
        assert warn["references"]

        # trace runs were persisted via the event-bus callback AND the runner
        runs = (await client.get("/api/runs")).json()["runs"]
        assert len(runs) >= 2
        scen = [r for r in runs if r["source"] == "scenario"]
        assert scen and scen[0]["provider"] == "stub"

        # query language
        stub_runs = (await client.get("/api/runs", params={"q": "provider:stub"})).json()["runs"]
        assert stub_runs
        none_runs = (await client.get("/api/runs", params={"q": "provider:doesnotexist"})).json()["runs"]
        assert none_runs == []

        # run detail with span waterfall
        detail = (await client.get(f"/api/runs/{scen[0]['id']}")).json()
        names = [sp["name"] for sp in detail["spans"]]
        assert "scenario.run" in names and "warn_policy.call" in names
        child = [sp for sp in detail["spans"] if sp["depth"] == 1]
        assert child and all(0 <= sp["pct_width"] <= 100 for sp in child)

        # feedback
        fb = await client.post(f"/api/runs/{scen[0]['id']}/feedback", json={"thumb": 1, "label": "good"})
        assert fb.json()["ok"]
        detail = (await client.get(f"/api/runs/{scen[0]['id']}")).json()
        assert detail["feedback"][0]["thumb"] == 1

        # warnings analytics
        analytics = (await client.get("/api/warnings/analytics")).json()
        assert analytics["total"] >= 2
        assert analytics["per_app"].get("app-A") == 1
    await cluster.aclose()


async def test_playground_and_experiments(tmp_path):
    cluster = await _cluster(tmp_path)
    async with _client(cluster) as client:
        await _login(client)
        resp = await client.post(
            "/api/playground/run",
            json={"prompt": "hello there", "experiment": "exp-1"},
        )
        body = resp.json()
        assert body["ok"] and body["tokens_in"] >= 1 and body["cost_usd_micro"] >= 0

        exps = (await client.get("/api/experiments")).json()["experiments"]
        exp = [e for e in exps if e["name"] == "exp-1"][0]
        assert exp["runs"] == 1 and "stub" in exp["providers"]
    await cluster.aclose()


async def test_datasets_and_eval(tmp_path):
    cluster = await _cluster(tmp_path)
    async with _client(cluster) as client:
        await _login(client)
        ds = (await client.post("/api/datasets", json={"name": "citations"})).json()
        resp = await client.post(
            f"/api/datasets/{ds['id']}/examples",
            json={
                "examples": [
                    {"input": PROMPT},
                    {"input": "What's 2+2?"},
                ]
            },
        )
        assert resp.json()["examples"] == 2

        preview = (await client.post(f"/api/datasets/{ds['id']}/run_example", json={})).json()
        assert preview["ok"] and preview["output"]

        ev = (await client.post("/eval/run", json={"dataset_id": ds["id"]})).json()
        assert ev["ok"]
        summary = ev["summary"]
        # deterministic stub ALWAYS emits citations: the citation-asking
        # example fails, the other passes -> pass rate 0.5
        assert summary["examples"] == 2
        assert abs(summary["pass_rate"] - 0.5) < 1e-6
        assert summary["p95_ms"] >= summary["p50_ms"] >= 0

        evals = (await client.get("/api/evals")).json()["evals"]
        assert evals and evals[0]["summary"]["examples"] == 2
    await cluster.aclose()


async def test_prompts_versioning(tmp_path):
    cluster = await _cluster(tmp_path)
    async with _client(cluster) as client:
        await _login(client)
        v1 = (await client.post("/api/prompts", json={"name": "p1", "content": "v one"})).json()
        v2 = (await client.post("/api/prompts", json={"name": "p1", "content": "v two"})).json()
        assert (v1["version"], v2["version"]) == (1, 2)
        versions = (await client.get(f"/api/prompts/{v1['prompt_id']}/versions")).json()["versions"]
        assert [v["version"] for v in versions] == [1, 2]
    await cluster.aclose()


async def test_agents_and_projects(tmp_path):
    cluster = await _cluster(tmp_path)
    # register the echo agent app on the in-process transport
    from kakveda_amd.services import agent_echo as echo_mod

    echo = echo_mod.create_app()
    cluster.tx.register_local("http://agent-echo:8120", echo)
    async with _client(cluster) as client:
        await _login(client)
        reg = (
            await client.post(
                "/admin/agents/register",
                json={"name": "agent-echo", "base_url": "http://agent-echo:8120"},
            )
        ).json()
        assert reg["ok"] and "echo" in reg["capabilities"]

        agents = (await client.get("/api/agents")).json()["agents"]
        assert agents[0]["name"] == "agent-echo"
        test = (await client.post(f"/api/agents/{agents[0]['id']}/test")).json()
        assert test["ok"]

        # project + API key + heartbeat + external ingest + budget
        proj = (await client.post("/api/projects", json={"name": "proj1"})).json()
        key = (await client.post(f"/api/projects/{proj['id']}/keys", json={})).json()
        assert key["api_key"].startswith("kv-")

        hb = await client.post(
            "/api/agents/heartbeat",
            json={"name": "agent-echo"},
            headers={"X-Api-Key": key["api_key"]},
        )
        assert hb.json()["ok"]

        ing = (
            await client.post(
                "/api/ingest/run",
                json={"prompt": "x" * 400, "response": "y" * 400},
                headers={"X-Api-Key": key["api_key"]},
            )
        ).json()
        assert ing["ok"] and ing["cost_usd_micro"] > 0

        # tight budget rejects the next ingest
        await client.post(f"/api/projects/{proj['id']}/budget", json={"monthly_usd_micro": 1})
        ing2 = (
            await client.post(
                "/api/ingest/run",
                json={"prompt": "x" * 4000, "response": "y" * 4000},
                headers={"X-Api-Key": key["api_key"]},
            )
        ).json()
        assert ing2["ok"] is False and ing2["error"] == "budget_exceeded"

        # bad key rejected
        bad = (
            await client.post("/api/ingest/run", json={"prompt": "p"}, headers={"X-Api-Key": "nope"})
        ).json()
        assert bad["ok"] is False
    await cluster.aclose()


async def test_rbac_and_admin(tmp_path):
    cluster = await _cluster(tmp_path)
    async with _client(cluster) as client:
        # viewer cannot create projects
        await _login(client, "viewer@kakveda.local", "viewer123")
        resp = await client.post("/api/projects", json={"name": "nope"})
        assert resp.status_code == 403

    async with _client(cluster) as client:
        await _login(client)
        # impersonate viewer -> project creation denied
        resp = await client.post("/admin/impersonate", json={"role": "viewer"})
        assert resp.json()["ok"]
        client.cookies.set("kv_view_as", "viewer")
        resp = await client.post("/api/projects", json={"name": "nope2"})
        assert resp.status_code == 403
        client.cookies.delete("kv_view_as")

        # seed pipeline data then purge it
        await client.post("/scenarios/run", json={"app_id": "app-A", "prompt": PROMPT})
        purge = (await client.post("/admin/purge_demo", json={"app_ids": ["app-A"]})).json()
        assert purge["ok"] and purge["removed"]["runs"] >= 1

        audit = (await client.get("/admin/audit")).json()["events"]
        actions = [e["action"] for e in audit]
        assert "admin.purge_demo" in actions and "login.ok" in actions
    await cluster.aclose()


async def test_health_test_injector(tmp_path):
    cluster = await _cluster(tmp_path)
    async with _client(cluster) as client:
        await _login(client)
        resp = (await client.post("/health/test", json={"app_id": "app-Z"})).json()
        assert resp["ok"]
        scorer = cluster.health_scoring.state.scorer
        points = scorer.timeline("app-Z", limit=5)
        assert points and points[-1]["score"] < 100.0
    await cluster.aclose()


async def test_html_pages(tmp_path):
    cluster = await _cluster(tmp_path)
    async with _client(cluster) as client:
        login_page = await client.get("/login")
        assert login_page.status_code == 200 and "Sign in" in login_page.text
        for public in ("/register", "/forgot", "/reset"):
            resp = await client.get(public)
            assert resp.status_code == 200, public
        await _login(client)
        home = await client.get("/")
        assert "Failure Intelligence" in home.text
        for page in ("/warnings", "/runs", "/playground", "/agents",
                     "/datasets", "/prompts", "/experiments", "/health",
                     "/scenarios", "/projects", "/evals", "/admin/users"):
            resp = await client.get(page)
            assert resp.status_code == 200, page
    await cluster.aclose()


async def test_failure_detail_version_addressing(tmp_path):
    cluster = await _cluster(tmp_path)
    async with _client(cluster) as client:
        await _login(client)
        # create two versions of the same failure identity
        await client.post("/scenarios/run", json={"app_id": "app-A", "prompt": PROMPT})
        await client.post("/scenarios/run", json={"app_id": "app-B", "prompt": PROMPT})
        latest = await client.get("/failure/F-0001")
        assert latest.status_code == 200
        assert "v2" in latest.text and "HALLUCINATION_CITATION" in latest.text
        v1 = await client.get("/failure/F-0001v1")
        assert "v1" in v1.text
        missing = await client.get("/failure/F-9999")
        assert missing.status_code == 404
    await cluster.aclose()


async def test_api_routes_require_auth(tmp_path):
    """JSON API routes reject anonymous callers with 401;
    API-key-guarded and contract-carrying endpoints stay reachable."""
    cluster = await _cluster(tmp_path)
    async with _client(cluster) as client:
        protected = [
            ("get", "/api/runs", None),
            ("get", "/api/datasets", None),
            ("post", "/api/datasets", {"name": "x"}),
            ("post", "/api/playground/run", {"prompt": "p"}),
            ("post", "/eval/run", {"dataset_id": 1}),
            ("post", "/api/prompts", {"name": "p", "content": "c"}),
            ("post", "/api/experiments", {"name": "e"}),
            ("get", "/api/agents", None),
            ("get", "/api/projects", None),
            ("get", "/api/warnings", None),
        ]
        for method, path, body in protected:
            resp = await getattr(client, method)(
                path, **({"json": body} if body is not None else {})
            )
            assert resp.status_code == 401, (path, resp.status_code)
            assert resp.json()["error"] == "auth_required", path
        # API-key-guarded endpoints answer with their own contract, not 401-auth
        resp = await client.post("/api/ingest/run", json={"prompt": "p"})
        assert resp.json()["error"] != "auth_required"
        resp = await client.post("/api/agents/heartbeat", json={})
        assert resp.json()["error"] == "invalid_api_key"
        # and a logged-in caller passes the guard
        await _login(client)
        resp = await client.get("/api/runs")
        assert resp.status_code == 200
    await cluster.aclose()


async def test_bootstrap_password_repair_gated(tmp_path, monkeypatch):
    """Rotating a demo password survives a restart unless the operator
    opts into DASHBOARD_BOOTSTRAP_FORCE_PASSWORDS=1."""
    from kakveda_amd.services.dashboard import db as dbm
    from kakveda_amd.services.dashboard.app import bootstrap
    from kakveda_amd.services.dashboard.auth import hash_password, verify_password

    cluster = await _cluster(tmp_path)
    ctx = cluster.dashboard.state.ctx
    with ctx.Session() as s:
        admin = s.query(dbm.User).filter_by(email="admin@kakveda.local").first()
        admin.password_hash = hash_password("rotated-password-1")
        s.commit()

    monkeypatch.delenv("DASHBOARD_BOOTSTRAP_FORCE_PASSWORDS", raising=False)
    bootstrap(ctx)  # simulated restart: must NOT revert
    with ctx.Session() as s:
        admin = s.query(dbm.User).filter_by(email="admin@kakveda.local").first()
        assert verify_password("rotated-password-1", admin.password_hash)

    monkeypatch.setenv("DASHBOARD_BOOTSTRAP_FORCE_PASSWORDS", "1")
    bootstrap(ctx)  # explicit opt-in: reverts to the documented default
    with ctx.Session() as s:
        admin = s.query(dbm.User).filter_by(email="admin@kakveda.local").first()
        assert verify_password("admin123", admin.password_hash)
    await cluster.aclose()


async def test_theme_assets_served(tmp_path):
    """The theme ships as real static assets (stylesheet, logo, JS) and
    pages link them (round-2 presentation pass)."""
    cluster = await _cluster(tmp_path)
    async with _client(cluster) as client:
        css = await client.get("/static/style.css")
        assert css.status_code == 200
        assert "text/css" in css.headers["content-type"]
        assert ".topbar" in css.text and ".tile" in css.text and ".chart" in css.text
        logo = await client.get("/static/logo.svg")
        assert logo.status_code == 200 and "svg" in logo.headers["content-type"]
        js = await client.get("/static/app.js")
        assert js.status_code == 200 and "pg-form" in js.text
        login = await client.get("/login")
        assert '/static/style.css' in login.text and 'class="brand"' in login.text
        await _login(client)
        home = await client.get("/")
        assert 'class="tiles"' in home.text and 'class="topbar"' in home.text
    await cluster.aclose()
