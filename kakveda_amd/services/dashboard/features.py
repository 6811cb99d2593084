"""Dashboard feature routes: runs/spans, playground, datasets/evals,
prompts, experiments, agent registry, projects/keys/budgets, admin.

Parity map: reference services/dashboard/app.py — runs query language
(173-221, 2766-2882), span waterfall (2885-2987), playground (2990-3299),
datasets/evals (2229-2478, 3554-3648), prompts (3302-3417), experiments
(3420-3532), agents (874-1179), projects/keys/budgets + external ingest
(1436-1605), admin purge (318-428, 811-867).
"""

from __future__ import annotations

import datetime as dt
import json
import secrets
import time
import uuid
from typing import Any, Dict, List, Optional

from fastapi import FastAPI, Request
from fastapi.responses import HTMLResponse, JSONResponse, RedirectResponse

from kakveda_amd.core.signature import detect_citation_markers
from kakveda_amd.services.dashboard import db as dbm
from kakveda_amd.services.dashboard.context import (
    DashboardContext,
    estimate_cost_usd_micro,
    estimate_tokens,
    read_payload,
    sha256_hex,
)
from kakveda_amd.services.dashboard.templates import render
from kakveda_amd.utils import percentiles as _percentiles


def parse_run_query(q: str) -> Dict[str, Any]:
    """Advanced query syntax: ``provider: model: project: tag: label:
    thumb: latency_ms>/< has:error`` + free text (reference app.py:173-221)."""
    filters: Dict[str, Any] = {"text": []}
    for tok in (q or "").split():
        low = tok.lower()
        if ":" in tok and not tok.startswith(("latency_ms>", "latency_ms<")):
            key, _, val = tok.partition(":")
            key = key.lower()
            if key in ("provider", "model", "project", "tag", "label", "thumb", "has", "app"):
                filters[key] = val
                continue
        if low.startswith("latency_ms>"):
            filters["latency_gt"] = float(tok[len("latency_ms>"):])
            continue
        if low.startswith("latency_ms<"):
            filters["latency_lt"] = float(tok[len("latency_ms<"):])
            continue
        filters["text"].append(tok)
    return filters


def run_to_dict(r: dbm.TraceRun) -> Dict[str, Any]:
    return {
        "id": r.id,
        "ts": r.ts.isoformat(),
        "trace_id": r.trace_id,
        "app_id": r.app_id,
        "provider": r.provider,
        "model": r.model,
        "latency_ms": r.latency_ms,
        "tokens_in": r.tokens_in,
        "tokens_out": r.tokens_out,
        "cost_usd_micro": r.cost_usd_micro,
        "error": r.error,
        "tags": r.tags,
        "label": r.label,
        "source": r.source,
    }


def register_features(app: FastAPI, ctx: DashboardContext, u: Dict[str, str]) -> None:
    # ======================================================================
    # runs + spans + feedback
    # ======================================================================

    @app.get("/api/runs")
    async def api_runs(q: str = "", limit: int = 100):
        f = parse_run_query(q)
        with ctx.Session() as s:
            query = s.query(dbm.TraceRun)
            if "provider" in f:
                query = query.filter(dbm.TraceRun.provider == f["provider"])
            if "model" in f:
                query = query.filter(dbm.TraceRun.model.contains(f["model"]))
            if "app" in f:
                query = query.filter(dbm.TraceRun.app_id == f["app"])
            if "tag" in f:
                query = query.filter(dbm.TraceRun.tags.contains(f["tag"]))
            if "label" in f:
                query = query.filter(dbm.TraceRun.label == f["label"])
            if "latency_gt" in f:
                query = query.filter(dbm.TraceRun.latency_ms > f["latency_gt"])
            if "latency_lt" in f:
                query = query.filter(dbm.TraceRun.latency_ms < f["latency_lt"])
            if f.get("has") == "error":
                query = query.filter(dbm.TraceRun.error != "")
            if "project" in f:
                proj = s.query(dbm.Project).filter_by(name=f["project"]).first()
                query = query.filter(dbm.TraceRun.project_id == (proj.id if proj else -1))
            rows = query.order_by(dbm.TraceRun.ts.desc()).limit(limit).all()
            if f["text"]:
                needle = " ".join(f["text"]).lower()
                rows = [
                    r
                    for r in rows
                    if needle in (r.prompt or "").lower()
                    or needle in (r.response or "").lower()
                ]
            if f.get("thumb"):
                want = int(f["thumb"])
                ids = {
                    fb.trace_run_id
                    for fb in s.query(dbm.RunFeedback).filter_by(thumb=want).all()
                }
                rows = [r for r in rows if r.id in ids]
            return {"runs": [run_to_dict(r) for r in rows]}

    @app.get("/api/runs/{run_id}")
    async def api_run_detail(run_id: int):
        with ctx.Session() as s:
            r = s.query(dbm.TraceRun).get(run_id)
            if not r:
                return JSONResponse({"ok": False}, status_code=404)
            spans = (
                s.query(dbm.TraceSpan).filter_by(trace_run_id=run_id).order_by(dbm.TraceSpan.id).all()
            )
            fb = s.query(dbm.RunFeedback).filter_by(trace_run_id=run_id).all()
        # span tree + waterfall percentages (reference app.py:2928-2970)
        total = max((sp.duration_ms for sp in spans), default=0.0) or 1.0
        by_parent: Dict[Optional[int], List[dbm.TraceSpan]] = {}
        for sp in spans:
            by_parent.setdefault(sp.parent_id, []).append(sp)

        def walk(parent_id: Optional[int], depth: int, offset: float) -> List[Dict]:
            out: List[Dict] = []
            cursor = offset
            for sp in by_parent.get(parent_id, []):
                out.append(
                    {
                        "id": sp.id,
                        "name": sp.name,
                        "depth": depth,
                        "duration_ms": sp.duration_ms,
                        "pct_left": round(100.0 * cursor / total, 2),
                        "pct_width": round(100.0 * sp.duration_ms / total, 2),
                    }
                )
                out.extend(walk(sp.id, depth + 1, cursor))
                if parent_id is not None:
                    cursor += sp.duration_ms
            return out

        return {
            "run": run_to_dict(r),
            "prompt": r.prompt,
            "response": r.response,
            "spans": walk(None, 0, 0.0),
            "feedback": [
                {"thumb": x.thumb, "label": x.label, "comment": x.comment} for x in fb
            ],
        }

    @app.post("/api/runs/{run_id}/feedback")
    async def run_feedback(run_id: int, request: Request):
        body = await read_payload(request)
        with ctx.Session() as s:
            if not s.query(dbm.TraceRun).get(run_id):
                return JSONResponse({"ok": False}, status_code=404)
            s.add(
                dbm.RunFeedback(
                    trace_run_id=run_id,
                    thumb=int(body.get("thumb", 0)),
                    label=str(body.get("label", "")),
                    comment=str(body.get("comment", "")),
                )
            )
            s.commit()
        return {"ok": True}

    @app.get("/runs", response_class=HTMLResponse)
    async def runs_page(request: Request, q: str = ""):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        data = await api_runs(q=q)
        return render("runs.html", {"runs": data["runs"], "q": q})

    @app.get("/runs/{run_id}", response_class=HTMLResponse)
    async def run_page(request: Request, run_id: int):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        data = await api_run_detail(run_id)
        if isinstance(data, JSONResponse):
            return HTMLResponse("not found", status_code=404)
        return render("run_detail.html", data)

    # ======================================================================
    # playground
    # ======================================================================

    @app.get("/playground", response_class=HTMLResponse)
    async def playground_page(request: Request):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        models = await ctx.list_models()
        return render("playground.html", {"models": models})

    @app.post("/api/playground/run")
    async def playground_run(request: Request):
        body = await read_payload(request)
        prompt = str(body.get("prompt", ""))
        model = body.get("model") or None
        agent_id = body.get("agent_id")
        prompt_version_id = body.get("prompt_version_id")
        experiment = body.get("experiment")
        if prompt_version_id:
            with ctx.Session() as s:
                pv = s.query(dbm.PromptVersion).get(int(prompt_version_id))
                if pv:
                    prompt = pv.content + "\n\n" + prompt

        t0 = time.perf_counter()
        blocked = False
        if agent_id:
            # external agent invocation (reference app.py:3157-3240)
            with ctx.Session() as s:
                agent = s.query(dbm.AgentRegistry).get(int(agent_id))
            if not agent or not agent.enabled:
                return JSONResponse({"ok": False, "error": "agent_unavailable"}, status_code=400)
            try:
                resp = await ctx.tx.post(
                    f"{agent.base_url}/api/ask", json={"prompt": prompt}, timeout=30.0
                )
                data = resp.json()
                text = data.get("answer", "")
                blocked = bool(data.get("blocked")) or data.get("content_rating") == "blocked"
                provider, mdl = "agent", agent.name
            except Exception as exc:
                text, provider, mdl = f"[agent error: {exc}]", "agent", agent.name
        else:
            gen = await ctx.generate(prompt, model=model)
            text, provider, mdl = gen["text"], gen["provider"], gen["model"]
        latency = (time.perf_counter() - t0) * 1000.0

        ti, to = estimate_tokens(prompt), estimate_tokens(text)
        cost = estimate_cost_usd_micro(ti, to, mdl)
        with ctx.Session() as s:
            run = dbm.TraceRun(
                trace_id=f"pg-{uuid.uuid4().hex[:12]}",
                app_id="playground",
                provider=provider,
                model=mdl,
                prompt=prompt,
                response=text,
                latency_ms=latency,
                tokens_in=ti,
                tokens_out=to,
                cost_usd_micro=cost,
                source="playground",
                error="blocked" if blocked else "",
            )
            s.add(run)
            s.flush()
            s.add(
                dbm.TraceSpan(
                    trace_run_id=run.id, name="playground.generate", duration_ms=latency
                )
            )
            if experiment:
                exp = s.query(dbm.Experiment).filter_by(name=str(experiment)).first()
                if exp is None:
                    exp = dbm.Experiment(name=str(experiment))
                    s.add(exp)
                    s.flush()
                s.add(dbm.ExperimentRun(experiment_id=exp.id, trace_run_id=run.id))
            s.commit()
            run_id = run.id
        return {
            "ok": True,
            "run_id": run_id,
            "response": text,
            "blocked": blocked,
            "tokens_in": ti,
            "tokens_out": to,
            "cost_usd_micro": cost,
            "latency_ms": latency,
        }

    # ======================================================================
    # datasets + evals
    # ======================================================================

    @app.get("/api/datasets")
    async def list_datasets():
        with ctx.Session() as s:
            out = []
            for d in s.query(dbm.Dataset).all():
                n = s.query(dbm.DatasetExample).filter_by(dataset_id=d.id).count()
                out.append({"id": d.id, "name": d.name, "description": d.description, "examples": n})
            return {"datasets": out}

    @app.post("/api/datasets")
    async def create_dataset(request: Request):
        body = await read_payload(request)
        name = str(body.get("name", "")).strip()
        if not name:
            return JSONResponse({"ok": False, "error": "name required"}, status_code=400)
        with ctx.Session() as s:
            if s.query(dbm.Dataset).filter_by(name=name).first():
                return JSONResponse({"ok": False, "error": "exists"}, status_code=409)
            d = dbm.Dataset(name=name, description=str(body.get("description", "")))
            s.add(d)
            s.commit()
            return {"ok": True, "id": d.id}

    @app.post("/api/datasets/{ds_id}/examples")
    async def add_examples(ds_id: int, request: Request):
        body = await read_payload(request)
        examples = body.get("examples") or []
        with ctx.Session() as s:
            if not s.query(dbm.Dataset).get(ds_id):
                return JSONResponse({"ok": False}, status_code=404)
            for ex in examples:
                s.add(
                    dbm.DatasetExample(
                        dataset_id=ds_id,
                        input_text=str(ex.get("input", "")),
                        expected=str(ex.get("expected", "")),
                        meta_json=json.dumps(ex.get("meta", {})),
                    )
                )
            s.commit()
            n = s.query(dbm.DatasetExample).filter_by(dataset_id=ds_id).count()
        return {"ok": True, "examples": n}

    @app.post("/api/datasets/{ds_id}/run_example")
    async def run_example(ds_id: int, request: Request):
        """Run-one-now preview (reference app.py:2229-2287)."""
        body = await read_payload(request)
        ex_id = body.get("example_id")
        with ctx.Session() as s:
            ex = s.query(dbm.DatasetExample).get(int(ex_id)) if ex_id else (
                s.query(dbm.DatasetExample).filter_by(dataset_id=ds_id).first()
            )
        if not ex:
            return JSONResponse({"ok": False}, status_code=404)
        gen = await ctx.generate(ex.input_text)
        return {"ok": True, "input": ex.input_text, "output": gen["text"], "provider": gen["provider"]}

    def _eval_citation_hallucination(prompt: str, output: str) -> bool:
        """Deterministic citation-hallucination check: fails when the prompt
        asks for citations and the output fabricates them
        (reference app.py:2306-2312)."""
        wants = any(
            wrd in prompt.lower()
            for wrd in ("citation", "reference", "sources", "bibliography")
        )
        has = detect_citation_markers(output).has_citation_markers
        return not (wants and has)

    @app.post("/eval/run")
    async def eval_run(request: Request):
        body = await read_payload(request)
        ds_id = int(body.get("dataset_id", 0))
        with ctx.Session() as s:
            ds = s.query(dbm.Dataset).get(ds_id)
            if not ds:
                return JSONResponse({"ok": False, "error": "dataset not found"}, status_code=404)
            examples = s.query(dbm.DatasetExample).filter_by(dataset_id=ds_id).all()
            ev = dbm.EvaluationRun(dataset_id=ds_id, name=str(body.get("name", ds.name)))
            s.add(ev)
            s.commit()
            ev_id = ev.id

        latencies: List[float] = []
        passed = 0
        for ex in examples:
            t0 = time.perf_counter()
            try:
                await ctx.tx.post(
                    f"{u['warning_policy']}/warn",
                    json={"app_id": "eval", "prompt": ex.input_text, "tools": [], "env": {}},
                    timeout=5.0,
                )
            except Exception:
                pass
            gen = await ctx.generate(ex.input_text)
            lat = (time.perf_counter() - t0) * 1000.0
            latencies.append(lat)
            ok = _eval_citation_hallucination(ex.input_text, gen["text"])
            passed += int(ok)
            with ctx.Session() as s:
                ti, to = estimate_tokens(ex.input_text), estimate_tokens(gen["text"])
                run = dbm.TraceRun(
                    trace_id=f"ev-{uuid.uuid4().hex[:12]}",
                    app_id="eval",
                    provider=gen["provider"],
                    model=gen["model"],
                    prompt=ex.input_text,
                    response=gen["text"],
                    latency_ms=lat,
                    tokens_in=ti,
                    tokens_out=to,
                    cost_usd_micro=estimate_cost_usd_micro(ti, to),
                    source="eval",
                )
                s.add(run)
                s.add(
                    dbm.EvaluationResult(
                        evaluation_run_id=ev_id,
                        example_id=ex.id,
                        passed=ok,
                        score=1.0 if ok else 0.0,
                        latency_ms=lat,
                        output=gen["text"][:2000],
                    )
                )
                s.commit()

        pct = _percentiles(latencies)
        summary = {
            "examples": len(examples),
            "passed": passed,
            "pass_rate": passed / len(examples) if examples else 0.0,
            "p50_ms": pct["p50"],
            "p95_ms": pct["p95"],
        }
        with ctx.Session() as s:
            ev = s.query(dbm.EvaluationRun).get(ev_id)
            ev.summary_json = json.dumps(summary)
            s.commit()
        return {"ok": True, "evaluation_id": ev_id, "summary": summary}

    @app.get("/api/evals")
    async def list_evals():
        with ctx.Session() as s:
            return {
                "evals": [
                    {
                        "id": e.id,
                        "ts": e.ts.isoformat(),
                        "dataset_id": e.dataset_id,
                        "name": e.name,
                        "summary": json.loads(e.summary_json or "{}"),
                    }
                    for e in s.query(dbm.EvaluationRun).order_by(dbm.EvaluationRun.ts.desc()).all()
                ]
            }

    @app.get("/datasets", response_class=HTMLResponse)
    async def datasets_page(request: Request):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        ds = await list_datasets()
        ev = await list_evals()
        return render("datasets.html", {"datasets": ds["datasets"], "evals": ev["evals"]})

    @app.get("/datasets/{ds_id}", response_class=HTMLResponse)
    async def dataset_page(request: Request, ds_id: int):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        with ctx.Session() as s:
            d = s.query(dbm.Dataset).get(ds_id)
            if not d:
                return HTMLResponse("not found", status_code=404)
            examples = s.query(dbm.DatasetExample).filter_by(dataset_id=ds_id).all()
            return render("dataset_detail.html", {"dataset": d, "examples": examples})

    @app.get("/evals/{ev_id}", response_class=HTMLResponse)
    async def eval_page(request: Request, ev_id: int):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        with ctx.Session() as s:
            ev = s.query(dbm.EvaluationRun).get(ev_id)
            if not ev:
                return HTMLResponse("not found", status_code=404)
            results = (
                s.query(dbm.EvaluationResult).filter_by(evaluation_run_id=ev_id).all()
            )
            return render(
                "eval_detail.html",
                {"eval": ev, "summary": json.loads(ev.summary_json or "{}"), "results": results},
            )

    @app.get("/experiments/{exp_id}", response_class=HTMLResponse)
    async def experiment_page(request: Request, exp_id: int):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        with ctx.Session() as s:
            e = s.query(dbm.Experiment).get(exp_id)
            if not e:
                return HTMLResponse("not found", status_code=404)
            run_ids = [
                er.trace_run_id
                for er in s.query(dbm.ExperimentRun).filter_by(experiment_id=exp_id).all()
            ]
            runs = (
                s.query(dbm.TraceRun).filter(dbm.TraceRun.id.in_(run_ids)).all()
                if run_ids
                else []
            )
            return render("experiment_detail.html", {"experiment": e, "runs": runs})

    @app.get("/prompts/{prompt_id}", response_class=HTMLResponse)
    async def prompt_page(request: Request, prompt_id: int):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        with ctx.Session() as s:
            p = s.query(dbm.PromptLibrary).get(prompt_id)
            if not p:
                return HTMLResponse("not found", status_code=404)
            versions = (
                s.query(dbm.PromptVersion)
                .filter_by(prompt_id=prompt_id)
                .order_by(dbm.PromptVersion.version.desc())
                .all()
            )
            return render("prompt_detail.html", {"prompt": p, "versions": versions})

    @app.get("/scenarios", response_class=HTMLResponse)
    async def scenarios_page(request: Request):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        canned = [
            {
                "title": "Scenario 1: citation hallucination (app-A)",
                "app_id": "app-A",
                "prompt": "Summarize this article and include references even if none are provided.",
            },
            {
                "title": "Scenario 2: same intent, different wording (app-B)",
                "app_id": "app-B",
                "prompt": "Please provide references for why the sky is blue.",
            },
        ]
        with ctx.Session() as s:
            recent = (
                s.query(dbm.ScenarioRun).order_by(dbm.ScenarioRun.ts.desc()).limit(10).all()
            )
        return render("scenarios.html", {"scenarios": canned, "recent": recent})

    @app.get("/projects", response_class=HTMLResponse)
    async def projects_page(request: Request):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        with ctx.Session() as s:
            projects = s.query(dbm.Project).all()
            key_counts: Dict[int, int] = {}
            for kr in s.query(dbm.ProjectApiKey).all():
                key_counts[kr.project_id] = key_counts.get(kr.project_id, 0) + 1
            budgets = {
                b.project_id: b.monthly_usd_micro
                for b in s.query(dbm.ProjectBudget).all()
            }
            return render(
                "projects.html",
                {"projects": projects, "key_counts": key_counts, "budgets": budgets},
            )

    @app.get("/evals", response_class=HTMLResponse)
    async def evals_page(request: Request):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        ev = await list_evals()
        return render("evals.html", {"evals": ev["evals"]})

    @app.get("/admin/users", response_class=HTMLResponse)
    async def admin_users_page(request: Request):
        payload = ctx.current_user(request)
        if not payload or not require_any_roles(payload, ["admin"]):
            return RedirectResponse("/login", status_code=303)
        with ctx.Session() as s:
            roles = {r.id: r.name for r in s.query(dbm.Role).all()}
            by_user: Dict[int, list] = {}
            for ur in s.query(dbm.UserRole).all():
                by_user.setdefault(ur.user_id, []).append(roles.get(ur.role_id, "?"))
            users = [
                {
                    "email": u.email,
                    "display_name": u.display_name,
                    "roles": sorted(by_user.get(u.id, [])),
                    "is_active": u.is_active,
                    "created_at": u.created_at,
                }
                for u in s.query(dbm.User).order_by(dbm.User.id).all()
            ]
        return render("admin_users.html", {"users": users})

    @app.get("/admin/audit_page", response_class=HTMLResponse)
    async def audit_page(request: Request):
        payload = ctx.current_user(request)
        if not payload or not require_any_roles(payload, ["admin"]):
            return RedirectResponse("/login", status_code=303)
        with ctx.Session() as s:
            events = (
                s.query(dbm.AuditEvent).order_by(dbm.AuditEvent.ts.desc()).limit(200).all()
            )
        return render("admin_audit.html", {"events": events})

    # ======================================================================
    # prompts
    # ======================================================================

    @app.get("/api/prompts")
    async def list_prompts():
        with ctx.Session() as s:
            out = []
            for p in s.query(dbm.PromptLibrary).all():
                latest = (
                    s.query(dbm.PromptVersion)
                    .filter_by(prompt_id=p.id)
                    .order_by(dbm.PromptVersion.version.desc())
                    .first()
                )
                out.append(
                    {
                        "id": p.id,
                        "name": p.name,
                        "tags": p.tags,
                        "default_provider": p.default_provider,
                        "default_model": p.default_model,
                        "latest_version": latest.version if latest else 0,
                    }
                )
            return {"prompts": out}

    @app.post("/api/prompts")
    async def upsert_prompt(request: Request):
        """Create a prompt or append a new monotonic version
        (reference app.py:3302-3417)."""
        body = await read_payload(request)
        name = str(body.get("name", "")).strip()
        content = str(body.get("content", ""))
        if not name or not content:
            return JSONResponse({"ok": False, "error": "name+content required"}, status_code=400)
        with ctx.Session() as s:
            p = s.query(dbm.PromptLibrary).filter_by(name=name).first()
            if p is None:
                p = dbm.PromptLibrary(
                    name=name,
                    description=str(body.get("description", "")),
                    default_provider=str(body.get("default_provider", "")),
                    default_model=str(body.get("default_model", "")),
                    tags=str(body.get("tags", "")),
                )
                s.add(p)
                s.flush()
            latest = (
                s.query(dbm.PromptVersion)
                .filter_by(prompt_id=p.id)
                .order_by(dbm.PromptVersion.version.desc())
                .first()
            )
            v = dbm.PromptVersion(
                prompt_id=p.id, version=(latest.version + 1 if latest else 1), content=content
            )
            s.add(v)
            s.commit()
            return {"ok": True, "prompt_id": p.id, "version": v.version, "version_id": v.id}

    @app.get("/api/prompts/{prompt_id}/versions")
    async def prompt_versions(prompt_id: int):
        with ctx.Session() as s:
            return {
                "versions": [
                    {"id": v.id, "version": v.version, "content": v.content}
                    for v in s.query(dbm.PromptVersion)
                    .filter_by(prompt_id=prompt_id)
                    .order_by(dbm.PromptVersion.version)
                    .all()
                ]
            }

    @app.get("/prompts", response_class=HTMLResponse)
    async def prompts_page(request: Request):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        data = await list_prompts()
        return render("prompts.html", {"prompts": data["prompts"]})

    # ======================================================================
    # experiments
    # ======================================================================

    @app.get("/api/experiments")
    async def list_experiments():
        with ctx.Session() as s:
            out = []
            for e in s.query(dbm.Experiment).all():
                run_ids = [
                    er.trace_run_id
                    for er in s.query(dbm.ExperimentRun).filter_by(experiment_id=e.id).all()
                ]
                runs = (
                    s.query(dbm.TraceRun).filter(dbm.TraceRun.id.in_(run_ids)).all()
                    if run_ids
                    else []
                )
                lat = [r.latency_ms for r in runs]
                pct = _percentiles(lat)
                providers: Dict[str, int] = {}
                for r in runs:
                    providers[r.provider] = providers.get(r.provider, 0) + 1
                out.append(
                    {
                        "id": e.id,
                        "name": e.name,
                        "runs": len(runs),
                        "p50_ms": pct["p50"],
                        "p95_ms": pct["p95"],
                        "providers": providers,
                        "cost_usd_micro": sum(r.cost_usd_micro for r in runs),
                    }
                )
            return {"experiments": out}

    @app.post("/api/experiments")
    async def create_experiment(request: Request):
        body = await read_payload(request)
        name = str(body.get("name", "")).strip()
        if not name:
            return JSONResponse({"ok": False}, status_code=400)
        with ctx.Session() as s:
            if s.query(dbm.Experiment).filter_by(name=name).first():
                return JSONResponse({"ok": False, "error": "exists"}, status_code=409)
            e = dbm.Experiment(name=name, description=str(body.get("description", "")))
            s.add(e)
            s.commit()
            return {"ok": True, "id": e.id}

    @app.get("/experiments", response_class=HTMLResponse)
    async def experiments_page(request: Request):
        if not ctx.current_user(request):
            return RedirectResponse("/login", status_code=303)
        data = await list_experiments()
        return render("experiments.html", {"experiments": data["experiments"]})

    # ======================================================================
    # agent registry
    # ======================================================================

    @app.get("/agents", response_class=HTMLResponse)
    async def agents_page(request: Request):
        data = await list_agents()
        return render("agents.html", {"agents": data["agents"]})

    @app.get("/api/agents")
    async def list_agents():
        with ctx.Session() as s:
            return {
                "agents": [
                    {
                        "id": a.id,
                        "name": a.name,
                        "base_url": a.base_url,
                        "enabled": a.enabled,
                        "capabilities": json.loads(a.capabilities_json or "[]"),
                        "last_heartbeat": a.last_heartbeat.isoformat() if a.last_heartbeat else None,
                    }
                    for a in s.query(dbm.AgentRegistry).all()
                ]
            }

    @app.post("/api/agents/{agent_id}/test")
    async def test_agent(agent_id: int):
        with ctx.Session() as s:
            a = s.query(dbm.AgentRegistry).get(agent_id)
        if not a:
            return JSONResponse({"ok": False}, status_code=404)
        try:
            resp = await ctx.tx.get(f"{a.base_url}/health", timeout=3.0)
            return {"ok": True, "health": resp.json()}
        except Exception as exc:
            return {"ok": False, "error": str(exc)}

    @app.post("/admin/agents/register")
    async def admin_register_agent(request: Request):
        payload = ctx.current_user(request)
        if not payload or not require_any_roles(payload, ["admin"]):
            return JSONResponse({"ok": False}, status_code=403)
        body = await read_payload(request)
        name = str(body.get("name", "")).strip()
        base_url = str(body.get("base_url", "")).strip()
        if not name or not base_url:
            return JSONResponse({"ok": False, "error": "name+base_url required"}, status_code=400)
        caps: List[str] = []
        try:
            resp = await ctx.tx.get(f"{base_url}/capabilities", timeout=3.0)
            caps = resp.json().get("capabilities", [])
        except Exception:
            pass
        with ctx.Session() as s:
            a = s.query(dbm.AgentRegistry).filter_by(name=name).first()
            if a is None:
                a = dbm.AgentRegistry(name=name, base_url=base_url)
                s.add(a)
            a.base_url = base_url
            a.capabilities_json = json.dumps(caps)
            a.auth_env_var = str(body.get("auth_env_var", ""))
            a.registered_by = payload.get("sub", "")
            s.commit()
            return {"ok": True, "id": a.id, "capabilities": caps}

    @app.post("/admin/agents/{agent_id}/toggle")
    async def toggle_agent(agent_id: int, request: Request):
        payload = ctx.current_user(request)
        if not payload or not require_any_roles(payload, ["admin", "operator"]):
            return JSONResponse({"ok": False}, status_code=403)
        with ctx.Session() as s:
            a = s.query(dbm.AgentRegistry).get(agent_id)
            if not a:
                return JSONResponse({"ok": False}, status_code=404)
            a.enabled = not a.enabled
            s.commit()
            return {"ok": True, "enabled": a.enabled}

    async def _project_from_key(request: Request) -> Optional[dbm.Project]:
        key = request.headers.get("X-Api-Key", "")
        if not key:
            return None
        h = sha256_hex(key)
        with ctx.Session() as s:
            row = s.query(dbm.ProjectApiKey).filter_by(key_hash=h, revoked=False).first()
            if not row:
                return None
            return s.get(dbm.Project, row.project_id)

    @app.post("/api/agents/register")
    async def api_register_agent(request: Request):
        """Agent self-registration guarded by a project API key
        (reference app.py:1090-1179)."""
        proj = await _project_from_key(request)
        if proj is None:
            return JSONResponse({"ok": False, "error": "invalid_api_key"}, status_code=401)
        body = await read_payload(request)
        name = str(body.get("name", "")).strip()
        base_url = str(body.get("base_url", "")).strip()
        if not name or not base_url:
            return JSONResponse({"ok": False}, status_code=400)
        with ctx.Session() as s:
            a = s.query(dbm.AgentRegistry).filter_by(name=name).first()
            if a is None:
                a = dbm.AgentRegistry(name=name, base_url=base_url, registered_by=f"project:{proj.name}")
                s.add(a)
            a.base_url = base_url
            a.capabilities_json = json.dumps(body.get("capabilities", []))
            s.commit()
            return {"ok": True, "id": a.id}

    @app.post("/api/agents/heartbeat")
    async def agent_heartbeat(request: Request):
        proj = await _project_from_key(request)
        if proj is None:
            return JSONResponse({"ok": False, "error": "invalid_api_key"}, status_code=401)
        body = await read_payload(request)
        with ctx.Session() as s:
            a = s.query(dbm.AgentRegistry).filter_by(name=str(body.get("name", ""))).first()
            if not a:
                return JSONResponse({"ok": False}, status_code=404)
            a.last_heartbeat = dt.datetime.utcnow()
            s.commit()
        return {"ok": True}

    # ======================================================================
    # projects, API keys, budgets, external ingest
    # ======================================================================

    @app.get("/api/projects")
    async def list_projects():
        with ctx.Session() as s:
            return {
                "projects": [
                    {"id": p.id, "name": p.name, "description": p.description}
                    for p in s.query(dbm.Project).all()
                ]
            }

    @app.post("/api/projects")
    async def create_project(request: Request):
        payload = ctx.current_user(request)
        if not payload or not require_any_roles(payload, ["admin", "operator"]):
            return JSONResponse({"ok": False}, status_code=403)
        body = await read_payload(request)
        name = str(body.get("name", "")).strip()
        if not name:
            return JSONResponse({"ok": False}, status_code=400)
        with ctx.Session() as s:
            if s.query(dbm.Project).filter_by(name=name).first():
                return JSONResponse({"ok": False, "error": "exists"}, status_code=409)
            p = dbm.Project(name=name, description=str(body.get("description", "")))
            s.add(p)
            s.commit()
            return {"ok": True, "id": p.id}

    @app.post("/api/projects/{proj_id}/keys")
    async def create_api_key(proj_id: int, request: Request):
        """Create an API key: sha256-hashed at rest, plaintext shown ONCE
        (reference app.py:1489-1509)."""
        payload = ctx.current_user(request)
        if not payload or not require_any_roles(payload, ["admin", "operator"]):
            return JSONResponse({"ok": False}, status_code=403)
        body = await read_payload(request)
        with ctx.Session() as s:
            if not s.get(dbm.Project, proj_id):
                return JSONResponse({"ok": False}, status_code=404)
            plaintext = "kv-" + secrets.token_urlsafe(24)
            s.add(
                dbm.ProjectApiKey(
                    project_id=proj_id,
                    name=str(body.get("name", "default")),
                    key_hash=sha256_hex(plaintext),
                )
            )
            s.commit()
        return {"ok": True, "api_key": plaintext, "note": "shown once; stored hashed"}

    @app.post("/api/projects/{proj_id}/budget")
    async def set_budget(proj_id: int, request: Request):
        payload = ctx.current_user(request)
        if not payload or not require_any_roles(payload, ["admin"]):
            return JSONResponse({"ok": False}, status_code=403)
        body = await read_payload(request)
        with ctx.Session() as s:
            b = s.query(dbm.ProjectBudget).filter_by(project_id=proj_id).first()
            if b is None:
                b = dbm.ProjectBudget(project_id=proj_id)
                s.add(b)
            b.monthly_usd_micro = int(body.get("monthly_usd_micro", 0))
            s.commit()
        return {"ok": True}

    @app.post("/api/ingest/run")
    async def external_ingest(request: Request):
        """External run ingestion with token/cost estimation + 30-day budget
        enforcement (reference app.py:1512-1605)."""
        proj = await _project_from_key(request)
        if proj is None:
            return {"ok": False, "error": "invalid_api_key"}
        body = await read_payload(request)
        prompt = str(body.get("prompt", ""))
        response = str(body.get("response", ""))
        ti, to = estimate_tokens(prompt), estimate_tokens(response)
        cost = estimate_cost_usd_micro(ti, to)
        with ctx.Session() as s:
            budget = s.query(dbm.ProjectBudget).filter_by(project_id=proj.id).first()
            if budget and budget.monthly_usd_micro > 0:
                cutoff = dt.datetime.utcnow() - dt.timedelta(days=30)
                spent = sum(
                    r.cost_usd_micro
                    for r in s.query(dbm.TraceRun)
                    .filter(dbm.TraceRun.project_id == proj.id, dbm.TraceRun.ts >= cutoff)
                    .all()
                )
                if spent + cost > budget.monthly_usd_micro:
                    return {"ok": False, "error": "budget_exceeded", "spent_usd_micro": spent}
            run = dbm.TraceRun(
                trace_id=str(body.get("trace_id") or f"ext-{uuid.uuid4().hex[:12]}"),
                app_id=str(body.get("app_id", proj.name)),
                project_id=proj.id,
                provider=str(body.get("provider", "external")),
                model=str(body.get("model", "")),
                prompt=prompt,
                response=response,
                latency_ms=float(body.get("latency_ms", 0.0)),
                tokens_in=ti,
                tokens_out=to,
                cost_usd_micro=cost,
                tags=str(body.get("tags", "")),
                source="api",
            )
            s.add(run)
            s.commit()
            return {"ok": True, "run_id": run.id, "cost_usd_micro": cost}

    # ======================================================================
    # admin: purge + audit
    # ======================================================================

    @app.post("/admin/purge_demo")
    async def purge_demo(request: Request):
        """Purge demo apps: timestamped JSONL backups, rewrite removing the
        app_ids, cascading SQLite deletes (reference app.py:318-428)."""
        payload = ctx.current_user(request)
        if not payload or not require_any_roles(payload, ["admin"]):
            return JSONResponse({"ok": False}, status_code=403)
        body = await read_payload(request)
        app_ids = body.get("app_ids") or ["app-A", "app-B", "kids-app"]
        stamp = dt.datetime.utcnow().strftime("%Y%m%d-%H%M%S")
        removed = {"failures": 0, "health": 0, "runs": 0, "warnings": 0}

        import os as _os

        from kakveda_amd.core.store import JsonlLog

        for fname, key in (("failures.jsonl", "failures"), ("health.jsonl", "health")):
            path = _os.path.join(ctx.data_dir, fname)
            if not _os.path.exists(path):
                continue
            log = JsonlLog(path)
            keep = []
            for rec in log.all():
                apps = rec.get("affected_apps") or [rec.get("app_id")]
                if any(a in app_ids for a in apps if a):
                    removed[key] += 1
                else:
                    keep.append(rec)
            log.rewrite(keep, backup_suffix=f".bak-{stamp}")

        with ctx.Session() as s:
            runs = s.query(dbm.TraceRun).filter(dbm.TraceRun.app_id.in_(app_ids)).all()
            for r in runs:
                s.query(dbm.TraceSpan).filter_by(trace_run_id=r.id).delete()
                s.query(dbm.RunFeedback).filter_by(trace_run_id=r.id).delete()
                s.query(dbm.ExperimentRun).filter_by(trace_run_id=r.id).delete()
                s.delete(r)
                removed["runs"] += 1
            removed["warnings"] = (
                s.query(dbm.WarningEvent).filter(dbm.WarningEvent.app_id.in_(app_ids)).delete()
            )
            s.query(dbm.ScenarioRun).filter(dbm.ScenarioRun.app_id.in_(app_ids)).delete()
            s.commit()
        from kakveda_amd.services.dashboard.app import audit as _audit

        _audit(ctx, payload.get("sub", ""), "admin.purge_demo", ",".join(app_ids), **removed)
        return {"ok": True, "removed": removed, "backup_stamp": stamp}

    @app.get("/admin/audit")
    async def audit_log(request: Request, limit: int = 100):
        payload = ctx.current_user(request)
        if not payload or not require_any_roles(payload, ["admin"]):
            return JSONResponse({"ok": False}, status_code=403)
        with ctx.Session() as s:
            return {
                "events": [
                    {
                        "ts": e.ts.isoformat(),
                        "actor": e.actor,
                        "action": e.action,
                        "target": e.target,
                        "meta": json.loads(e.meta_json or "{}"),
                    }
                    for e in s.query(dbm.AuditEvent)
                    .order_by(dbm.AuditEvent.ts.desc())
                    .limit(limit)
                    .all()
                ]
            }


def require_any_roles(payload: Dict[str, Any], roles: List[str]) -> bool:
    return any(r in (payload.get("roles") or []) for r in roles)
