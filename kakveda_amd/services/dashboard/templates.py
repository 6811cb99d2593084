"""Jinja2 templates for the dashboard HTML pages.

Kept in-package via a DictLoader (parity target: the reference's 24
template files; here each page is intentionally minimal markup over the
same data the /api routes expose — the API surface is the contract)."""

from __future__ import annotations

from typing import Any, Dict

from jinja2 import DictLoader, Environment, select_autoescape

from kakveda_amd.services.dashboard.static import LOGO_SVG

_NAV = [
    ("/", "Dashboard"),
    ("/warnings", "Warnings"),
    ("/scenarios", "Scenarios"),
    ("/runs", "Runs"),
    ("/playground", "Playground"),
    ("/datasets", "Datasets"),
    ("/evals", "Evals"),
    ("/prompts", "Prompts"),
    ("/experiments", "Experiments"),
    ("/agents", "Agents"),
    ("/projects", "Projects"),
    ("/health", "Health"),
]

_BASE = (
    """<!doctype html><html><head><meta charset="utf-8">
<meta name="viewport" content="width=device-width,initial-scale=1">
<title>{% block title %}kakveda-amd{% endblock %}</title>
<link rel="stylesheet" href="/static/style.css">
<link rel="icon" type="image/svg+xml" href="/static/logo.svg">
<script src="/static/app.js" defer></script>
</head><body>
<header class="topbar">
<a class="brand" href="/">"""
    + LOGO_SVG
    + """<span>kakveda<span class="amd">-amd</span></span></a>
<nav class="main">
{% for href, label in nav %}<a href="{{ href }}"{% if active == href %} class="active"{% endif %}>{{ label }}</a>{% endfor %}
</nav>
<div class="spacer"></div>
{% if user %}<span class="who">{{ user }}</span>
<nav class="main"><a href="/logout">Logout</a></nav>{% endif %}
</header>
<main>
{% block content %}{% endblock %}
</main>
<footer>kakveda-amd — MI355X-native failure intelligence ·
<a href="/healthz">healthz</a> · <a href="/admin/audit_page">audit</a></footer>
</body></html>"""
)

_TEMPLATES = {
    "base.html": _BASE,
    "login.html": """{% extends "base.html" %}{% block content %}
<div class="card auth"><div class="brand">""" + LOGO_SVG + """</div>
<h2>Sign in</h2>
<form method="post" action="/login">
<input name="email" placeholder="email" value="admin@kakveda.local">
<input name="password" type="password" placeholder="password">
<button type="submit">Login</button></form>
<p class="dim">Demo users: admin/operator/viewer/demo @kakveda.local</p>
<p><a href="/register">Create account</a> · <a href="/forgot">Forgot password</a></p>
</div>{% endblock %}""",
    "home.html": """{% extends "base.html" %}{% block content %}
<h1>Failure Intelligence</h1>
<p class="dim">Signed in as {{ user }} ({{ roles|join(", ") }})</p>
<div class="tiles">
<div class="tile"><div class="n">{{ failures|length }}</div><div class="l">recent failures</div></div>
<div class="tile"><div class="n">{{ patterns|length }}</div><div class="l">patterns</div></div>
<div class="tile"><div class="n">{{ warnings|length }}</div><div class="l">latest warnings</div></div>
<div class="tile"><div class="n">{{ runs|length }}</div><div class="l">recent runs</div></div>
</div>
<div class="grid2">
<div class="card"><h3>Recent failures (GFKB)</h3><table>
<tr><th>id</th><th>v</th><th>type</th><th>apps</th><th>occ</th></tr>
{% for f in failures %}<tr>
<td><a href="/failure/{{ f.failure_id }}">{{ f.failure_id }}</a></td>
<td>{{ f.version }}</td>
<td>{{ f.failure_type }}</td><td>{{ f.affected_apps|join(", ") }}</td>
<td>{{ f.occurrences }}</td></tr>{% endfor %}</table></div>
<div class="card"><h3>Patterns</h3><table>
<tr><th>id</th><th>name</th><th>apps</th><th>failures</th></tr>
{% for p in patterns %}<tr><td>{{ p.pattern_id }}</td><td>{{ p.name }}</td>
<td>{{ p.affected_apps|join(", ") }}</td><td>{{ p.failure_ids|length }}</td></tr>
{% endfor %}</table></div>
</div>
<div class="card"><h3>Latest warnings</h3><table>
<tr><th>ts</th><th>app</th><th>action</th><th>confidence</th></tr>
{% for w in warnings %}<tr><td>{{ w.ts }}</td><td>{{ w.app_id }}</td>
<td><span class="badge {{ w.action }}">{{ w.action }}</span></td>
<td>{{ "%.2f"|format(w.confidence) }}</td></tr>{% endfor %}</table></div>
<div class="card"><h3>Run a scenario</h3>
<form class="inline" method="post" action="/scenarios/run">
<input name="app_id" value="app-A">
<input name="prompt" size="60"
 value="Summarize this and include references even if none are provided.">
<button type="submit">Run</button></form></div>
{% endblock %}""",
    "warnings.html": """{% extends "base.html" %}{% block content %}
<h1>Warnings</h1>
<div class="tiles">
<div class="tile"><div class="n">{{ analytics.total }}</div>
<div class="l">last {{ analytics.days }} days</div></div>
<div class="tile"><div class="n">{{ analytics.per_app|length }}</div>
<div class="l">apps affected</div></div>
<div class="tile"><div class="n">{{ analytics.per_pattern|length }}</div>
<div class="l">patterns hit</div></div>
<div class="tile"><div class="n">{{ analytics.est_cost_impact_usd_micro }}</div>
<div class="l">est. cost µUSD</div></div>
</div>
{% if analytics.daily %}
{% set peak = analytics.daily.values()|max %}
<div class="card"><h3>Daily warnings</h3>
<div class="chart">
{% for day, n in analytics.daily.items() %}
<div class="col"><div class="bar" style="height:{{ (100 * n / peak)|round }}%"
 title="{{ day }}: {{ n }}"></div><span class="lab">{{ day[5:] }}</span></div>
{% endfor %}
</div></div>
{% endif %}
<div class="grid2">
<div class="card"><h3>Per app</h3><table><tr><th>app</th><th>count</th></tr>
{% for app, n in analytics.per_app.items() %}
<tr><td>{{ app }}</td><td>{{ n }}</td></tr>{% endfor %}</table></div>
<div class="card"><h3>Per pattern</h3><table><tr><th>pattern</th><th>count</th></tr>
{% for p, n in analytics.per_pattern.items() %}
<tr><td>{{ p }}</td><td>{{ n }}</td></tr>{% endfor %}</table></div>
</div>
<div class="card"><table>
<tr><th>ts</th><th>app</th><th>action</th><th>conf</th><th>pattern</th><th>message</th></tr>
{% for w in warnings %}<tr id="w-{{ w.id }}"><td>{{ w.ts }}</td><td>{{ w.app_id }}</td>
<td>{{ w.action }}</td><td>{{ "%.2f"|format(w.confidence) }}</td>
<td>{{ w.pattern_id }}</td><td>{{ w.message[:120] }}</td></tr>{% endfor %}
</table></div>{% endblock %}""",
    "runs.html": """{% extends "base.html" %}{% block content %}
<h1>Runs</h1>
<form method="get" action="/runs"><input name="q" size="60" value="{{ q }}"
 placeholder="provider:stub model:llama latency_ms>100 has:error free text">
<button type="submit">Filter</button></form>
<div class="card"><table>
<tr><th>id</th><th>ts</th><th>app</th><th>provider</th><th>model</th>
<th>latency</th><th>tokens</th><th>cost µUSD</th></tr>
{% for r in runs %}<tr><td><a href="/runs/{{ r.id }}">{{ r.id }}</a></td>
<td>{{ r.ts }}</td><td>{{ r.app_id }}</td><td>{{ r.provider }}</td>
<td>{{ r.model }}</td><td>{{ "%.0f"|format(r.latency_ms) }} ms</td>
<td>{{ r.tokens_in }}/{{ r.tokens_out }}</td><td>{{ r.cost_usd_micro }}</td></tr>
{% endfor %}</table></div>{% endblock %}""",
    "run_detail.html": """{% extends "base.html" %}{% block content %}
<h1>Run {{ run.id }} — {{ run.trace_id }}</h1>
<div class="card"><b>{{ run.provider }}/{{ run.model }}</b> ·
{{ "%.0f"|format(run.latency_ms) }} ms · {{ run.cost_usd_micro }} µUSD
<h4>Prompt</h4><pre>{{ prompt }}</pre><h4>Response</h4><pre>{{ response }}</pre></div>
<div class="card"><h3>Span waterfall</h3>
{% for sp in spans %}
<div style="margin-left:{{ sp.depth * 20 }}px">
<small>{{ sp.name }} — {{ "%.1f"|format(sp.duration_ms) }} ms</small>
<div class="bar" style="margin-left:{{ sp.pct_left }}%;width:{{ sp.pct_width }}%"></div>
</div>{% endfor %}</div>{% endblock %}""",
    "playground.html": """{% extends "base.html" %}{% block content %}
<h1>Playground</h1>
<div class="card">
<form id="pg-form">
<select name="model">
{% for m in models %}<option value="{{ m }}">{{ m }}</option>{% endfor %}
</select><br>
<textarea name="prompt" rows="5" style="width:100%"
 placeholder="Ask something… the deterministic stub answers with citations when no model backend is reachable."></textarea><br>
<button type="submit">Run</button>
</form>
<h3>Output</h3><pre id="pg-out" class="dim">—</pre>
<p class="dim">API: POST /api/playground/run with {"prompt", "model",
"agent_id", "prompt_version_id", "experiment"}.</p>
</div>{% endblock %}""",
    "agents.html": """{% extends "base.html" %}{% block content %}
<h1>Agent registry</h1>
<div class="card"><table>
<tr><th>id</th><th>name</th><th>url</th><th>enabled</th><th>capabilities</th>
<th>heartbeat</th></tr>
{% for a in agents %}<tr><td>{{ a.id }}</td><td>{{ a.name }}</td>
<td>{{ a.base_url }}</td><td>{{ a.enabled }}</td>
<td>{{ a.capabilities|join(", ") }}</td><td>{{ a.last_heartbeat }}</td></tr>
{% endfor %}</table></div>{% endblock %}""",
    "failure_detail.html": """{% extends "base.html" %}{% block content %}
<h1>{{ record.failure_id }} v{{ record.version }} — {{ record.failure_type }}</h1>
<div class="card">
<p><b>Severity:</b> {{ record.impact_severity }} ·
<b>Occurrences:</b> {{ record.occurrences }} ·
<b>Apps:</b> {{ record.affected_apps|join(", ") }}</p>
<p><b>Root cause:</b> {{ record.root_cause }}</p>
<p><b>Resolution:</b> {{ record.resolution }}</p>
<p><b>Signature:</b> <code>{{ record.signature_text }}</code></p>
</div>
<div class="card"><h3>Version history</h3><table>
<tr><th>version</th><th>updated</th><th>occurrences</th><th>apps</th></tr>
{% for v in versions %}<tr>
<td><a href="/failure/{{ fid }}v{{ v.version }}">v{{ v.version }}</a></td>
<td>{{ v.updated_at }}</td><td>{{ v.occurrences }}</td>
<td>{{ v.affected_apps|join(", ") }}</td></tr>{% endfor %}
</table></div>{% endblock %}""",
    "datasets.html": """{% extends "base.html" %}{% block content %}
<h1>Datasets & evals</h1>
<div class="card"><table><tr><th>id</th><th>name</th><th>examples</th></tr>
{% for d in datasets %}<tr><td>{{ d.id }}</td><td>{{ d.name }}</td>
<td>{{ d.examples }}</td></tr>{% endfor %}</table></div>
<div class="card"><h3>Evaluation runs</h3>
<table><tr><th>id</th><th>ts</th><th>name</th><th>pass rate</th><th>p50/p95 ms</th></tr>
{% for e in evals %}<tr><td>{{ e.id }}</td><td>{{ e.ts }}</td><td>{{ e.name }}</td>
<td>{{ "%.0f%%"|format(100 * e.summary.get("pass_rate", 0)) }}</td>
<td>{{ "%.0f"|format(e.summary.get("p50_ms", 0)) }}/{{ "%.0f"|format(e.summary.get("p95_ms", 0)) }}</td></tr>
{% endfor %}</table></div>{% endblock %}""",
    "prompts.html": """{% extends "base.html" %}{% block content %}
<h1>Prompt library</h1>
<div class="card"><table>
<tr><th>id</th><th>name</th><th>latest v</th><th>default model</th><th>tags</th></tr>
{% for p in prompts %}<tr><td>{{ p.id }}</td><td>{{ p.name }}</td>
<td>v{{ p.latest_version }}</td><td>{{ p.default_model }}</td><td>{{ p.tags }}</td></tr>
{% endfor %}</table></div>{% endblock %}""",
    "experiments.html": """{% extends "base.html" %}{% block content %}
<h1>Experiments</h1>
<div class="card"><table>
<tr><th>id</th><th>name</th><th>runs</th><th>p50/p95 ms</th><th>providers</th><th>cost µUSD</th></tr>
{% for e in experiments %}<tr><td>{{ e.id }}</td><td>{{ e.name }}</td><td>{{ e.runs }}</td>
<td>{{ "%.0f"|format(e.p50_ms) }}/{{ "%.0f"|format(e.p95_ms) }}</td>
<td>{% for k, v in e.providers.items() %}{{ k }}:{{ v }} {% endfor %}</td>
<td>{{ e.cost_usd_micro }}</td></tr>{% endfor %}</table></div>{% endblock %}""",
    "health.html": """{% extends "base.html" %}{% block content %}
<h1>App health</h1>
{% for app_id, points in apps.items() %}
{% set latest = points[-1] %}
<div class="card"><h3>{{ app_id }}</h3>
<div class="score">
<b>{{ "%.1f"|format(latest.score) }}</b>
<div class="track"><div class="bar
 {% if latest.score >= 80 %}good{% elif latest.score >= 50 %}mid{% else %}low{% endif %}"
 style="width:{{ [latest.score, 0]|max }}%"></div></div>
</div>
<table>
<tr><th>ts</th><th>score</th><th>failure rate</th><th>penalty</th></tr>
{% for p in points %}<tr><td>{{ p.ts }}</td><td>{{ "%.1f"|format(p.score) }}</td>
<td>{{ "%.2f"|format(p.failure_rate) }}</td><td>{{ p.recurrent_penalty }}</td></tr>
{% endfor %}</table></div>
{% else %}<div class="card">No health points yet — ingest failures or use
POST /health/test.</div>{% endfor %}{% endblock %}""",
    "dataset_detail.html": """{% extends "base.html" %}{% block content %}
<h1>Dataset: {{ dataset.name }}</h1>
<p>{{ dataset.description }}</p>
<div class="card"><table><tr><th>id</th><th>input</th><th>expected</th></tr>
{% for e in examples %}<tr><td>{{ e.id }}</td><td>{{ e.input_text[:100] }}</td>
<td>{{ e.expected[:60] }}</td></tr>{% endfor %}</table></div>{% endblock %}""",
    "eval_detail.html": """{% extends "base.html" %}{% block content %}
<h1>Evaluation {{ eval.id }} — {{ eval.name }}</h1>
<div class="card"><p><b>Pass rate:</b>
{{ "%.0f%%"|format(100 * summary.get("pass_rate", 0)) }} ·
<b>p50/p95:</b> {{ "%.0f"|format(summary.get("p50_ms", 0)) }}/{{ "%.0f"|format(summary.get("p95_ms", 0)) }} ms</p>
<table><tr><th>example</th><th>passed</th><th>latency</th><th>output</th></tr>
{% for r in results %}<tr><td>{{ r.example_id }}</td><td>{{ r.passed }}</td>
<td>{{ "%.0f"|format(r.latency_ms) }} ms</td><td>{{ r.output[:80] }}</td></tr>
{% endfor %}</table></div>{% endblock %}""",
    "experiment_detail.html": """{% extends "base.html" %}{% block content %}
<h1>Experiment: {{ experiment.name }}</h1>
<div class="card"><table>
<tr><th>run</th><th>provider</th><th>model</th><th>latency</th><th>cost µUSD</th></tr>
{% for r in runs %}<tr><td><a href="/runs/{{ r.id }}">{{ r.id }}</a></td>
<td>{{ r.provider }}</td><td>{{ r.model }}</td>
<td>{{ "%.0f"|format(r.latency_ms) }} ms</td><td>{{ r.cost_usd_micro }}</td></tr>
{% endfor %}</table></div>{% endblock %}""",
    "prompt_detail.html": """{% extends "base.html" %}{% block content %}
<h1>Prompt: {{ prompt.name }}</h1>
<p>{{ prompt.description }} · default {{ prompt.default_provider }}/{{ prompt.default_model }}</p>
{% for v in versions %}<div class="card"><h3>v{{ v.version }}</h3>
<pre>{{ v.content }}</pre></div>{% endfor %}{% endblock %}""",
    "admin_audit.html": """{% extends "base.html" %}{% block content %}
<h1>Audit log</h1>
<div class="card"><table>
<tr><th>ts</th><th>actor</th><th>action</th><th>target</th></tr>
{% for e in events %}<tr><td>{{ e.ts }}</td><td>{{ e.actor }}</td>
<td>{{ e.action }}</td><td>{{ e.target }}</td></tr>{% endfor %}
</table></div>{% endblock %}""",
    "scenarios.html": """{% extends "base.html" %}{% block content %}
<h1>Scenario runner</h1>
{% for sc in scenarios %}<div class="card"><h3>{{ sc.title }}</h3>
<form method="post" action="/scenarios/run">
<input type="hidden" name="app_id" value="{{ sc.app_id }}">
<input name="prompt" size="70" value="{{ sc.prompt }}">
<button type="submit">Run scenario</button></form></div>{% endfor %}
<div class="card"><h3>Recent scenario runs</h3><table>
<tr><th>ts</th><th>app</th><th>action</th><th>confidence</th></tr>
{% for r in recent %}<tr><td>{{ r.ts }}</td><td>{{ r.app_id }}</td>
<td>{{ r.warn_action }}</td><td>{{ "%.2f"|format(r.warn_confidence) }}</td></tr>
{% endfor %}</table></div>{% endblock %}""",
    "register.html": """{% extends "base.html" %}{% block content %}
<div class="card auth"><h2>Create account</h2>
<form method="post" action="/register">
<input name="email" placeholder="email"><br>
<input name="password" type="password" placeholder="password (8+ chars)"><br>
<button type="submit">Register</button></form>
<p><a href="/login">Back to sign in</a></p></div>{% endblock %}""",
    "forgot.html": """{% extends "base.html" %}{% block content %}
<div class="card auth"><h2>Forgot password</h2>
<form method="post" action="/forgot">
<input name="email" placeholder="email"><br>
<button type="submit">Send reset token</button></form>
<p><a href="/login">Back to sign in</a></p></div>{% endblock %}""",
    "reset.html": """{% extends "base.html" %}{% block content %}
<div class="card auth"><h2>Reset password</h2>
<form method="post" action="/reset">
<input name="token" placeholder="reset token" value="{{ token }}"><br>
<input name="password" type="password" placeholder="new password"><br>
<button type="submit">Reset</button></form></div>{% endblock %}""",
    "projects.html": """{% extends "base.html" %}{% block content %}
<h1>Projects</h1>
{% for p in projects %}<div class="card"><h3>{{ p.name }}</h3>
<p>{{ p.description }}</p>
<p>API keys: {{ key_counts.get(p.id, 0) }} ·
monthly budget: {{ "%.2f"|format(budgets.get(p.id, 0) / 1000000) }} USD</p>
<form method="post" action="/api/projects/{{ p.id }}/keys">
<input name="name" placeholder="key name" value="default">
<button type="submit">New API key</button></form></div>
{% else %}<div class="card">No projects yet.</div>{% endfor %}
<div class="card"><h3>New project</h3>
<form method="post" action="/api/projects">
<input name="name" placeholder="name">
<input name="description" placeholder="description">
<button type="submit">Create</button></form></div>{% endblock %}""",
    "admin_users.html": """{% extends "base.html" %}{% block content %}
<h1>Users</h1>
<div class="card"><table>
<tr><th>email</th><th>display</th><th>roles</th><th>active</th><th>created</th></tr>
{% for u in users %}<tr><td>{{ u.email }}</td><td>{{ u.display_name }}</td>
<td>{{ u.roles|join(", ") }}</td><td>{{ u.is_active }}</td>
<td>{{ u.created_at }}</td></tr>{% endfor %}
</table></div>{% endblock %}""",
    "evals.html": """{% extends "base.html" %}{% block content %}
<h1>Evaluations</h1>
<div class="card"><table>
<tr><th>ts</th><th>name</th><th>dataset</th><th>pass rate</th><th>p50 ms</th></tr>
{% for e in evals %}<tr><td>{{ e.ts }}</td>
<td><a href="/evals/{{ e.id }}">{{ e.name }}</a></td>
<td>{{ e.dataset_id }}</td>
<td>{{ e.summary.get("pass_rate", "-") }}</td>
<td>{{ e.summary.get("p50_ms", "-") }}</td></tr>{% endfor %}
</table></div>{% endblock %}""",
}

_env = Environment(loader=DictLoader(_TEMPLATES), autoescape=select_autoescape(["html"]))

#: template -> nav item highlighted as active
_ACTIVE = {
    "home.html": "/",
    "warnings.html": "/warnings",
    "scenarios.html": "/scenarios",
    "runs.html": "/runs",
    "run_detail.html": "/runs",
    "playground.html": "/playground",
    "datasets.html": "/datasets",
    "dataset_detail.html": "/datasets",
    "evals.html": "/evals",
    "eval_detail.html": "/evals",
    "prompts.html": "/prompts",
    "prompt_detail.html": "/prompts",
    "experiments.html": "/experiments",
    "experiment_detail.html": "/experiments",
    "agents.html": "/agents",
    "projects.html": "/projects",
    "health.html": "/health",
}


def render(name: str, ctx: Dict[str, Any]) -> str:
    base = {"nav": _NAV, "active": _ACTIVE.get(name)}
    base.update(ctx)
    return _env.get_template(name).render(**base)
