"""Core subsystem tests: config hot-reload, JSONL store, health engine,
runtime logging, CLI arg surface."""

import json
import os
import time

import pytest

from kakveda_amd.core.config import ConfigStore
from kakveda_amd.core.store import JsonlLog
from kakveda_amd.health.scoring import HealthScorer


def test_config_defaults_without_file(tmp_path):
    cfg = ConfigStore(path=str(tmp_path / "missing.yaml"))
    assert cfg.get("failure_matching.similarity_threshold") == 0.8
    assert cfg.get("warning_policy.default_action") == "warn"
    assert cfg.get("health_score.severity_weights.medium") == 3
    assert cfg.get("nope.nothere", 42) == 42


def test_config_hot_reload(tmp_path):
    path = tmp_path / "config.yaml"
    path.write_text(
        "failure_matching:\n  similarity_threshold: 0.5\n"
        "hot_reload:\n  enabled: true\n  poll_seconds: 0\n"
    )
    cfg = ConfigStore(path=str(path))
    assert cfg.get("failure_matching.similarity_threshold") == 0.5
    time.sleep(0.05)
    path.write_text(
        "failure_matching:\n  similarity_threshold: 0.9\n"
        "hot_reload:\n  enabled: true\n  poll_seconds: 0\n"
    )
    os.utime(path, (time.time() + 5, time.time() + 5))  # force mtime change
    assert cfg.get("failure_matching.similarity_threshold") == 0.9
    # deep-merge keeps untouched defaults
    assert cfg.get("warning_policy.default_action") == "warn"


def test_jsonl_log_roundtrip(tmp_path):
    path = tmp_path / "x.jsonl"
    log = JsonlLog(path)
    for i in range(5):
        log.append({"i": i, "app_id": f"a{i % 2}"})
    assert len(log) == 5
    # durable: a new instance reloads from disk
    log2 = JsonlLog(path)
    assert [r["i"] for r in log2.all()] == [0, 1, 2, 3, 4]
    # tail with filter
    assert [r["i"] for r in log2.tail(2, where=lambda r: r["app_id"] == "a0")] == [2, 4]
    # rewrite with backup
    log2.rewrite([{"i": 99}], backup_suffix=".bak-test")
    assert len(JsonlLog(path)) == 1
    assert (tmp_path / "x.jsonl.bak-test").exists()


def test_jsonl_tolerates_torn_line(tmp_path):
    path = tmp_path / "t.jsonl"
    path.write_text('{"a": 1}\n{"broken...\n{"a": 2}\n')
    log = JsonlLog(path)
    assert [r["a"] for r in log.all()] == [1, 2]


def test_health_formula_reference_parity():
    """score = max(0, 100 - 5*sum(weights) - 2.5*sum(recur-1)) — reference
    health_scoring/app.py:60-91."""
    scorer = HealthScorer()
    p1 = scorer.observe({"app_id": "a", "severity": "medium", "failure_type": "X"})
    # one medium failure: 100 - 3*5 = 85, no recurrence
    assert p1.score == 85.0
    assert p1.failure_rate == 0.1
    assert p1.recurrent_penalty == 0.0
    assert p1.avg_recovery_time_sec == 30.0

    p2 = scorer.observe({"app_id": "a", "severity": "medium", "failure_type": "X"})
    # two mediums: 100 - 30 - 2.5 = 67.5; recurrence penalty 2.5
    assert p2.score == 67.5
    assert p2.recurrent_penalty == 2.5
    assert p2.avg_recovery_time_sec == 55.0
    assert p2.notes["window_failures"] == 2
    assert p2.notes["top_failure"] == "X"

    # unknown severity -> weight 1 (reference default)
    p3 = scorer.observe({"app_id": "b", "severity": "weird", "failure_type": "Y"})
    assert p3.score == 95.0


def test_health_window_eviction():
    scorer = HealthScorer(window_size=3)
    for i in range(5):
        scorer.observe({"app_id": "a", "severity": "high", "failure_type": f"T{i}"})
    win = scorer._windows["a"]
    assert len(win.events) == 3
    assert win.weighted == 21.0  # 3 * 7
    # distinct types within window -> no recurrence
    assert win.recurrent_penalty == 0.0


def test_health_score_floor_at_zero():
    scorer = HealthScorer()
    for _ in range(20):
        p = scorer.observe({"app_id": "a", "severity": "high", "failure_type": "X"})
    assert p.score == 0.0


def test_cli_surface(capsys):
    from kakveda_amd.cli.main import main

    with pytest.raises(SystemExit):
        main(["--help"])
    assert main(["version"]) == 0
    out = capsys.readouterr().out
    assert "kakveda-amd 0" in out

    with pytest.raises(SystemExit):
        main(["logs"])  # missing service arg

    assert main(["status"]) == 0


def test_cli_init(tmp_path, monkeypatch):
    from kakveda_amd.cli.main import main

    monkeypatch.chdir(tmp_path)
    assert main(["init", "--data-dir", str(tmp_path / "d")]) == 0
    env = (tmp_path / ".env").read_text()
    assert "KAKVEDA_JWT_SECRET=" in env
    # refuses overwrite without --force
    assert main(["init"]) == 1
    assert main(["init", "--force"]) == 0


async def test_warning_threshold_hot_reload(tmp_path):
    """warning_policy reads the threshold per request via the hot-reload
    config (reference warning_policy/app.py:21-22)."""
    cfg_path = tmp_path / "config.yaml"
    cfg_path.write_text(
        "failure_matching:\n  similarity_threshold: 0.99\n"
        "hot_reload:\n  enabled: true\n  poll_seconds: 0\n"
    )
    from kakveda_amd.services.cluster import LocalCluster

    cluster = LocalCluster(
        data_dir=str(tmp_path / "data"), config=ConfigStore(path=str(cfg_path))
    )
    await cluster.start()
    prompt = "Summarize and include references even if none are provided."
    from datetime import datetime, timezone

    await cluster.ingest(
        {
            "trace_id": "t1",
            "ts": datetime.now(timezone.utc).isoformat(),
            "app_id": "a",
            "prompt": prompt,
            "response": "text [1] References",
            "tools": [],
            "env": {},
        }
    )
    # threshold 0.99: near-exact match still passes (cos ~ 1.0)
    w = await cluster.warn("a", prompt)
    assert w["references"], w
    # raise impossible threshold -> silent/warn with no refs
    cfg_path.write_text(
        "failure_matching:\n  similarity_threshold: 1.01\n"
        "hot_reload:\n  enabled: true\n  poll_seconds: 0\n"
    )
    os.utime(cfg_path, (time.time() + 5, time.time() + 5))
    w2 = await cluster.warn("a", prompt)
    assert w2["references"] == []
    await cluster.aclose()


async def test_event_bus_drop_on_error(tmp_path):
    """Fan-out is best-effort: an unreachable subscriber is dropped without
    failing the publish (reference event_bus/app.py:48-51)."""
    import httpx

    from kakveda_amd.services import event_bus as eb
    from kakveda_amd.services.wiring import Transport

    tx = Transport(timeout=0.2)
    app = eb.create_app(transport=tx)
    client = httpx.AsyncClient(
        transport=httpx.ASGITransport(app=app), base_url="http://bus"
    )
    await client.post(
        "/subscribe",
        json={"topic": "t", "callback_url": "http://127.0.0.1:59999/nope"},
    )
    resp = await client.post("/publish", json={"topic": "t", "payload": {"x": 1}})
    body = resp.json()
    assert body["ok"] and body["subscribers"] == 1 and body["delivered"] == 0
    # duplicate subscription dedups
    await client.post(
        "/subscribe",
        json={"topic": "t", "callback_url": "http://127.0.0.1:59999/nope"},
    )
    topics = (await client.get("/topics")).json()["topics"]
    assert len(topics["t"]) == 1
    await client.aclose()
    await tx.aclose()


def test_stopwatch_and_percentiles():
    from kakveda_amd.utils import Stopwatch, percentiles

    sw = Stopwatch()
    with sw.span("a"):
        pass
    with sw.span("b"):
        pass
    assert [s["name"] for s in sw.spans] == ["a", "b"]
    assert sw.total_ms() >= 0
    p = percentiles([3.0, 1.0, 2.0, 4.0])
    assert p["p50"] == 3.0 and p["p95"] == 4.0
    assert percentiles([]) == {"p50": 0.0, "p95": 0.0}


async def test_prometheus_metrics_endpoint(tmp_path):
    """Every served app exposes /metrics with request counters
    (exceeds reference parity — SURVEY.md 5.5 notes 'No Prometheus')."""
    import httpx

    from kakveda_amd.core.metrics import instrument
    from kakveda_amd.services import event_bus as eb

    app = eb.create_app()
    assert instrument(app, "event-bus-test")
    client = httpx.AsyncClient(
        transport=httpx.ASGITransport(app=app), base_url="http://x"
    )
    await client.get("/topics")
    metrics = (await client.get("/metrics")).text
    assert "kakveda_http_requests_total" in metrics
    assert 'service="event-bus-test"' in metrics
    assert "kakveda_http_request_seconds" in metrics
    await client.aclose()
