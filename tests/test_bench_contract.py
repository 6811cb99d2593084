"""Driver-contract test for bench.py: launch it exactly the way the
driver does (torch.distributed.run, one process per "GPU", 127.0.0.1
rendezvous) in CPU dry-run mode and validate the single JSON line it
prints against the contract fields (see BASELINE.json / the bench
contract in bench.py's docstring)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(cmd, timeout=240):
    return subprocess.run(
        cmd, cwd=REPO, capture_output=True, text=True, timeout=timeout
    )


def _last_json(stdout: str) -> dict:
    lines = [l for l in stdout.strip().splitlines() if l.startswith("{")]
    assert lines, f"no JSON line in output: {stdout[-2000:]}"
    return json.loads(lines[-1])


def _check_contract(out: dict, n_gpus: int, steps: int, warmup: int):
    assert out["metric"] == "preflight_warning_lookups_per_sec"
    assert out["unit"] == "lookups/s"
    assert out["n_gpus"] == n_gpus
    assert out["steps"] == steps and out["warmup"] == warmup
    assert out["higher_is_better"] is True
    assert out["scaling"] == "strong"
    assert out["value"] > 0 and out["ms_per_step"] > 0
    cfg = out["config"]
    assert cfg["model"] == "gfkb-cosine-knn-768d"
    assert cfg["top_k"] == 5
    assert out["data"] == "synthetic"


def test_bench_single_process_cpu_dryrun():
    r = _run(
        [
            sys.executable,
            "bench.py",
            "--steps",
            "2",
            "--warmup",
            "1",
            "--entries",
            "20000",
            "--batch",
            "64",
        ]
    )
    assert r.returncode == 0, r.stderr[-2000:]
    _check_contract(_last_json(r.stdout), n_gpus=1, steps=2, warmup=1)


def test_bench_torchrun_world2_cpu_dryrun():
    """The exact driver launch: torch.distributed.run, nproc 2, master
    127.0.0.1. On CPU this uses gloo + the same ShardedStore all-gather
    merge the RCCL path uses."""
    r = _run(
        [
            sys.executable,
            "-m",
            "torch.distributed.run",
            "--nnodes=1",
            "--nproc-per-node",
            "2",
            "--master-addr",
            "127.0.0.1",
            "--master-port",
            "29531",
            "bench.py",
            "--gpus",
            "2",
            "--steps",
            "2",
            "--warmup",
            "1",
            "--entries",
            "20000",
            "--batch",
            "64",
        ],
        timeout=300,
    )
    assert r.returncode == 0, (r.stdout[-1500:], r.stderr[-1500:])
    out = _last_json(r.stdout)
    _check_contract(out, n_gpus=2, steps=2, warmup=1)
    assert out["config"]["parallelism"] == "shard2"
