"""Service-level serving throughput over REAL HTTP with process isolation.

The in-process harness (serve_latency.py) shares one event loop between
the server and all clients, which caps the measured number at the
client-side overhead (ROUND3.md: 644-860 req/s while the engine+batcher
layer sustains 3,694 req/s at 10M).  This benchmark removes that
artifact: uvicorn serves the warning-policy composition over TCP in its
own process(es), and load comes from separate client processes, each
driving C concurrent keep-alive connections.

  python benchmarks/serve_http_bench.py --entries 50000 --seconds 10 \
      --procs 4 --conns 16 --workers 2

Prints one JSON line: total rps, merged p50/p99 (exact, from pooled
per-request latencies), per-worker config.
"""

from __future__ import annotations

import argparse
import json
import os
import signal
import subprocess
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

BODY = {
    "app_id": "app-A",
    "prompt": "please provide references for why the sky is blue.",
    "tools": [],
    "env": {"e2e": "1", "source": "none"},
}


def client_main(url: str, seconds: float, conns: int) -> None:
    import asyncio

    import httpx

    async def run() -> None:
        lats: list[float] = []
        matched = 0
        errors = 0
        async with httpx.AsyncClient(
            base_url=url,
            timeout=120.0,
            limits=httpx.Limits(max_connections=conns,
                                max_keepalive_connections=conns),
        ) as cli:
            # concurrent warmup (sequential per-connection warmup at high
            # fan-in takes minutes against a saturated server)
            await asyncio.gather(
                *(cli.post("/warn", json=BODY) for _ in range(min(conns, 32))),
                return_exceptions=True)
            stop_at = time.perf_counter() + seconds

            sheds = 0

            async def loop() -> None:
                nonlocal matched, errors, sheds
                while time.perf_counter() < stop_at:
                    t0 = time.perf_counter()
                    try:
                        r = await cli.post("/warn", json=BODY)
                        if r.status_code == 200:
                            lats.append(time.perf_counter() - t0)
                            if r.json().get("references"):
                                matched += 1
                        elif r.status_code == 503:
                            # honour admission control: back off as a
                            # well-behaved client would
                            sheds += 1
                            await asyncio.sleep(
                                float(r.headers.get("retry-after", "1")))
                        else:
                            errors += 1
                    except Exception:
                        errors += 1

            t0 = time.perf_counter()
            res = await asyncio.gather(*(loop() for _ in range(conns)),
                                       return_exceptions=True)
            errors += sum(1 for x in res if isinstance(x, Exception))
            wall = time.perf_counter() - t0
        print(json.dumps({
            "count": len(lats),
            "wall": wall,
            "matched": matched,
            "errors": errors,
            "sheds": sheds,
            "lats_ms": [round(x * 1000, 3) for x in lats],
        }))

    asyncio.run(run())


def main() -> int:
    ap = argparse.ArgumentParser()
    ap.add_argument("--entries", type=int, default=50_000)
    ap.add_argument("--dim", type=int, default=768)
    ap.add_argument("--seconds", type=float, default=10.0)
    ap.add_argument("--procs", type=int, default=4, help="client processes")
    ap.add_argument("--conns", type=int, default=16,
                    help="concurrent connections per client process")
    ap.add_argument("--workers", type=int, default=1, help="uvicorn workers")
    ap.add_argument("--port", type=int, default=8199)
    ap.add_argument("--client", action="store_true", help=argparse.SUPPRESS)
    ap.add_argument("--url", default=None, help=argparse.SUPPRESS)
    args = ap.parse_args()

    if args.client:
        client_main(args.url, args.seconds, args.conns)
        return 0

    env = dict(os.environ)
    env["KAKVEDA_BENCH_ENTRIES"] = str(args.entries)
    env["KAKVEDA_BENCH_DIM"] = str(args.dim)
    env.setdefault("PYTHONPATH", ROOT)
    url = f"http://127.0.0.1:{args.port}"
    server = subprocess.Popen(
        [sys.executable, "-m", "uvicorn", "serve_http_app:create_app",
         "--factory", "--app-dir", os.path.join(ROOT, "benchmarks"),
         "--host", "127.0.0.1", "--port", str(args.port),
         "--workers", str(args.workers), "--log-level", "warning"],
        env=env, cwd=ROOT, start_new_session=True,
    )
    try:
        # readiness: corpus build can take minutes at 10M x workers
        import httpx

        deadline = time.time() + 600
        while True:
            if server.poll() is not None:
                print("server exited early", file=sys.stderr)
                return 1
            try:
                r = httpx.post(f"{url}/warn", json=BODY, timeout=5.0)
                if r.status_code == 200:
                    break
            except Exception:
                pass
            if time.time() > deadline:
                print("server never became ready", file=sys.stderr)
                return 1
            time.sleep(1.0)

        clients = [
            subprocess.Popen(
                [sys.executable, os.path.abspath(__file__), "--client",
                 "--url", url, "--seconds", str(args.seconds),
                 "--conns", str(args.conns)],
                stdout=subprocess.PIPE, text=True, env=env, cwd=ROOT,
            )
            for _ in range(args.procs)
        ]
        lats: list[float] = []
        total = matched = errors = sheds = 0
        wall = 0.0
        for c in clients:
            out, _ = c.communicate(timeout=args.seconds + 600)
            if not out.strip():
                print("client produced no output (crashed)", file=sys.stderr)
                errors += 1
                continue
            d = json.loads(out.splitlines()[-1])
            total += d["count"]
            matched += d["matched"]
            errors += d.get("errors", 0)
            sheds += d.get("sheds", 0)
            wall = max(wall, d["wall"])
            lats.extend(d["lats_ms"])
        lats.sort()
        print(json.dumps({
            "metric": "serve_warn_http",
            "unit": "req/s",
            "value": round(total / wall, 1),
            "requests": total,
            "matched": matched,
            "errors": errors,
            "sheds": sheds,
            "p50_ms": lats[len(lats) // 2] if lats else None,
            "p99_ms": lats[max(0, int(len(lats) * 0.99) - 1)] if lats else None,
            "entries": args.entries,
            "procs": args.procs,
            "conns": args.conns,
            "uvicorn_workers": args.workers,
            "transport": "tcp",
            "data": "synthetic",
        }))
        return 0
    finally:
        try:
            os.killpg(server.pid, signal.SIGTERM)
        except Exception:
            server.terminate()
        try:
            server.wait(timeout=20)
        except Exception:
            server.kill()
    return 0


if __name__ == "__main__":
    sys.exit(main())
