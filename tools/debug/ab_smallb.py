"""Same-box A/B of the small-batch streaming kernel lane remap
(KAKVEDA_SMALLB=4, 8 lanes per row) against the default per-lane-row
kernel (v2/v3), at request-level serving shapes (B=1 and B=4).

Usage: python tools/debug/ab_smallb.py [N ...]   (default: 10M)
"""
import os
import subprocess
import sys

NS = [int(float(a)) for a in sys.argv[1:]] or [10_000_000]

CODE = r"""
import torch, sys
from kakveda_amd import ops
N = int(sys.argv[1])
D, k = 768, 5
for B in (1, 4):
    g = torch.Generator(device='cuda').manual_seed(3)
    q = torch.randn(B, D, generator=g, device='cuda')
    q = (q / q.norm(dim=-1, keepdim=True)).to(torch.bfloat16)
    c = torch.empty(N, D, dtype=torch.bfloat16, device='cuda')
    g2 = torch.Generator(device='cuda').manual_seed(4)
    fill = 1 << 20
    for s in range(0, N, fill):
        e = min(s + fill, N)
        t = torch.randn(e - s, D, generator=g2, device='cuda')
        c[s:e] = (t / t.norm(dim=-1, keepdim=True)).to(torch.bfloat16)
    for _ in range(3):
        s_, i_ = ops.cosine_topk(q, c, k)
    torch.cuda.synchronize()
    ev0, ev1 = torch.cuda.Event(True), torch.cuda.Event(True)
    ts = []
    for _ in range(20):
        ev0.record()
        s_, i_ = ops.cosine_topk(q, c, k)
        ev1.record()
        torch.cuda.synchronize()
        ts.append(ev0.elapsed_time(ev1))
    ts.sort()
    med = ts[len(ts) // 2]
    bw = B and (2.0 * N * D / (med * 1e-3) / 1e12)
    print(f"RES B={B} med_ms={med:.3f} min_ms={ts[0]:.3f} tbps={bw:.2f} "
          f"csum={float(s_.double().sum()):.4f}", flush=True)
    del c
    torch.cuda.empty_cache()
"""

for N in NS:
    res = {}
    # after the default flip: "v2" forces the legacy kernel, "4" is default
    for sel in ("v2", "4"):
        env = dict(os.environ)
        env.pop("KAKVEDA_SMALLB", None)
        if sel == "v2":
            env["KAKVEDA_SMALLB"] = "2"
        r = subprocess.run([sys.executable, "-c", CODE, str(N)], env=env,
                           capture_output=True, text=True, timeout=900)
        if r.returncode != 0:
            print(f"N={N} {sel}: FAILED\n{r.stderr[-2000:]}", flush=True)
            continue
        for line in r.stdout.splitlines():
            if line.startswith("RES"):
                print(f"N={N:>9d} sel={sel:3s} {line}", flush=True)
                b = line.split("B=")[1].split()[0]
                res[(sel, b)] = line
    for b in ("1", "4"):
        a, c = res.get(("v2", b)), res.get(("4", b))
        if a and c:
            m0 = float(a.split("med_ms=")[1].split()[0])
            m1 = float(c.split("med_ms=")[1].split()[0])
            c0 = float(a.split("csum=")[1].split()[0])
            c1 = float(c.split("csum=")[1].split()[0])
            print(f"N={N} B={b}: v4 vs v2 speedup {m0 / m1:.3f}x  csum "
                  f"match: {abs(c0 - c1) < 5e-3} ({c0:.4f} vs {c1:.4f})",
                  flush=True)
