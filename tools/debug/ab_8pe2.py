"""Same-box A/B of the emission kernel hot-sweep rework (EPI_MODE 15,
KAKVEDA_KNN_KERNEL=8pe2) against the production emission kernel (8pe).

The kernel selector is read once per process, so each variant runs in a
subprocess; both run back-to-back on the same box/GPU to stay inside the
±3% box-to-box band.  Shapes mirror bench.py's headline config
(B=2048, D=768, k=5) at the corpus sizes passed on the command line.

Usage: python tools/debug/ab_8pe2.py [N ...]   (default: 1M 10M)
"""
import os
import subprocess
import sys

NS = [int(float(a)) for a in sys.argv[1:]] or [1_000_000, 10_000_000]

CODE = r"""
import torch, time, sys
from kakveda_amd import ops
N = int(sys.argv[1])
B, D, k = 2048, 768, 5
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
for _ in range(10):
    ev0.record()
    s_, i_ = ops.cosine_topk(q, c, k)
    ev1.record()
    torch.cuda.synchronize()
    ts.append(ev0.elapsed_time(ev1))
ts.sort()
med = ts[len(ts) // 2]
tf = 2.0 * B * N * D / (med * 1e-3) / 1e12
print(f"RES med_ms={med:.3f} min_ms={ts[0]:.3f} tf={tf:.0f} csum={float(s_.double().sum()):.4f}")
"""

for N in NS:
    res = {}
    for ksel in ("8pe", "8pe2"):
        env = dict(os.environ)
        env["KAKVEDA_KNN_KERNEL"] = ksel
        r = subprocess.run([sys.executable, "-c", CODE, str(N)], env=env,
                           capture_output=True, text=True, timeout=900)
        if r.returncode != 0:
            print(f"N={N} {ksel}: FAILED\n{r.stderr[-2000:]}", flush=True)
            res[ksel] = None
            continue
        line = [l for l in r.stdout.splitlines() if l.startswith("RES")][0]
        res[ksel] = line
        print(f"N={N:>9d} {ksel:5s} {line}", flush=True)
    if res.get("8pe") and res.get("8pe2"):
        m0 = float(res["8pe"].split("med_ms=")[1].split()[0])
        m1 = float(res["8pe2"].split("med_ms=")[1].split()[0])
        c0 = float(res["8pe"].split("csum=")[1].split()[0])
        c1 = float(res["8pe2"].split("csum=")[1].split()[0])
        print(f"N={N}: 8pe2 vs 8pe speedup {m0 / m1:.3f}x   csum match: "
              f"{abs(c0 - c1) < 1e-2} ({c0:.4f} vs {c1:.4f})", flush=True)
