"""Fixed workload for PMC capture of the emission search at the bench
shape: 3x fused search at B=2048 x 10M x 768 bf16 (prepass + main +
merge). Used to quantify corpus L2 reuse under the XCD-aware dispatch
remap (each corpus chunk is read by 8 row-tile blocks pinned to the
same XCD's L2)."""
import sys
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parents[2]))

from kakveda_amd import ops  # noqa: E402

N, D, B = 10_000_000, 768, 2048
g = torch.Generator(device="cuda").manual_seed(3)
q = torch.randn(B, D, generator=g, device="cuda")
q = (q / q.norm(dim=-1, keepdim=True)).to(torch.bfloat16)
c = torch.empty(N, D, dtype=torch.bfloat16, device="cuda")
g2 = torch.Generator(device="cuda").manual_seed(4)
for s in range(0, N, 1 << 20):
    e = min(s + (1 << 20), N)
    t = torch.randn(e - s, D, generator=g2, device="cuda")
    c[s:e] = (t / t.norm(dim=-1, keepdim=True)).to(torch.bfloat16)
torch.cuda.synchronize()
for _ in range(3):
    s_, i_ = ops.cosine_topk(q, c, 5)
torch.cuda.synchronize()
print("csum", float(s_.double().sum()))
