import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))
import torch
import numpy as np
from kakveda_amd import ops
from kakveda_amd.encoder.featurizer import featurize_batch
from kakveda_amd.encoder.model import TraceEncoder
ext = ops._require_ext()
dev = "cuda"
D, B, N = 768, 2048, 1_000_000
enc = TraceEncoder(dim=D, hash_dim=1 << 16, seed=1234, device=dev)
texts = [f"intent_tags:intent:citations_required | prompt_hint:synthetic probe {i} explain with sources | tools:t{i % 7} | env_keys:e2e,k{i % 5}" for i in range(1024)]
idx_np, w_np = featurize_batch(texts, hash_dim=enc.hash_dim, max_features=64)
fi = torch.from_numpy(idx_np).to(dev).repeat(2, 1)[:B].contiguous()
fw = torch.from_numpy(w_np).to(dev).repeat(2, 1)[:B].contiguous()
q = enc.encode_features(fi, fw).to(torch.bfloat16)
g2 = torch.Generator(device=dev).manual_seed(55)
c = torch.randn(N, D, generator=g2, device=dev, dtype=torch.float32)
c = (c / c.norm(dim=-1, keepdim=True)).to(torch.bfloat16)
cc, cand, rowthr = ext.emit_counts_probe(q, c, 64)
torch.cuda.synchronize()
cc = cc.long()
bad = (cc > 1000).nonzero().flatten()
print("bad rows:", bad.tolist()[:10], "counts:", cc[bad].tolist()[:10], flush=True)

def dec_u32(encs):
    encs = encs.astype("uint32")
    b = np.where(encs & 0x80000000, encs ^ np.uint32(0x80000000),
                 np.invert(encs)).astype("uint32")
    return b.view(np.float32)

if bad.numel():
    r0 = int(bad[0])
    e = cand[r0, :200].cpu().numpy().astype("uint64")
    encs = (e >> 32).astype("uint32")
    cols = (0x7FFFFFFF - (e & 0xFFFFFFFF)).astype("int64")
    scores = dec_u32(encs)
    print("row", r0, "cols[:16]:", cols[:16].tolist(), flush=True)
    print("claimed scores[:16]:", [round(float(x), 4) for x in scores[:16]], flush=True)
    valid = (cols >= 0) & (cols < N)
    print("valid cols in 200:", int(valid.sum()), flush=True)
    vcols = torch.from_numpy(cols[:64].clip(0, N - 1)).to(dev)
    ts = (q[r0:r0 + 1].float() @ c[vcols].float().t())[0]
    print("true sims there:", [round(float(x), 4) for x in ts[:16]], flush=True)
    print("col dupes in 200:", 200 - len(set(cols.tolist())), flush=True)
    thr_u = int(rowthr[r0].item()) & 0xFFFFFFFF
    print("row floor:", float(dec_u32(np.array([thr_u], dtype="uint64"))[0]), flush=True)
    # a clean row for contrast
    goodr = int((cc <= 1000).nonzero().flatten()[0])
    eg = cand[goodr, :8].cpu().numpy().astype("uint64")
    print("good row", goodr, "count", int(cc[goodr]), "scores:",
          [round(float(x), 4) for x in dec_u32((eg >> 32).astype("uint32"))], flush=True)
