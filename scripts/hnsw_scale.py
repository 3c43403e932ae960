# hnsw 1M-scale build + serve rates (BASELINE.md evidence)
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from distributed_faiss_amd.hip_engine import HipEngine
from oracle.core import OracleHNSWSearch

def clustered(n, d, seed=0, centers=4096, sigma=0.3):
    crng = np.random.default_rng(1000)
    cent = crng.standard_normal((centers, d)).astype(np.float32) * 3.0
    rng = np.random.default_rng(seed)
    lbl = rng.integers(0, centers, n)
    return (cent[lbl] + sigma * rng.standard_normal((n, d)).astype(np.float32)).astype(np.float32)

import os
n = int(os.environ.get("HN_N", "1000000"))
refine = int(os.environ.get("HN_REFINE", "1"))
d, M, efc = 64, 32, 100
xb = clustered(n, d, seed=7)
spec = {"type": "hnswsq", "dim": d, "metric": 1, "m": M,
        "ef_construction": efc, "nprobe": 64, "seed": 11,
        "hnsw_refine": refine}
eng = HipEngine(spec=spec)
eng.train(xb[:100_000])
t0 = time.time()
eng.add(xb)
tb = time.time() - t0
print(f"build {n} pts in {tb:.1f}s ({n/tb/1e3:.0f}k pts/s)")
vmin, vdiff = eng.get_sq_params()
codes = OracleHNSWSearch.encode(xb, vmin, vdiff)
scale = vdiff.astype(np.float32)/np.float32(255.0)
dec = vmin[None,:] + (codes.astype(np.float32)+0.5)*scale[None,:]
q = clustered(10_000, d, seed=42)
dec_t = torch.as_tensor(dec).cuda(); q_t = torch.as_tensor(q).cuda()
gt = None
# chunked GT
best = torch.full((10_000, 10), float("inf"), device="cuda"); bidx = torch.zeros((10_000,10), dtype=torch.int64, device="cuda")
for b0 in range(0, n, 250_000):
    d2 = torch.cdist(q_t, dec_t[b0:b0+250_000])**2
    v, i = torch.topk(d2, 10, largest=False)
    cv = torch.cat([best, v], 1); ci = torch.cat([bidx, i+b0], 1)
    best, sel = torch.topk(cv, 10, largest=False); bidx = torch.gather(ci, 1, sel)
gt = bidx.cpu().numpy()
for ef in (32, 64, 128):
    eng.nprobe = ef
    for _ in range(2): eng.search(q[:1000], 10)  # warm
    t0 = time.time(); D, I = eng.search(q, 10); dt = time.time()-t0
    hits = np.mean([len(set(I[i]) & set(gt[i]))/10.0 for i in range(10_000)])
    print(f"ef={ef}: recall@10={hits:.3f} QPS={10_000/dt:.0f}")
