# hnsw graph-quality diagnostic: recall vs ef + degree stats
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch
from distributed_faiss_amd.hip_engine import HipEngine
from oracle.core import OracleHNSWSearch

def clustered(n, d, seed=0, centers=64, sigma=0.3):
    crng = np.random.default_rng(1000)  # centers SHARED across calls
    cent = crng.standard_normal((centers, d)).astype(np.float32) * 3.0
    rng = np.random.default_rng(seed)
    lbl = rng.integers(0, centers, n)
    return (cent[lbl] + sigma * rng.standard_normal((n, d)).astype(np.float32)).astype(np.float32)

n, d, M, efc = 50_000, 64, 32, 100
xb = clustered(n, d, seed=7)
spec = {"type": "hnswsq", "dim": d, "metric": 1, "m": M,
        "ef_construction": efc, "nprobe": 64, "seed": 11}
eng = HipEngine(spec=spec)
eng.train(xb[:12500]); 
import time; t0=time.time()
for s in range(0, n, 17000): eng.add(xb[s:s+17000])
print(f"build {time.time()-t0:.1f}s")
g = eng.hnsw_dump()
print("deg0 mean/min/max:", g["cnt0"].mean(), g["cnt0"].min(), g["cnt0"].max(),
      "isolated:", int((g["cnt0"] == 0).sum()), "maxlevel:", g["maxlevel"])
vmin, vdiff = eng.get_sq_params()
codes = OracleHNSWSearch.encode(xb, vmin, vdiff)
scale = vdiff.astype(np.float32) / np.float32(255.0)
dec = vmin[None, :] + (codes.astype(np.float32) + 0.5) * scale[None, :]
q = clustered(200, d, seed=42)
d2 = torch.cdist(torch.as_tensor(q).cuda(), torch.as_tensor(dec).cuda()) ** 2
gt = torch.topk(d2, 10, largest=False).indices.cpu().numpy()
# level-0 connectivity from the entry (host BFS on the dump)
from collections import deque
adj = g["nbr0"]; cnt = g["cnt0"]
seen = np.zeros(n, dtype=bool)
dq = deque([int(g["entry"])]); seen[int(g["entry"])] = True
while dq:
    u = dq.popleft()
    for v in adj[u, :cnt[u]]:
        if not seen[v]:
            seen[v] = True; dq.append(int(v))
print(f"level-0 reachable from entry: {seen.mean()*100:.1f}%")
for ef in (32, 64, 128, 256):
    eng.nprobe = ef
    t0=time.time(); D, I = eng.search(q, 10); dt=time.time()-t0
    hits = np.mean([len(set(I[i]) & set(gt[i])) / 10.0 for i in range(len(q))])
    print(f"ef={ef}: recall@10={hits:.3f} ({dt*1000:.0f} ms/200q)")
