import os, sys, time
import torch
sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from distributed_faiss_amd.hip_engine import HipEngine

def stage(n, nlist, m, d=128, nq=1000, tag=""):
    print(f"=== {tag} n={n} nlist={nlist} m={m} d={d}", flush=True)
    g = torch.Generator(device="cuda").manual_seed(0)
    xb = torch.randn(n, d, generator=g, device="cuda")
    eng = HipEngine(spec={"type": "ivfpq", "dim": d, "metric": 1,
                          "nlist": nlist, "m": m, "nbits": 8,
                          "nprobe": 16, "seed": 1234})
    t0 = time.time(); eng.train_dev(xb); print(f"  train ok {time.time()-t0:.1f}s", flush=True)
    t0 = time.time(); eng.add_dev(xb); torch.cuda.synchronize()
    print(f"  add ok {time.time()-t0:.1f}s ntotal={eng.ntotal}", flush=True)
    t0 = time.time()
    D, I = eng.search_dev(xb[:nq].contiguous(), 10)
    torch.cuda.synchronize()
    print(f"  search ok {time.time()-t0:.3f}s Imin={int(I.min())}", flush=True)

stage(300_000, 256, 8, d=64, tag="A")      # n alone
stage(262_144, 1024, 8, d=64, tag="B")     # nlist=1024, small d
stage(131_073, 1024, 16, d=128, tag="C")   # minimal 2-chunk assign
stage(300_000, 1024, 16, d=128, tag="D")   # the failing combo
print("ALL OK", flush=True)
