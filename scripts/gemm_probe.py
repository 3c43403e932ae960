# Measure the distance-GEMM rate (fp32 MFMA vs bf16 MFMA path) at the
# config-4 shapes, via the engine's own event timing.
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from distributed_faiss_amd.hip_engine import HipEngine  # noqa: E402


def probe(d, nlist, nq, bf16):
    g = torch.Generator(device="cuda").manual_seed(0)
    cent = torch.randn(nlist, d, generator=g, device="cuda")
    q = torch.randn(nq, d, generator=g, device="cuda")
    eng = HipEngine(spec={"type": "ivf_flat", "dim": d, "metric": 1,
                          "nlist": nlist, "nprobe": 1, "seed": 1,
                          "coarse_bf16": bf16})
    eng.set_trained(cent.cpu().numpy())
    eng.set_timing(True)
    eng.get_timing()
    import ctypes
    lib = eng.lib
    probes = torch.empty(nq, 1, dtype=torch.int32, device="cuda")
    keys = torch.empty(nq, 1, dtype=torch.float32, device="cuda")
    stream = ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
    for _ in range(5):
        rc = lib.dfann_coarse(eng.h, nq, ctypes.c_void_p(q.data_ptr()), 1,
                              ctypes.c_void_p(probes.data_ptr()),
                              ctypes.c_void_p(keys.data_ptr()), stream)
        assert rc == 0
    torch.cuda.synchronize()
    t = eng.get_timing()
    tf = (t["gemm_flops"] / 1e12) / (t["gemm_ms"] / 1e3) if t["gemm_ms"] else 0
    print(f"d={d:4d} nlist={nlist:6d} nq={nq:6d} bf16={bf16} "
          f"gemm={t['gemm_ms']:.2f}ms {tf:7.1f} TF", flush=True)
    eng.set_timing(False)


for bf16 in (0, 1):
    probe(768, 4096, 65536, bf16)
    probe(768, 65536, 8192, bf16)
    probe(128, 1024, 65536, bf16)

# round-2: BK=32 3-buffer counted-vmcnt ring A/B at the coarse shapes
print("--- p3 ring A/B (bf16) ---", flush=True)
for p3 in ("0", "1"):
    os.environ["DFANN_GEMM_P3"] = p3
    print(f"DFANN_GEMM_P3={p3}")
    probe(768, 16384, 10000, 1)   # headline coarse (nlist=16384)
    probe(768, 65536, 8192, 1)    # old headline coarse (VERDICT item 7 shape)
    probe(128, 16384, 65536, 1)   # config-5 assign shape
