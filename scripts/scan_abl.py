# Scan-kernel ablation: where do the microseconds go?
# Hypotheses: (a) list-length skew (clustered vs uniform data),
# (b) extraction cost (k=1 vs k=10), (c) block-count scaling (nprobe).
# Run on the GPU box: python scripts/scan_abl.py
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from distributed_faiss_amd.hip_engine import HipEngine  # noqa: E402


def build(n, d, nlist, m, clustered):
    g = torch.Generator(device="cuda").manual_seed(0)
    if clustered:
        centers = torch.randn(10000, d, generator=g, device="cuda")
        lbl = torch.randint(0, 10000, (n,), generator=g, device="cuda")
        xb = centers[lbl] + 0.5 * torch.randn(n, d, generator=g, device="cuda")
    else:
        xb = torch.randn(n, d, generator=g, device="cuda")
    eng = HipEngine(spec={"type": "ivfpq", "dim": d, "metric": 1,
                          "nlist": nlist, "m": m, "nbits": 8, "nprobe": 1,
                          "seed": 1234})
    eng.train_dev(xb)
    eng.add_dev(xb)
    torch.cuda.synchronize()
    return eng, xb


def probe(eng, q, k, nprobe, tag):
    eng.nprobe = nprobe
    eng.set_timing(True)
    eng.get_timing()
    for _ in range(5):
        eng.search_dev(q, k)
    torch.cuda.synchronize()
    t = eng.get_timing()
    eng.set_timing(False)
    gbs = (t["scan_bytes"] / 1e9) / (t["scan_ms"] / 1e3) if t["scan_ms"] else 0
    print(f"{tag:34s} k={k:3d} nprobe={nprobe:3d} "
          f"scan={t['scan_ms']/t['scan_launches']:7.3f}ms/launch "
          f"rows/launch={t['scan_rows']/t['scan_launches']:10.0f} "
          f"{gbs:7.0f} GB/s  gemm={t['gemm_ms']:6.2f}ms merge={t['merge_ms']:6.2f}ms",
          flush=True)


def main():
    n, d, m = 1_000_000, 128, 16
    q = torch.randn(10000, d, generator=torch.Generator(device="cuda").manual_seed(9),
                    device="cuda")
    for clustered in (True, False):
        eng, xb = build(n, d, 1024, m, clustered)
        tag = "clustered" if clustered else "uniform"
        for k in (1, 10):
            for nprobe in (1, 4, 16):
                probe(eng, q, k, nprobe, f"{tag}")
        del eng, xb
        torch.cuda.empty_cache()


if __name__ == "__main__":
    main()
