# Pinpoint which part of the serving step breaks HIP stream capture.
# Tries capturing progressively larger step slices on a small index.
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

from distributed_faiss_amd.dist import allgather_shard_topk, merge_gathered  # noqa: E402
from distributed_faiss_amd.hip_engine import HipEngine  # noqa: E402


def try_capture(name, fn):
    try:
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            fn()
        g.replay()
        torch.cuda.synchronize()
        print(f"[capture OK] {name}", flush=True)
        return True
    except Exception as e:
        torch.cuda.synchronize()
        print(f"[capture FAIL] {name}: {str(e).splitlines()[0]}", flush=True)
        return False


def main():
    gen = torch.Generator(device="cuda").manual_seed(0)
    xb = torch.randn(100_000, 64, generator=gen, device="cuda")
    q = torch.randn(2000, 64, generator=gen, device="cuda")
    eng = HipEngine(spec={"type": "ivfpq", "dim": 64, "metric": 1,
                          "nlist": 256, "m": 8, "nbits": 8, "nprobe": 16,
                          "seed": 1})
    eng.train_dev(xb)
    eng.add_dev(xb)
    eng.nprobe = 16
    for _ in range(3):
        D, I = eng.search_dev(q, 10)
        Da, Ia = allgather_shard_topk(D, I)
        merge_gathered(Da, Ia, 10, False)
    torch.cuda.synchronize()

    Ds = torch.empty((2000, 10), dtype=torch.float32, device="cuda")
    Is = torch.empty((2000, 10), dtype=torch.int64, device="cuda")
    try_capture("search_dev into preallocated out", lambda: eng.search_dev(q, 10, D=Ds, I=Is))
    try_capture("search_dev fresh out", lambda: eng.search_dev(q, 10))
    try_capture("coarse-only (dfann_coarse)", lambda: eng.search_preassigned.__self__ and None)

    from distributed_faiss_amd.hip_engine import merge_topk_dev
    D, I = eng.search_dev(q, 10)
    Da, Ia = allgather_shard_topk(D, I)
    torch.cuda.synchronize()
    try_capture("merge_topk_dev only", lambda: merge_topk_dev(Da, Ia, 10, False))
    Dm, slots = merge_topk_dev(Da, Ia, 10, False)
    torch.cuda.synchronize()

    def post():
        s_idx = torch.div(slots, 2000 * 10, rounding_mode="floor")
        local = Ia.reshape(-1)[slots]
        return s_idx * 100_000 + local
    try_capture("post indexing only", post)
    try_capture("merge_gathered", lambda: merge_gathered(Da, Ia, 10, False))

    def full():
        D, I = eng.search_dev(q, 10)
        Da2, Ia2 = allgather_shard_topk(D, I)
        merge_gathered(Da2, Ia2, 10, False)
    try_capture("full step", full)


if __name__ == "__main__":
    main()
