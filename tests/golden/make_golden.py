# Golden-fixture generator (the reference ships no
# known-answer vectors for IVF numerics, so WE freeze them from the
# oracle). Run from the repo root:  python3 tests/golden/make_golden.py
# Commits small .npz fixtures pinning oracle outputs (trained artifacts +
# search results) for fixed seeds; tests/test_golden.py replays them.
import os
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, REPO)

from oracle import make_oracle_engine  # noqa: E402

HERE = os.path.dirname(os.path.abspath(__file__))

D, N, NQ, K = 32, 2000, 16, 10

CONFIGS = {
    "flat_ip": {"type": "flat", "dim": D, "metric": 0},
    "flat_l2": {"type": "flat", "dim": D, "metric": 1},
    "ivfflat_l2": {"type": "ivf_flat", "dim": D, "metric": 1, "nlist": 8, "nprobe": 8},
    "ivfflat_ip": {"type": "ivf_flat", "dim": D, "metric": 0, "nlist": 8, "nprobe": 8},
    "ivfpq_l2": {"type": "ivfpq", "dim": D, "metric": 1, "nlist": 8, "m": 8, "nprobe": 8},
    "ivfpq_ip": {"type": "ivfpq", "dim": D, "metric": 0, "nlist": 8, "m": 8, "nprobe": 8},
    "ivfsq8_l2": {"type": "ivfsq", "dim": D, "metric": 1, "nlist": 8,
                  "sq_type": "8bit", "nprobe": 8},
    "ivfsq8_ip": {"type": "ivfsq", "dim": D, "metric": 0, "nlist": 8,
                  "sq_type": "8bit", "nprobe": 8},
    "ivfsqf_l2": {"type": "ivfsq", "dim": D, "metric": 1, "nlist": 8,
                  "sq_type": "fp16", "nprobe": 8},
}


def data():
    rng = np.random.default_rng(42)
    xb = rng.standard_normal((N, D), dtype=np.float32)
    q = rng.standard_normal((NQ, D), dtype=np.float32)
    return xb, q


def main():
    xb, q = data()
    for name, spec in CONFIGS.items():
        eng = make_oracle_engine(dict(spec, seed=1234))
        eng.train(xb)
        eng.add(xb)
        D_, I_ = eng.search(q, K)
        out = {"D": D_, "I": I_}
        if spec["type"] != "flat":
            out["centroids"] = eng.centroids
        if spec["type"] == "ivfpq":
            out["codebooks"] = eng.codebooks
        if spec["type"] == "ivfsq" and spec.get("sq_type") == "8bit":
            out["vmin"] = eng.vmin
            out["vdiff"] = eng.vdiff
        path = os.path.join(HERE, f"{name}.npz")
        with open(path, "wb") as f:
            np.savez(f, **out)
        print("wrote", path, "D[0,:3] =", D_[0, :3])


if __name__ == "__main__":
    main()
