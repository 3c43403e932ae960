# Oracle vs frozen golden fixtures: pins the oracle's numerics across
# refactors (the reference ships no IVF known-answer vectors — SURVEY.md
# §8c — so these frozen outputs ARE the parity anchor for the restatement).
import os

import numpy as np
import pytest

from oracle import make_oracle_engine

HERE = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")


def _configs():
    import importlib.util

    spec = importlib.util.spec_from_file_location(
        "make_golden", os.path.join(HERE, "make_golden.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


MG = _configs()


@pytest.mark.parametrize("name", sorted(MG.CONFIGS))
def test_oracle_matches_golden(name):
    path = os.path.join(HERE, f"{name}.npz")
    if not os.path.exists(path):
        pytest.skip("fixture not generated")
    xb, q = MG.data()
    spec = dict(MG.CONFIGS[name], seed=1234)
    eng = make_oracle_engine(spec)
    eng.train(xb)
    eng.add(xb)
    D_, I_ = eng.search(q, MG.K)
    with np.load(path) as z:
        np.testing.assert_array_equal(I_, z["I"])
        np.testing.assert_array_equal(D_, z["D"])  # bitwise: same code path
        if "centroids" in z.files:
            np.testing.assert_array_equal(eng.centroids, z["centroids"])
