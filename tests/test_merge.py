# Client merge semantics.
#
# test_result_aggregation_kat ports the reference's ONE golden-vector test
# verbatim (reference tests/test_integration.py:181-203) — the hard-coded
# two-shard (D, I) inputs and winner assertions pin merge ordering and the
# dot-negation convention (reference client.py:291-294, quirk 2).
import numpy as np

from distributed_faiss_amd.client import IndexClient
from oracle import aggregate_results


def _mock_results():
    # values from reference tests/test_integration.py:183-194 (the KAT inputs)
    return [
        (
            np.array([[12.1, 13.2, 13.3, 14.3]], dtype=np.float32),
            [[1465, 1460, 443197, 1340]],
            None,
        ),
        (
            np.array([[8.1, 12.6, 13.1, 17.4]], dtype=np.float32),
            [[0, 14, 3, 1]],
            None,
        ),
    ]


def test_result_aggregation_kat():
    D, i_minimize = IndexClient._aggregate_results(_mock_results(), 4, 1, False, False)
    _, i_maximize = IndexClient._aggregate_results(_mock_results(), 4, 1, True, False)
    assert i_maximize != i_minimize
    assert i_minimize[0][0] == 0  # the smallest distance
    assert D[0][0] < D[0][1]
    assert i_maximize[0][0] == 1  # the largest distance
    assert 0 in i_minimize[0]


def test_merge_exact_order_minimize():
    D, meta = IndexClient._aggregate_results(_mock_results(), 4, 1, False, False)
    assert meta[0] == [0, 1465, 14, 3]
    np.testing.assert_allclose(D[0], [8.1, 12.1, 12.6, 13.1])


def test_merge_exact_order_maximize_negated():
    # dot: returned scores are NEGATED (quirk 2) and sorted ascending
    D, meta = IndexClient._aggregate_results(_mock_results(), 4, 1, True, False)
    assert meta[0] == [1, 1340, 443197, 1460]
    np.testing.assert_allclose(D[0], [-17.4, -14.3, -13.3, -13.2])


def test_oracle_merge_matches_client():
    shard_D = [m[0] for m in _mock_results()]
    shard_meta = [m[1] for m in _mock_results()]
    for maximize in (False, True):
        Dc, mc = IndexClient._aggregate_results(_mock_results(), 4, 1, maximize, False)
        Do, mo = aggregate_results(shard_D, shard_meta, 4, maximize)
        np.testing.assert_array_equal(Dc, Do)
        assert mc == mo


def test_merge_padding_loses():
    # a shard with fewer than k results pads D=+FLT_MAX / id slots with
    # meta None (faiss heap-init convention); padded slots must lose
    fmax = np.float32(3.4028235e38)
    results = [
        (
            np.array([[1.0, fmax, fmax]], dtype=np.float32),
            [["a", None, None]],
            None,
        ),
        (
            np.array([[2.0, 3.0, fmax]], dtype=np.float32),
            [["b", "c", None]],
            None,
        ),
    ]
    D, meta = IndexClient._aggregate_results(results, 3, 1, False, False)
    assert meta[0] == ["a", "b", "c"]
    np.testing.assert_allclose(D[0], [1.0, 2.0, 3.0])


def test_merge_multi_query():
    rng = np.random.default_rng(7)
    nq, k, S = 5, 4, 3
    shard_D = [rng.random((nq, k)).astype(np.float32) for _ in range(S)]
    shard_meta = [[[f"s{s}q{i}k{j}" for j in range(k)] for i in range(nq)] for s in range(S)]
    results = [(shard_D[s], shard_meta[s], None) for s in range(S)]
    D, meta = IndexClient._aggregate_results(results, k, nq, False, False)
    # brute force reference
    for i in range(nq):
        allpairs = sorted(
            (float(shard_D[s][i][j]), s, j) for s in range(S) for j in range(k)
        )[:k]
        np.testing.assert_allclose(D[i], [p[0] for p in allpairs], rtol=0, atol=0)
        assert meta[i] == [shard_meta[p[1]][i][p[2]] for p in allpairs]
