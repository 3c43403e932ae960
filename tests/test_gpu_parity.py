# GPU parity: the HIP engine against the CPU oracle.
#
# Two tiers:
#  * bit-exact: PQ / SQ8 / SQfp16 scans with FULLY shared state (injected
#    trained artifacts, unambiguous assignment, shared probe lists through
#    search_preassigned) — distances bitwise equal, ids bitwise equal.
#  * tolerance: end-to-end paths where the reduction order legitimately
#    differs (coarse/flat GEMM = MFMA fmaf chain vs BLAS; IVF-Flat 16-lane
#    tree reduction): ids equal on tie-free data, distances <= 1e-4 rel.
import os
import sys

import numpy as np
import pytest

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)

torch = pytest.importorskip("torch")
if not torch.cuda.is_available():
    pytest.skip("needs an MI355X", allow_module_level=True)

from distributed_faiss_amd.hip_engine import (  # noqa: E402
    HipEngine,
    HipProvider,
    merge_topk_dev,
)
from oracle import make_oracle_engine  # noqa: E402

IP, L2 = 0, 1


def _rand(n, d, seed=0):
    return np.random.default_rng(seed).standard_normal((n, d), dtype=np.float32)


def _clustered(nlist, per, d, seed=0, sep=20.0, sigma=0.05):
    """Well-separated clusters: assignment unambiguous for engine & oracle."""
    rng = np.random.default_rng(seed)
    cent = rng.standard_normal((nlist, d)).astype(np.float32) * sep
    lbl = rng.integers(0, nlist, nlist * per)
    x = cent[lbl] + sigma * rng.standard_normal((nlist * per, d)).astype(np.float32)
    return cent, x.astype(np.float32)


# ---------------------------------------------------------------------------
# flat (GEMM + top-k)
# ---------------------------------------------------------------------------


@pytest.mark.parametrize("metric", [IP, L2])
def test_flat_matches_oracle(metric):
    d, n, nq, k = 48, 3000, 33, 10
    xb, q = _rand(n, d, 1), _rand(nq, d, 2)
    spec = {"type": "flat", "dim": d, "metric": metric}
    eng = HipEngine(spec=spec)
    eng.train(xb)
    eng.add(xb[:1700])
    eng.add(xb[1700:])  # multi-add arrival ids
    D, I = eng.search(q, k)
    orc = make_oracle_engine(spec)
    orc.add(xb)
    Do, Io = orc.search(q, k)
    assert (I == Io).mean() > 0.999, f"id mismatch {(I != Io).sum()}"
    np.testing.assert_allclose(D, Do, rtol=1e-4, atol=1e-4)


def test_flat_transpose_guard():
    # asymmetric structured data (guide G9): xb[i] = (i+1) * e_{i % d};
    # an operand/output transpose in the GEMM cannot pass this
    d, n = 16, 64
    xb = np.zeros((n, d), dtype=np.float32)
    for i in range(n):
        xb[i, i % d] = float(i + 1)
    eng = HipEngine(spec={"type": "flat", "dim": d, "metric": IP})
    eng.train(xb)
    eng.add(xb)
    q = np.zeros((d, d), dtype=np.float32)
    np.fill_diagonal(q, 1.0)
    D, I = eng.search(q, 1)
    for j in range(d):
        expect = j + 48 if j + 48 < n else j + 48 - d  # largest i with i%d==j
        best = max(range(j, n, d))
        assert I[j, 0] == best
        assert D[j, 0] == float(best + 1)
        del expect


def test_flat_padding_and_empty():
    d = 16
    eng = HipEngine(spec={"type": "flat", "dim": d, "metric": L2})
    eng.train(np.zeros((1, d), np.float32))
    D, I = eng.search(_rand(3, d), 4)
    assert (I == -1).all()
    eng.add(_rand(2, d, 5))
    D, I = eng.search(_rand(3, d), 4)
    assert (I[:, :2] >= 0).all() and (I[:, 2:] == -1).all()
    assert (D[:, 2:] == np.float32(3.4028235e38)).all()


def test_tie_break_ascending_ids():
    d = 16
    v = _rand(1, d, 7)
    xb = np.repeat(v, 50, axis=0)  # 50 identical vectors
    eng = HipEngine(spec={"type": "flat", "dim": d, "metric": IP})
    eng.train(xb)
    eng.add(xb)
    D, I = eng.search(v, 10)
    np.testing.assert_array_equal(I[0], np.arange(10))


# ---------------------------------------------------------------------------
# IVF end-to-end (engine trains itself; oracle gets the engine's artifacts)
# ---------------------------------------------------------------------------


def _mk_oracle_with_engine_artifacts(eng, spec, xb):
    orc = make_oracle_engine(spec)
    orc.centroids = eng.get_centroids()
    if spec["type"] == "ivfpq":
        orc.codebooks = eng.get_codebooks()
    if spec["type"] == "ivfsq" and spec.get("sq_type") == "8bit":
        vmin, vdiff = eng.get_sq_params()
        orc.vmin, orc.vdiff = vmin, vdiff
        orc.scale = (vdiff / np.float32(255.0)).astype(np.float32)
    orc.is_trained = True
    orc.add(xb)
    orc.nprobe = spec["nprobe"]
    return orc


E2E_SPECS = [
    {"type": "ivf_flat", "nlist": 16, "nprobe": 16},
    {"type": "ivfpq", "nlist": 16, "m": 8, "nprobe": 16},
    {"type": "ivfsq", "nlist": 16, "sq_type": "8bit", "nprobe": 16},
    {"type": "ivfsq", "nlist": 16, "sq_type": "fp16", "nprobe": 16},
]


@pytest.mark.parametrize("base", E2E_SPECS)
@pytest.mark.parametrize("metric", [IP, L2])
def test_ivf_end_to_end_vs_oracle(base, metric):
    d, n, nq, k = 32, 4000, 25, 10
    xb, q = _rand(n, d, 3), _rand(nq, d, 4)
    spec = dict(base, dim=d, metric=metric, seed=1234)
    eng = HipEngine(spec=spec)
    eng.train(xb)
    eng.add(xb)
    D, I = eng.search(q, k)
    orc = _mk_oracle_with_engine_artifacts(eng, spec, xb)
    Do, Io = orc.search(q, k)
    m = (I == Io).mean()
    assert m > 0.98, f"id match {m} ({spec})"
    same = I == Io
    np.testing.assert_allclose(D[same], Do[same], rtol=1e-4, atol=1e-4)


# ---------------------------------------------------------------------------
# bit-exact scans (shared artifacts + shared probes + unambiguous encode)
# ---------------------------------------------------------------------------


def _bitexact_setup(typ, metric, sq_type=None, d=32, nlist=8, per=400):
    cent, xb = _clustered(nlist, per, d, seed=10)
    spec = {"type": typ, "dim": d, "metric": metric, "nlist": nlist,
            "nprobe": nlist, "seed": 1234}
    rng = np.random.default_rng(11)
    kwargs = {}
    if typ == "ivfpq":
        # points = centroid + exact codebook words (+noise << codeword
        # margins) so the per-subspace argmin is unambiguous: engine
        # (GEMM-decomposition encode) and oracle (sequential encode) pick
        # identical codes despite different rounding orders
        spec["m"] = 8
        cb = rng.standard_normal((8, 256, d // 8)).astype(np.float32)
        kwargs["codebooks"] = cb
        lbl = rng.integers(0, nlist, xb.shape[0])
        cw = rng.integers(0, 256, (xb.shape[0], 8))
        dec = np.concatenate([cb[j][cw[:, j]] for j in range(8)], axis=1)
        xb = (cent[lbl] + dec
              + 1e-3 * rng.standard_normal(xb.shape)).astype(np.float32)
    if typ == "ivfsq":
        spec["sq_type"] = sq_type
        if sq_type == "8bit":
            kwargs["vmin"] = np.full(d, -2.0, np.float32)
            kwargs["vdiff"] = np.full(d, 4.0, np.float32)
    eng = HipEngine(spec=spec)
    eng.set_trained(cent, kwargs.get("codebooks"), kwargs.get("vmin"),
                    kwargs.get("vdiff"))
    eng.add(xb)
    orc = make_oracle_engine(spec)
    orc.centroids = cent
    if "codebooks" in kwargs:
        orc.codebooks = kwargs["codebooks"]
    if "vmin" in kwargs:
        orc.vmin = kwargs["vmin"]
        orc.vdiff = kwargs["vdiff"]
        orc.scale = (kwargs["vdiff"] / np.float32(255.0)).astype(np.float32)
    orc.is_trained = True
    orc.add(xb)
    orc.nprobe = nlist
    q = xb[::37][:20] + 0.01 * rng.standard_normal((20, d)).astype(np.float32)
    return eng, orc, q.astype(np.float32)


@pytest.mark.parametrize("metric", [IP, L2])
@pytest.mark.parametrize("cfg", [("ivfpq", None), ("ivfsq", "8bit"),
                                 ("ivfsq", "fp16")])
def test_scan_bitexact(cfg, metric):
    typ, sq = cfg
    eng, orc, q = _bitexact_setup(typ, metric, sq)
    k = 10
    probes, keys = eng.coarse(q, orc.nlist)
    D, I = eng.search_preassigned(q, probes, keys, k)
    Do, Io = orc.search_preassigned(q, probes, keys, k)
    np.testing.assert_array_equal(I, Io)
    np.testing.assert_array_equal(D, Do)  # BITWISE


def test_ivfflat_tolerance_parity():
    # IVF-Flat uses a 16-lane tree reduction (documented deviation):
    # tolerance only
    d, nlist = 32, 8
    cent, xb = _clustered(nlist, 300, d, seed=12)
    spec = {"type": "ivf_flat", "dim": d, "metric": L2, "nlist": nlist,
            "nprobe": nlist, "seed": 1234}
    eng = HipEngine(spec=spec)
    eng.set_trained(cent)
    eng.add(xb)
    orc = make_oracle_engine(spec)
    orc.centroids = cent
    orc.is_trained = True
    orc.add(xb)
    orc.nprobe = nlist
    q = xb[:15] + 0.01 * _rand(15, d, 13)
    D, I = eng.search(q, 10)
    Do, Io = orc.search(q, 10)
    assert (I == Io).mean() > 0.99
    same = I == Io
    np.testing.assert_allclose(D[same], Do[same], rtol=1e-4, atol=1e-4)


@pytest.mark.parametrize("metric", [IP, L2])
def test_coarse_bf16_path(metric):
    # approximate bf16 assign/coarse (spec coarse_bf16): on well-separated
    # clusters the probe sets match fp32 exactly, so end-to-end results are
    # identical — and a wrong MFMA fragment layout would wreck every probe
    d, nlist = 64, 16
    cent, xb = _clustered(nlist, 300, d, seed=21)
    q = xb[::11][:30] + 0.01 * _rand(30, d, 22)
    res = []
    for bf16 in (0, 1):
        spec = {"type": "ivf_flat", "dim": d, "metric": metric,
                "nlist": nlist, "nprobe": 4, "seed": 3, "coarse_bf16": bf16}
        eng = HipEngine(spec=spec)
        eng.set_trained(cent)
        eng.add(xb)
        res.append(eng.search(q, 8))
    np.testing.assert_array_equal(res[0][1], res[1][1])
    np.testing.assert_array_equal(res[0][0], res[1][0])


def test_coarse_bf16_trains():
    # full train/add/search with bf16 k-means assignment: sane recall
    d, nlist = 32, 8
    cent, xb = _clustered(nlist, 400, d, seed=23)
    spec = {"type": "ivfpq", "dim": d, "metric": L2, "nlist": nlist, "m": 8,
            "nprobe": 8, "seed": 3, "coarse_bf16": 1}
    eng = HipEngine(spec=spec)
    eng.train(xb)
    eng.add(xb)
    q = xb[:20] + 0.01 * _rand(20, d, 24)
    D, I = eng.search(q, 5)
    # PQ-quantized near-duplicates can out-tie the query's own id at
    # rank 1 (lower arrival id wins a shared code), so require the self
    # match anywhere in the top-5
    hits = (I == np.arange(20)[:, None]).any(axis=1).mean()
    assert hits > 0.9, f"bf16-trained recall {hits}"


@pytest.mark.parametrize("metric", [IP, L2])
def test_large_k_and_nprobe_bitonic_paths(metric):
    # k=64 / nprobe=32 exercise the LDS-bitonic selection kernels (the
    # register path covers only k<=16)
    d, n = 32, 5000
    xb, q = _rand(n, d, 50), _rand(15, d, 51)
    spec = {"type": "ivf_flat", "dim": d, "metric": metric, "nlist": 32,
            "nprobe": 32, "seed": 9}
    eng = HipEngine(spec=spec)
    eng.train(xb)
    eng.add(xb)
    D, I = eng.search(q, 64)
    orc = _mk_oracle_with_engine_artifacts(eng, spec, xb)
    Do, Io = orc.search(q, 64)
    m = (I == Io).mean()
    assert m > 0.98, f"id match {m}"
    same = I == Io
    np.testing.assert_allclose(D[same], Do[same], rtol=1e-4, atol=1e-4)
    # flat large-k too
    fe = HipEngine(spec={"type": "flat", "dim": d, "metric": metric})
    fe.train(xb)
    fe.add(xb)
    Df, If = fe.search(q, 100)
    fo = make_oracle_engine({"type": "flat", "dim": d, "metric": metric})
    fo.add(xb)
    Dfo, Ifo = fo.search(q, 100)
    assert (If == Ifo).mean() > 0.99
    np.testing.assert_allclose(Df, Dfo, rtol=1e-4, atol=1e-4)


def test_pq_precomputed_table_path():
    # faiss-style term2/term3 PQ-L2 scan vs the direct-LUT path: same
    # math, different rounding — the decomposition carries a cancellation
    # error ~||c||^2 * eps (the known faiss precomputed-table caveat), so
    # the data keeps ||c|| moderate and the comparison allows rank swaps
    # between near-equal neighbors. Artifacts are SHARED (train once).
    d, nlist, m = 64, 16, 8
    cent, xb = _clustered(nlist, 400, d, seed=31, sep=3.0)
    q = xb[::13][:25] + 0.01 * _rand(25, d, 32)
    trainer = HipEngine(spec={"type": "ivfpq", "dim": d, "metric": L2,
                              "nlist": nlist, "m": m, "nprobe": 8, "seed": 5})
    trainer.train(xb)
    cents, cbs = trainer.get_centroids(), trainer.get_codebooks()
    res = []
    for pre in (0, 1):
        spec = {"type": "ivfpq", "dim": d, "metric": L2, "nlist": nlist,
                "m": m, "nprobe": 8, "seed": 5, "pq_precomputed": pre}
        eng = HipEngine(spec=spec)
        eng.set_trained(cents, cbs)
        eng.add(xb)
        res.append(eng.search(q, 10))
    (D0, I0), (D1, I1) = res
    overlap = np.mean([len(set(a) & set(b)) / len(a)
                       for a, b in zip(I0, I1)])
    assert overlap > 0.97, f"pre vs direct id overlap {overlap}"
    same = I0 == I1
    assert same.mean() > 0.9
    np.testing.assert_allclose(D0[same], D1[same], rtol=2e-3, atol=2e-3)


@pytest.mark.parametrize("metric", [IP, L2])
@pytest.mark.parametrize("m", [8, 32])
def test_pq_glut_path_bitexact(m, metric):
    # HBM-LUT scan path (k_pq_lut + GLUT staging) vs the in-kernel LUT
    # build AND vs the oracle: k_pq_lut uses the identical sequential-t
    # accumulation, so all three must agree BITWISE. m=32 also covers
    # the auto-gate (pq_lut_global=-1 turns the path on at m >= 32);
    # ws_mb=1 at m=8 forces the query-chunk loop (multiple k_pq_lut +
    # scan launches per search).
    d, nlist, per = 64, 8, 300
    cent, xb = _clustered(nlist, per, d, seed=41)
    rng = np.random.default_rng(42)
    cb = rng.standard_normal((m, 256, d // m)).astype(np.float32)
    lbl = rng.integers(0, nlist, xb.shape[0])
    cw = rng.integers(0, 256, (xb.shape[0], m))
    dec = np.concatenate([cb[j][cw[:, j]] for j in range(m)], axis=1)
    xb = (cent[lbl] + dec
          + 1e-3 * rng.standard_normal(xb.shape)).astype(np.float32)
    q = xb[::17][:30] + 0.01 * rng.standard_normal((30, d)).astype(np.float32)
    q = q.astype(np.float32)
    res = []
    for glut, ws in ((0, 512), (1, 512), (1, 1)):
        spec = {"type": "ivfpq", "dim": d, "metric": metric, "nlist": nlist,
                "m": m, "nprobe": nlist, "seed": 7, "pq_lut_global": glut,
                "ws_mb": ws}
        eng = HipEngine(spec=spec)
        eng.set_trained(cent, cb)
        eng.add(xb)
        probes, keys = eng.coarse(q, nlist)
        res.append(eng.search_preassigned(q, probes, keys, 10))
    (D0, I0), (D1, I1), (D2, I2) = res
    np.testing.assert_array_equal(I0, I1)
    np.testing.assert_array_equal(D0, D1)  # BITWISE vs in-kernel build
    np.testing.assert_array_equal(I1, I2)
    np.testing.assert_array_equal(D1, D2)  # BITWISE across chunk sizes
    spec_o = {"type": "ivfpq", "dim": d, "metric": metric, "nlist": nlist,
              "m": m, "nprobe": nlist, "seed": 7}
    orc = make_oracle_engine(spec_o)
    orc.centroids, orc.codebooks, orc.is_trained = cent, cb, True
    orc.add(xb)
    orc.nprobe = nlist
    eng = HipEngine(spec=dict(spec_o, pq_lut_global=1))
    eng.set_trained(cent, cb)
    eng.add(xb)
    probes, keys = eng.coarse(q, nlist)
    D, I = eng.search_preassigned(q, probes, keys, 10)
    Do, Io = orc.search_preassigned(q, probes, keys, 10)
    np.testing.assert_array_equal(I, Io)
    np.testing.assert_array_equal(D, Do)  # BITWISE vs oracle


def test_pq_lut_f16_tolerance_path():
    # fp16 ADC tables (pq_lut_f16 — the faiss GpuIndexIVFPQ
    # useFloat16LookupTables equivalent): documented approximation, so
    # tolerance + rank-overlap parity against the exact path on SHARED
    # artifacts, plus bitwise self-determinism.
    d, nlist, m = 64, 8, 32
    cent, xb = _clustered(nlist, 300, d, seed=51)
    rng = np.random.default_rng(52)
    cb = rng.standard_normal((m, 256, d // m)).astype(np.float32)
    lbl = rng.integers(0, nlist, xb.shape[0])
    cw = rng.integers(0, 256, (xb.shape[0], m))
    dec = np.concatenate([cb[j][cw[:, j]] for j in range(m)], axis=1)
    xb = (cent[lbl] + dec
          + 1e-3 * rng.standard_normal(xb.shape)).astype(np.float32)
    q = (xb[::17][:30]
         + 0.01 * rng.standard_normal((30, d))).astype(np.float32)
    res = []
    for f16 in (0, 1):
        spec = {"type": "ivfpq", "dim": d, "metric": L2, "nlist": nlist,
                "m": m, "nprobe": nlist, "seed": 7, "pq_lut_f16": f16}
        eng = HipEngine(spec=spec)
        eng.set_trained(cent, cb)
        eng.add(xb)
        probes, keys = eng.coarse(q, nlist)
        res.append(eng.search_preassigned(q, probes, keys, 10))
        if f16:  # determinism of the approximation path
            D2, I2 = eng.search_preassigned(q, probes, keys, 10)
            np.testing.assert_array_equal(res[-1][0], D2)
            np.testing.assert_array_equal(res[-1][1], I2)
    (D0, I0), (D1, I1) = res
    overlap = np.mean([len(set(a) & set(b)) / len(a)
                       for a, b in zip(I0, I1)])
    assert overlap > 0.97, f"f16 vs exact id overlap {overlap}"
    same = I0 == I1
    assert same.mean() > 0.9
    np.testing.assert_allclose(D0[same], D1[same], rtol=5e-3, atol=5e-3)


def test_chunked_assign_matches_unchunked():
    # ws_mb=1 forces the multi-chunk assign/coarse paths (the round-1
    # negative-OOB regression lived there): results must be identical to
    # the single-chunk engine
    d, n = 32, 30000
    xb, q = _rand(n, d, 40), _rand(20, d, 41)
    specs = [dict(type="ivf_flat", dim=d, metric=L2, nlist=64, nprobe=64,
                  seed=7, ws_mb=w) for w in (1, 512)]
    res = []
    for spec in specs:
        eng = HipEngine(spec=spec)
        eng.train(xb)
        eng.add(xb)
        res.append(eng.search(q, 10))
    np.testing.assert_array_equal(res[0][1], res[1][1])
    np.testing.assert_array_equal(res[0][0], res[1][0])


def test_flat_multichunk_matches_oracle():
    # > one 65536-column chunk: chunk-winner merge path
    d, n = 16, 150000
    xb, q = _rand(n, d, 42), _rand(9, d, 43)
    eng = HipEngine(spec={"type": "flat", "dim": d, "metric": L2})
    eng.train(xb)
    eng.add(xb)
    D, I = eng.search(q, 7)
    orc = make_oracle_engine({"type": "flat", "dim": d, "metric": L2})
    orc.add(xb)
    Do, Io = orc.search(q, 7)
    assert (I == Io).mean() > 0.999
    np.testing.assert_allclose(D, Do, rtol=1e-4, atol=1e-4)


# ---------------------------------------------------------------------------
# decode / persistence / merge
# ---------------------------------------------------------------------------


def test_search_and_reconstruct_gpu():
    eng, orc, q = _bitexact_setup("ivfpq", L2)
    D, I, R = eng.search_and_reconstruct(q, 5)
    Ro = orc.decode_ids(I)
    np.testing.assert_allclose(R, Ro, rtol=1e-6, atol=1e-6)


def test_save_load_roundtrip_gpu(tmp_path):
    for base in E2E_SPECS + [{"type": "flat"}]:
        spec = dict(base, dim=32, metric=L2, seed=1)
        if base["type"] == "flat":
            spec["metric"] = IP
        xb, q = _rand(2000, 32, 20), _rand(8, 32, 21)
        eng = HipEngine(spec=spec)
        eng.train(xb)
        eng.add(xb)
        if "nprobe" in spec:
            eng.nprobe = spec["nprobe"]
        D1, I1 = eng.search(q, 6)
        p = str(tmp_path / f"{base['type']}{base.get('sq_type','')}.dfann")
        eng.save(p)
        eng2 = HipProvider().load(p)
        eng2.nprobe = spec.get("nprobe", 1)
        D2, I2 = eng2.search(q, 6)
        np.testing.assert_array_equal(I1, I2)
        np.testing.assert_array_equal(D1, D2)
        assert eng2.ntotal == eng.ntotal
        # incremental add after load keeps working
        eng2.add(xb[:100])
        assert eng2.ntotal == eng.ntotal + 100
        eng2.search(q, 6)


def test_merge_topk_gpu_matches_reference_semantics():
    # same mock results as the reference merge KAT
    # (tests/test_integration.py:183-194) through the GPU merge kernel
    D1 = np.array([[12.1, 13.2, 13.3, 14.3]], dtype=np.float32)
    D2 = np.array([[8.1, 12.6, 13.1, 17.4]], dtype=np.float32)
    Dall = torch.as_tensor(np.stack([D1, D2])).cuda()
    Iall = torch.zeros((2, 1, 4), dtype=torch.int64, device="cuda")
    Dm, Im = merge_topk_dev(Dall, Iall, 4, maximize=False)
    np.testing.assert_allclose(Dm.cpu().numpy()[0], [8.1, 12.1, 12.6, 13.1],
                               rtol=1e-6)
    # slots: shard*nq*k + q*k + j with nq=1,k=4
    np.testing.assert_array_equal(Im.cpu().numpy()[0], [4, 0, 5, 6])
    Dm, Im = merge_topk_dev(Dall, Iall, 4, maximize=True)
    np.testing.assert_allclose(Dm.cpu().numpy()[0], [-17.4, -14.3, -13.3, -13.2],
                               rtol=1e-6)
    np.testing.assert_array_equal(Im.cpu().numpy()[0], [7, 3, 2, 1])


def test_golden_fixture_parity_gpu():
    # engine with ORACLE-trained artifacts reproduces the committed golden
    # (D, I) for the bit-exact families
    import importlib.util

    here = os.path.join(os.path.dirname(os.path.abspath(__file__)), "golden")
    s = importlib.util.spec_from_file_location(
        "make_golden", os.path.join(here, "make_golden.py"))
    MG = importlib.util.module_from_spec(s)
    s.loader.exec_module(MG)
    xb, q = MG.data()
    for name in ["ivfsq8_l2", "ivfsqf_l2", "ivfpq_l2"]:
        path = os.path.join(here, f"{name}.npz")
        if not os.path.exists(path):
            pytest.skip("fixtures missing")
        z = np.load(path)
        spec = dict(MG.CONFIGS[name], seed=1234)
        eng = HipEngine(spec=spec)
        eng.set_trained(z["centroids"],
                        z["codebooks"] if "codebooks" in z.files else None,
                        z["vmin"] if "vmin" in z.files else None,
                        z["vdiff"] if "vdiff" in z.files else None)
        eng.add(xb)
        eng.nprobe = spec["nprobe"]
        D, I = eng.search(q, MG.K)
        m = (I == z["I"]).mean()
        assert m > 0.98, f"{name}: id match vs golden {m}"
        same = I == z["I"]
        np.testing.assert_allclose(D[same], z["D"][same], rtol=1e-4, atol=1e-4)


def test_client_server_sharded_equals_single_gpu(tmp_path):
    # the reference's strongest invariant (tests/test_integration.py:205-265)
    # on the HIP backend: sharded over 2 in-process GPU shards == single
    from distributed_faiss_amd import IndexCfg, IndexClient, IndexServer

    prov = HipProvider()
    s1 = IndexServer(0, str(tmp_path / "a"), provider=prov)
    s2 = IndexServer(1, str(tmp_path / "a"), provider=prov)
    single = IndexServer(0, str(tmp_path / "b"), provider=prov)
    cli = IndexClient(servers=[s1, s2])
    sc = IndexClient(servers=[single])
    cfg = IndexCfg(index_builder_type="flat", dim=32)
    cli.create_index("x", cfg)
    sc.create_index("x", cfg)
    rng = np.random.default_rng(30)
    for _ in range(4):
        emb = rng.random((500, 32), dtype=np.float32)
        meta = [f"m{i}" for i in range(500)]
        cli.add_index_data("x", emb, meta, train_async_if_triggered=False)
        sc.add_index_data("x", emb, meta, train_async_if_triggered=False)
    cli.sync_train("x")
    sc.sync_train("x")
    import time

    for c in (cli, sc):
        for _ in range(200):
            from distributed_faiss_amd import IndexState

            if c.get_state("x") == IndexState.TRAINED:
                break
            time.sleep(0.05)
    q = rng.random((8, 32), dtype=np.float32)
    Da, Ma = cli.search(q, 5, "x")
    Ds, Ms = sc.search(q, 5, "x")
    assert (Da == Ds).all()
    assert Ma == Ms


def test_client_search_dev_matches_search(tmp_path):
    # the bench's timed step (IndexClient.search_dev: device-resident
    # local search -> all-gather -> dfann_merge_topk) must agree with the
    # full reference API (IndexClient.search: host arrays + metadata),
    # which wraps the same pipeline — incl. the quirk-2 dot negation
    from distributed_faiss_amd import IndexCfg, IndexClient, IndexServer

    for metric_name in ("dot", "l2"):
        prov = HipProvider()
        srv = IndexServer(0, str(tmp_path / metric_name), provider=prov)
        cli = IndexClient(servers=[srv])
        cfg = (IndexCfg(index_builder_type="flat", dim=32, metric="dot")
               if metric_name == "dot"
               else IndexCfg(faiss_factory="Flat", dim=32, metric="l2"))
        cli.create_index("x", cfg)
        cli.cfg = cfg
        rng = np.random.default_rng(31)
        emb = rng.random((800, 32), dtype=np.float32)
        cli.add_index_data("x", emb, [("m", i) for i in range(800)],
                           train_async_if_triggered=False)
        cli.sync_train("x")
        import time

        from distributed_faiss_amd import IndexState

        for _ in range(200):
            if (cli.get_state("x") == IndexState.TRAINED
                    and cli.get_ntotal("x") == 800):
                break
            time.sleep(0.05)
        q = rng.random((6, 32), dtype=np.float32)
        Dh, Mh = cli.search(q, 5, "x")
        qt = torch.as_tensor(q).cuda()
        Dd, s_idx, local = cli.search_dev(qt, 5, "x")
        np.testing.assert_allclose(np.asarray(Dh), Dd.cpu().numpy(),
                                   rtol=1e-6, atol=1e-6)
        # winners map to the same metadata: shard 0, id -> ("m", id)
        loc = local.cpu().numpy()
        for i in range(6):
            for j in range(5):
                assert Mh[i][j] == ("m", int(loc[i, j]))
