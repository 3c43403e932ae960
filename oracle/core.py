# ORACLE core — numpy restatement of the faiss-cpu arithmetic consumed by
# the reference at distributed_faiss/index.py:217 (train), :425 (add),
# :257 (search), :255 (search_and_reconstruct), :350 (reconstruct_n) and
# the client merge at distributed_faiss/client.py:29-54,265-310.
# See oracle/__init__.py for pinning status and deviations.
#
# Numeric conventions (mirrored bit-for-bit by the HIP kernels where the
# parity tests demand it — DESIGN.md §numerics):
#   * all arithmetic fp32;
#   * list-scan accumulations are SEQUENTIAL over the reduced axis
#     (subspace j ascending for PQ ADC, dim j ascending for IVFFlat/IVFSQ),
#     with separate multiply and add (no fma) — the HIP scan kernels use
#     the same order with fp contraction off, so given shared LUTs /
#     artifacts the distances are bitwise equal;
#   * ties order by (distance, ascending global id); global id = arrival
#     position within the shard (reference quirk: ids are implicit,
#     SURVEY.md §2 item 9);
#   * unfilled result slots: I = -1 and D = +FLT_MAX (L2) / -FLT_MAX (IP),
#     faiss heap-initialization convention;
#   * L2 distances are SQUARED L2, computed by the BLAS decomposition
#     ||q||^2 - 2 q.x + ||x||^2 for flat/coarse (tiny negatives possible),
#     and by direct sequential accumulation inside list scans.

import json
import os

import numpy as np

METRIC_INNER_PRODUCT = 0
METRIC_L2 = 1

FLT_MAX = np.float32(3.4028235e38)

# ---------------------------------------------------------------------------
# Deterministic RNG shared with the C++ engine (splitmix64; same constants
# in csrc/dfann_engine.hip) so oracle and GPU k-means start from identical
# initial centroids for the same seed.
# ---------------------------------------------------------------------------

_SM64_GAMMA = 0x9E3779B97F4A7C15
_MASK64 = (1 << 64) - 1


def splitmix64_seq(seed: int, n: int) -> np.ndarray:
    """First n outputs of splitmix64 seeded with `seed` (uint64)."""
    out = np.empty(n, dtype=np.uint64)
    x = seed & _MASK64
    for i in range(n):
        x = (x + _SM64_GAMMA) & _MASK64
        z = x
        z = ((z ^ (z >> 30)) * 0xBF58476D1CE4E5B9) & _MASK64
        z = ((z ^ (z >> 27)) * 0x94D049BB133111EB) & _MASK64
        z = z ^ (z >> 31)
        out[i] = z
    return out


def partial_shuffle_indices(n: int, k: int, seed: int) -> np.ndarray:
    """First k entries of a seeded partial Fisher-Yates shuffle of range(n).

    Used to pick k-means initial centroids. Identical algorithm in the C++
    engine (dfann_engine: pick_init_centroids).
    """
    k = min(k, n)
    idx = np.arange(n, dtype=np.int64)
    r = splitmix64_seq(seed, k)
    for i in range(k):
        j = i + int(r[i] % np.uint64(n - i))
        idx[i], idx[j] = idx[j], idx[i]
    return idx[:k]


# ---------------------------------------------------------------------------
# Pairwise scores (BLAS decomposition) + top-k selection
# ---------------------------------------------------------------------------


def pairwise_scores(q: np.ndarray, x: np.ndarray, metric: int) -> np.ndarray:
    """(nq, n) score matrix. L2: squared distance (minimize); IP: dot (maximize)."""
    q = np.ascontiguousarray(q, dtype=np.float32)
    x = np.ascontiguousarray(x, dtype=np.float32)
    ip = q @ x.T  # fp32 BLAS
    if metric == METRIC_INNER_PRODUCT:
        return ip
    qn = (q * q).sum(axis=1, dtype=np.float32)[:, None]
    xn = (x * x).sum(axis=1, dtype=np.float32)[None, :]
    return qn - np.float32(2.0) * ip + xn


def topk_rows(scores: np.ndarray, ids: np.ndarray, k: int, metric: int):
    """Per-row top-k with (score, ascending id) tie-break and faiss padding.

    scores: (nq, n) fp32; ids: (n,) int64 global ids. Returns D (nq,k) f32,
    I (nq,k) int64 (-1 padded).
    """
    nq, n = scores.shape
    D = np.full((nq, k), FLT_MAX if metric == METRIC_L2 else -FLT_MAX, dtype=np.float32)
    I = np.full((nq, k), -1, dtype=np.int64)
    if n == 0:
        return D, I
    kk = min(k, n)
    key = scores if metric == METRIC_L2 else -scores
    # lexsort: primary = key ascending, secondary = id ascending
    for i in range(nq):
        order = np.lexsort((ids, key[i]))[:kk]
        D[i, :kk] = scores[i, order]
        I[i, :kk] = ids[order]
    return D, I


# ---------------------------------------------------------------------------
# k-means (restating faiss v1.7 Clustering semantics; deviations in
# oracle/__init__.py header)
# ---------------------------------------------------------------------------


def kmeans(
    x: np.ndarray,
    k: int,
    metric: int = METRIC_L2,
    seed: int = 1234,
    niter: int = 25,
    max_points_per_centroid: int = 256,
):
    """Train k centroids. Returns (k, d) fp32 centroids.

    faiss Clustering restated: niter=25, subsample to k*256 points
    (strided here — deterministic deviation), random init by seeded
    partial shuffle, assignment by the index metric, centroid = mean of
    assigned points, empty cluster takes a split of the largest cluster.
    """
    x = np.ascontiguousarray(x, dtype=np.float32)
    n, d = x.shape
    cap = k * max_points_per_centroid
    if n > cap:
        sel = (np.arange(cap, dtype=np.int64) * n) // cap  # strided subsample
        x = x[sel]
        n = cap
    if n < k:
        raise ValueError(f"kmeans: n={n} < k={k}")
    init = partial_shuffle_indices(n, k, seed)
    cent = x[init].copy()
    for _ in range(niter):
        assign = assign_batch(x, cent, metric)
        sums = np.zeros((k, d), dtype=np.float64)
        np.add.at(sums, assign, x.astype(np.float64))
        counts = np.bincount(assign, minlength=k)
        nonz = counts > 0
        cent[nonz] = (sums[nonz] / counts[nonz, None]).astype(np.float32)
        # deterministic empty-cluster split: largest donor (lowest index on ties)
        empties = np.flatnonzero(~nonz)
        if empties.size:
            counts_work = counts.copy()
            eps = np.float32(1.0 / 1024.0)
            for ci in empties:
                cj = int(np.argmax(counts_work))
                cent[ci] = cent[cj] * (np.float32(1.0) + eps)
                cent[cj] = cent[cj] * (np.float32(1.0) - eps)
                counts_work[ci] = counts_work[cj] // 2
                counts_work[cj] -= counts_work[cj] // 2
    return cent


def assign_batch(x: np.ndarray, cent: np.ndarray, metric: int, chunk: int = 65536) -> np.ndarray:
    """Nearest-centroid assignment (argmin L2 / argmax IP, lowest index ties)."""
    n = x.shape[0]
    out = np.empty(n, dtype=np.int64)
    for s in range(0, n, chunk):
        sc = pairwise_scores(x[s : s + chunk], cent, metric)
        if metric == METRIC_L2:
            out[s : s + chunk] = np.argmin(sc, axis=1)
        else:
            out[s : s + chunk] = np.argmax(sc, axis=1)
    return out


# ---------------------------------------------------------------------------
# Sequential-accumulation scan primitives (the bit-exactness contract)
# ---------------------------------------------------------------------------


def seq_l2(q: np.ndarray, xs: np.ndarray) -> np.ndarray:
    """||q - xs_i||^2 accumulated sequentially over dims. xs: (n, d)."""
    acc = np.zeros(xs.shape[0], dtype=np.float32)
    for j in range(xs.shape[1]):
        diff = np.float32(q[j]) - xs[:, j]
        acc = acc + diff * diff  # separate mul + add, fp32
    return acc


def seq_ip(q: np.ndarray, xs: np.ndarray) -> np.ndarray:
    """q . xs_i accumulated sequentially over dims."""
    acc = np.zeros(xs.shape[0], dtype=np.float32)
    for j in range(xs.shape[1]):
        acc = acc + np.float32(q[j]) * xs[:, j]
    return acc


def adc_scan(lut: np.ndarray, codes: np.ndarray) -> np.ndarray:
    """ADC distances: sum_j lut[j, codes[i, j]], sequential over j.

    lut: (m, 256) fp32; codes: (n, m) uint8. fp32 accumulation, j ascending
    — mirrored exactly by the HIP ivfpq scan kernel.
    """
    n = codes.shape[0]
    acc = np.zeros(n, dtype=np.float32)
    for j in range(lut.shape[0]):
        acc = acc + lut[j, codes[:, j]]
    return acc


# ---------------------------------------------------------------------------
# Index implementations
# ---------------------------------------------------------------------------


class _OracleIndexBase:
    """Common shell: arrival-order ids, faiss result conventions."""

    def __init__(self, d: int, metric: int):
        self.d = int(d)
        self.metric = int(metric)
        self.is_trained = False
        self.ntotal = 0
        self.nprobe = 1

    def _check_q(self, q):
        q = np.ascontiguousarray(q, dtype=np.float32)
        assert q.ndim == 2 and q.shape[1] == self.d, f"bad query shape {q.shape}"
        return q

    def save(self, path: str):
        save_oracle_engine(self, path)


class OracleFlat(_OracleIndexBase):
    """faiss IndexFlatIP / IndexFlatL2 restated (reference index.py:28-30,94).

    NB reference quirk 3: the builder string "flat" always
    constructs the IP variant regardless of cfg.metric — that quirk lives in
    the factory (engine_factory), not here.
    """

    def __init__(self, d: int, metric: int):
        super().__init__(d, metric)
        self.is_trained = True  # flat needs no training
        self.xb = np.empty((0, d), dtype=np.float32)

    def train(self, x):
        self.is_trained = True  # no-op, faiss IndexFlat.train is a no-op

    def add(self, x):
        x = np.ascontiguousarray(x, dtype=np.float32)
        self.xb = np.concatenate([self.xb, x], axis=0)
        self.ntotal = self.xb.shape[0]

    def search(self, q, k):
        q = self._check_q(q)
        sc = pairwise_scores(q, self.xb, self.metric)
        ids = np.arange(self.ntotal, dtype=np.int64)
        return topk_rows(sc, ids, k, self.metric)

    def search_and_reconstruct(self, q, k):
        D, I = self.search(q, k)
        R = np.zeros((q.shape[0], k, self.d), dtype=np.float32)
        valid = I >= 0
        R[valid] = self.xb[I[valid]]
        return D, I, R

    def get_centroids(self):
        raise RuntimeError("flat index has no quantizer")  # mirrors AttributeError path

    # -- persistence (oracle-private .npz format) --
    def state_dict(self):
        return {"kind": "flat", "d": self.d, "metric": self.metric, "xb": self.xb}

    def load_state(self, st):
        self.xb = st["xb"].astype(np.float32)
        self.ntotal = self.xb.shape[0]
        self.is_trained = True


class _OracleIVFBase(_OracleIndexBase):
    """Shared IVF machinery: coarse quantizer, arrival-order lists."""

    def __init__(self, d, nlist, metric, seed=1234):
        super().__init__(d, metric)
        self.nlist = int(nlist)
        self.seed = int(seed)
        self.centroids = None  # (nlist, d) fp32
        self.list_ids = [np.empty(0, dtype=np.int64) for _ in range(int(nlist))]

    def get_centroids(self):
        if not self.is_trained:
            raise RuntimeError("index not trained")
        return self.centroids.copy()

    def coarse_topn(self, q, nprobe):
        """(nq, nprobe) probe list ids in rank order + (nq, nprobe) coarse scores."""
        sc = pairwise_scores(q, self.centroids, self.metric)
        nprobe = min(nprobe, self.nlist)
        key = sc if self.metric == METRIC_L2 else -sc
        probes = np.empty((q.shape[0], nprobe), dtype=np.int64)
        for i in range(q.shape[0]):
            order = np.lexsort((np.arange(self.nlist), key[i]))[:nprobe]
            probes[i] = order
        rows = np.arange(q.shape[0])[:, None]
        return probes, sc[rows, probes]

    # -- common search plumbing (per-class _scan_one(qi, li, bias)) --------
    # bias: None -> compute internally (seq_ip of q and the list centroid,
    # IP metric only); else the externally supplied coarse bias (parity
    # tests feed the engine's coarse output so the scan compares
    # bit-for-bit regardless of coarse rounding).

    def search(self, q, k):
        q = self._check_q(q)
        probes, _ = self.coarse_topn(q, self.nprobe)
        return self._search_probes(q, probes, None, k)

    def search_preassigned(self, q, probes, keys, k):
        """probes: (nq, nprobe) list ids; keys: coarse minimize-keys
        (IP bias = -key), may be None for L2."""
        q = self._check_q(q)
        probes = np.asarray(probes)
        keys = None if keys is None else np.asarray(keys, dtype=np.float32)
        return self._search_probes(q, probes, keys, k)

    def _search_probes(self, q, probes, keys, k):
        cand_d, cand_i = [], []
        for i in range(q.shape[0]):
            ds, ids = [], []
            for pi in range(probes.shape[1]):
                li = int(probes[i, pi])
                bias = None if keys is None else np.float32(-keys[i, pi])
                res = self._scan_one(q[i], li, bias)
                if res is None:
                    continue
                ds.append(res[0])
                ids.append(res[1])
            cand_d.append(ds)
            cand_i.append(ids)
        return self._merge_candidates(cand_d, cand_i, k)

    def _merge_candidates(self, cand_d, cand_i, k):
        """cand_*: per-query lists of fp32/int64 arrays -> (D, I)."""
        nq = len(cand_d)
        D = np.full((nq, k), FLT_MAX if self.metric == METRIC_L2 else -FLT_MAX, np.float32)
        I = np.full((nq, k), -1, np.int64)
        for i in range(nq):
            dd = np.concatenate(cand_d[i]) if cand_d[i] else np.empty(0, np.float32)
            ii = np.concatenate(cand_i[i]) if cand_i[i] else np.empty(0, np.int64)
            if dd.size == 0:
                continue
            key = dd if self.metric == METRIC_L2 else -dd
            order = np.lexsort((ii, key))[: min(k, dd.size)]
            D[i, : order.size] = dd[order]
            I[i, : order.size] = ii[order]
        return D, I


class OracleIVFFlat(_OracleIVFBase):
    """faiss IndexIVFFlat restated (reference index.py:36-40)."""

    def __init__(self, d, nlist, metric, seed=1234):
        super().__init__(d, nlist, metric, seed)
        self.list_data = [np.empty((0, d), dtype=np.float32) for _ in range(self.nlist)]

    def train(self, x):
        self.centroids = kmeans(x, self.nlist, self.metric, self.seed)
        self.is_trained = True

    def add(self, x):
        x = np.ascontiguousarray(x, dtype=np.float32)
        assign = assign_batch(x, self.centroids, self.metric)
        base = self.ntotal
        for li in range(self.nlist):
            mask = assign == li
            if mask.any():
                self.list_data[li] = np.concatenate([self.list_data[li], x[mask]])
                self.list_ids[li] = np.concatenate(
                    [self.list_ids[li], base + np.flatnonzero(mask).astype(np.int64)]
                )
        self.ntotal += x.shape[0]

    def _scan_one(self, qi, li, bias):
        xs = self.list_data[li]
        if xs.shape[0] == 0:
            return None
        if self.metric == METRIC_L2:
            return seq_l2(qi, xs), self.list_ids[li]
        return seq_ip(qi, xs), self.list_ids[li]

    def reconstruct_ids(self, I):
        """Gather stored vectors for global ids (for search_and_reconstruct)."""
        lut = {}
        for li in range(self.nlist):
            for pos, gid in enumerate(self.list_ids[li]):
                lut[int(gid)] = (li, pos)
        R = np.zeros(I.shape + (self.d,), dtype=np.float32)
        for idx, gid in np.ndenumerate(I):
            if gid >= 0:
                li, pos = lut[int(gid)]
                R[idx] = self.list_data[li][pos]
        return R

    def search_and_reconstruct(self, q, k):
        D, I = self.search(q, k)
        return D, I, self.reconstruct_ids(I)

    def state_dict(self):
        return {
            "kind": "ivf_flat", "d": self.d, "metric": self.metric,
            "nlist": self.nlist, "seed": self.seed, "centroids": self.centroids,
            "list_data": np.array(self.list_data, dtype=object),
            "list_ids": np.array(self.list_ids, dtype=object),
            "ntotal": self.ntotal,
        }

    def load_state(self, st):
        self.centroids = st["centroids"].astype(np.float32)
        self.list_data = list(st["list_data"])
        self.list_ids = list(st["list_ids"])
        self.ntotal = int(st["ntotal"])
        self.is_trained = True


class OracleIVFPQ(_OracleIVFBase):
    """faiss IndexIVFPQ restated (reference index.py:43-48 'knnlm' builder).

    Residual PQ (by_residual=true, faiss default), ksub=256, nbits=8 only.
    codebooks: (m, 256, dsub) fp32. ADC LUT scan, LUT built per (query,
    probe) from the residual q - centroid[probe].
    """

    def __init__(self, d, nlist, m, metric, nbits=8, seed=1234):
        super().__init__(d, nlist, metric, seed)
        assert nbits == 8, "only 8-bit PQ codes supported (ksub=256)"
        assert d % m == 0, f"dim {d} not divisible by m={m}"
        self.m = int(m)
        self.dsub = d // m
        self.codebooks = None
        self.list_codes = [np.empty((0, m), dtype=np.uint8) for _ in range(self.nlist)]

    def train(self, x):
        x = np.ascontiguousarray(x, dtype=np.float32)
        self.centroids = kmeans(x, self.nlist, self.metric, self.seed)
        assign = assign_batch(x, self.centroids, self.metric)
        resid = x - self.centroids[assign]
        cb = np.empty((self.m, 256, self.dsub), dtype=np.float32)
        for j in range(self.m):
            sub = resid[:, j * self.dsub : (j + 1) * self.dsub]
            cb[j] = kmeans(sub, 256, METRIC_L2, self.seed + 1 + j)
        self.codebooks = cb
        self.is_trained = True

    def encode(self, x):
        """(assign, codes): coarse assignment + per-subspace argmin codes."""
        x = np.ascontiguousarray(x, dtype=np.float32)
        assign = assign_batch(x, self.centroids, self.metric)
        resid = x - self.centroids[assign]
        codes = np.empty((x.shape[0], self.m), dtype=np.uint8)
        # sequential-accumulation subspace distances (mirrors k_pq_encode:
        # fp32, mul+add, t ascending — not the BLAS decomposition), chunked
        for j in range(self.m):
            sub = resid[:, j * self.dsub : (j + 1) * self.dsub]
            for s in range(0, x.shape[0], 65536):
                ss = sub[s : s + 65536]
                acc = np.zeros((ss.shape[0], 256), dtype=np.float32)
                for t in range(self.dsub):
                    diff = ss[:, t, None] - self.codebooks[j][None, :, t]
                    acc = acc + diff * diff
                codes[s : s + 65536, j] = np.argmin(acc, axis=1).astype(np.uint8)
        return assign, codes

    def add(self, x):
        assign, codes = self.encode(x)
        base = self.ntotal
        for li in range(self.nlist):
            mask = assign == li
            if mask.any():
                self.list_codes[li] = np.concatenate([self.list_codes[li], codes[mask]])
                self.list_ids[li] = np.concatenate(
                    [self.list_ids[li], base + np.flatnonzero(mask).astype(np.int64)]
                )
        self.ntotal += x.shape[0]

    def build_lut(self, qi: np.ndarray, li: int):
        """(lut, bias) for one (query, probe). L2: lut[j,c] = ||r_j - cb||^2
        seq over dsub, bias=0. IP: lut[j,c] = q_j . cb, bias = q . centroid.
        """
        if self.metric == METRIC_L2:
            r = qi - self.centroids[li]
            lut = np.empty((self.m, 256), dtype=np.float32)
            for j in range(self.m):
                rs = r[j * self.dsub : (j + 1) * self.dsub]
                acc = np.zeros(256, dtype=np.float32)
                for t in range(self.dsub):
                    diff = np.float32(rs[t]) - self.codebooks[j][:, t]
                    acc = acc + diff * diff
                lut[j] = acc
            return lut, np.float32(0.0)
        lut = np.empty((self.m, 256), dtype=np.float32)
        for j in range(self.m):
            qs = qi[j * self.dsub : (j + 1) * self.dsub]
            acc = np.zeros(256, dtype=np.float32)
            for t in range(self.dsub):
                acc = acc + np.float32(qs[t]) * self.codebooks[j][:, t]
            lut[j] = acc
        bias = np.float32(seq_ip(qi, self.centroids[li][None, :])[0])
        return lut, bias

    def _scan_one(self, qi, li, bias):
        codes = self.list_codes[li]
        if codes.shape[0] == 0:
            return None
        lut, bias0 = self.build_lut(qi, li)
        dist = adc_scan(lut, codes)
        if self.metric == METRIC_INNER_PRODUCT:
            dist = (bias0 if bias is None else bias) + dist
        return dist, self.list_ids[li]

    def decode_ids(self, I):
        lut = {}
        for li in range(self.nlist):
            for pos, gid in enumerate(self.list_ids[li]):
                lut[int(gid)] = (li, pos)
        R = np.zeros(I.shape + (self.d,), dtype=np.float32)
        for idx, gid in np.ndenumerate(I):
            if gid >= 0:
                li, pos = lut[int(gid)]
                code = self.list_codes[li][pos]
                dec = np.concatenate([self.codebooks[j][code[j]] for j in range(self.m)])
                R[idx] = self.centroids[li] + dec
        return R

    def search_and_reconstruct(self, q, k):
        D, I = self.search(q, k)
        return D, I, self.decode_ids(I)

    def state_dict(self):
        return {
            "kind": "ivfpq", "d": self.d, "metric": self.metric, "nlist": self.nlist,
            "m": self.m, "seed": self.seed, "centroids": self.centroids,
            "codebooks": self.codebooks,
            "list_codes": np.array(self.list_codes, dtype=object),
            "list_ids": np.array(self.list_ids, dtype=object), "ntotal": self.ntotal,
        }

    def load_state(self, st):
        self.centroids = st["centroids"].astype(np.float32)
        self.codebooks = st["codebooks"].astype(np.float32)
        self.list_codes = list(st["list_codes"])
        self.list_ids = list(st["list_ids"])
        self.ntotal = int(st["ntotal"])
        self.is_trained = True


class OracleIVFSQ(_OracleIVFBase):
    """faiss IndexIVFScalarQuantizer restated (reference index.py:63-68:
    QT_fp16 builder 'ivfsq'; QT_8bit reachable via the factory string path,
    e.g. "IVF{centroids},SQ8" — reference tests/test_index_config.json).

    Residual encoding (faiss ctor default encode_residual=true). 8-bit codec:
    per-dim [vmin, vmin+vdiff] trained on residual min/max; code =
    clip(int(255 * (x-vmin)/vdiff), 0, 255); decode = vmin + (code+0.5) *
    (vdiff/255). fp16 codec: IEEE half round-trip of the residual.
    """

    def __init__(self, d, nlist, metric, qtype="fp16", seed=1234):
        super().__init__(d, nlist, metric, seed)
        assert qtype in ("fp16", "8bit")
        self.qtype = qtype
        self.vmin = None    # (d,) fp32, 8bit only
        self.scale = None   # (d,) fp32 = vdiff/255
        self.vdiff = None
        code_bytes = 2 * d if qtype == "fp16" else d
        self.code_bytes = code_bytes
        self.list_codes = [np.empty((0, code_bytes), dtype=np.uint8) for _ in range(self.nlist)]

    def train(self, x):
        x = np.ascontiguousarray(x, dtype=np.float32)
        self.centroids = kmeans(x, self.nlist, self.metric, self.seed)
        if self.qtype == "8bit":
            assign = assign_batch(x, self.centroids, self.metric)
            resid = x - self.centroids[assign]
            vmin = resid.min(axis=0).astype(np.float32)
            vmax = resid.max(axis=0).astype(np.float32)
            self.vmin = vmin
            self.vdiff = (vmax - vmin).astype(np.float32)
            self.vdiff[self.vdiff == 0] = np.float32(1.0)  # degenerate dim guard
            self.scale = (self.vdiff / np.float32(255.0)).astype(np.float32)
        self.is_trained = True

    def _encode_resid(self, resid):
        if self.qtype == "fp16":
            return resid.astype(np.float16).view(np.uint8).reshape(resid.shape[0], -1)
        xi = (resid - self.vmin[None, :]) / self.vdiff[None, :]
        code = np.clip((np.float32(255.0) * xi).astype(np.int32), 0, 255).astype(np.uint8)
        return code

    def _decode_codes(self, codes):
        if self.qtype == "fp16":
            return codes.view(np.float16).astype(np.float32).reshape(codes.shape[0], self.d)
        return self.vmin[None, :] + (codes.astype(np.float32) + np.float32(0.5)) * self.scale[None, :]

    def add(self, x):
        x = np.ascontiguousarray(x, dtype=np.float32)
        assign = assign_batch(x, self.centroids, self.metric)
        resid = x - self.centroids[assign]
        codes = self._encode_resid(resid)
        base = self.ntotal
        for li in range(self.nlist):
            mask = assign == li
            if mask.any():
                self.list_codes[li] = np.concatenate([self.list_codes[li], codes[mask]])
                self.list_ids[li] = np.concatenate(
                    [self.list_ids[li], base + np.flatnonzero(mask).astype(np.int64)]
                )
        self.ntotal += x.shape[0]

    def _scan_one(self, qi, li, bias):
        codes = self.list_codes[li]
        if codes.shape[0] == 0:
            return None
        if self.qtype == "8bit":
            # folded decode algebra + 8-lane butterfly order, op-for-op the
            # HIP scan kernel (csrc/kernels.hip FAM==2 subgroup-8 branch):
            # L2 diff = u - c*v with u = (q-cent-vmin) - 0.5*scale,
            # v = scale; IP term = u + c*v with u = q*vmin + 0.5*(q*scale),
            # v = q*scale. Lane l sums dims {l*16 + 128*c + b} sequentially;
            # the 8 partials combine as ((p0+p4)+(p2+p6)) + ((p1+p5)+(p3+p7)).
            # Equal to the faiss codec in exact arithmetic; this is the
            # shared fp32 rounding order of the bit-exact tier.
            cf = codes.astype(np.float32)
            if self.metric == METRIC_L2:
                r = (qi - self.centroids[li]).astype(np.float32)
                u = (r - self.vmin) - np.float32(0.5) * self.scale
                v = self.scale

                def term(t):
                    diff = u[t] - cf[:, t] * v[t]
                    return diff * diff
            else:
                qsc = (qi * self.scale).astype(np.float32)
                uip = qi * self.vmin + np.float32(0.5) * qsc

                def term(t):
                    return uip[t] + cf[:, t] * qsc[t]

            n = codes.shape[0]
            parts = [np.zeros(n, dtype=np.float32) for _ in range(8)]
            for lane in range(8):
                for t0 in range(lane * 16, self.d, 128):
                    for b in range(16):
                        t = t0 + b
                        if t < self.d:
                            parts[lane] = parts[lane] + term(t)
            acc = ((parts[0] + parts[4]) + (parts[2] + parts[6])) + (
                (parts[1] + parts[5]) + (parts[3] + parts[7]))
            if self.metric == METRIC_L2:
                return acc, self.list_ids[li]
            if bias is None:
                bias = np.float32(seq_ip(qi, self.centroids[li][None, :])[0])
            return bias + acc, self.list_ids[li]
        dec = self._decode_codes(codes)  # fp16: residual values
        if self.metric == METRIC_L2:
            r = (qi - self.centroids[li]).astype(np.float32)
            return seq_l2(r, dec), self.list_ids[li]
        if bias is None:
            bias = np.float32(seq_ip(qi, self.centroids[li][None, :])[0])
        return bias + seq_ip(qi, dec), self.list_ids[li]

    def decode_ids(self, I):
        lut = {}
        for li in range(self.nlist):
            for pos, gid in enumerate(self.list_ids[li]):
                lut[int(gid)] = (li, pos)
        R = np.zeros(I.shape + (self.d,), dtype=np.float32)
        for idx, gid in np.ndenumerate(I):
            if gid >= 0:
                li, pos = lut[int(gid)]
                dec = self._decode_codes(self.list_codes[li][pos : pos + 1])[0]
                R[idx] = self.centroids[li] + dec
        return R

    def search_and_reconstruct(self, q, k):
        D, I = self.search(q, k)
        return D, I, self.decode_ids(I)

    def state_dict(self):
        return {
            "kind": "ivfsq", "d": self.d, "metric": self.metric, "nlist": self.nlist,
            "qtype": self.qtype, "seed": self.seed, "centroids": self.centroids,
            "vmin": self.vmin if self.vmin is not None else np.empty(0, np.float32),
            "vdiff": self.vdiff if self.vdiff is not None else np.empty(0, np.float32),
            "list_codes": np.array(self.list_codes, dtype=object),
            "list_ids": np.array(self.list_ids, dtype=object), "ntotal": self.ntotal,
        }

    def load_state(self, st):
        self.centroids = st["centroids"].astype(np.float32)
        vmin = st["vmin"]
        if vmin.size:
            self.vmin = vmin.astype(np.float32)
            self.vdiff = st["vdiff"].astype(np.float32)
            self.scale = (self.vdiff / np.float32(255.0)).astype(np.float32)
        self.list_codes = list(st["list_codes"])
        self.list_ids = list(st["list_ids"])
        self.ntotal = int(st["ntotal"])
        self.is_trained = True


# ---------------------------------------------------------------------------
# Engine factory + persistence: the backend duck-type consumed by
# distributed_faiss_amd.index.Index (injected from tests only).
# ---------------------------------------------------------------------------


def make_oracle_engine(spec: dict):
    """Build an oracle index from the engine spec dict.

    spec keys: type (flat|ivf_flat|ivfpq|ivfsq), dim, metric (0|1), nlist,
    m, nbits, sq_type, nprobe, seed.
    """
    t = spec["type"]
    d = int(spec["dim"])
    metric = int(spec["metric"])
    seed = int(spec.get("seed", 1234))
    if t == "flat":
        eng = OracleFlat(d, metric)
    elif t == "ivf_flat":
        eng = OracleIVFFlat(d, int(spec["nlist"]), metric, seed)
    elif t == "ivfpq":
        eng = OracleIVFPQ(d, int(spec["nlist"]), int(spec["m"]), metric,
                          int(spec.get("nbits", 8)), seed)
    elif t == "ivfsq":
        eng = OracleIVFSQ(d, int(spec["nlist"]), metric, spec.get("sq_type", "fp16"), seed)
    else:
        raise ValueError(f"unknown engine type {t}")
    eng.nprobe = int(spec.get("nprobe", 1))
    eng.spec = dict(spec)
    return eng


def save_oracle_engine(eng, path: str):
    st = eng.state_dict()
    st["spec_json"] = json.dumps(eng.spec)
    st["nprobe"] = eng.nprobe
    with open(path, "wb") as f:  # keep the exact filename (np.savez appends .npz to str paths)
        np.savez(f, **{k: v for k, v in st.items()})


class OracleProvider:
    """Engine provider duck-type (create/load) for injecting the oracle as
    the Index backend — tests only (oracle/__init__.py header)."""

    def create(self, spec: dict):
        return make_oracle_engine(spec)

    def load(self, path: str):
        return load_oracle_engine(path)


def load_oracle_engine(path: str):
    with np.load(path, allow_pickle=True) as z:
        st = {k: z[k] for k in z.files}
    spec = json.loads(str(st["spec_json"]))
    eng = make_oracle_engine(spec)
    eng.load_state(st)
    eng.nprobe = int(st["nprobe"])
    return eng


# ---------------------------------------------------------------------------
# Client-side merge restated (reference client.py:29-54 ResultHeap +
# client.py:265-310 _aggregate_results). Used by the merge KAT; the product
# client has its own numpy implementation (distributed_faiss_amd/client.py)
# tested against this one and against the reference's hard-coded KAT values.
# ---------------------------------------------------------------------------


def aggregate_results(shard_D, shard_I_meta, topk, maximize):
    """Merge S shards' (D, meta) like the reference client.

    shard_D: list of (nq, k) fp32; shard_I_meta: list of nq x k metadata
    lists. Returns (D_merged, meta_matrix). For maximize (dot) the returned
    distances are NEGATED — reference quirk 2 (client.py:291-294: -D pushed
    into the min-merge and never flipped back).
    """
    nq, k = shard_D[0].shape
    flat_meta = []
    parts = []
    for s, D in enumerate(shard_D):
        for i in range(nq):
            flat_meta.extend(shard_I_meta[s][i])
        parts.append(-D if maximize else D)
    allD = np.stack(parts, axis=1).reshape(nq, -1).astype(np.float32)  # (nq, S*k)
    synth = np.arange(len(shard_D) * nq * k, dtype=np.int64).reshape(len(shard_D), nq, k)
    synth = np.transpose(synth, (1, 0, 2)).reshape(nq, -1)
    Dout = np.empty((nq, topk), dtype=np.float32)
    meta_out = []
    for i in range(nq):
        order = np.lexsort((synth[i], allD[i]))[:topk]
        Dout[i] = allD[i, order]
        meta_out.append([flat_meta[synth[i, o]] for o in order])
    return Dout, meta_out


# ---------------------------------------------------------------------------
# HNSW search restatement (shared-graph parity tier — DESIGN.md §hnsw).
#
# Mirrors csrc/hnsw.hip k_hnsw_search OP-FOR-OP over a graph imported
# from the engine (hip_engine.HipEngine.hnsw_dump): the 8-lane partial
# sums + fixed butterfly of the distance, the 8192-slot visited hash
# with probe cap 32, the Sel frontier semantics (top-ef of everything
# appended, expansion = first unexpanded in (dist, flagged-id) order),
# and the two-stage greedy argmin tie structure. Distances are bitwise
# equal and result (D, I) exactly equal for any graph. The BUILD side is
# not restated (the batched wave insertion is engine-defined; build
# quality is gated by recall property tests and build determinism by the
# dump — oracle/__init__.py pinning notes).
# ---------------------------------------------------------------------------

HNSW_HASH = 16384
HNSW_PROBES = 64
HNSW_MAXL = 8
_FLT_MAX = np.float32(3.402823466e+38)


class OracleHNSWSearch:
    def __init__(self, d, graph, vmin, vdiff, codes):
        self.d = d
        self.g = graph  # dict from hnsw_dump
        self.vmin = vmin.astype(np.float32)
        self.vdiff = vdiff.astype(np.float32)
        self.scale = (self.vdiff / np.float32(255.0)).astype(np.float32)
        self.codes = codes  # (n, d) uint8, non-residual SQ8

    @staticmethod
    def encode(x, vmin, vdiff):
        """k_sq_encode restated (trunc toward zero, f32 ops)."""
        x = x.astype(np.float32)
        xi = (x - vmin[None, :].astype(np.float32)) / vdiff[None, :].astype(np.float32)
        c = (np.float32(255.0) * xi).astype(np.int32)  # trunc toward zero
        return np.clip(c, 0, 255).astype(np.uint8)

    def _dist_q(self, u, cid):
        # 8-lane partials (lane g8 covers bytes [g8*16,g8*16+16) stride
        # 128, sequential within), then the fixed xor-butterfly 4,2,1
        d = self.d
        s = self.scale
        row = self.codes[cid]
        parts = np.zeros(8, dtype=np.float32)
        for g8 in range(8):
            acc = np.float32(0.0)
            t0 = g8 * 16
            while t0 < d:
                for b in range(16):
                    t = t0 + b
                    if t < d:
                        diff = np.float32(u[t] - np.float32(row[t]) * s[t])
                        acc = np.float32(acc + np.float32(diff * diff))
                t0 += 128
            parts[g8] = acc
        b1 = np.array([np.float32(parts[i] + parts[i ^ 4]) for i in range(8)],
                      dtype=np.float32)
        b2 = np.array([np.float32(b1[i] + b1[i ^ 2]) for i in range(8)],
                      dtype=np.float32)
        b3 = np.float32(b2[0] + b2[1])
        return b3

    def _nbrs(self, node, level):
        g = self.g
        if level == 0:
            c = int(g["cnt0"][node])
            return g["nbr0"][node, :c]
        slot = int(g["upslot"][node])
        c = int(g["cntU"][slot, level - 1])
        return g["nbrU"][slot, level - 1, :c]

    def _greedy(self, u, cur, cur_d, level):
        NG = 32  # blockDim 256 / 8
        while True:
            nb = self._nbrs(cur, level)
            cnt = len(nb)
            # two-stage argmin: per-group best over passes, then groups
            gd = np.full(NG, _FLT_MAX, dtype=np.float32)
            gi = np.full(NG, -1, dtype=np.int64)
            for c0 in range(0, cnt, NG):
                for grp in range(NG):
                    ci = c0 + grp
                    if ci < cnt:
                        nid = int(nb[ci])
                        dd = self._dist_q(u, nid)
                        if dd < gd[grp] or (dd == gd[grp] and nid < gi[grp]):
                            gd[grp] = dd
                            gi[grp] = nid
            bd, bi = gd[0], gi[0]
            for i in range(1, NG):
                if gd[i] < bd or (gd[i] == bd and gi[i] < bi):
                    bd, bi = gd[i], gi[i]
            if bi >= 0 and bd < cur_d:
                cur, cur_d = int(bi), bd
            else:
                return cur, cur_d

    @staticmethod
    def _hash(ident):
        x = (ident * 0x9E3779B9) & 0xFFFFFFFF
        x ^= x >> 16
        return x & (HNSW_HASH - 1)

    @classmethod
    def _visited_insert(cls, tab, ident):
        h = cls._hash(ident)
        for p in range(HNSW_PROBES):
            slot = (h + p) & (HNSW_HASH - 1)
            v = tab.get(slot, 0)
            if v == ident + 1:
                return True
            if v == 0:
                tab[slot] = ident + 1
                return False
        return True  # cluster full: treated as visited

    def _beam(self, u, entry, entry_d, level, ef):
        tab = {}
        entries = []  # (dist, id, expanded)
        self._visited_insert(tab, entry)
        entries.append([np.float32(entry_d), entry, False])
        while True:
            entries.sort(key=lambda e: (e[0], (e[1] | (1 << 31)) if e[2] else e[1]))
            del entries[ef:]  # Sel truncation to top-ef
            pick = None
            for e in entries:
                if not e[2]:
                    pick = e
                    break
            if pick is None:
                return entries
            pick[2] = True
            for nid in self._nbrs(pick[1], level):
                nid = int(nid)
                dd = self._dist_q(u, nid)
                if not self._visited_insert(tab, nid):
                    entries.append([dd, nid, False])

    def search(self, q, k, ef):
        q = np.ascontiguousarray(q, dtype=np.float32)
        nq = q.shape[0]
        ef = max(ef, k)
        g = self.g
        D = np.full((nq, k), _FLT_MAX, dtype=np.float32)
        I = np.full((nq, k), -1, dtype=np.int64)
        for qi in range(nq):
            u = ((q[qi] - self.vmin) - np.float32(0.5) * self.scale).astype(np.float32)
            cur = int(g["entry"])
            cur_d = self._dist_q(u, cur)
            for l in range(int(g["maxlevel"]), 0, -1):
                cur, cur_d = self._greedy(u, cur, cur_d, l)
            res = self._beam(u, cur, cur_d, 0, ef)
            res.sort(key=lambda e: (e[0], (e[1] | (1 << 31)) if e[2] else e[1]))
            for j, e in enumerate(res[:k]):
                D[qi, j] = e[0]
                I[qi, j] = e[1]
        return D, I
