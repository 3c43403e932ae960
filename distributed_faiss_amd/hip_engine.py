# HIP engine binding — the PRODUCT backend.
#
# ctypes binding of libdfann.so (C-ABI in include/dfann.h; kernels in
# csrc/). This is the only compute path the product ships: if the native
# library is missing, creation raises immediately — there is no CPU
# fallback (the CPU oracle under oracle/ is test infrastructure and is
# never imported from here).
#
# torch is used for device memory and streams only (plumbing): inputs are
# uploaded to HBM as torch.cuda tensors and the engine is handed raw
# device pointers + the current stream.

import ctypes
import json
import os

import numpy as np

_LIB = None


class DfannTiming(ctypes.Structure):
    _fields_ = [
        ("scan_ms", ctypes.c_double),
        ("scan_launches", ctypes.c_int64),
        ("scan_rows", ctypes.c_int64),
        ("scan_bytes", ctypes.c_int64),
        ("gemm_ms", ctypes.c_double),
        ("gemm_flops", ctypes.c_int64),
        ("merge_ms", ctypes.c_double),
        ("merge_launches", ctypes.c_int64),
        ("lut_ms", ctypes.c_double),
        ("lut_launches", ctypes.c_int64),
    ]


def lib_path():
    return os.path.join(os.path.dirname(os.path.abspath(__file__)), "libdfann.so")


def load_lib():
    global _LIB
    if _LIB is not None:
        return _LIB
    # Load torch FIRST so its bundled libamdhip64 owns the soname: loading
    # libdfann.so first binds the system ROCm runtime, which on the GPU
    # boxes reports "no ROCm-capable device" while torch's runtime works
    # (two HIP runtimes in one process otherwise).
    try:
        import torch  # noqa: F401
    except ImportError:
        pass
    p = lib_path()
    if not os.path.exists(p):
        raise RuntimeError(
            f"native engine library not found at {p} — build it with "
            "`python -c \"import __graft_entry__; __graft_entry__.build()\"` "
            "(hipcc --offload-arch=gfx950). The product path has no CPU fallback."
        )
    lib = ctypes.CDLL(p)
    c = ctypes
    P = c.POINTER
    lib.dfann_create.argtypes = [c.c_char_p, P(c.c_void_p)]
    lib.dfann_destroy.argtypes = [c.c_void_p]
    lib.dfann_train.argtypes = [c.c_void_p, c.c_int64, c.c_void_p, c.c_void_p]
    lib.dfann_add.argtypes = [c.c_void_p, c.c_int64, c.c_void_p, c.c_void_p]
    lib.dfann_search.argtypes = [c.c_void_p, c.c_int64, c.c_void_p, c.c_int,
                                 c.c_void_p, c.c_void_p, c.c_void_p]
    lib.dfann_search_reconstruct.argtypes = [c.c_void_p, c.c_int64, c.c_void_p,
                                             c.c_int, c.c_void_p, c.c_void_p,
                                             c.c_void_p, c.c_void_p]
    lib.dfann_coarse.argtypes = [c.c_void_p, c.c_int64, c.c_void_p, c.c_int,
                                 c.c_void_p, c.c_void_p, c.c_void_p]
    lib.dfann_search_preassigned.argtypes = [c.c_void_p, c.c_int64, c.c_void_p,
                                             c.c_int, c.c_void_p, c.c_void_p,
                                             c.c_int, c.c_void_p, c.c_void_p,
                                             c.c_void_p]
    lib.dfann_set_nprobe.argtypes = [c.c_void_p, c.c_int]
    lib.dfann_ntotal.argtypes = [c.c_void_p]
    lib.dfann_ntotal.restype = c.c_int64
    lib.dfann_nlist.argtypes = [c.c_void_p]
    lib.dfann_is_trained.argtypes = [c.c_void_p]
    lib.dfann_dim.argtypes = [c.c_void_p]
    lib.dfann_spec_json.argtypes = [c.c_void_p]
    lib.dfann_spec_json.restype = c.c_char_p
    lib.dfann_get_centroids.argtypes = [c.c_void_p, c.c_void_p]
    lib.dfann_get_codebooks.argtypes = [c.c_void_p, c.c_void_p]
    lib.dfann_get_sq_params.argtypes = [c.c_void_p, c.c_void_p, c.c_void_p]
    lib.dfann_set_trained.argtypes = [c.c_void_p] + [c.c_void_p] * 4
    lib.dfann_save.argtypes = [c.c_void_p, c.c_char_p]
    lib.dfann_load.argtypes = [c.c_char_p, P(c.c_void_p)]
    lib.dfann_merge_topk.argtypes = [c.c_int64, c.c_int, c.c_int, c.c_void_p,
                                     c.c_void_p, c.c_int, c.c_void_p,
                                     c.c_void_p, c.c_void_p]
    lib.dfann_set_timing.argtypes = [c.c_void_p, c.c_int]
    lib.dfann_get_timing.argtypes = [c.c_void_p, P(DfannTiming)]
    lib.dfann_hnsw_info.argtypes = [c.c_void_p, c.c_void_p]
    lib.dfann_hnsw_dump.argtypes = [c.c_void_p] + [c.c_void_p] * 6
    lib.dfann_last_error.restype = c.c_char_p
    _LIB = lib
    return lib


def _check(lib, rc):
    if rc != 0:
        raise RuntimeError("dfann: " + lib.dfann_last_error().decode())


def _torch():
    import torch

    if not torch.cuda.is_available():
        raise RuntimeError(
            "HIP engine requires a GPU (torch.cuda.is_available() is False); "
            "the product path has no CPU fallback"
        )
    return torch


class HipEngine:
    """Engine backend (duck-type shared with the test oracle): numpy in/out
    at this layer; device-resident entry points (`*_dev`) for the bench."""

    def __init__(self, spec=None, _handle=None):
        self.lib = load_lib()
        if _handle is not None:
            self.h = _handle
            self.spec = json.loads(self.lib.dfann_spec_json(self.h).decode())
        else:
            self.spec = dict(spec)
            h = ctypes.c_void_p()
            _check(self.lib, self.lib.dfann_create(
                json.dumps(self.spec).encode(), ctypes.byref(h)))
            self.h = h
        self._nprobe = int(self.spec.get("nprobe", 1))

    def __del__(self):
        if getattr(self, "h", None) is not None and self.lib is not None:
            self.lib.dfann_destroy(self.h)
            self.h = None

    # -- properties --------------------------------------------------------

    @property
    def ntotal(self):
        return int(self.lib.dfann_ntotal(self.h))

    @property
    def nlist(self):
        return int(self.lib.dfann_nlist(self.h))

    @property
    def is_trained(self):
        return bool(self.lib.dfann_is_trained(self.h))

    @property
    def nprobe(self):
        return self._nprobe

    @nprobe.setter
    def nprobe(self, v):
        self._nprobe = int(v)
        _check(self.lib, self.lib.dfann_set_nprobe(self.h, int(v)))

    @property
    def d(self):
        return int(self.lib.dfann_dim(self.h))

    def _stream(self, torch):
        return ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)

    # -- build path --------------------------------------------------------

    def train(self, x):
        torch = _torch()
        xt = torch.as_tensor(np.ascontiguousarray(x, dtype=np.float32)).cuda()
        _check(self.lib, self.lib.dfann_train(
            self.h, xt.shape[0], ctypes.c_void_p(xt.data_ptr()),
            self._stream(torch)))
        torch.cuda.synchronize()

    def train_dev(self, xt):
        torch = _torch()
        _check(self.lib, self.lib.dfann_train(
            self.h, xt.shape[0], ctypes.c_void_p(xt.data_ptr()),
            self._stream(torch)))
        torch.cuda.synchronize()

    def add(self, x):
        torch = _torch()
        xt = torch.as_tensor(np.ascontiguousarray(x, dtype=np.float32)).cuda()
        self.add_dev(xt)

    def add_dev(self, xt):
        torch = _torch()
        _check(self.lib, self.lib.dfann_add(
            self.h, xt.shape[0], ctypes.c_void_p(xt.data_ptr()),
            self._stream(torch)))

    # -- search path -------------------------------------------------------

    def search(self, q, k):
        torch = _torch()
        qt = torch.as_tensor(np.ascontiguousarray(q, dtype=np.float32)).cuda()
        D, I = self.search_dev(qt, k)
        return D.cpu().numpy(), I.cpu().numpy()

    def search_dev(self, qt, k, D=None, I=None):
        torch = _torch()
        nq = qt.shape[0]
        if D is None:
            D = torch.empty((nq, k), dtype=torch.float32, device="cuda")
        if I is None:
            I = torch.empty((nq, k), dtype=torch.int64, device="cuda")
        _check(self.lib, self.lib.dfann_search(
            self.h, nq, ctypes.c_void_p(qt.data_ptr()), k,
            ctypes.c_void_p(D.data_ptr()), ctypes.c_void_p(I.data_ptr()),
            self._stream(torch)))
        return D, I

    def search_and_reconstruct(self, q, k):
        torch = _torch()
        qt = torch.as_tensor(np.ascontiguousarray(q, dtype=np.float32)).cuda()
        nq = qt.shape[0]
        D = torch.empty((nq, k), dtype=torch.float32, device="cuda")
        I = torch.empty((nq, k), dtype=torch.int64, device="cuda")
        R = torch.empty((nq, k, self.d), dtype=torch.float32, device="cuda")
        _check(self.lib, self.lib.dfann_search_reconstruct(
            self.h, nq, ctypes.c_void_p(qt.data_ptr()), k,
            ctypes.c_void_p(D.data_ptr()), ctypes.c_void_p(I.data_ptr()),
            ctypes.c_void_p(R.data_ptr()), self._stream(torch)))
        return D.cpu().numpy(), I.cpu().numpy(), R.cpu().numpy()

    def coarse(self, q, nprobe):
        torch = _torch()
        qt = torch.as_tensor(np.ascontiguousarray(q, dtype=np.float32)).cuda()
        nq = qt.shape[0]
        probes = torch.empty((nq, nprobe), dtype=torch.int32, device="cuda")
        keys = torch.empty((nq, nprobe), dtype=torch.float32, device="cuda")
        _check(self.lib, self.lib.dfann_coarse(
            self.h, nq, ctypes.c_void_p(qt.data_ptr()), nprobe,
            ctypes.c_void_p(probes.data_ptr()), ctypes.c_void_p(keys.data_ptr()),
            self._stream(torch)))
        return probes.cpu().numpy(), keys.cpu().numpy()

    def search_preassigned(self, q, probes, keys, k):
        torch = _torch()
        qt = torch.as_tensor(np.ascontiguousarray(q, dtype=np.float32)).cuda()
        pt = torch.as_tensor(np.ascontiguousarray(probes, dtype=np.int32)).cuda()
        kt = torch.as_tensor(np.ascontiguousarray(keys, dtype=np.float32)).cuda()
        nq, nprobe = pt.shape
        D = torch.empty((nq, k), dtype=torch.float32, device="cuda")
        I = torch.empty((nq, k), dtype=torch.int64, device="cuda")
        _check(self.lib, self.lib.dfann_search_preassigned(
            self.h, nq, ctypes.c_void_p(qt.data_ptr()), nprobe,
            ctypes.c_void_p(pt.data_ptr()), ctypes.c_void_p(kt.data_ptr()), k,
            ctypes.c_void_p(D.data_ptr()), ctypes.c_void_p(I.data_ptr()),
            self._stream(torch)))
        return D.cpu().numpy(), I.cpu().numpy()

    # -- introspection / artifacts ----------------------------------------

    def get_centroids(self):
        out = np.empty((self.nlist, self.d), dtype=np.float32)
        _check(self.lib, self.lib.dfann_get_centroids(
            self.h, out.ctypes.data_as(ctypes.c_void_p)))
        return out

    def get_codebooks(self):
        m = int(self.spec["m"])
        dsub = self.d // m
        out = np.empty((m, 256, dsub), dtype=np.float32)
        _check(self.lib, self.lib.dfann_get_codebooks(
            self.h, out.ctypes.data_as(ctypes.c_void_p)))
        return out

    def get_sq_params(self):
        vmin = np.empty(self.d, dtype=np.float32)
        vdiff = np.empty(self.d, dtype=np.float32)
        _check(self.lib, self.lib.dfann_get_sq_params(
            self.h, vmin.ctypes.data_as(ctypes.c_void_p),
            vdiff.ctypes.data_as(ctypes.c_void_p)))
        return vmin, vdiff

    def get_lists(self):
        """(off, ids, codes) — finalized inverted lists on host; codes are
        stride-padded rows (stride = dfann_code_stride)."""
        self.lib.dfann_get_lists.argtypes = [ctypes.c_void_p] + [ctypes.c_void_p] * 3
        self.lib.dfann_code_stride.argtypes = [ctypes.c_void_p]
        stride = int(self.lib.dfann_code_stride(self.h))
        n = self.ntotal
        off = np.empty(self.nlist + 1, dtype=np.int64)
        ids = np.empty(max(n, 1), dtype=np.int64)
        codes = np.empty((max(n, 1), stride), dtype=np.uint8)
        _check(self.lib, self.lib.dfann_get_lists(
            self.h, off.ctypes.data_as(ctypes.c_void_p),
            ids.ctypes.data_as(ctypes.c_void_p),
            codes.ctypes.data_as(ctypes.c_void_p)))
        return off, ids[:n], codes[:n]

    def set_trained(self, centroids, codebooks=None, vmin=None, vdiff=None):
        def ptr(a):
            if a is None:
                return None
            return np.ascontiguousarray(a, dtype=np.float32).ctypes.data_as(
                ctypes.c_void_p)

        cent = np.ascontiguousarray(centroids, dtype=np.float32) \
            if centroids is not None else None
        cb = np.ascontiguousarray(codebooks, dtype=np.float32) \
            if codebooks is not None else None
        vm = np.ascontiguousarray(vmin, dtype=np.float32) if vmin is not None else None
        vd = np.ascontiguousarray(vdiff, dtype=np.float32) if vdiff is not None else None
        _check(self.lib, self.lib.dfann_set_trained(
            self.h,
            cent.ctypes.data_as(ctypes.c_void_p) if cent is not None else None,
            cb.ctypes.data_as(ctypes.c_void_p) if cb is not None else None,
            vm.ctypes.data_as(ctypes.c_void_p) if vm is not None else None,
            vd.ctypes.data_as(ctypes.c_void_p) if vd is not None else None))

    # -- hnsw introspection (test plumbing) -------------------------------

    HNSW_MAXL = 8

    def hnsw_info(self):
        """(M, deg0, nslots, entry, maxlevel, ef_construction)."""
        out = np.empty(6, dtype=np.int64)
        _check(self.lib, self.lib.dfann_hnsw_info(
            self.h, out.ctypes.data_as(ctypes.c_void_p)))
        return tuple(int(v) for v in out)

    def hnsw_dump(self):
        """Full graph to host: dict of levels/cnt0/nbr0/upslot/cntU/nbrU
        + info tuple. Deterministic given (data, seed, wave schedule)."""
        M, deg0, nslots, entry, maxlevel, efc = self.hnsw_info()
        n = self.ntotal
        L = self.HNSW_MAXL
        levels = np.empty(n, dtype=np.int32)
        cnt0 = np.empty(n, dtype=np.int32)
        nbr0 = np.empty((n, deg0), dtype=np.int32)
        upslot = np.empty(n, dtype=np.int32)
        cntU = np.empty((max(nslots, 1), L), dtype=np.int32)
        nbrU = np.empty((max(nslots, 1), L, M), dtype=np.int32)
        _check(self.lib, self.lib.dfann_hnsw_dump(
            self.h, *(a.ctypes.data_as(ctypes.c_void_p)
                      for a in (levels, cnt0, nbr0, upslot, cntU, nbrU))))
        return {"levels": levels, "cnt0": cnt0, "nbr0": nbr0,
                "upslot": upslot, "cntU": cntU, "nbrU": nbrU,
                "M": M, "deg0": deg0, "nslots": nslots, "entry": entry,
                "maxlevel": maxlevel, "efc": efc}

    # -- persistence / timing ---------------------------------------------

    def save(self, path):
        _check(self.lib, self.lib.dfann_save(self.h, path.encode()))

    def set_timing(self, enabled):
        _check(self.lib, self.lib.dfann_set_timing(self.h, 1 if enabled else 0))

    def get_timing(self):
        t = DfannTiming()
        _check(self.lib, self.lib.dfann_get_timing(self.h, ctypes.byref(t)))
        return {f[0]: getattr(t, f[0]) for f in DfannTiming._fields_}


def merge_topk_dev(Dall, Iall, k, maximize):
    """On-GPU shard merge (torch cuda tensors (S,nq,k)) -> (Dout, Iout).
    Output ids are global slots s*nq*k + q*k + j (ref client.py merge)."""
    torch = _torch()
    lib = load_lib()
    S, nq, kk = Dall.shape
    Dout = torch.empty((nq, k), dtype=torch.float32, device="cuda")
    Iout = torch.empty((nq, k), dtype=torch.int64, device="cuda")
    stream = ctypes.c_void_p(torch.cuda.current_stream().cuda_stream)
    _check(lib, lib.dfann_merge_topk(
        nq, S, kk, ctypes.c_void_p(Dall.data_ptr()),
        ctypes.c_void_p(Iall.data_ptr()), 1 if maximize else 0,
        ctypes.c_void_p(Dout.data_ptr()), ctypes.c_void_p(Iout.data_ptr()),
        stream))
    return Dout, Iout


class HipProvider:
    """Engine provider (create/load) — the product default for Index."""

    def __init__(self):
        load_lib()  # fail loudly now if the native library is absent

    def create(self, spec: dict):
        return HipEngine(spec=spec)

    def load(self, path: str):
        lib = load_lib()
        h = ctypes.c_void_p()
        _check(lib, lib.dfann_load(path.encode(), ctypes.byref(h)))
        return HipEngine(_handle=h)
