# C-ABI surface: the shared library loads on a GPU-less box and exports
# every symbol include/dfann.h declares (no compute calls without a GPU).
import ctypes
import os
import re
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
HDR = os.path.join(REPO, "include", "dfann.h")
SO = os.path.join(REPO, "distributed_faiss_amd", "libdfann.so")


def _header_symbols():
    src = open(HDR).read()
    return sorted(set(re.findall(r"\b(dfann_\w+)\s*\(", src)))


@pytest.fixture(scope="module")
def lib():
    if not os.path.exists(SO):
        import __graft_entry__

        __graft_entry__.build()
    return ctypes.CDLL(SO)


def test_header_has_symbols():
    syms = _header_symbols()
    assert "dfann_create" in syms and "dfann_search" in syms
    assert len(syms) >= 15


def test_all_header_symbols_exported(lib):
    for sym in _header_symbols():
        assert hasattr(lib, sym), f"missing export {sym}"


def test_create_destroy_without_gpu(lib):
    # spec parsing + handle lifecycle touch no device
    h = ctypes.c_void_p()
    lib.dfann_create.argtypes = [ctypes.c_char_p, ctypes.POINTER(ctypes.c_void_p)]
    rc = lib.dfann_create(
        b'{"type": "ivfpq", "dim": 64, "metric": 1, "nlist": 8, "m": 8, '
        b'"nbits": 8, "nprobe": 4, "seed": 1234}', ctypes.byref(h))
    assert rc == 0
    lib.dfann_ntotal.argtypes = [ctypes.c_void_p]
    lib.dfann_ntotal.restype = ctypes.c_int64
    assert lib.dfann_ntotal(h) == 0
    lib.dfann_is_trained.argtypes = [ctypes.c_void_p]
    assert lib.dfann_is_trained(h) == 0
    lib.dfann_destroy.argtypes = [ctypes.c_void_p]
    assert lib.dfann_destroy(h) == 0


def test_bad_spec_rejected(lib):
    h = ctypes.c_void_p()
    lib.dfann_create.argtypes = [ctypes.c_char_p, ctypes.POINTER(ctypes.c_void_p)]
    lib.dfann_last_error.restype = ctypes.c_char_p
    rc = lib.dfann_create(b'{"type": "bogus", "dim": 8}', ctypes.byref(h))
    assert rc != 0
    assert b"bogus" in lib.dfann_last_error()
    # dim % m != 0
    rc = lib.dfann_create(
        b'{"type": "ivfpq", "dim": 10, "metric": 1, "nlist": 4, "m": 4}',
        ctypes.byref(h))
    assert rc != 0


def test_product_provider_requires_library(monkeypatch):
    # HipProvider must fail loudly when the .so is absent (no CPU fallback)
    import distributed_faiss_amd.hip_engine as he

    monkeypatch.setattr(he, "lib_path", lambda: "/nonexistent/libdfann.so")
    monkeypatch.setattr(he, "_LIB", None)
    with pytest.raises(RuntimeError, match="no CPU fallback"):
        he.HipProvider()
