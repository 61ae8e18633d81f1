"""ctypes wrapper for the CPU parity oracle (oracle/liboracle.so).

TEST INFRASTRUCTURE ONLY: importable from tests/, __graft_entry__.smoke()
(as the checker) and bench.py's cpu_baseline leg. The product GPU path never
touches this module.
"""
import ctypes
import os
import subprocess

import numpy as np

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIBPATH = os.path.join(ROOT, "oracle", "liboracle.so")

_lib = None

i64p = ctypes.POINTER(ctypes.c_int64)
u64p = ctypes.POINTER(ctypes.c_uint64)
f64p = ctypes.POINTER(ctypes.c_double)


def _p(a):
    return a.ctypes.data_as(ctypes.c_void_p)


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(LIBPATH):
            subprocess.check_call(["make", "-C", os.path.join(ROOT, "oracle")])
        _lib = ctypes.CDLL(LIBPATH)
        L = _lib
        L.oracle_hash_i64.restype = ctypes.c_uint64
        L.oracle_hash_i64.argtypes = [ctypes.c_int64]
        L.oracle_partition_of.restype = ctypes.c_uint32
        L.oracle_partition_of.argtypes = [ctypes.c_int64, ctypes.c_uint32]
        L.oracle_checksum_pairs_i64.restype = ctypes.c_uint64
        for name in ("oracle_reduce_by_key_i64", "oracle_group_count_i64"):
            getattr(L, name).restype = ctypes.c_int64
        L.oracle_reduce_by_key_f64.restype = ctypes.c_int64
        L.oracle_group_by_key_i64.restype = ctypes.c_int64
        L.oracle_join_i64.restype = ctypes.c_int64
        L.oracle_distinct_i64.restype = ctypes.c_int64
    return _lib


def slice_bounds(n, nparts):
    out = np.zeros(nparts + 1, dtype=np.uint64)
    lib().oracle_slice_bounds(ctypes.c_uint64(n), ctypes.c_uint32(nparts), _p(out))
    return out


def hash_i64(k):
    return lib().oracle_hash_i64(ctypes.c_int64(int(k)))


def partition_of(k, nparts):
    return lib().oracle_partition_of(ctypes.c_int64(int(k)), ctypes.c_uint32(nparts))


def checksum_pairs(keys, vals):
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    vals = np.ascontiguousarray(vals, dtype=np.int64)
    return lib().oracle_checksum_pairs_i64(_p(keys), _p(vals), ctypes.c_uint64(len(keys)))


def _agg(name, keys, vals, pin, pout, vdtype=np.int64):
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    vals = np.ascontiguousarray(vals, dtype=vdtype)
    assert len(keys) == len(vals)
    cap = len(keys) + 16
    while True:
        ok = np.empty(cap, dtype=np.int64)
        ov = np.empty(cap, dtype=vdtype)
        r = getattr(lib(), name)(
            _p(keys), _p(vals), ctypes.c_uint64(len(keys)),
            ctypes.c_uint32(pin), ctypes.c_uint32(pout),
            _p(ok), _p(ov), ctypes.c_uint64(cap))
        if r >= 0:
            return ok[:r], ov[:r]
        cap *= 2


def reduce_by_key_i64(keys, vals, pin, pout):
    return _agg("oracle_reduce_by_key_i64", keys, vals, pin, pout)


def group_count_i64(keys, vals, pin, pout):
    return _agg("oracle_group_count_i64", keys, vals, pin, pout)


def reduce_by_key_f64(keys, vals, pin, pout):
    return _agg("oracle_reduce_by_key_f64", keys, vals, pin, pout, vdtype=np.float64)


def group_by_key_i64(keys, vals, pin, pout):
    """returns (group_keys, offsets, concatenated_values)"""
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    vals = np.ascontiguousarray(vals, dtype=np.int64)
    n = len(keys)
    kcap = n + 16
    ok = np.empty(kcap, dtype=np.int64)
    off = np.empty(kcap + 1, dtype=np.uint64)
    ov = np.empty(n + 16, dtype=np.int64)
    r = lib().oracle_group_by_key_i64(
        _p(keys), _p(vals), ctypes.c_uint64(n),
        ctypes.c_uint32(pin), ctypes.c_uint32(pout),
        _p(ok), _p(off), _p(ov),
        ctypes.c_uint64(kcap), ctypes.c_uint64(n + 16))
    assert r >= 0
    return ok[:r], off[:r + 1], ov[:int(off[r])]


def sort_by_key_i64(keys, vals):
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    vals = np.ascontiguousarray(vals, dtype=np.int64)
    ok = np.empty(len(keys), dtype=np.int64)
    ov = np.empty(len(keys), dtype=np.int64)
    lib().oracle_sort_by_key_i64(_p(keys), _p(vals), ctypes.c_uint64(len(keys)), _p(ok), _p(ov))
    return ok, ov


def join_i64(ak, av, bk, bv, pin, pout):
    ak = np.ascontiguousarray(ak, dtype=np.int64)
    av = np.ascontiguousarray(av, dtype=np.int64)
    bk = np.ascontiguousarray(bk, dtype=np.int64)
    bv = np.ascontiguousarray(bv, dtype=np.int64)
    cap = (len(ak) + len(bk)) * 4 + 64
    while True:
        ok = np.empty(cap, dtype=np.int64)
        ova = np.empty(cap, dtype=np.int64)
        ovb = np.empty(cap, dtype=np.int64)
        r = lib().oracle_join_i64(
            _p(ak), _p(av), ctypes.c_uint64(len(ak)),
            _p(bk), _p(bv), ctypes.c_uint64(len(bk)),
            ctypes.c_uint32(pin), ctypes.c_uint32(pout),
            _p(ok), _p(ova), _p(ovb), ctypes.c_uint64(cap))
        if r >= 0:
            return ok[:r], ova[:r], ovb[:r]
        cap *= 2


def distinct_i64(keys, pin, pout):
    keys = np.ascontiguousarray(keys, dtype=np.int64)
    cap = len(keys) + 16
    ok = np.empty(cap, dtype=np.int64)
    r = lib().oracle_distinct_i64(_p(keys), ctypes.c_uint64(len(keys)),
                                  ctypes.c_uint32(pin), ctypes.c_uint32(pout),
                                  _p(ok), ctypes.c_uint64(cap))
    assert r >= 0
    return ok[:r]
