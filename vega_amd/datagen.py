"""ctypes wrapper for the deterministic synthetic-input generators
(vega_amd/csrc/datagen.c -> libvega_datagen.so).

Seeds follow SURVEY.md §8d: 0xC0FFEE + config index.
"""
import ctypes
import os
import subprocess

import numpy as np

HERE = os.path.dirname(os.path.abspath(__file__))
LIBPATH = os.path.join(HERE, "csrc", "libvega_datagen.so")

SEED_BASE = 0xC0FFEE

_lib = None


def _p(a):
    return a.ctypes.data_as(ctypes.c_void_p)


def lib():
    global _lib
    if _lib is None:
        if not os.path.exists(LIBPATH):
            subprocess.check_call(
                ["gcc", "-O2", "-fopenmp", "-fPIC", "-Wall", "-shared",
                 os.path.join(HERE, "csrc", "datagen.c"), "-o", LIBPATH, "-lm"])
        _lib = ctypes.CDLL(LIBPATH)
    return _lib


def uniform_pairs(seed, n, key_bits=63, start=0):
    k = np.empty(n, dtype=np.int64)
    v = np.empty(n, dtype=np.int64)
    lib().vega_gen_uniform_pairs_i64(
        ctypes.c_uint64(seed), ctypes.c_uint64(start), ctypes.c_uint64(n),
        ctypes.c_int(key_bits), _p(k), _p(v))
    return k, v


def uniform_pairs_f64(seed, n, key_bits=63, start=0):
    k = np.empty(n, dtype=np.int64)
    v = np.empty(n, dtype=np.float64)
    lib().vega_gen_uniform_pairs_f64(
        ctypes.c_uint64(seed), ctypes.c_uint64(start), ctypes.c_uint64(n),
        ctypes.c_int(key_bits), _p(k), _p(v))
    return k, v


def uniform_range_pairs(seed, n, key_range, start=0):
    k = np.empty(n, dtype=np.int64)
    v = np.empty(n, dtype=np.int64)
    lib().vega_gen_uniform_range_pairs_i64(
        ctypes.c_uint64(seed), ctypes.c_uint64(start), ctypes.c_uint64(n),
        ctypes.c_uint64(key_range), _p(k), _p(v))
    return k, v


def zipf_pairs(seed, n, s=1.1, keyspace=100_000_000, start=0):
    k = np.empty(n, dtype=np.int64)
    v = np.empty(n, dtype=np.int64)
    lib().vega_gen_zipf_pairs_i64(
        ctypes.c_uint64(seed), ctypes.c_uint64(start), ctypes.c_uint64(n),
        ctypes.c_double(s), ctypes.c_uint64(keyspace), _p(k), _p(v))
    return k, v
