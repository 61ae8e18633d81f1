"""Cross-check the C oracle against the independent pure-Python restatement
(tests/pyref.py) on random inputs, plus structural properties (partition
invariance, slicing, checksum order-independence)."""
import numpy as np
import pytest

import oracle_ctypes as oc
import pyref

import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from vega_amd import datagen


def rand_pairs(seed, n, key_bits=8):
    return datagen.uniform_pairs(seed, n, key_bits=key_bits)


@pytest.mark.parametrize("seed,n,bits,pin,pout", [
    (1, 1000, 6, 4, 4),
    (2, 1000, 16, 3, 7),
    (3, 5000, 4, 8, 2),
    (4, 1, 8, 1, 1),
    (5, 0, 8, 2, 2),      # empty input
    (6, 7, 63, 4, 256),   # more partitions than rows; huge key space
    (7, 4096, 2, 16, 1),  # 4 distinct keys, heavy combine
])
def test_reduce_by_key_matches_pyref(seed, n, bits, pin, pout):
    k, v = rand_pairs(seed, n, bits)
    ok, ov = oc.reduce_by_key_i64(k, v, pin, pout)
    ref = pyref.reduce_by_key(k, v)
    got = dict(zip(ok.tolist(), ov.tolist()))
    assert len(ok) == len(set(ok.tolist()))  # keys unique
    assert got == ref


@pytest.mark.parametrize("seed,n,bits", [(11, 2000, 5), (12, 333, 10)])
def test_group_by_key_matches_pyref(seed, n, bits):
    k, v = rand_pairs(seed, n, bits)
    gk, off, gv = oc.group_by_key_i64(k, v, 4, 8)
    ref = pyref.group_by_key(k, v)
    got = {int(gk[i]): gv[int(off[i]):int(off[i + 1])].tolist() for i in range(len(gk))}
    assert got == ref  # including per-group value ORDER (global row order)


@pytest.mark.parametrize("seed,na,nb,bits", [(21, 300, 400, 5), (22, 50, 50, 3)])
def test_join_matches_pyref(seed, na, nb, bits):
    ak, av = rand_pairs(seed, na, bits)
    bk, bv = rand_pairs(seed + 100, nb, bits)
    k, va, vb = oc.join_i64(ak, av, bk, bv, 4, 4)
    got = sorted(zip(k.tolist(), va.tolist(), vb.tolist()))
    assert got == pyref.join(ak, av, bk, bv)


def test_sort_by_key_matches_pyref():
    k, v = rand_pairs(31, 5000, 12)
    ok, ov = oc.sort_by_key_i64(k, v)
    ref = pyref.sort_by_key(k, v)
    got = list(zip(ok.tolist(), ov.tolist()))
    # stable sort by key: equal-key values in row order — exact match
    assert got == ref


def test_reduce_partition_invariance():
    k, v = rand_pairs(41, 3000, 7)
    base = None
    for pin, pout in [(1, 1), (4, 4), (16, 3), (7, 256)]:
        ok, ov = oc.reduce_by_key_i64(k, v, pin, pout)
        got = sorted(zip(ok.tolist(), ov.tolist()))
        base = got if base is None else base
        assert got == base


def test_reduce_f64_close_to_numpy():
    k = np.random.RandomState(5).randint(0, 64, size=2000).astype(np.int64)
    v = np.random.RandomState(6).rand(2000)
    ok, ov = oc.reduce_by_key_f64(k, v, 4, 4)
    got = dict(zip(ok.tolist(), ov.tolist()))
    for key in np.unique(k):
        ref = v[k == key].sum()
        assert abs(got[int(key)] - ref) <= 1e-9 * max(1.0, abs(ref))


def test_reduce_wrapping_overflow():
    # Rust release-mode i64 add wraps; the engine and oracle must match that
    k = np.zeros(4, dtype=np.int64)
    v = np.array([2**62, 2**62, 2**62, 2**62], dtype=np.int64)
    ok, ov = oc.reduce_by_key_i64(k, v, 2, 2)
    assert ok.tolist() == [0] and ov.tolist() == [0]  # 4*2^62 wraps to 0


def test_slice_bounds_reference_formula():
    # parallel_collection_rdd.rs:116-145: partition p = [p*n/P, (p+1)*n/P)
    for n, p in [(15, 4), (9, 4), (9, 2), (7, 2), (10, 3), (0, 4), (5, 7), (10**9, 256)]:
        b = oc.slice_bounds(n, p).tolist()
        assert b[0] == 0 and b[-1] == n
        assert b == [(i * n) // p for i in range(p + 1)]


def test_checksum_order_independent():
    k, v = rand_pairs(51, 1000, 8)
    perm = np.random.RandomState(0).permutation(1000)
    assert oc.checksum_pairs(k, v) == oc.checksum_pairs(k[perm], v[perm])
    assert oc.checksum_pairs(k, v) != oc.checksum_pairs(k, v + 1)


def test_hash_is_splitmix64():
    # pin the hash so GPU/oracle/python all agree; splitmix64(0 + golden) etc.
    def sm64(x):
        x = (x + 0x9E3779B97F4A7C15) & (1 << 64) - 1
        x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & (1 << 64) - 1
        x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & (1 << 64) - 1
        return x ^ (x >> 31)
    for k in [0, 1, -1, 12345, -987654321, 2**62]:
        assert oc.hash_i64(k) == sm64(k & (1 << 64) - 1)
        assert oc.partition_of(k, 256) == oc.hash_i64(k) % 256


def test_datagen_deterministic_and_sliceable():
    k1, v1 = datagen.uniform_pairs(123, 100)
    k2, v2 = datagen.uniform_pairs(123, 100)
    assert (k1 == k2).all() and (v1 == v2).all()
    # rank-sliced generation matches the global stream
    ka, va = datagen.uniform_pairs(123, 40, start=0)
    kb, vb = datagen.uniform_pairs(123, 60, start=40)
    assert (np.concatenate([ka, kb]) == k1).all()
    assert (np.concatenate([va, vb]) == v1).all()
    z, _ = datagen.zipf_pairs(7, 20000, s=1.1, keyspace=1000)
    assert z.min() >= 0 and z.max() < 1000
    # Zipf skew: most-frequent key should dominate
    _, counts = np.unique(z, return_counts=True)
    assert counts.max() > 20000 * 0.05


def test_oracle_fuzz_vs_pyref():
    rng = np.random.RandomState(1234)
    for _ in range(20):
        n = int(rng.randint(1, 3000))
        bits = int(rng.randint(1, 20))
        pin = int(rng.randint(1, 17))
        pout = int(rng.randint(1, 300))
        seed = int(rng.randint(0, 1 << 30))
        k, v = rand_pairs(seed, n, bits)
        ok, ov = oc.reduce_by_key_i64(k, v, pin, pout)
        assert dict(zip(ok.tolist(), ov.tolist())) == pyref.reduce_by_key(k, v)
