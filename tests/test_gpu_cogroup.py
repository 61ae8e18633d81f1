"""GPU tests for cogroup and its derivatives (pair_rdd.rs:123-155,
co_grouped_rdd.rs:206-249; rdd.rs intersection/subtract compositions), the
deterministic f64 reduction, and the join >2^32 output guard.
"""
import os
import sys

import numpy as np
import pytest

import pyref
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from vega_amd import datagen

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ctx():
    from vega_amd import gpu
    with gpu.VegaContext() as c:
        yield c


def pyref_cogroup(ak, av, bk, bv):
    ga = pyref.group_by_key(ak, av)
    gb = pyref.group_by_key(bk, bv)
    out = {}
    for k in set(ga) | set(gb):
        out[k] = (sorted(ga.get(k, [])), sorted(gb.get(k, [])))
    return out


# the join golden's input sides (test_pair_rdd.rs:40-82 shape, i64-encoded)
# exercised through cogroup — cogroup is what the reference's join composes
def _sides(seed, na, nb, bits=6):
    ak, av = datagen.uniform_pairs(seed, na, key_bits=bits)
    bk, bv = datagen.uniform_pairs(seed + 1, nb, key_bits=bits)
    return ak, av, bk, bv


@pytest.mark.parametrize("na,nb,bits,seed", [
    (6, 4, 3, 21),            # tiny, reference-test scale
    (50_000, 30_000, 8, 22),  # dense overlap
    (100_000, 100_000, 20, 23),  # sparse overlap
    (5000, 0, 8, 24),         # empty side
    (100_000, 77, 4, 25),     # heavy skew one side
])
def test_cogroup_vs_pyref(ctx, na, nb, bits, seed):
    ak, av, bk, bv = _sides(seed, na, nb, bits)
    ra = ctx.make_rdd(ak, av)
    rb = ctx.make_rdd(bk, bv)
    keys, offa, lena, offb, lenb, vala, valb = ra.cogroup(rb)
    got = {}
    for i in range(len(keys)):
        k = int(keys[i])
        assert k not in got, "duplicate key in cogroup output"
        got[k] = (sorted(vala[offa[i]:offa[i] + lena[i]].tolist()),
                  sorted(valb[offb[i]:offb[i] + lenb[i]].tolist()))
    assert got == pyref_cogroup(ak, av, bk, bv)
    ra.free(); rb.free()


def test_cogroup_value_order_stable(ctx):
    """value order within each group = row order (reference per-partition
    append order; aggregator.rs:33-53)"""
    ak = np.array([5, 5, 5, 9, 5], dtype=np.int64)
    av = np.array([10, 11, 12, 13, 14], dtype=np.int64)
    bk = np.array([9, 5], dtype=np.int64)
    bv = np.array([1, 2], dtype=np.int64)
    ra = ctx.make_rdd(ak, av)
    rb = ctx.make_rdd(bk, bv)
    keys, offa, lena, offb, lenb, vala, valb = ra.cogroup(rb)
    m = {int(keys[i]): (vala[offa[i]:offa[i] + lena[i]].tolist(),
                        valb[offb[i]:offb[i] + lenb[i]].tolist())
         for i in range(len(keys))}
    assert m[5] == ([10, 11, 12, 14], [2])
    assert m[9] == ([13], [1])
    ra.free(); rb.free()


@pytest.mark.parametrize("na,nb,bits,seed", [
    (100_000, 80_000, 10, 31),
    (50_000, 50_000, 4, 32),
    (1000, 100_000, 16, 33),
])
def test_intersection_subtract(ctx, na, nb, bits, seed):
    ak, av, bk, bv = _sides(seed, na, nb, bits)
    ra = ctx.make_rdd(ak, av)
    rb = ctx.make_rdd(bk, bv)
    sa, sb = set(ak.tolist()), set(bk.tolist())
    inter = ra.intersection(rb)
    ik, iv = inter.collect()
    assert sorted(ik.tolist()) == sorted(sa & sb)
    assert (iv == 0).all()
    sub = ra.subtract(rb)
    sk, _ = sub.collect()
    assert sorted(sk.tolist()) == sorted(sa - sb)
    ra.free(); rb.free(); inter.free(); sub.free()


# ---------------- deterministic f64 reduction ----------------

def test_f64_reduce_bit_stable(ctx):
    """SUM_F64 is deterministic: repeated runs on the same device data give
    BITWISE-identical sums (fixed-shape chunk-ordered summation — no atomics;
    pair_rdd.rs:74-78 f64 closures)."""
    from vega_amd import gpu
    n = 2_000_000
    k = np.random.RandomState(7).randint(0, 500, size=n).astype(np.int64)
    v = np.random.RandomState(8).standard_normal(n) * 1e6
    rdd = ctx.make_rdd(k, v)
    runs = []
    for _ in range(3):
        red = rdd.reduce_by_key(gpu.OP_SUM_F64)
        gk, gv = red.collect()
        order = np.argsort(gk, kind="stable")
        runs.append((gk[order].tolist(), gv[order].view(np.int64).tolist()))
        red.free()
    assert runs[0] == runs[1] == runs[2], "f64 sums not bit-stable across runs"
    # and within 1e-6 relative of the oracle's sequential sums
    import oracle_ctypes as oc
    ok, ov = oc.reduce_by_key_f64(k, v, 256, 256)
    gk, gv = np.array(runs[0][0]), np.array(runs[0][1]).view(np.float64)
    oorder = np.argsort(ok, kind="stable")
    assert (gk == ok[oorder]).all()
    ref = ov[oorder]
    denom = np.maximum(np.abs(ref), 1e-30)
    assert (np.abs(gv - ref) / denom < 1e-6).all()
    rdd.free()


def test_f64_single_hot_key_long_run(ctx):
    """one key spanning many chunks: the chunk-ordered combine path"""
    from vega_amd import gpu
    n = 1_000_000
    k = np.zeros(n, dtype=np.int64)
    v = np.random.RandomState(9).standard_normal(n)
    rdd = ctx.make_rdd(k, v)
    sums = set()
    for _ in range(2):
        red = rdd.reduce_by_key(gpu.OP_SUM_F64)
        gk, gv = red.collect()
        assert len(gk) == 1 and gk[0] == 0
        sums.add(float(gv[0]).hex())
    assert len(sums) == 1, "hot-key f64 sum not bit-stable"
    assert abs(float.fromhex(next(iter(sums))) - v.sum()) < 1e-6 * max(abs(v.sum()), 1)
    rdd.free()


# ---------------- join output overflow guard ----------------

def test_join_overflow_guard(ctx):
    """a hot-key join whose output exceeds 2^32-1 rows: the count query
    reports the exact u64 total and the emit path refuses loudly instead of
    silently wrapping the u32 scan."""
    import torch
    from vega_amd import gpu
    n = 70_000  # 70k x 70k one-key join -> 4.9e9 > 2^32 output rows
    dev = torch.device("cuda")
    ak = torch.zeros(n, dtype=torch.int64, device=dev)
    av = torch.arange(n, dtype=torch.int64, device=dev)
    bk = torch.zeros(n, dtype=torch.int64, device=dev)
    bv = torch.arange(n, dtype=torch.int64, device=dev)
    ws = gpu.alloc_ws(n)
    gpu.dev_sort_pairs(ak, av, ws)
    gpu.dev_sort_pairs(bk, bv, ws)
    import ctypes
    lib = gpu.lib()
    # count-only query: exact u64 total
    nout = ctypes.c_uint64(0)
    rc = lib.vega_dev_join_sorted(
        gpu._stream(), gpu._t(ak), gpu._t(av), ctypes.c_uint64(n),
        gpu._t(bk), gpu._t(bv), ctypes.c_uint64(n),
        None, None, None, ctypes.c_uint64(0), ctypes.byref(nout),
        gpu._t(ws), ctypes.c_size_t(ws.numel()))
    assert rc == 0 and nout.value == n * n, (rc, nout.value)
    # emit attempt must refuse (VEGA_ERR_UNSUPPORTED = -4), not corrupt
    out = torch.empty(16, dtype=torch.int64, device=dev)
    rc = lib.vega_dev_join_sorted(
        gpu._stream(), gpu._t(ak), gpu._t(av), ctypes.c_uint64(n),
        gpu._t(bk), gpu._t(bv), ctypes.c_uint64(n),
        gpu._t(out), gpu._t(out), gpu._t(out), ctypes.c_uint64(out.numel()),
        ctypes.byref(nout), gpu._t(ws), ctypes.c_size_t(ws.numel()))
    assert rc == -4, rc
