"""Long hash-run coverage: equal-h32 runs longer than the 1024-row walk cap
go to the cooperative k_group_cleanup_long kernel (one workgroup per run —
the bound that keeps a Zipf hot key from parking an O(run) walk on one lane,
ADVICE r01). Clean, dirty-relaxed and strict variants, all vs the oracle."""
import os
import sys

import numpy as np
import pytest

import oracle_ctypes as oc
import pyref
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ctx():
    from vega_amd import gpu
    with gpu.VegaContext() as c:
        yield c


def _colliding_pair():
    """two distinct wide keys with equal low-32 splitmix64 hashes"""
    from vega_amd import shuffle
    cand = np.arange(1_500_000, dtype=np.int64) + (1 << 61)
    h32 = (shuffle.hash_u64_np(cand) & np.uint64(0xFFFFFFFF)).astype(np.uint64)
    order = np.argsort(h32, kind="stable")
    hs = h32[order]
    dup = np.flatnonzero(hs[1:] == hs[:-1])
    assert len(dup) > 0
    i = dup[0]
    return int(cand[order[i]]), int(cand[order[i + 1]])


def test_group_count_hot_key_long_run(ctx):
    """300k duplicates of one wide key: the hash path's equal-h32 run spans
    ~300 walk caps; the cooperative kernel must find it CLEAN (0 breaks)"""
    rng = np.random.RandomState(33)
    hot = np.int64(0x0123456789ABCDEF)
    k = np.concatenate([np.full(300_000, hot, dtype=np.int64),
                        rng.randint(1, 1 << 62, size=700_000).astype(np.int64)])
    rng.shuffle(k)
    rdd = ctx.make_rdd(k, np.ones_like(k))
    red = rdd.group_count()
    gk, gv = red.collect()
    got = dict(zip(gk.tolist(), gv.tolist()))
    assert got == pyref.group_count(k)
    rdd.free(); red.free()


def test_group_count_long_dirty_run_falls_back(ctx):
    """a colliding key INTERLEAVED inside the hot run (>2 segments) forces
    the full-key-sort fallback from the cooperative path — results must
    still match the oracle exactly"""
    ka, kb = _colliding_pair()
    rng = np.random.RandomState(34)
    k = np.full(200_000, ka, dtype=np.int64)
    k[rng.choice(200_000, size=50, replace=False)] = kb  # interleaved
    k = np.concatenate([k, rng.randint(1, 1 << 62, size=300_000).astype(np.int64)])
    rng.shuffle(k)
    v = rng.randint(-1000, 1000, size=len(k)).astype(np.int64)
    rdd = ctx.make_rdd(k, v)
    from vega_amd import gpu
    red = rdd.reduce_by_key(gpu.OP_SUM_I64)
    gk, gv = red.collect()
    ok, ov = oc.reduce_by_key_i64(k, v, 8, 8)
    assert sorted(zip(gk.tolist(), gv.tolist())) == sorted(zip(ok.tolist(), ov.tolist()))
    rdd.free(); red.free()


def test_join_hot_key_long_run_strict(ctx):
    """join forces the STRICT cleanup contract; a 100k-duplicate key makes a
    long strict run (all-equal: passes without fallback) whose matches must
    all be emitted"""
    rng = np.random.RandomState(35)
    hot = np.int64(0x7EDCBA9876543210)
    ak = np.concatenate([np.full(100_000, hot, dtype=np.int64),
                         rng.randint(1, 1 << 62, size=200_000).astype(np.int64)])
    rng.shuffle(ak)
    av = np.arange(len(ak), dtype=np.int64)
    bk = np.concatenate([[hot], rng.randint(1, 1 << 62, size=5000).astype(np.int64)])
    bv = np.arange(len(bk), dtype=np.int64)
    a = ctx.make_rdd(ak, av)
    b = ctx.make_rdd(bk, bv)
    j = a.join(b)
    k, va, vb = j.collect_join()
    ok, ova, ovb = oc.join_i64(ak, av, bk.astype(np.int64), bv, 4, 4)
    assert len(k) == len(ok)
    assert sorted(zip(k.tolist(), va.tolist(), vb.tolist())) == \
        sorted(zip(ok.tolist(), ova.tolist(), ovb.tolist()))
    a.free(); b.free(); j.free()
