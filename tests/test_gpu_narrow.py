"""Device-resident narrow ops (map/filter — SURVEY §8f f4): parity vs numpy
and composition into the shuffle without host round-trips."""
import os
import sys

import numpy as np
import pytest

import oracle_ctypes as oc
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from vega_amd import datagen

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def ctx():
    from vega_amd import gpu
    with gpu.VegaContext() as c:
        yield c


def test_map_ops(ctx):
    from vega_amd import gpu
    k, v = datagen.uniform_pairs(61, 100_000, key_bits=16)
    rdd = ctx.make_rdd(k, v)
    for op, exp_k, exp_v in [
        (gpu.MAP_VALUES_ADD, k, v + 7),
        (gpu.MAP_VALUES_MUL, k, v * 3),
        (gpu.MAP_KEYS_ADD, k + 7, v),
        (gpu.MAP_SWAP, v, k),
    ]:
        m = rdd.map(op, p0=7 if op != gpu.MAP_VALUES_MUL else 3)
        gk, gv = m.collect()
        assert (gk == exp_k).all() and (gv == exp_v).all()
        m.free()
    rdd.free()


def test_filter_stable_order(ctx):
    from vega_amd import gpu
    k, v = datagen.uniform_pairs(62, 300_000, key_bits=12)
    rdd = ctx.make_rdd(k, v)
    for pred, p0, p1, mask in [
        (gpu.PRED_KEY_MOD_EQ, 5, 2, (k % 5) == 2),
        (gpu.PRED_VAL_GT, 0, 0, v > 0),
        (gpu.PRED_KEY_IN_RANGE, 100, 2000, (k >= 100) & (k < 2000)),
    ]:
        f = rdd.filter(pred, p0, p1)
        gk, gv = f.collect()
        # stable compaction: exact row order preserved
        assert (gk == k[mask]).all() and (gv == v[mask]).all()
        f.free()
    rdd.free()


def test_map_filter_into_shuffle(ctx):
    """narrow chain feeding reduce_by_key entirely on device, vs the oracle
    on the equivalently transformed host arrays"""
    from vega_amd import gpu
    k, v = datagen.uniform_pairs(63, 400_000, key_bits=10)
    rdd = ctx.make_rdd(k, v)
    chain = rdd.map(gpu.MAP_VALUES_ADD, 5).filter(gpu.PRED_KEY_MOD_EQ, 3, 1)
    red = chain.reduce_by_key(gpu.OP_SUM_I64)
    gk, gv = red.collect()
    mask = (k % 3) == 1
    ok, ov = oc.reduce_by_key_i64(k[mask], (v + 5)[mask], 8, 8)
    assert sorted(zip(gk.tolist(), gv.tolist())) == sorted(zip(ok.tolist(), ov.tolist()))
    rdd.free(); red.free()


def test_distinct_composition(ctx):
    """distinct (rdd.rs:501-531) as map+reduce over the device chain"""
    from vega_amd import gpu
    k, _ = datagen.uniform_pairs(64, 100_000, key_bits=8)
    rdd = ctx.make_rdd(k, np.zeros_like(k))
    gk, _ = rdd.group_count().collect()
    assert sorted(gk.tolist()) == sorted(np.unique(k).tolist())
    rdd.free()
