"""GPU inner-join (K4) parity vs the CPU oracle and the reference's golden
join test vectors."""
import json
import os
import sys

import numpy as np
import pytest

import oracle_ctypes as oc
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from vega_amd import datagen

pytestmark = pytest.mark.gpu
HERE = os.path.dirname(os.path.abspath(__file__))


@pytest.fixture(scope="module")
def ctx():
    from vega_amd import gpu
    with gpu.VegaContext() as c:
        yield c


def test_join_golden(ctx):
    g = json.load(open(os.path.join(HERE, "golden", "join.json")))
    a = ctx.make_rdd(g["a_keys"], g["a_vals"], nparts=g["nparts_in"])
    b = ctx.make_rdd(g["b_keys"], g["b_vals"], nparts=g["nparts_in"])
    j = a.join(b, nparts=g["nparts_out"])
    k, va, vb = j.collect_join()
    got = sorted(zip(k.tolist(), va.tolist(), vb.tolist()))
    assert got == [tuple(x) for x in g["expected_sorted"]]
    a.free(); b.free(); j.free()


@pytest.mark.parametrize("na,nb,bits,seeds", [
    (100_000, 120_000, 10, (41, 141)),   # many matches, small cross products
    (50_000, 50_000, 4, (42, 142)),      # 16 keys -> big cross products
    (1000, 0, 8, (43, 143)),             # empty side
    (10_000, 10_000, 30, (44, 144)),     # sparse overlap
])
def test_join_random(ctx, na, nb, bits, seeds):
    ak, av = datagen.uniform_pairs(seeds[0], na, key_bits=bits)
    bk, bv = datagen.uniform_pairs(seeds[1], nb, key_bits=bits)
    a = ctx.make_rdd(ak, av)
    b = ctx.make_rdd(bk, bv)
    j = a.join(b)
    k, va, vb = j.collect_join()
    got = sorted(zip(k.tolist(), va.tolist(), vb.tolist()))
    ok, ova, ovb = oc.join_i64(ak, av, bk, bv, 4, 4)
    exp = sorted(zip(ok.tolist(), ova.tolist(), ovb.tolist()))
    assert got == exp
    a.free(); b.free(); j.free()


def test_join_c4_shape_scaled(ctx):
    # C4 shape scaled down: keys uniform in [0, n) both sides
    n = 2_000_000
    ak, av = datagen.uniform_range_pairs(0xC0FFEE + 4, n, n)
    bk, bv = datagen.uniform_range_pairs(0xC0FFEE + 40, n, n)
    a = ctx.make_rdd(ak, av)
    b = ctx.make_rdd(bk, bv)
    j = a.join(b)
    k, va, vb = j.collect_join()
    ok, ova, ovb = oc.join_i64(ak, av, bk, bv, 16, 16)
    assert len(k) == len(ok)
    got = sorted(zip(k.tolist(), va.tolist(), vb.tolist()))
    exp = sorted(zip(ok.tolist(), ova.tolist(), ovb.tolist()))
    assert got == exp
    a.free(); b.free(); j.free()


def test_join_mixed_order_tags(ctx):
    """one narrow-key side (grouping falls back to the skipped key sort, tag
    0) joined with a wide-key side (pinned hash order, tag 4): the engine
    must harmonize both to one comparator before the merge"""
    na, nb = 300_000, 200_000
    ak, av = datagen.uniform_pairs(51, na, key_bits=8)     # narrow: tag 0
    bk_w, bv = datagen.uniform_pairs(52, nb, key_bits=64)  # wide (negatives): tag 4
    # overlap: map half of b's keys into a's narrow range
    bk = np.where(np.arange(nb) % 2 == 0, bk_w & 0xFF, bk_w)
    a = ctx.make_rdd(ak, av)
    b = ctx.make_rdd(bk, bv)
    j = a.join(b)
    k, va, vb = j.collect_join()
    got = sorted(zip(k.tolist(), va.tolist(), vb.tolist()))
    ok, ova, ovb = oc.join_i64(ak, av, bk, bv, 4, 4)
    exp = sorted(zip(ok.tolist(), ova.tolist(), ovb.tolist()))
    assert got == exp
    a.free(); b.free(); j.free()
