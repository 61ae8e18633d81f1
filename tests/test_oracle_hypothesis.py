"""Property-based fuzzing of the CPU oracle against the independent
pure-Python restatement (bounded so the CPU suite stays fast)."""
import numpy as np
from hypothesis import given, settings, strategies as st

import oracle_ctypes as oc
import pyref

i64 = st.integers(min_value=-(2**63), max_value=2**63 - 1)
small_key = st.integers(min_value=-5, max_value=5)
key = st.one_of(small_key, i64)  # bias toward collisions, keep extremes

rows = st.lists(st.tuples(key, i64), min_size=0, max_size=300)
parts = st.integers(min_value=1, max_value=32)


@settings(max_examples=60, deadline=None)
@given(rows, parts, parts)
def test_reduce_by_key_property(data, pin, pout):
    k = np.array([a for a, _ in data], dtype=np.int64)
    v = np.array([b for _, b in data], dtype=np.int64)
    ok, ov = oc.reduce_by_key_i64(k, v, pin, pout)
    assert dict(zip(ok.tolist(), ov.tolist())) == pyref.reduce_by_key(k, v)


@settings(max_examples=40, deadline=None)
@given(rows, parts)
def test_group_by_key_property(data, pin):
    k = np.array([a for a, _ in data], dtype=np.int64)
    v = np.array([b for _, b in data], dtype=np.int64)
    gk, off, gv = oc.group_by_key_i64(k, v, pin, 4)
    got = {int(gk[i]): gv[int(off[i]):int(off[i + 1])].tolist() for i in range(len(gk))}
    assert got == pyref.group_by_key(k, v)


@settings(max_examples=40, deadline=None)
@given(rows, rows)
def test_join_property(da, db):
    ak = np.array([a for a, _ in da], dtype=np.int64)
    av = np.array([b for _, b in da], dtype=np.int64)
    bk = np.array([a for a, _ in db], dtype=np.int64)
    bv = np.array([b for _, b in db], dtype=np.int64)
    k, va, vb = oc.join_i64(ak, av, bk, bv, 3, 5)
    assert sorted(zip(k.tolist(), va.tolist(), vb.tolist())) == pyref.join(ak, av, bk, bv)


@settings(max_examples=40, deadline=None)
@given(rows)
def test_sort_by_key_property(data):
    k = np.array([a for a, _ in data], dtype=np.int64)
    v = np.array([b for _, b in data], dtype=np.int64)
    ok, ov = oc.sort_by_key_i64(k, v)
    assert list(zip(ok.tolist(), ov.tolist())) == pyref.sort_by_key(k, v)
