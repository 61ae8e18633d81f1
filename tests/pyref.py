"""Independent pure-Python restatement of the hot-path semantics.

Used ONLY to cross-check the C oracle on random inputs (double-entry
bookkeeping: the C oracle follows the reference's partitioned structure
(dependency.rs / shuffled_rdd.rs), this module computes the same results the
simplest possible way). Python dicts preserve insertion order, and because the
map side iterates partitions in row order and the reduce side merges chunks in
partition order (see oracle/oracle.c header), a single global-row-order dict
produces identical group value order.
"""
from collections import defaultdict


def reduce_by_key(keys, vals):
    d = {}
    for k, v in zip(keys, vals):
        k = int(k); v = int(v)
        d[k] = (d.get(k, 0) + v) & 0xFFFFFFFFFFFFFFFF if k in d else v & 0xFFFFFFFFFFFFFFFF
    # wrap to signed i64
    def s64(x):
        x &= 0xFFFFFFFFFFFFFFFF
        return x - (1 << 64) if x >= (1 << 63) else x
    return {k: s64(v) for k, v in d.items()}


def group_by_key(keys, vals):
    d = defaultdict(list)
    for k, v in zip(keys, vals):
        d[int(k)].append(int(v))
    return dict(d)


def group_count(keys):
    d = defaultdict(int)
    for k in keys:
        d[int(k)] += 1
    return dict(d)


def join(ak, av, bk, bv):
    ga = group_by_key(ak, av)
    gb = group_by_key(bk, bv)
    out = []
    for k in ga:
        if k in gb:
            for x in ga[k]:
                for y in gb[k]:
                    out.append((k, x, y))
    return sorted(out)


def sort_by_key(keys, vals):
    # stable by key: equal-key values keep row order (python sort is stable)
    return sorted(zip([int(k) for k in keys], [int(v) for v in vals]),
                  key=lambda t: t[0])
