"""Transcribes the reference's own golden test vectors into JSON fixtures.

Sources (literal inputs and expected outputs transcribed by hand — the
reference is Rust and cannot be compiled/run in this environment):
  - /root/reference/tests/test_pair_rdd.rs:9-37   (group_by_key)
  - /root/reference/tests/test_pair_rdd.rs:40-82  (join)
  - /root/reference/tests/test_pair_rdd.rs:85-109 (count_by_value, 4 and 2 parts)
  - /root/reference/tests/test_pair_rdd.rs:112-135(group_by -> group_by_key on sign)
  - /root/reference/tests/test_rdd.rs:46-55       (distinct count, reduce sum)
  - /root/reference/tests/test_rdd.rs:286-322     (distinct with 3/2/10 partitions)

Non-i64 key/value types are mapped to i64 as documented in each fixture's
"mapping" field; this preserves the pinned semantics (grouping, counting,
join cross-products, value order) exactly — see tests/golden/README.md.

Run: python tests/golden/make_golden.py   (rewrites the *.json fixtures)
"""
import json
import os

HERE = os.path.dirname(os.path.abspath(__file__))


def write(name, obj):
    with open(os.path.join(HERE, name), "w") as f:
        json.dump(obj, f, indent=1, sort_keys=True)
        f.write("\n")


# test_group_by_key (test_pair_rdd.rs:9-37): keys "x"->120, "y"->121
write("group_by_key.json", {
    "source": "/root/reference/tests/test_pair_rdd.rs:9-37",
    "mapping": "string keys 'x'->120, 'y'->121 (ord)",
    "op": "group_by_key", "nparts_in": 4, "nparts_out": 4,
    "keys": [120] * 7 + [121] * 8,
    "vals": [1, 2, 3, 4, 5, 6, 7, 1, 2, 3, 4, 5, 6, 7, 8],
    "expected_groups": {"120": [1, 2, 3, 4, 5, 6, 7],
                        "121": [1, 2, 3, 4, 5, 6, 7, 8]},
})

# test_join (test_pair_rdd.rs:40-82): col2.join(col1, 4).
# col1 value pairs (A,B)->12 etc (A=1..H=8, enc v1*10+v2); col2 A1->101 etc.
write("join.json", {
    "source": "/root/reference/tests/test_pair_rdd.rs:40-82",
    "mapping": "col1 ('A','B')->12, ('C','D')->34, ('E','F')->56, ('G','H')->78; "
               "col2 'A1'->101,'A2'->102,'B1'->201,'B2'->202,'C1'->301,'C2'->302; "
               "left side = col2 (the reference calls col2.join(col1))",
    "op": "join", "nparts_in": 4, "nparts_out": 4,
    "a_keys": [1, 1, 2, 2, 3, 3], "a_vals": [101, 102, 201, 202, 301, 302],
    "b_keys": [1, 2, 3, 4], "b_vals": [12, 34, 56, 78],
    "expected_sorted": [[1, 101, 12], [1, 102, 12], [2, 201, 34],
                        [2, 202, 34], [3, 301, 56], [3, 302, 56]],
})

# test_count_by_value (test_pair_rdd.rs:85-109): count_by_value = map(x->(x,1))
# + reduce_by_key(+, number_of_splits) (rdd.rs:449-459)
for parts in (4, 2):
    write(f"count_by_value_p{parts}.json", {
        "source": "/root/reference/tests/test_pair_rdd.rs:85-109 (+rdd.rs:449-459)",
        "mapping": "i32 -> i64",
        "op": "reduce_by_key", "nparts_in": parts, "nparts_out": parts,
        "keys": [1, 2, 1, 3, 2, 3, 3, 2, 3],
        "vals": [1] * 9,
        "expected_sorted": [[1, 2], [2, 3], [3, 4]],
    })

# test_group_by (test_pair_rdd.rs:112-135): group_by(sign) == map(x->(sign(x),x))
# .group_by_key; keys neg->-1, zero->0, pos->1.
write("group_by_sign.json", {
    "source": "/root/reference/tests/test_pair_rdd.rs:112-135",
    "mapping": "'neg'->-1, 'zero'->0, 'pos'->1",
    "op": "group_by_key", "nparts_in": 2, "nparts_out": 2,
    "keys": [-1, -1, -1, 0, 1, 1, 1],
    "vals": [-3, -2, -1, 0, 1, 2, 3],
    "expected_groups": {"-1": [-3, -2, -1], "0": [0], "1": [1, 2, 3]},
})

# test_basic_operations (test_rdd.rs:46-55): distinct().count() == 4;
# reduce(+) == 10 == a 1-key reduce_by_key over the same rows.
write("distinct_dups.json", {
    "source": "/root/reference/tests/test_rdd.rs:52-53",
    "mapping": "i32 -> i64",
    "op": "distinct", "nparts_in": 2, "nparts_out": 2,
    "keys": [1, 1, 2, 2, 3, 3, 4, 4],
    "expected_sorted": [1, 2, 3, 4],
})
write("reduce_sum.json", {
    "source": "/root/reference/tests/test_rdd.rs:54 (reduce == single-key reduce_by_key)",
    "mapping": "reduce(+) on [1,2,3,4] == reduce_by_key on key 0",
    "op": "reduce_by_key", "nparts_in": 2, "nparts_out": 1,
    "keys": [0, 0, 0, 0], "vals": [1, 2, 3, 4],
    "expected_sorted": [[0, 10]],
})

# test_distinct (test_rdd.rs:286-322): 10 rows, 3 input partitions; distinct
# with default(3), 2 and 10 output partitions -> same 5-element set.
for pout in (3, 2, 10):
    write(f"distinct_p{pout}.json", {
        "source": "/root/reference/tests/test_rdd.rs:286-322",
        "mapping": "i32 -> i64",
        "op": "distinct", "nparts_in": 3, "nparts_out": pout,
        "keys": [1, 2, 2, 2, 3, 3, 3, 4, 4, 5],
        "expected_sorted": [1, 2, 3, 4, 5],
    })

print("golden fixtures written to", HERE)
