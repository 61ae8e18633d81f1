/* vega_cli — C++ host self-test binary over vega_core.hpp / the C ABI.
 *
 * `vega_cli selftest` runs the reference's golden scenarios (transcribed in
 * tests/golden/, literals repeated here) plus a randomized reduce checked
 * against an in-binary std::map reference (independent of oracle/ — this
 * binary is test tooling for the C++ host layer). Exit 0 = all green.
 * Without a GPU it must FAIL LOUDLY (nonzero, message) — there is no CPU
 * fallback anywhere in the product path.
 */
#include <algorithm>
#include <cstdio>
#include <cstring>
#include <map>
#include <vector>

#include "../../include/vega_common.h"
#include "vega_core.hpp"

using pairs_t = std::vector<std::pair<int64_t, int64_t>>;

static int fails = 0;
#define CHECK_EQ(a, b, what)                                                   \
    do {                                                                       \
        if (!((a) == (b))) {                                                   \
            fprintf(stderr, "FAIL %s (%s:%d)\n", what, __FILE__, __LINE__);    \
            fails++;                                                           \
        } else {                                                               \
            printf("ok   %s\n", what);                                         \
        }                                                                      \
    } while (0)

static pairs_t sorted(pairs_t v) {
    std::sort(v.begin(), v.end());
    return v;
}

int main(int argc, char **argv) {
    if (argc < 2 || strcmp(argv[1], "selftest") != 0) {
        fprintf(stderr, "usage: vega_cli selftest\n");
        return 2;
    }
    setvbuf(stdout, nullptr, _IONBF, 0); /* keep progress visible on a crash */
    try {
        vega::Context sc;

        /* count_by_value golden (test_pair_rdd.rs:85-109): both as the
         * explicit map v->(v,1) + reduce_by_key composition (rdd.rs:449-459)
         * and through the dedicated count_by_value entry */
        {
            pairs_t in;
            for (int64_t x : {1, 2, 1, 3, 2, 3, 3, 2, 3}) in.push_back({x, 1});
            for (uint32_t parts : {4u, 2u}) {
                auto r = sc.make_rdd(in, parts).reduce_by_key(VEGA_OP_SUM_I64, parts);
                CHECK_EQ(sorted(r.collect()), (pairs_t{{1, 2}, {2, 3}, {3, 4}}),
                         "count_by_value golden (composition)");
            }
            pairs_t in2;
            for (int64_t x : {1, 2, 1, 3, 2, 3, 3, 2, 3}) in2.push_back({0, x});
            auto r2 = sc.make_rdd(in2, 4).count_by_value(4);
            CHECK_EQ(sorted(r2.collect()), (pairs_t{{1, 2}, {2, 3}, {3, 4}}),
                     "count_by_value golden (API)");
        }
        /* group_by_key golden counts (test_pair_rdd.rs:9-37; x->120, y->121) */
        {
            pairs_t in;
            for (int i = 1; i <= 7; i++) in.push_back({120, i});
            for (int i = 1; i <= 8; i++) in.push_back({121, i});
            auto g = sc.make_rdd(in, 4).group_by_key_count(4);
            CHECK_EQ(sorted(g.collect()), (pairs_t{{120, 7}, {121, 8}}),
                     "group_by_key golden counts");
        }
        /* distinct golden (test_rdd.rs:286-322) */
        {
            pairs_t in;
            for (int64_t x : {1, 2, 2, 2, 3, 3, 3, 4, 4, 5}) in.push_back({x, 0});
            for (uint32_t pout : {3u, 2u, 10u}) {
                auto d = sc.make_rdd(in, 3).distinct(pout);
                CHECK_EQ(d.count(), (uint64_t)5, "distinct golden count");
            }
        }
        /* reduce golden (test_rdd.rs:54): reduce(+) == single-key r_b_k */
        {
            auto r = sc.make_rdd({{0, 1}, {0, 2}, {0, 3}, {0, 4}}, 2)
                         .reduce_by_key(VEGA_OP_SUM_I64, 1);
            CHECK_EQ(r.collect(), (pairs_t{{0, 10}}), "reduce(+)==10 golden");
        }
        /* randomized reduce vs in-binary std::map (datagen formula inline) */
        {
            const uint64_t n = 250000, seed = 4242;
            pairs_t in(n);
            std::map<int64_t, uint64_t> ref;
            for (uint64_t i = 0; i < n; i++) {
                int64_t k = (int64_t)(vega_rand_u64(seed, 2 * i) & 0x3FFF);
                int64_t v = (int64_t)vega_rand_u64(seed, 2 * i + 1);
                in[i] = {k, v};
                ref[k] += (uint64_t)v; /* wrapping, like Rust release i64 add */
            }
            auto got = sorted(sc.make_rdd(in, 16).reduce_by_key(VEGA_OP_SUM_I64, 16).collect());
            pairs_t exp;
            for (auto &kv : ref) exp.push_back({kv.first, (int64_t)kv.second});
            CHECK_EQ(got, exp, "randomized reduce vs std::map (n=250k)");
        }
        /* join golden (test_pair_rdd.rs:40-82; col2.join(col1)) */
        {
            pairs_t c1{{1, 12}, {2, 34}, {3, 56}, {4, 78}};
            pairs_t c2{{1, 101}, {1, 102}, {2, 201}, {2, 202}, {3, 301}, {3, 302}};
            auto a = sc.parallelize(c2, 4);
            auto b = sc.parallelize(c1, 4);
            auto j = a.join(b, 4);
            CHECK_EQ(j.count(), (uint64_t)6, "join golden count (6 tuples)");
        }
        /* sort_by_key: signed order + stability surrogate (sortedness) */
        {
            const uint64_t n = 100000, seed = 777;
            pairs_t in(n);
            for (uint64_t i = 0; i < n; i++)
                in[i] = {(int64_t)vega_rand_u64(seed, 2 * i),
                         (int64_t)vega_rand_u64(seed, 2 * i + 1)};
            auto s = sc.make_rdd(in, 8).sort_by_key().collect();
            bool ok = s.size() == n;
            for (size_t i = 1; i < s.size() && ok; i++) ok = s[i - 1].first <= s[i].first;
            CHECK_EQ(ok, true, "sort_by_key signed ascending (n=100k, full i64)");
        }
    } catch (const std::exception &e) {
        fprintf(stderr, "vega_cli: FATAL: %s\n", e.what());
        return 1;
    }
    if (fails) {
        fprintf(stderr, "vega_cli: %d check(s) FAILED\n", fails);
        return 1;
    }
    printf("vega_cli selftest: all green\n");
    return 0;
}
