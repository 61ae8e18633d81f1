/* vega_core.hpp — C++ host mirror of the reference's driver surface above
 * the C ABI (include/vega_gpu.h).
 *
 * The reference host is compiled Rust (no Rust toolchain in this image —
 * SURVEY.md §0), so this is the compiled-host embodiment of the same API
 * semantics: Context (context.rs:147-164), make_rdd/parallelize
 * (context.rs:406-442), PairRdd ops (pair_rdd.rs:20-171), actions
 * collect/count (rdd.rs:420-447). Same names, same argument meaning, same
 * error behavior (throws on failure — the reference Results/panics).
 *
 * Header-only; link against libvega_gpu.so.
 */
#pragma once

#include <cstdint>
#include <stdexcept>
#include <string>
#include <utility>
#include <vector>

#include "../../include/vega_gpu.h"

namespace vega {

class VegaError : public std::runtime_error {
  public:
    explicit VegaError(const std::string &m) : std::runtime_error(m) {}
};

inline void check(int rc, const char *what, vega_ctx_t *c = nullptr) {
    if (rc != VEGA_OK)
        throw VegaError(std::string(what) + " failed rc=" + std::to_string(rc) +
                        (c ? std::string(": ") + vega_gpu_last_error(c) : ""));
}

class Context;

/* Rdd<(i64, i64|f64)> — handle + the PairRdd method surface */
class PairRdd {
  public:
    PairRdd() = default;
    PairRdd(vega_ctx_t *c, vega_rdd_t h, bool f64vals) : c_(c), h_(h), f64_(f64vals) {}

    /* pair_rdd.rs:54-80 (op = the aggregator closure triple) */
    PairRdd reduce_by_key(vega_op_t op = VEGA_OP_SUM_I64, uint32_t num_splits = 256) const {
        vega_rdd_t out = 0;
        check(vega_gpu_reduce_by_key(c_, h_, op, num_splits, &out), "reduce_by_key", c_);
        return PairRdd(c_, out, op == VEGA_OP_SUM_F64);
    }
    /* pair_rdd.rs:35-52; group sizes (full groups: sort_by_key + scan host-side) */
    PairRdd group_by_key_count(uint32_t num_splits = 256) const {
        vega_rdd_t out = 0;
        check(vega_gpu_group_count(c_, h_, num_splits, &out), "group_by_key", c_);
        return PairRdd(c_, out, false);
    }
    /* Spark-semantics ascending sort (absent from the reference) */
    PairRdd sort_by_key() const {
        vega_rdd_t out = 0;
        check(vega_gpu_sort_by_key(c_, h_, &out), "sort_by_key", c_);
        return PairRdd(c_, out, f64_);
    }
    /* pair_rdd.rs:104-121 inner join */
    PairRdd join(const PairRdd &other, uint32_t num_splits = 256) const {
        vega_rdd_t out = 0;
        check(vega_gpu_join(c_, h_, other.h_, num_splits, &out), "join", c_);
        return PairRdd(c_, out, false);
    }
    /* rdd.rs:199-235 narrow transforms as op-enums (device-resident) */
    PairRdd map(vega_map_op_t op, int64_t p0 = 0) const {
        vega_rdd_t out = 0;
        check(vega_gpu_map(c_, h_, op, p0, &out), "map", c_);
        return PairRdd(c_, out, false);
    }
    PairRdd filter(vega_pred_t pred, int64_t p0 = 0, int64_t p1 = 0) const {
        vega_rdd_t out = 0;
        check(vega_gpu_filter(c_, h_, pred, p0, p1, &out), "filter", c_);
        return PairRdd(c_, out, false);
    }
    /* rdd.rs:449-459: counts over the VALUE column */
    PairRdd count_by_value(uint32_t num_splits = 256) const {
        vega_rdd_t out = 0;
        check(vega_gpu_count_by_value(c_, h_, num_splits, &out), "count_by_value", c_);
        return PairRdd(c_, out, false);
    }
    /* rdd.rs:501-531 */
    PairRdd distinct(uint32_t num_splits = 256) const {
        vega_rdd_t out = 0;
        check(vega_gpu_distinct(c_, h_, num_splits, &out), "distinct", c_);
        return PairRdd(c_, out, false);
    }

    /* actions (rdd.rs:420-447) */
    uint64_t count() const {
        uint64_t n = 0;
        check(vega_gpu_count(c_, h_, &n), "count", c_);
        return n;
    }
    std::vector<std::pair<int64_t, int64_t>> collect() const {
        uint64_t n = count();
        std::vector<int64_t> k(n), v(n);
        if (n) check(vega_gpu_collect(c_, h_, k.data(), v.data(), &n), "collect", c_);
        std::vector<std::pair<int64_t, int64_t>> out(n);
        for (uint64_t i = 0; i < n; i++) out[i] = {k[i], v[i]};
        return out;
    }
    std::vector<std::pair<int64_t, double>> collect_f64() const {
        uint64_t n = count();
        std::vector<int64_t> k(n);
        std::vector<double> v(n);
        if (n) check(vega_gpu_collect(c_, h_, k.data(), v.data(), &n), "collect", c_);
        std::vector<std::pair<int64_t, double>> out(n);
        for (uint64_t i = 0; i < n; i++) out[i] = {k[i], v[i]};
        return out;
    }
    void free() {
        if (c_ && h_) vega_gpu_free_rdd(c_, h_);
        h_ = 0;
    }
    vega_rdd_t handle() const { return h_; }

  private:
    vega_ctx_t *c_ = nullptr;
    vega_rdd_t h_ = 0;
    bool f64_ = false;
};

/* Context::new (context.rs:147-164); local mode, one GPU per process */
class Context {
  public:
    Context() { check(vega_gpu_init(1, &c_), "Context::new (vega_gpu_init)"); }
    ~Context() {
        if (c_) vega_gpu_shutdown(c_);
    }
    Context(const Context &) = delete;
    Context &operator=(const Context &) = delete;

    /* context.rs:433-442 (parallelize) / :406-417 (make_rdd = parallelize) */
    PairRdd parallelize(const std::vector<std::pair<int64_t, int64_t>> &data,
                        uint32_t num_splits) {
        std::vector<int64_t> k(data.size()), v(data.size());
        for (size_t i = 0; i < data.size(); i++) { k[i] = data[i].first; v[i] = data[i].second; }
        vega_rdd_t h = 0;
        check(vega_gpu_make_rdd(c_, k.data(), v.data(), data.size(), num_splits, &h),
              "parallelize", c_);
        return PairRdd(c_, h, false);
    }
    PairRdd make_rdd(const std::vector<std::pair<int64_t, int64_t>> &data,
                     uint32_t num_splits) {
        return parallelize(data, num_splits);
    }
    PairRdd gen_uniform(uint64_t n, uint64_t seed, int key_bits, uint64_t start = 0,
                        uint32_t num_splits = 256) {
        vega_rdd_t h = 0;
        check(vega_gpu_gen_rdd_uniform(c_, n, seed, key_bits, start, num_splits, &h),
              "gen_uniform", c_);
        return PairRdd(c_, h, false);
    }
    vega_ctx_t *raw() { return c_; }

  private:
    vega_ctx_t *c_ = nullptr;
};

} // namespace vega
