/* vega_gpu.h — the drop-in C ABI of the MI355X-native shuffle/sort/aggregate
 * engine (libvega_gpu.so).
 *
 * DROP-IN BOUNDARY: the reference's GPU-replaceable seam is
 *   - map side:  ShuffleDependencyTrait::do_shuffle_task(rdd, partition) -> uri
 *                (/root/reference/src/dependency.rs:92-97, hot loop :164-229)
 *   - reduce side: Rdd::compute(split) -> iterator
 *                (/root/reference/src/rdd/rdd.rs:179; shuffled_rdd.rs:149-170)
 * A Rust host keeps the Rdd/PairRdd trait surface and calls these entry
 * points over plain FFI (see INTEGRATION.md for the exact `extern "C"` block
 * a vega maintainer would add). No torch types, no C++ types: pointers,
 * sizes, int error codes. Handles are opaque.
 *
 * Two API levels:
 *  1. RDD-handle API — mirrors Context::parallelize/make_rdd
 *     (context.rs:406-442) + PairRdd ops (pair_rdd.rs:20-171) for a
 *     single-process host. One context drives ONE GPU (vega local mode is
 *     one process; multi-GPU runs one process per GPU, rank model below).
 *  2. Device-pointer API — for a rank-per-GPU launcher (torch.distributed /
 *     RCCL over xGMI): caller owns device buffers and the stream; these are
 *     the raw kernel entries (map-side radix partition replacing
 *     dependency.rs:191-210, sort+segmented-reduce replacing
 *     shuffled_rdd.rs:154-164's HashMap merge).
 *
 * All functions return 0 on success, negative VEGA_ERR_* otherwise.
 */
#ifndef VEGA_GPU_H
#define VEGA_GPU_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

#define VEGA_OK               0
#define VEGA_ERR_INVALID     -1
#define VEGA_ERR_NOMEM       -2
#define VEGA_ERR_HIP         -3
#define VEGA_ERR_UNSUPPORTED -4
#define VEGA_ERR_CAP         -5   /* output capacity too small */

/* Aggregator op-enum: the fixed-function replacements for the reference's
 * boxed closures (aggregator.rs:8-16; reduce_by_key closures
 * pair_rdd.rs:74-78; group default aggregator.rs:33-53). */
typedef enum {
    VEGA_OP_SUM_I64 = 0,   /* reduce_by_key(+) on i64 (wrapping, bit-exact) */
    VEGA_OP_COUNT   = 1,   /* group_by_key -> per-key count */
    VEGA_OP_SUM_F64 = 2,   /* reduce_by_key(+) on f64 (1e-6 rel tolerance) */
    VEGA_OP_MIN_I64 = 3,
    VEGA_OP_MAX_I64 = 4,
} vega_op_t;

/* narrow transforms (rdd.rs:199-235 map/filter, pair_rdd.rs:84-101
 * map_values) as fixed op-enums — the device analogue of the reference's
 * closures (like vega_op_t for the Aggregator). Outputs stay device-resident
 * and feed the shuffle with no host round-trip. */
typedef enum {
    VEGA_MAP_VALUES_ADD = 0, /* v' = v + p0 (wrapping) */
    VEGA_MAP_VALUES_MUL = 1, /* v' = v * p0 (wrapping) */
    VEGA_MAP_KEYS_ADD   = 2, /* k' = k + p0 (wrapping) */
    VEGA_MAP_SWAP       = 3, /* (k,v) -> (v,k) */
} vega_map_op_t;
typedef enum {
    VEGA_PRED_KEY_MOD_EQ   = 0, /* keep if k % p0 == p1 (C signed rem) */
    VEGA_PRED_VAL_GT       = 1, /* keep if v > p0 */
    VEGA_PRED_KEY_IN_RANGE = 2, /* keep if p0 <= k < p1 */
} vega_pred_t;

typedef struct vega_ctx vega_ctx_t;
typedef uint64_t vega_rdd_t;      /* opaque RDD handle, 0 = invalid */

/* ---------------- context ---------------- */
/* ngpus: number of GPUs this process drives. ngpus == 1 = the rank-per-GPU
 * model (device = current HIP device). ngpus > 1 = the in-process local-mode
 * analogue: one process drives all ngpus devices, sharding rows a9-style and
 * exchanging buckets over RCCL/xGMI (reduce_by_key / group_count /
 * distinct; other ops on a G>1 context return VEGA_ERR_UNSUPPORTED).
 *
 * PER-CALL ROW LIMIT: every op is bounded to n < 2^32 rows per call (u32
 * histogram/scan plumbing); larger inputs must be sharded (the rank-per-GPU
 * launcher always does). Calls beyond the limit return
 * VEGA_ERR_UNSUPPORTED — never a silent wrap. A join whose OUTPUT would
 * exceed 2^32-1 rows likewise refuses to emit (count queries still return
 * the exact u64 total). */
int vega_gpu_init(int ngpus, vega_ctx_t **out);
int vega_gpu_shutdown(vega_ctx_t *ctx);
int vega_gpu_synchronize(vega_ctx_t *ctx);
const char *vega_gpu_last_error(vega_ctx_t *ctx);

/* ---------------- RDD construction ---------------- */
/* make_rdd: host (k,v) arrays copied to device; nparts = logical partition
 * count with ParallelCollection::slice chunking
 * (parallel_collection_rdd.rs:116-145): partition p = rows [pn/P,(p+1)n/P). */
int vega_gpu_make_rdd(vega_ctx_t *ctx, const int64_t *keys, const int64_t *vals,
                      uint64_t n, uint32_t nparts, vega_rdd_t *out);
int vega_gpu_make_rdd_f64(vega_ctx_t *ctx, const int64_t *keys, const double *vals,
                          uint64_t n, uint32_t nparts, vega_rdd_t *out);
/* device-side deterministic generation (bit-identical to
 * vega_gen_uniform_pairs_i64 in datagen.c); start = global row offset so each
 * rank generates its own shard of one global stream. */
int vega_gpu_gen_rdd_uniform(vega_ctx_t *ctx, uint64_t n, uint64_t seed,
                             int key_bits, uint64_t start, uint32_t nparts,
                             vega_rdd_t *out);

/* ---------------- PairRdd ops (pair_rdd.rs names) ---------------- */
/* reduce_by_key (pair_rdd.rs:54-80) with op as the aggregator */
int vega_gpu_reduce_by_key(vega_ctx_t *ctx, vega_rdd_t rdd, vega_op_t op,
                           uint32_t nparts, vega_rdd_t *out);
/* group_by_key -> (key, group size) (cfg C2; full group materialization is
 * host-side via collect of the sorted pairs) */
int vega_gpu_group_count(vega_ctx_t *ctx, vega_rdd_t rdd, uint32_t nparts,
                         vega_rdd_t *out);
/* sort_by_key ascending, signed i64 order, stable (absent from the
 * reference — SURVEY.md §8a a8; Spark semantics) */
int vega_gpu_sort_by_key(vega_ctx_t *ctx, vega_rdd_t rdd, vega_rdd_t *out);
/* inner join via sorted runs (co_grouped_rdd.rs:206-249 + pair_rdd.rs:104-121) */
int vega_gpu_join(vega_ctx_t *ctx, vega_rdd_t a, vega_rdd_t b, uint32_t nparts,
                  vega_rdd_t *out);
/* distinct (rdd.rs:501-531): the deduplicated KEY SET (the element column of
 * this typed engine is the key). Result values are zeroed — use
 * vega_gpu_group_count for (key, count). */
int vega_gpu_distinct(vega_ctx_t *ctx, vega_rdd_t rdd, uint32_t nparts,
                      vega_rdd_t *out);
/* group_by_key (pair_rdd.rs:35-52; aggregator.rs:33-53 Vec-collect): groups
 * materialized IN THE ENGINE — result is a grouped rdd holding distinct
 * keys, u64 offsets and the values column in grouped order (value order
 * within a group = row order; the grouping sort is stable). Collect with
 * vega_gpu_collect_groups. */
int vega_gpu_group_by_key(vega_ctx_t *ctx, vega_rdd_t rdd, uint32_t nparts,
                          vega_rdd_t *out);
/* cogroup (pair_rdd.rs:123-155 via co_grouped_rdd.rs:206-249): for every key
 * in EITHER side, the (Vec<V>, Vec<W>) ranges. Single call, host outputs:
 * keys/offa/lena/offb/lenb sized cap (na+nb always suffices; the needed
 * count is returned in *nk even on VEGA_ERR_CAP), vala[na] and valb[nb] are
 * the two value columns in grouped order; key i's A values are
 * vala[offa[i] .. offa[i]+lena[i]), likewise B. */
int vega_gpu_cogroup_collect(vega_ctx_t *ctx, vega_rdd_t a, vega_rdd_t b,
                             int64_t *keys, uint64_t *offa, uint64_t *lena,
                             uint64_t *offb, uint64_t *lenb, int64_t *vala,
                             int64_t *valb, uint64_t cap, uint64_t *nk);
/* intersection / subtract (rdd.rs compositions over CoGroupedRdd): key-set
 * semantics — distinct keys present in both / in a only. Values zeroed. */
int vega_gpu_intersection(vega_ctx_t *ctx, vega_rdd_t a, vega_rdd_t b,
                          uint32_t nparts, vega_rdd_t *out);
int vega_gpu_subtract(vega_ctx_t *ctx, vega_rdd_t a, vega_rdd_t b,
                      uint32_t nparts, vega_rdd_t *out);
/* count_by_value (rdd.rs:449-459 = map(x->(x,1)) + reduce_by_key(+)):
 * counts over the VALUE column; result rows are (value, count) */
int vega_gpu_count_by_value(vega_ctx_t *ctx, vega_rdd_t rdd, uint32_t nparts,
                            vega_rdd_t *out);

/* narrow transforms (device-resident; stable order for filter) */
int vega_gpu_map(vega_ctx_t *ctx, vega_rdd_t rdd, vega_map_op_t op, int64_t p0,
                 vega_rdd_t *out);
int vega_gpu_filter(vega_ctx_t *ctx, vega_rdd_t rdd, vega_pred_t pred,
                    int64_t p0, int64_t p1, vega_rdd_t *out);

/* ---------------- actions ---------------- */
int vega_gpu_count(vega_ctx_t *ctx, vega_rdd_t rdd, uint64_t *n);
/* collect (rdd.rs:420-434): D2H of the rows. Call with keys==NULL to query n. */
int vega_gpu_collect(vega_ctx_t *ctx, vega_rdd_t rdd, int64_t *keys, void *vals,
                     uint64_t *n);
/* collect of a join result (K,(V,W)) — pair_rdd.rs:104-121's output shape */
int vega_gpu_collect_join(vega_ctx_t *ctx, vega_rdd_t rdd, int64_t *keys,
                          int64_t *va, int64_t *vb, uint64_t *n);
/* collect of a grouped rdd (vega_gpu_group_by_key): keys[nk],
 * offsets[nk+1] (u64), values[nvals] (int64 or double per the source rdd).
 * Query sizes with keys == NULL. */
int vega_gpu_collect_groups(vega_ctx_t *ctx, vega_rdd_t rdd, int64_t *keys,
                            uint64_t *offsets, void *values, uint64_t *nk,
                            uint64_t *nvals);
int vega_gpu_free_rdd(vega_ctx_t *ctx, vega_rdd_t rdd);

/* ---------------- profiling ---------------- */
int vega_gpu_set_profiling(vega_ctx_t *ctx, int enabled);
/* JSON {"kernel":{"ms":total_ms,"n":launches},...} since last enable */
int vega_gpu_kernel_stats(vega_ctx_t *ctx, char *buf, size_t buflen);

/* =========== device-pointer API (rank-per-GPU launcher) =========== */
/* stream: a hipStream_t cast to void*; all calls async on that stream unless
 * noted. Buffers are caller-owned DEVICE pointers (e.g. torch tensors). */

/* workspace bytes required for n rows (sort temps + scan scratch) */
size_t vega_dev_ws_bytes(uint64_t n);

/* deterministic uniform generator kernel (rows [start, start+n) of stream
 * `seed`, keys masked to key_bits) */
int vega_dev_gen_uniform_i64(void *stream, int64_t *keys, int64_t *vals,
                             uint64_t n, uint64_t seed, int key_bits,
                             uint64_t start);
/* f64-value variant (values exact dyadic uniform [0,1), bit-identical to
 * datagen.c's vega_gen_uniform_pairs_f64) */
int vega_dev_gen_uniform_f64(void *stream, int64_t *keys, double *vals,
                             uint64_t n, uint64_t seed, int key_bits,
                             uint64_t start);

/* map-side radix partition (replaces the per-row get_partition + HashMap of
 * dependency.rs:191-210): bucket of row = splitmix64(key) % nparts; rows
 * scattered bucket-contiguous into out_k/out_v (each n rows), per-bucket
 * counts written to h_counts[nparts] ON THE HOST after an internal stream
 * sync (the exchange plan needs them). nparts <= 256. */
int vega_dev_partition_i64(void *stream, const int64_t *keys, const int64_t *vals,
                           uint64_t n, uint32_t nparts,
                           int64_t *out_k, int64_t *out_v,
                           uint64_t *h_counts, void *d_ws, size_t ws_bytes);

/* range partition for sort_by_key's exchange (Spark-style range
 * partitioner): bucket = # splitters <= key (signed order); d_splitters is
 * a DEVICE array of nparts-1 ascending splitters. */
int vega_dev_partition_range_i64(void *stream, const int64_t *keys, const int64_t *vals,
                                 uint64_t n, uint32_t nparts, const int64_t *d_splitters,
                                 int64_t *out_k, int64_t *out_v, uint64_t *h_counts,
                                 void *d_ws, size_t ws_bytes);

/* reduce-side grouping + segmented aggregate (replaces
 * shuffled_rdd.rs:154-164's HashMap merge_combiners): adaptive grouping sort
 * (skipped key radix for narrow keys, else 32/40-bit splitmix64-hash radix
 * with an exact collision cleanup) then one combiner per equal-key run.
 * in_k/in_v are NOT modified. out arrays must hold n rows (worst case
 * all-distinct). *h_nout = #distinct keys (host, after internal sync).
 * vals/out_v are int64 for SUM_I64/COUNT/MIN/MAX, double for SUM_F64. */
int vega_dev_sort_reduce(void *stream, const int64_t *in_k, const void *in_v,
                         uint64_t n, int op, int64_t *out_k, void *out_v,
                         uint64_t *h_nout, void *d_ws, size_t ws_bytes);

/* stable LSB radix sort by signed-i64 key (sort_by_key). In-place semantics:
 * result lands back in keys/vals. */
int vega_dev_sort_pairs_i64(void *stream, int64_t *keys, int64_t *vals,
                            uint64_t n, void *d_ws, size_t ws_bytes);

/* sort-merge inner join of two KEY-SORTED sides (K4): counts pass + emit
 * pass. *h_nout = rows emitted (<= cap). out_* sized cap. */
int vega_dev_join_sorted(void *stream,
                         const int64_t *ak, const int64_t *av, uint64_t na,
                         const int64_t *bk, const int64_t *bv, uint64_t nb,
                         int64_t *out_k, int64_t *out_va, int64_t *out_vb,
                         uint64_t cap, uint64_t *h_nout,
                         void *d_ws, size_t ws_bytes);

/* bring rows into the GROUPING order in place ((h32(key), key) unsigned
 * lexicographic — 4-5 hash radix passes + collision cleanup; cheaper than a
 * full signed sort when only co-grouping matters, e.g. before a join) */
int vega_dev_group_pairs_i64(void *stream, int64_t *keys, int64_t *vals, uint64_t n,
                             int *h_order_tag, void *d_ws, size_t ws_bytes);

/* sort-merge inner join; order_mode must match how BOTH sides were sorted:
 * 0 = signed-key order, 1 = unsigned-key order, 2 = grouping order (tag 4) */
int vega_dev_join_grouped(void *stream, const int64_t *ak, const int64_t *av, uint64_t na,
                          const int64_t *bk, const int64_t *bv, uint64_t nb,
                          int order_mode,
                          int64_t *out_k, int64_t *out_va, int64_t *out_vb,
                          uint64_t cap, uint64_t *h_nout, void *d_ws, size_t ws_bytes);

/* order-independent multiset checksum of device rows (same formula as
 * oracle_checksum_pairs_i64) — large-size parity checks without D2H */
int vega_dev_checksum_pairs(void *stream, const int64_t *keys, const int64_t *vals,
                            uint64_t n, uint64_t *h_sum, void *d_ws, size_t ws_bytes);

#ifdef __cplusplus
}
#endif
#endif /* VEGA_GPU_H */
