/* oracle.c — CPU restatement of vega's shuffle/aggregate hot-path semantics.
 *
 * !!! TEST INFRASTRUCTURE ONLY !!!
 * This library is the parity oracle and the CPU baseline timer. Only tests/,
 * __graft_entry__.smoke() (as the checker) and bench.py's cpu_baseline leg may
 * load or call it. The product GPU path must never route through this code.
 *
 * It restates, for i64 keys with i64/f64 values, exactly:
 *   - map-side bucketed HashMap combine ..... /root/reference/src/dependency.rs:164-229
 *     (one HashMap per output split :176-178; per-row get_partition + merge_value
 *      :191-210; per-bucket emission :212-223)
 *   - reduce_by_key aggregator ............. /root/reference/src/rdd/pair_rdd.rs:54-80
 *     (create_combiner = identity, merge_value = merge_combiners = f, :74-78)
 *   - group_by_key default aggregator ...... /root/reference/src/aggregator.rs:33-53
 *     (create = vec![v], merge_value = push, merge_combiners = append)
 *   - reduce-side HashMap merge ............ /root/reference/src/rdd/shuffled_rdd.rs:149-170
 *     (per reduce partition: merge_combiners over fetched map chunks :154-164)
 *   - cogroup / join ....................... /root/reference/src/rdd/co_grouped_rdd.rs:206-249
 *     + cross-product flat_map_values ...... /root/reference/src/rdd/pair_rdd.rs:104-121
 *   - input slicing ........................ /root/reference/src/rdd/parallel_collection_rdd.rs:116-145
 *   - sort_by_key .......................... ABSENT from the reference (verified by grep;
 *     nearest is take_ordered, rdd.rs:1106-1153). We implement Spark-semantics ascending
 *     sort; the oracle is a stable CPU sort by key.
 *
 * PARITY PINNING: golden vectors transcribed from the reference's own tests
 * (tests/test_pair_rdd.rs:9-135, tests/test_rdd.rs) live in tests/golden/ and
 * pin this restatement; see tests/golden/README.md. Partition ASSIGNMENT is
 * unpinned (reference hash = fasthash::MetroHash64, not vendored, no test
 * asserts concrete values — SURVEY.md §8c); result-set parity is invariant
 * under any deterministic total partition function, and comparisons are done
 * on sorted collected output exactly as the reference tests do
 * (test_pair_rdd.rs:30-36). The reference CANNOT be compiled here (no
 * rustc/cargo, toolchain pinned to nightly-2020-05-31, no network), so there
 * is no oracle/_ref build; parity is pinned by the golden vectors plus an
 * independent numpy restatement in tests/.
 *
 * Ordering semantics this restatement preserves (and the GPU path must match
 * for group values): within a group, values appear in global row order —
 * map side appends in row order within each input partition (dependency.rs
 * :191-210 iterates the split in order), reduce side merges map chunks in map
 * partition order (shuffled_rdd.rs:154-164 with merge_combiners = append).
 * Reduce emission order is unpinned (Rust HashMap iteration); we canonicalize
 * all outputs by sorting by key (ascending; values in group order).
 *
 * f64 sums: the reference's per-key merge order IS pinned (row order within a
 * partition, then partition order) but within-bucket chunk-internal order is
 * HashMap-arbitrary; our f64 path follows first-seen insertion order, and GPU
 * comparisons use the 1e-6 relative tolerance BASELINE.json names.
 *
 * Build: oracle/Makefile -> liboracle.so (gcc -O2 -fopenmp).
 */

#include <stdint.h>
#include <stdlib.h>
#include <string.h>
#include <stdio.h>
#ifdef _OPENMP
#include <omp.h>
#endif

#include "../include/vega_common.h"

/* ------------------------------------------------------------------ */
/* open-addressing hash table: i64 key -> slot index, insertion-ordered */

typedef struct {
    int64_t *keys;      /* key per slot */
    int64_t *iv;        /* i64 combiner per slot */
    double  *fv;        /* f64 combiner per slot */
    uint8_t *used;
    uint32_t *order;    /* slot -> insertion rank (unused) */
    uint64_t *ins;      /* insertion rank -> slot */
    uint64_t cap;       /* power of two */
    uint64_t count;
} table_t;

static void tbl_init(table_t *t, uint64_t expected, int want_f64) {
    uint64_t cap = 16;
    while (cap < expected * 2) cap <<= 1;
    t->cap = cap; t->count = 0;
    t->keys = (int64_t *)malloc(cap * sizeof(int64_t));
    t->iv   = want_f64 ? NULL : (int64_t *)malloc(cap * sizeof(int64_t));
    t->fv   = want_f64 ? (double *)malloc(cap * sizeof(double)) : NULL;
    t->used = (uint8_t *)calloc(cap, 1);
    t->order = NULL;
    t->ins  = (uint64_t *)malloc(cap * sizeof(uint64_t));
}

static void tbl_free(table_t *t) {
    free(t->keys); free(t->iv); free(t->fv); free(t->used); free(t->ins);
}

static void tbl_grow(table_t *t) {
    table_t n;
    n.cap = t->cap << 1; n.count = t->count;
    n.keys = (int64_t *)malloc(n.cap * sizeof(int64_t));
    n.iv = t->iv ? (int64_t *)malloc(n.cap * sizeof(int64_t)) : NULL;
    n.fv = t->fv ? (double *)malloc(n.cap * sizeof(double)) : NULL;
    n.used = (uint8_t *)calloc(n.cap, 1);
    n.order = NULL;
    n.ins = (uint64_t *)malloc(n.cap * sizeof(uint64_t));
    for (uint64_t r = 0; r < t->count; r++) {
        uint64_t s = t->ins[r];
        uint64_t h = vega_hash_u64((uint64_t)t->keys[s]) & (n.cap - 1);
        while (n.used[h]) h = (h + 1) & (n.cap - 1);
        n.used[h] = 1; n.keys[h] = t->keys[s];
        if (n.iv) n.iv[h] = t->iv[s];
        if (n.fv) n.fv[h] = t->fv[s];
        n.ins[r] = h;
    }
    tbl_free(t);
    *t = n;
}

/* find-or-insert; returns slot, sets *fresh */
static uint64_t tbl_probe(table_t *t, int64_t k, int *fresh) {
    if (t->count * 2 >= t->cap) tbl_grow(t);
    uint64_t h = vega_hash_u64((uint64_t)k) & (t->cap - 1);
    for (;;) {
        if (!t->used[h]) {
            t->used[h] = 1; t->keys[h] = k;
            t->ins[t->count++] = h;
            *fresh = 1;
            return h;
        }
        if (t->keys[h] == k) { *fresh = 0; return h; }
        h = (h + 1) & (t->cap - 1);
    }
}

/* ------------------------------------------------------------------ */
/* kv vector */
typedef struct { int64_t *k; int64_t *iv; double *fv; uint64_t n, cap; } kvec_t;

static void kv_push(kvec_t *v, int64_t k, int64_t iv, double fv, int want_f64) {
    if (v->n == v->cap) {
        v->cap = v->cap ? v->cap * 2 : 64;
        v->k = (int64_t *)realloc(v->k, v->cap * sizeof(int64_t));
        if (want_f64) v->fv = (double *)realloc(v->fv, v->cap * sizeof(double));
        else v->iv = (int64_t *)realloc(v->iv, v->cap * sizeof(int64_t));
    }
    v->k[v->n] = k;
    if (want_f64) v->fv[v->n] = fv; else v->iv[v->n] = iv;
    v->n++;
}

/* ------------------------------------------------------------------ */
/* exported helpers */

uint64_t oracle_hash_i64(int64_t k) { return vega_hash_u64((uint64_t)k); }
uint32_t oracle_partition_of(int64_t k, uint32_t nparts) { return vega_partition_of(k, nparts); }

/* ParallelCollection::slice bounds (parallel_collection_rdd.rs:116-145):
 * bounds[p] = floor(p*n/P); caller provides bounds[nparts+1]. */
void oracle_slice_bounds(uint64_t n, uint32_t nparts, uint64_t *bounds) {
    for (uint32_t p = 0; p <= nparts; p++)
        bounds[p] = vega_slice_start(n, nparts, p);
}

/* order-independent multiset checksum of (k,v) rows */
uint64_t oracle_checksum_pairs_i64(const int64_t *k, const int64_t *v, uint64_t n) {
    uint64_t acc = 0;
#ifdef _OPENMP
#pragma omp parallel for reduction(+:acc) schedule(static)
#endif
    for (uint64_t i = 0; i < n; i++)
        acc += vega_hash_u64(vega_hash_u64((uint64_t)k[i]) ^ (uint64_t)v[i]);
    return acc;
}

/* ------------------------------------------------------------------ */
/* the hot path: map-side bucketed combine + reduce-side merge.
 *
 * op: 0 = SUM_I64 (reduce_by_key(+), pair_rdd.rs:74-78, wrapping i64 add —
 *        Rust release mode wraps), 1 = COUNT (group_by_key then count,
 *        aggregator.rs:33-53 with only the group SIZE materialized),
 *        2 = SUM_F64 (f64 values).
 *
 * Returns number of output rows, or -1 if cap is too small (callers size cap
 * generously; -1 means "retry with bigger buffers" in tests).
 */
typedef struct { kvec_t *bucket; } map_out_t; /* [nparts_out] per input part */

static int64_t shuffle_agg(const int64_t *keys, const void *vals, uint64_t n,
                           uint32_t nparts_in, uint32_t nparts_out, int op,
                           int64_t *out_k, void *out_v, uint64_t cap) {
    int want_f64 = (op == 2);
    const int64_t *iv = (const int64_t *)vals;
    const double *dv = (const double *)vals;

    map_out_t *mo = (map_out_t *)calloc(nparts_in, sizeof(map_out_t));

    /* ---- map side: one task per input partition (dependency.rs:164-229);
     * vega runs these on a thread pool (local_scheduler.rs:336-352). */
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic, 1)
#endif
    for (uint32_t p = 0; p < nparts_in; p++) {
        uint64_t lo = vega_slice_start(n, nparts_in, p);
        uint64_t hi = vega_slice_start(n, nparts_in, p + 1);
        table_t t; tbl_init(&t, (hi - lo) / 2 + 16, want_f64);
        for (uint64_t i = lo; i < hi; i++) {
            int fresh;
            uint64_t s = tbl_probe(&t, keys[i], &fresh);
            /* aggregator closures (pair_rdd.rs:74-78 / aggregator.rs:33-53) */
            if (op == 0) t.iv[s] = fresh ? iv[i] : (int64_t)((uint64_t)t.iv[s] + (uint64_t)iv[i]);
            else if (op == 1) t.iv[s] = fresh ? 1 : t.iv[s] + 1;
            else t.fv[s] = fresh ? dv[i] : t.fv[s] + dv[i];
        }
        /* emit per output bucket, insertion order (dependency.rs:212-223) */
        mo[p].bucket = (kvec_t *)calloc(nparts_out, sizeof(kvec_t));
        for (uint64_t r = 0; r < t.count; r++) {
            uint64_t s = t.ins[r];
            uint32_t b = vega_partition_of(t.keys[s], nparts_out);
            kv_push(&mo[p].bucket[b], t.keys[s], want_f64 ? 0 : t.iv[s],
                    want_f64 ? t.fv[s] : 0, want_f64);
        }
        tbl_free(&t);
    }

    /* ---- reduce side: one task per output partition
     * (shuffled_rdd.rs:149-170), chunks merged in map-partition order. */
    int64_t total = 0;
    int overflow = 0;
    kvec_t *red = (kvec_t *)calloc(nparts_out, sizeof(kvec_t));
#ifdef _OPENMP
#pragma omp parallel for schedule(dynamic, 1)
#endif
    for (uint32_t b = 0; b < nparts_out; b++) {
        uint64_t exp = 16;
        for (uint32_t p = 0; p < nparts_in; p++) exp += mo[p].bucket[b].n;
        table_t t; tbl_init(&t, exp, want_f64);
        for (uint32_t p = 0; p < nparts_in; p++) {
            kvec_t *c = &mo[p].bucket[b];
            for (uint64_t i = 0; i < c->n; i++) {
                int fresh;
                uint64_t s = tbl_probe(&t, c->k[i], &fresh);
                /* merge_combiners (shuffled_rdd.rs:154-164) */
                if (op == 0) t.iv[s] = fresh ? c->iv[i] : (int64_t)((uint64_t)t.iv[s] + (uint64_t)c->iv[i]);
                else if (op == 1) t.iv[s] = fresh ? c->iv[i] : t.iv[s] + c->iv[i];
                else t.fv[s] = fresh ? c->fv[i] : t.fv[s] + c->fv[i];
            }
            free(c->k); free(c->iv); free(c->fv);
        }
        for (uint64_t r = 0; r < t.count; r++) {
            uint64_t s = t.ins[r];
            kv_push(&red[b], t.keys[s], want_f64 ? 0 : t.iv[s],
                    want_f64 ? t.fv[s] : 0, want_f64);
        }
        int64_t cnt = (int64_t)t.count;
        tbl_free(&t);
#ifdef _OPENMP
#pragma omp atomic
#endif
        total += cnt;
    }
    for (uint32_t p = 0; p < nparts_in; p++) free(mo[p].bucket);
    free(mo);

    if ((uint64_t)total > cap) overflow = 1;
    /* concat reduce partitions in partition order (collect semantics,
     * context.rs:457-473 gathers results by output partition id) */
    uint64_t off = 0;
    if (!overflow) {
        for (uint32_t b = 0; b < nparts_out; b++) {
            memcpy(out_k + off, red[b].k, red[b].n * sizeof(int64_t));
            if (want_f64) memcpy((double *)out_v + off, red[b].fv, red[b].n * sizeof(double));
            else memcpy((int64_t *)out_v + off, red[b].iv, red[b].n * sizeof(int64_t));
            off += red[b].n;
        }
    }
    for (uint32_t b = 0; b < nparts_out; b++) { free(red[b].k); free(red[b].iv); free(red[b].fv); }
    free(red);
    return overflow ? -1 : (int64_t)off;
}

int64_t oracle_reduce_by_key_i64(const int64_t *keys, const int64_t *vals, uint64_t n,
                                 uint32_t nparts_in, uint32_t nparts_out,
                                 int64_t *out_k, int64_t *out_v, uint64_t cap) {
    return shuffle_agg(keys, vals, n, nparts_in, nparts_out, 0, out_k, out_v, cap);
}

int64_t oracle_group_count_i64(const int64_t *keys, const int64_t *vals, uint64_t n,
                               uint32_t nparts_in, uint32_t nparts_out,
                               int64_t *out_k, int64_t *out_v, uint64_t cap) {
    (void)vals; /* counts ignore values */
    return shuffle_agg(keys, vals, n, nparts_in, nparts_out, 1, out_k, out_v, cap);
}

int64_t oracle_reduce_by_key_f64(const int64_t *keys, const double *vals, uint64_t n,
                                 uint32_t nparts_in, uint32_t nparts_out,
                                 int64_t *out_k, double *out_v, uint64_t cap) {
    return shuffle_agg(keys, vals, n, nparts_in, nparts_out, 2, out_k, out_v, cap);
}

/* ------------------------------------------------------------------ */
/* group_by_key with full value lists (aggregator.rs:33-53 default
 * aggregator). Output: keys (first-seen order per reduce partition,
 * concatenated over partitions), offsets[ngroups+1] into out_vals, values of
 * each group in global row order (see ordering note in the header). */
int64_t oracle_group_by_key_i64(const int64_t *keys, const int64_t *vals, uint64_t n,
                                uint32_t nparts_in, uint32_t nparts_out,
                                int64_t *out_k, uint64_t *out_off, int64_t *out_vals,
                                uint64_t key_cap, uint64_t val_cap) {
    /* map side: per input partition, per key a value vector */
    typedef struct { int64_t *v; uint64_t n, cap; } vvec_t;
    uint64_t total_keys = 0, total_vals = 0;

    /* simple single-threaded restatement (parity sizes are small/medium) */
    /* per (input partition, bucket): list of (key, vec) */
    typedef struct { int64_t key; vvec_t vv; } kgroup_t;
    typedef struct { kgroup_t *g; uint64_t n, cap; } glist_t;
    glist_t *mo = (glist_t *)calloc((uint64_t)nparts_in * nparts_out, sizeof(glist_t));

    for (uint32_t p = 0; p < nparts_in; p++) {
        uint64_t lo = vega_slice_start(n, nparts_in, p);
        uint64_t hi = vega_slice_start(n, nparts_in, p + 1);
        /* presize so the table NEVER grows (growth would invalidate the
         * slot-indexed per_slot vectors): cap >= 2*(rows+1) > 2*count always */
        table_t t; tbl_init(&t, (hi - lo) + 1, 0);
        uint64_t cap_guard = t.cap;
        vvec_t *per_slot = (vvec_t *)calloc(t.cap, sizeof(vvec_t));
        for (uint64_t i = lo; i < hi; i++) {
            int fresh;
            uint64_t s = tbl_probe(&t, keys[i], &fresh);
            if (t.cap != cap_guard) { fprintf(stderr, "oracle: table grew unexpectedly\n"); abort(); }
            vvec_t *vv = &per_slot[s];
            if (vv->n == vv->cap) {
                vv->cap = vv->cap ? vv->cap * 2 : 4;
                vv->v = (int64_t *)realloc(vv->v, vv->cap * sizeof(int64_t));
            }
            vv->v[vv->n++] = vals[i]; /* push in row order (aggregator.rs:38-41) */
        }
        for (uint64_t r = 0; r < t.count; r++) {
            uint64_t s = t.ins[r];
            uint32_t b = vega_partition_of(t.keys[s], nparts_out);
            glist_t *gl = &mo[(uint64_t)p * nparts_out + b];
            if (gl->n == gl->cap) {
                gl->cap = gl->cap ? gl->cap * 2 : 16;
                gl->g = (kgroup_t *)realloc(gl->g, gl->cap * sizeof(kgroup_t));
            }
            gl->g[gl->n].key = t.keys[s];
            gl->g[gl->n].vv = per_slot[s];
            gl->n++;
        }
        free(per_slot);
        tbl_free(&t);
    }

    /* reduce side: append chunk vectors in map-partition order
     * (merge_combiners = append, aggregator.rs:42-46) */
    uint64_t ko = 0, vo = 0;
    int overflow = 0;
    for (uint32_t b = 0; b < nparts_out && !overflow; b++) {
        uint64_t exp = 1;
        for (uint32_t p = 0; p < nparts_in; p++) exp += mo[(uint64_t)p * nparts_out + b].n;
        /* presize: growth never happens (count <= exp-1 < cap/2) */
        table_t t; tbl_init(&t, exp, 0);
        uint64_t cap_guard = t.cap;
        vvec_t *per_slot = (vvec_t *)calloc(t.cap, sizeof(vvec_t));
        for (uint32_t p = 0; p < nparts_in; p++) {
            glist_t *gl = &mo[(uint64_t)p * nparts_out + b];
            for (uint64_t i = 0; i < gl->n; i++) {
                int fresh;
                uint64_t s = tbl_probe(&t, gl->g[i].key, &fresh);
                if (t.cap != cap_guard) { fprintf(stderr, "oracle: table grew unexpectedly\n"); abort(); }
                vvec_t *dst = &per_slot[s];
                vvec_t *src = &gl->g[i].vv;
                if (fresh && dst->n == 0 && dst->cap == 0) {
                    *dst = *src; /* take ownership of first chunk */
                } else {
                    if (dst->n + src->n > dst->cap) {
                        dst->cap = dst->n + src->n;
                        dst->v = (int64_t *)realloc(dst->v, dst->cap * sizeof(int64_t));
                    }
                    memcpy(dst->v + dst->n, src->v, src->n * sizeof(int64_t));
                    dst->n += src->n;
                    free(src->v);
                }
            }
        }
        for (uint64_t r = 0; r < t.count; r++) {
            uint64_t s = t.ins[r];
            if (ko >= key_cap || vo + per_slot[s].n > val_cap) { overflow = 1; break; }
            out_k[ko] = t.keys[s];
            out_off[ko] = vo;
            memcpy(out_vals + vo, per_slot[s].v, per_slot[s].n * sizeof(int64_t));
            vo += per_slot[s].n;
            ko++;
            free(per_slot[s].v);
        }
        free(per_slot);
        tbl_free(&t);
    }
    for (uint64_t i = 0; i < (uint64_t)nparts_in * nparts_out; i++) free(mo[i].g);
    free(mo);
    if (overflow) return -1;
    out_off[ko] = vo;
    total_keys = ko; total_vals = vo; (void)total_vals;
    return (int64_t)total_keys;
}

/* ------------------------------------------------------------------ */
/* sort_by_key: absent from the reference. Spark-semantics
 * ascending stable sort by key; values keep row order within equal keys. */
static int cmp_pair(const void *a, const void *b) {
    const int64_t *x = (const int64_t *)a, *y = (const int64_t *)b;
    if (x[0] < y[0]) return -1;
    if (x[0] > y[0]) return 1;
    /* stability surrogate: original index stored in [2] */
    if (x[2] < y[2]) return -1;
    if (x[2] > y[2]) return 1;
    return 0;
}

void oracle_sort_by_key_i64(const int64_t *keys, const int64_t *vals, uint64_t n,
                            int64_t *out_k, int64_t *out_v) {
    int64_t *tmp = (int64_t *)malloc(n * 3 * sizeof(int64_t));
    for (uint64_t i = 0; i < n; i++) {
        tmp[i * 3] = keys[i]; tmp[i * 3 + 1] = vals[i]; tmp[i * 3 + 2] = (int64_t)i;
    }
    qsort(tmp, n, 3 * sizeof(int64_t), cmp_pair);
    for (uint64_t i = 0; i < n; i++) { out_k[i] = tmp[i * 3]; out_v[i] = tmp[i * 3 + 1]; }
    free(tmp);
}

/* ------------------------------------------------------------------ */
/* inner join via cogroup (co_grouped_rdd.rs:206-249 + pair_rdd.rs:104-121):
 * for each key present on both sides, emit the cross product va × vb, va in
 * side-a row order outer, vb inner (flat_map_values :109-115). */
int64_t oracle_join_i64(const int64_t *ak, const int64_t *av, uint64_t na,
                        const int64_t *bk, const int64_t *bv, uint64_t nb,
                        uint32_t nparts_in, uint32_t nparts_out,
                        int64_t *out_k, int64_t *out_va, int64_t *out_vb, uint64_t cap) {
    /* group each side (values in row order) */
    uint64_t kcapa = na + 1, kcapb = nb + 1;
    int64_t *gak = (int64_t *)malloc(kcapa * sizeof(int64_t));
    uint64_t *gao = (uint64_t *)malloc((kcapa + 1) * sizeof(uint64_t));
    int64_t *gav = (int64_t *)malloc(na * sizeof(int64_t));
    int64_t nga = oracle_group_by_key_i64(ak, av, na, nparts_in, nparts_out, gak, gao, gav, kcapa, na);
    int64_t *gbk = (int64_t *)malloc(kcapb * sizeof(int64_t));
    uint64_t *gbo = (uint64_t *)malloc((kcapb + 1) * sizeof(uint64_t));
    int64_t *gbv = (int64_t *)malloc(nb * sizeof(int64_t));
    int64_t ngb = oracle_group_by_key_i64(bk, bv, nb, nparts_in, nparts_out, gbk, gbo, gbv, kcapb, nb);
    if (nga < 0 || ngb < 0) return -1;

    /* index side b by key */
    table_t t; tbl_init(&t, (uint64_t)ngb + 16, 0);
    for (int64_t i = 0; i < ngb; i++) {
        int fresh;
        uint64_t s = tbl_probe(&t, gbk[i], &fresh);
        t.iv[s] = i;
    }
    uint64_t off = 0;
    int overflow = 0;
    for (int64_t i = 0; i < nga && !overflow; i++) {
        /* lookup without insert */
        uint64_t h = vega_hash_u64((uint64_t)gak[i]) & (t.cap - 1);
        int64_t j = -1;
        while (t.used[h]) {
            if (t.keys[h] == gak[i]) { j = t.iv[h]; break; }
            h = (h + 1) & (t.cap - 1);
        }
        if (j < 0) continue;
        for (uint64_t x = gao[i]; x < gao[i + 1] && !overflow; x++)
            for (uint64_t y = gbo[j]; y < gbo[j + 1]; y++) {
                if (off >= cap) { overflow = 1; break; }
                out_k[off] = gak[i]; out_va[off] = gav[x]; out_vb[off] = gbv[y]; off++;
            }
    }
    tbl_free(&t);
    free(gak); free(gao); free(gav); free(gbk); free(gbo); free(gbv);
    return overflow ? -1 : (int64_t)off;
}

/* ------------------------------------------------------------------ */
/* distinct (rdd.rs:501-531 = map to (k, None) + reduce_by_key + map back) and
 * count_by_value (rdd.rs:449-459 = map to (v,1) + reduce_by_key(+)) fall out
 * of reduce_by_key; exposed for the f1 "next rows" coverage. */
int64_t oracle_distinct_i64(const int64_t *keys, uint64_t n,
                            uint32_t nparts_in, uint32_t nparts_out,
                            int64_t *out_k, uint64_t cap) {
    int64_t *vals = (int64_t *)calloc(n ? n : 1, sizeof(int64_t));
    int64_t *ov = (int64_t *)malloc((cap ? cap : 1) * sizeof(int64_t));
    int64_t r = shuffle_agg(keys, vals, n, nparts_in, nparts_out, 0, out_k, ov, cap);
    free(vals); free(ov);
    return r;
}
