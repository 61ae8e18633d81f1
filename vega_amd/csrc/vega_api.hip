/* vega_api.hip — the C-ABI (include/vega_gpu.h) over the CDNA4 kernels.
 *
 * RDD-handle API: the single-process drop-in for the reference's
 * Context/PairRdd seam (context.rs:406-442, pair_rdd.rs:20-171); see
 * INTEGRATION.md for the Rust-side binding. Device-pointer API: raw entries
 * for the rank-per-GPU launcher (torch.distributed / RCCL over xGMI).
 *
 * Fails loudly: every entry returns a negative VEGA_ERR_* on any HIP error;
 * there is no CPU fallback anywhere in this library.
 */
#include <hip/hip_runtime.h>
#include <rccl/rccl.h>
#include <cstdio>
#include <cstring>
#include <map>
#include <string>
#include <vector>

#include "../../include/vega_common.h"
#include "../../include/vega_gpu.h"
#include "vega_internal.h"

using namespace vega;

struct RddImpl {
    uint64_t n = 0;
    int vtype = 0; /* 0 = i64 vals, 1 = f64 vals */
    int64_t *d_k = nullptr;
    void *d_v = nullptr;
    void *d_v2 = nullptr; /* second value column (join output (K,(V,W))) */
    uint64_t alloc_rows = 0;
    uint32_t nparts = 1;
    bool sorted = false;
    /* grouped rdd (group_by_key): d_k = distinct keys (n of them), d_v =
     * u64 offsets (n+1), d_v2 = values in grouped order (n2 rows) */
    bool grouped = false;
    uint64_t n2 = 0;
    /* multi-GPU (ngpus > 1): per-device shards; d_k/d_v above are device 0's
     * shard so the single-GPU code paths stay untouched for G == 1 */
    std::vector<int64_t *> mk;
    std::vector<void *> mv;
    std::vector<uint64_t> mn;
};

struct vega_ctx {
    int device = 0;
    hipStream_t stream = nullptr;
    std::map<uint64_t, RddImpl *> rdds;
    uint64_t next_id = 1;
    void *ws = nullptr;
    size_t ws_bytes = 0;
    char err[512] = {0};
    /* multi-GPU (one process drives all G devices; vega local-mode shape):
     * per-device streams, RCCL communicators, workspaces */
    int ngpus = 1;
    std::vector<hipStream_t> mstreams;
    std::vector<ncclComm_t> comms;
    std::vector<void *> mws;
    std::vector<size_t> mws_bytes;
};

#define CTX_TRY(ctx, x)                                                        \
    do {                                                                       \
        hipError_t _e = (x);                                                   \
        if (_e != hipSuccess) {                                                \
            snprintf((ctx)->err, sizeof (ctx)->err, "%s:%d: %s", __FILE__,     \
                     __LINE__, hipGetErrorString(_e));                         \
            return _e == hipErrorNotSupported ? VEGA_ERR_UNSUPPORTED           \
                                              : VEGA_ERR_HIP;                  \
        }                                                                      \
    } while (0)

static int ensure_ws(vega_ctx *c, uint64_t n) {
    size_t need = ws_bytes_for(n);
    if (c->ws_bytes >= need) return VEGA_OK;
    if (c->ws) (void)hipFree(c->ws);
    c->ws = nullptr;
    c->ws_bytes = 0;
    CTX_TRY(c, hipMalloc(&c->ws, need));
    c->ws_bytes = need;
    return VEGA_OK;
}

static RddImpl *get_rdd(vega_ctx *c, vega_rdd_t h) {
    auto it = c->rdds.find(h);
    return it == c->rdds.end() ? nullptr : it->second;
}

static int new_rdd(vega_ctx *c, uint64_t rows_alloc, int vtype, uint32_t nparts,
                   RddImpl **out, vega_rdd_t *hout) {
    RddImpl *r = new RddImpl();
    r->vtype = vtype;
    r->nparts = nparts ? nparts : 1;
    r->alloc_rows = rows_alloc;
    if (rows_alloc) {
        CTX_TRY(c, hipMalloc(&r->d_k, rows_alloc * 8));
        CTX_TRY(c, hipMalloc(&r->d_v, rows_alloc * 8));
    }
    uint64_t h = c->next_id++;
    c->rdds[h] = r;
    *out = r;
    *hout = h;
    return VEGA_OK;
}

extern "C" {

int vega_gpu_init(int ngpus, vega_ctx_t **out) {
    int ndev = 0;
    if (hipGetDeviceCount(&ndev) != hipSuccess || ndev < 1) return VEGA_ERR_HIP;
    if (ngpus < 1 || ngpus > ndev || ngpus > 256) return VEGA_ERR_INVALID;
    vega_ctx *c = new vega_ctx();
    c->ngpus = ngpus;
    (void)hipGetDevice(&c->device);
    if (hipStreamCreate(&c->stream) != hipSuccess) {
        delete c;
        return VEGA_ERR_HIP;
    }
    if (ngpus > 1) {
        /* one process drives all G devices over RCCL/xGMI (the in-process
         * analogue of the reference's local scheduler) */
        c->mstreams.resize(ngpus);
        c->comms.resize(ngpus);
        c->mws.assign(ngpus, nullptr);
        c->mws_bytes.assign(ngpus, 0);
        for (int g = 0; g < ngpus; ++g) {
            if (hipSetDevice(g) != hipSuccess ||
                hipStreamCreate(&c->mstreams[g]) != hipSuccess) {
                delete c;
                return VEGA_ERR_HIP;
            }
        }
        if (ncclCommInitAll(c->comms.data(), ngpus, nullptr) != ncclSuccess) {
            delete c;
            return VEGA_ERR_HIP;
        }
        (void)hipSetDevice(0);
    }
    *out = c;
    return VEGA_OK;
}

static void free_rdd_buffers(vega_ctx *c, RddImpl *r) {
    if (c->ngpus > 1 && !r->mk.empty()) {
        for (int g = 0; g < c->ngpus; ++g) {
            (void)hipSetDevice(g);
            if (r->mk[g]) (void)hipFree(r->mk[g]);
            if (r->mv[g]) (void)hipFree(r->mv[g]);
        }
        (void)hipSetDevice(0);
    } else {
        if (r->d_k) (void)hipFree(r->d_k);
        if (r->d_v) (void)hipFree(r->d_v);
    }
    if (r->d_v2) (void)hipFree(r->d_v2);
}

int vega_gpu_shutdown(vega_ctx_t *c) {
    if (!c) return VEGA_ERR_INVALID;
    (void)hipStreamSynchronize(c->stream);
    for (int g = 0; g < (int)c->mstreams.size(); ++g) {
        (void)hipSetDevice(g);
        (void)hipStreamSynchronize(c->mstreams[g]);
    }
    for (auto &kv : c->rdds) {
        free_rdd_buffers(c, kv.second);
        delete kv.second;
    }
    if (c->ws) (void)hipFree(c->ws);
    for (int g = 0; g < (int)c->mws.size(); ++g) {
        if (c->mws[g]) {
            (void)hipSetDevice(g);
            (void)hipFree(c->mws[g]);
        }
    }
    for (auto &cm : c->comms) (void)ncclCommDestroy(cm);
    for (int g = 0; g < (int)c->mstreams.size(); ++g) {
        (void)hipSetDevice(g);
        (void)hipStreamDestroy(c->mstreams[g]);
    }
    (void)hipSetDevice(c->device);
    (void)hipStreamDestroy(c->stream);
    delete c;
    return VEGA_OK;
}

int vega_gpu_synchronize(vega_ctx_t *c) {
    if (!c) return VEGA_ERR_INVALID;
    CTX_TRY(c, hipStreamSynchronize(c->stream));
    return VEGA_OK;
}

const char *vega_gpu_last_error(vega_ctx_t *c) { return c ? c->err : "null ctx"; }

static int ensure_mws(vega_ctx *c, int g, uint64_t n) {
    size_t need = ws_bytes_for(n);
    if (c->mws_bytes[g] >= need) return VEGA_OK;
    CTX_TRY(c, hipSetDevice(g));
    if (c->mws[g]) (void)hipFree(c->mws[g]);
    c->mws[g] = nullptr;
    c->mws_bytes[g] = 0;
    CTX_TRY(c, hipMalloc(&c->mws[g], need));
    c->mws_bytes[g] = need;
    return VEGA_OK;
}

/* new multi-sharded rdd: alloc_g rows per device */
static int new_mrdd(vega_ctx *c, const std::vector<uint64_t> &alloc_g, int vtype,
                    uint32_t nparts, RddImpl **out, vega_rdd_t *hout) {
    RddImpl *r = new RddImpl();
    r->vtype = vtype;
    r->nparts = nparts ? nparts : 1;
    r->mk.assign(c->ngpus, nullptr);
    r->mv.assign(c->ngpus, nullptr);
    r->mn.assign(c->ngpus, 0);
    for (int g = 0; g < c->ngpus; ++g) {
        uint64_t a = alloc_g[g] ? alloc_g[g] : 1;
        CTX_TRY(c, hipSetDevice(g));
        void *k = nullptr, *v = nullptr;
        CTX_TRY(c, hipMalloc(&k, a * 8));
        CTX_TRY(c, hipMalloc(&v, a * 8));
        r->mk[g] = (int64_t *)k;
        r->mv[g] = v;
    }
    (void)hipSetDevice(0);
    uint64_t h = c->next_id++;
    c->rdds[h] = r;
    *out = r;
    *hout = h;
    return VEGA_OK;
}

static int make_rdd_common(vega_ctx *c, const int64_t *keys, const void *vals,
                           uint64_t n, uint32_t nparts, int vtype, vega_rdd_t *out) {
    if (!c || (!keys && n) || (!vals && n)) return VEGA_ERR_INVALID;
    if (c->ngpus > 1) { /* shard rows [g*n/G,(g+1)*n/G) per device (a9) */
        std::vector<uint64_t> sizes(c->ngpus);
        for (int g = 0; g < c->ngpus; ++g)
            sizes[g] = vega_slice_start(n, c->ngpus, g + 1) - vega_slice_start(n, c->ngpus, g);
        RddImpl *r;
        int rc = new_mrdd(c, sizes, vtype, nparts, &r, out);
        if (rc) return rc;
        for (int g = 0; g < c->ngpus; ++g) {
            uint64_t lo = vega_slice_start(n, c->ngpus, g);
            r->mn[g] = sizes[g];
            if (!sizes[g]) continue;
            CTX_TRY(c, hipSetDevice(g));
            CTX_TRY(c, hipMemcpyAsync(r->mk[g], keys + lo, sizes[g] * 8,
                                      hipMemcpyHostToDevice, c->mstreams[g]));
            CTX_TRY(c, hipMemcpyAsync(r->mv[g], (const char *)vals + lo * 8, sizes[g] * 8,
                                      hipMemcpyHostToDevice, c->mstreams[g]));
        }
        for (int g = 0; g < c->ngpus; ++g) CTX_TRY(c, hipStreamSynchronize(c->mstreams[g]));
        (void)hipSetDevice(0);
        r->n = n;
        return VEGA_OK;
    }
    RddImpl *r;
    int rc = new_rdd(c, n ? n : 1, vtype, nparts, &r, out);
    if (rc) return rc;
    r->n = n;
    if (n) {
        CTX_TRY(c, hipMemcpyAsync(r->d_k, keys, n * 8, hipMemcpyHostToDevice, c->stream));
        CTX_TRY(c, hipMemcpyAsync(r->d_v, vals, n * 8, hipMemcpyHostToDevice, c->stream));
        CTX_TRY(c, hipStreamSynchronize(c->stream));
    }
    return VEGA_OK;
}

int vega_gpu_make_rdd(vega_ctx_t *c, const int64_t *keys, const int64_t *vals,
                      uint64_t n, uint32_t nparts, vega_rdd_t *out) {
    return make_rdd_common(c, keys, vals, n, nparts, 0, out);
}
int vega_gpu_make_rdd_f64(vega_ctx_t *c, const int64_t *keys, const double *vals,
                          uint64_t n, uint32_t nparts, vega_rdd_t *out) {
    return make_rdd_common(c, keys, vals, n, nparts, 1, out);
}

int vega_gpu_gen_rdd_uniform(vega_ctx_t *c, uint64_t n, uint64_t seed, int key_bits,
                             uint64_t start, uint32_t nparts, vega_rdd_t *out) {
    if (!c) return VEGA_ERR_INVALID;
    if (c->ngpus > 1) {
        std::vector<uint64_t> sizes(c->ngpus);
        for (int g = 0; g < c->ngpus; ++g)
            sizes[g] = vega_slice_start(n, c->ngpus, g + 1) - vega_slice_start(n, c->ngpus, g);
        RddImpl *r;
        int rc = new_mrdd(c, sizes, 0, nparts, &r, out);
        if (rc) return rc;
        for (int g = 0; g < c->ngpus; ++g) {
            r->mn[g] = sizes[g];
            if (!sizes[g]) continue;
            CTX_TRY(c, hipSetDevice(g));
            CTX_TRY(c, gen_uniform(c->mstreams[g], r->mk[g], (int64_t *)r->mv[g], sizes[g],
                                   seed, key_bits, start + vega_slice_start(n, c->ngpus, g), false));
        }
        (void)hipSetDevice(0);
        r->n = n;
        return VEGA_OK;
    }
    RddImpl *r;
    int rc = new_rdd(c, n ? n : 1, 0, nparts, &r, out);
    if (rc) return rc;
    r->n = n;
    CTX_TRY(c, gen_uniform(c->stream, r->d_k, (int64_t *)r->d_v, n, seed, key_bits, start, false));
    return VEGA_OK;
}

/* the hot path: sort + segmented aggregate. nparts is the logical output
 * partition count (pair_rdd.rs:54-80); the collected RESULT is invariant to
 * it (each key lands in exactly one reduce partition either way), so the
 * single-GPU engine computes the global aggregate directly. */
/* G>1 reduce: per-device radix partition into G buckets (bucket g2 owned by
 * device g2, replacing the MapOutputTracker+HTTP pull with host-side counts
 * + one RCCL grouped send/recv all-to-all-v over xGMI), then per-device
 * grouping sort + segmented aggregate. */
static int reduce_multi(vega_ctx *c, RddImpl *r, int op, uint32_t nparts,
                        vega_rdd_t *out) {
    const int G = c->ngpus;
    if ((op == VEGA_OP_SUM_F64) && r->vtype != 1) return VEGA_ERR_INVALID;
    /* per-device partition into G buckets */
    std::vector<int64_t *> pk(G, nullptr), pv(G, nullptr);
    std::vector<std::vector<uint64_t>> counts(G, std::vector<uint64_t>(G, 0));
    for (int g = 0; g < G; ++g) {
        int rc = ensure_mws(c, g, r->mn[g] + 1024);
        if (rc) return rc;
        CTX_TRY(c, hipSetDevice(g));
        uint64_t a = r->mn[g] ? r->mn[g] : 1;
        void *k = nullptr, *v = nullptr;
        CTX_TRY(c, hipMalloc(&k, a * 8));
        CTX_TRY(c, hipMalloc(&v, a * 8));
        pk[g] = (int64_t *)k;
        pv[g] = (int64_t *)v;
        Ws ws(c->mws[g], c->mws_bytes[g]);
        CTX_TRY(c, hash_partition(c->mstreams[g], (const uint64_t *)r->mk[g],
                                  (const uint64_t *)r->mv[g], r->mn[g], (uint32_t)G,
                                  (uint64_t *)pk[g], (uint64_t *)pv[g],
                                  counts[g].data(), ws));
    }
    /* recv shard sizes per destination device */
    std::vector<uint64_t> rn(G, 0);
    for (int g = 0; g < G; ++g)
        for (int p = 0; p < G; ++p) rn[p] += counts[g][p];
    std::vector<int64_t *> rk(G, nullptr), rv(G, nullptr);
    for (int p = 0; p < G; ++p) {
        CTX_TRY(c, hipSetDevice(p));
        uint64_t a = rn[p] ? rn[p] : 1;
        void *k = nullptr, *v = nullptr;
        CTX_TRY(c, hipMalloc(&k, a * 8));
        CTX_TRY(c, hipMalloc(&v, a * 8));
        rk[p] = (int64_t *)k;
        rv[p] = (int64_t *)v;
    }
    /* grouped all-to-all-v (keys, then values) */
    for (int pass = 0; pass < 2; ++pass) {
        if (ncclGroupStart() != ncclSuccess) return VEGA_ERR_HIP;
        std::vector<uint64_t> roff(G, 0);
        for (int g = 0; g < G; ++g) {
            uint64_t soff = 0;
            for (int p = 0; p < G; ++p) {
                uint64_t cnt = counts[g][p];
                const int64_t *sbase = pass == 0 ? pk[g] : pv[g];
                int64_t *rbase = pass == 0 ? rk[p] : rv[p];
                if (cnt) {
                    if (ncclSend(sbase + soff, cnt, ncclInt64, p, c->comms[g],
                                 c->mstreams[g]) != ncclSuccess ||
                        ncclRecv(rbase + roff[p], cnt, ncclInt64, g, c->comms[p],
                                 c->mstreams[p]) != ncclSuccess)
                        return VEGA_ERR_HIP;
                }
                soff += cnt;
                roff[p] += cnt;
            }
        }
        if (ncclGroupEnd() != ncclSuccess) return VEGA_ERR_HIP;
    }
    /* per-device grouping sort + segmented aggregate */
    std::vector<uint64_t> alloc_out(G);
    for (int p = 0; p < G; ++p) alloc_out[p] = rn[p] ? rn[p] : 1;
    RddImpl *o;
    int rc = new_mrdd(c, alloc_out, op == VEGA_OP_SUM_F64 ? 1 : 0, nparts, &o, out);
    if (rc) return rc;
    uint64_t total = 0;
    for (int p = 0; p < G; ++p) {
        rc = ensure_mws(c, p, rn[p] + 1024);
        if (rc) return rc;
        CTX_TRY(c, hipSetDevice(p));
        Ws ws(c->mws[p], c->mws_bytes[p]);
        uint64_t nout = 0;
        CTX_TRY(c, group_sort_reduce(c->mstreams[p], (const uint64_t *)rk[p],
                                     (const uint64_t *)rv[p], rn[p], op,
                                     (uint64_t *)o->mk[p], o->mv[p], &nout, ws));
        o->mn[p] = nout;
        total += nout;
    }
    for (int g = 0; g < G; ++g) { /* transients */
        CTX_TRY(c, hipSetDevice(g));
        (void)hipFree(pk[g]);
        (void)hipFree(pv[g]);
        (void)hipFree(rk[g]);
        (void)hipFree(rv[g]);
    }
    (void)hipSetDevice(0);
    o->n = total;
    o->sorted = false;
    return VEGA_OK;
}

static int reduce_common(vega_ctx *c, vega_rdd_t rdd, int op, uint32_t nparts,
                         vega_rdd_t *out) {
    RddImpl *r = get_rdd(c, rdd);
    if (!r) return VEGA_ERR_INVALID;
    if (c->ngpus > 1) return reduce_multi(c, r, op, nparts, out);
    /* SUM_F64 needs f64 values; COUNT ignores values; the i64 ops need i64 */
    if (op == VEGA_OP_SUM_F64 && r->vtype != 1) return VEGA_ERR_INVALID;
    if (op != VEGA_OP_SUM_F64 && op != VEGA_OP_COUNT && r->vtype == 1) return VEGA_ERR_INVALID;
    int rc = ensure_ws(c, r->n);
    if (rc) return rc;
    RddImpl *o;
    rc = new_rdd(c, r->n ? r->n : 1, op == VEGA_OP_SUM_F64 ? 1 : 0, nparts, &o, out);
    if (rc) return rc;
    Ws ws(c->ws, c->ws_bytes);
    uint64_t nout = 0;
    CTX_TRY(c, group_sort_reduce(c->stream, (const uint64_t *)r->d_k,
                                 (const uint64_t *)r->d_v, r->n, op,
                                 (uint64_t *)o->d_k, o->d_v, &nout, ws));
    o->n = nout;
    o->sorted = false; /* grouped (hash order), not key-sorted; collect compares sorted */
    return VEGA_OK;
}

int vega_gpu_reduce_by_key(vega_ctx_t *c, vega_rdd_t rdd, vega_op_t op,
                           uint32_t nparts, vega_rdd_t *out) {
    if (!c) return VEGA_ERR_INVALID;
    return reduce_common(c, rdd, op, nparts, out);
}

int vega_gpu_group_count(vega_ctx_t *c, vega_rdd_t rdd, uint32_t nparts, vega_rdd_t *out) {
    if (!c) return VEGA_ERR_INVALID;
    return reduce_common(c, rdd, VEGA_OP_COUNT, nparts, out);
}

/* distinct (rdd.rs:501-531): the deduplicated key set. The reference's
 * distinct returns ELEMENTS (map x->(x,None) + reduce_by_key + unwrap); here
 * the element column is the key, so values of the result are zeroed — the
 * result is the key set, not (key, count) (use group_count for counts). */
int vega_gpu_distinct(vega_ctx_t *c, vega_rdd_t rdd, uint32_t nparts, vega_rdd_t *out) {
    if (!c) return VEGA_ERR_INVALID;
    int rc = reduce_common(c, rdd, VEGA_OP_COUNT, nparts, out);
    if (rc) return rc;
    RddImpl *o = get_rdd(c, *out);
    if (o && c->ngpus > 1 && !o->mk.empty()) {
        for (int g = 0; g < c->ngpus; ++g) {
            if (!o->mn[g]) continue;
            CTX_TRY(c, hipSetDevice(g));
            CTX_TRY(c, hipMemsetAsync(o->mv[g], 0, o->mn[g] * 8, c->mstreams[g]));
        }
        (void)hipSetDevice(0);
    } else if (o && o->n) {
        CTX_TRY(c, hipMemsetAsync(o->d_v, 0, o->n * 8, c->stream));
    }
    return VEGA_OK;
}

/* count_by_value (rdd.rs:449-459): group-count with the VALUE column as the
 * key — the composition map(x->(x,1)) + reduce_by_key(+) collapses to the
 * same sort+segmented-count the reduce path runs */
int vega_gpu_count_by_value(vega_ctx_t *c, vega_rdd_t rdd, uint32_t nparts,
                            vega_rdd_t *out) {
    if (!c) return VEGA_ERR_INVALID;
    if (c->ngpus > 1) return VEGA_ERR_UNSUPPORTED; /* G>1: north-star ops only (this branch) */
    RddImpl *r = get_rdd(c, rdd);
    if (!r || r->vtype != 0) return VEGA_ERR_INVALID;
    int rc = ensure_ws(c, r->n);
    if (rc) return rc;
    RddImpl *o;
    rc = new_rdd(c, r->n ? r->n : 1, 0, nparts, &o, out);
    if (rc) return rc;
    Ws ws(c->ws, c->ws_bytes);
    uint64_t nout = 0;
    CTX_TRY(c, group_sort_reduce(c->stream, (const uint64_t *)r->d_v,
                                 (const uint64_t *)r->d_k, r->n, VEGA_OP_COUNT,
                                 (uint64_t *)o->d_k, o->d_v, &nout, ws));
    o->n = nout;
    return VEGA_OK;
}

int vega_gpu_sort_by_key(vega_ctx_t *c, vega_rdd_t rdd, vega_rdd_t *out) {
    if (!c) return VEGA_ERR_INVALID;
    if (c->ngpus > 1) return VEGA_ERR_UNSUPPORTED; /* G>1: north-star ops only (this branch) */
    RddImpl *r = get_rdd(c, rdd);
    if (!r) return VEGA_ERR_INVALID;
    int rc = ensure_ws(c, r->n);
    if (rc) return rc;
    RddImpl *o;
    rc = new_rdd(c, r->n ? r->n : 1, r->vtype, r->nparts, &o, out);
    if (rc) return rc;
    o->n = r->n;
    o->sorted = true;
    Ws ws(c->ws, c->ws_bytes);
    const uint64_t *sk, *sv;
    CTX_TRY(c, radix_sort_u64(c->stream, (const uint64_t *)r->d_k, (const uint64_t *)r->d_v,
                              r->n, true, true, ws, &sk, &sv));
    CTX_TRY(c, hipMemcpyAsync(o->d_k, sk, r->n * 8, hipMemcpyDeviceToDevice, c->stream));
    CTX_TRY(c, hipMemcpyAsync(o->d_v, sv, r->n * 8, hipMemcpyDeviceToDevice, c->stream));
    return VEGA_OK;
}

/* inner join (pair_rdd.rs:104-121 via cogroup co_grouped_rdd.rs:206-249):
 * bring both sides into the GROUPING order (4-5 hash passes instead of the
 * full 8-pass signed sort) and sort-merge with the (h32,key) comparator. */
/* grouped copies of both sides with a CONSISTENT comparator for the
 * sort-merge kernels: mode 2 ((h32,key) lex) when both group sorts kept the
 * pinned 4-byte hash order, else both harmonized to the full unsigned-key
 * order (mode 1). Caller frees *ha / *hb. */
static int group_two_sides(vega_ctx *c, RddImpl *ra, RddImpl *rb, uint32_t nparts,
                           vega_rdd_t *ha, vega_rdd_t *hb, RddImpl **sa_out,
                           RddImpl **sb_out, int *mode_out) {
    uint64_t nmax = ra->n > rb->n ? ra->n : rb->n;
    int rc = ensure_ws(c, nmax);
    if (rc) return rc;
    RddImpl *sa, *sb;
    *ha = 0; *hb = 0;
    rc = new_rdd(c, ra->n ? ra->n : 1, 0, nparts, &sa, ha);
    if (rc) return rc;
    rc = new_rdd(c, rb->n ? rb->n : 1, 0, nparts, &sb, hb);
    if (rc) { vega_gpu_free_rdd(c, *ha); *ha = 0; return rc; }
    sa->n = ra->n;
    sb->n = rb->n;
    int tag_a = 0, tag_b = 0;
    if (ra->n) {
        CTX_TRY(c, hipMemcpyAsync(sa->d_k, ra->d_k, ra->n * 8, hipMemcpyDeviceToDevice, c->stream));
        CTX_TRY(c, hipMemcpyAsync(sa->d_v, ra->d_v, ra->n * 8, hipMemcpyDeviceToDevice, c->stream));
        Ws wsa(c->ws, c->ws_bytes);
        CTX_TRY(c, group_pairs_inplace(c->stream, sa->d_k, (int64_t *)sa->d_v, sa->n, &tag_a, wsa));
    }
    if (rb->n) {
        CTX_TRY(c, hipMemcpyAsync(sb->d_k, rb->d_k, rb->n * 8, hipMemcpyDeviceToDevice, c->stream));
        CTX_TRY(c, hipMemcpyAsync(sb->d_v, rb->d_v, rb->n * 8, hipMemcpyDeviceToDevice, c->stream));
        Ws wsb(c->ws, c->ws_bytes);
        CTX_TRY(c, group_pairs_inplace(c->stream, sb->d_k, (int64_t *)sb->d_v, sb->n, &tag_b, wsb));
    }
    int mode = 2;
    if (tag_a != 4 || tag_b != 4) {
        mode = 1;
        const uint64_t *rk, *rv;
        if (sa->n && tag_a == 4) {
            Ws wsa(c->ws, c->ws_bytes);
            CTX_TRY(c, radix_sort_u64(c->stream, (const uint64_t *)sa->d_k,
                                      (const uint64_t *)sa->d_v, sa->n, true, false, wsa, &rk, &rv));
            if ((const uint64_t *)sa->d_k != rk) {
                CTX_TRY(c, hipMemcpyAsync(sa->d_k, rk, sa->n * 8, hipMemcpyDeviceToDevice, c->stream));
                CTX_TRY(c, hipMemcpyAsync(sa->d_v, rv, sa->n * 8, hipMemcpyDeviceToDevice, c->stream));
            }
        }
        if (sb->n && tag_b == 4) {
            Ws wsb(c->ws, c->ws_bytes);
            CTX_TRY(c, radix_sort_u64(c->stream, (const uint64_t *)sb->d_k,
                                      (const uint64_t *)sb->d_v, sb->n, true, false, wsb, &rk, &rv));
            if ((const uint64_t *)sb->d_k != rk) {
                CTX_TRY(c, hipMemcpyAsync(sb->d_k, rk, sb->n * 8, hipMemcpyDeviceToDevice, c->stream));
                CTX_TRY(c, hipMemcpyAsync(sb->d_v, rv, sb->n * 8, hipMemcpyDeviceToDevice, c->stream));
            }
        }
    }
    *sa_out = sa;
    *sb_out = sb;
    *mode_out = mode;
    return VEGA_OK;
}

int vega_gpu_join(vega_ctx_t *c, vega_rdd_t a, vega_rdd_t b, uint32_t nparts,
                  vega_rdd_t *out) {
    if (!c) return VEGA_ERR_INVALID;
    if (c->ngpus > 1) return VEGA_ERR_UNSUPPORTED; /* G>1: north-star ops only (this branch) */
    RddImpl *ra = get_rdd(c, a), *rb = get_rdd(c, b);
    if (!ra || !rb || ra->vtype || rb->vtype) return VEGA_ERR_INVALID;
    vega_rdd_t ha = 0, hb = 0;
    RddImpl *sa, *sb;
    int join_mode = 2;
    int rc = group_two_sides(c, ra, rb, nparts, &ha, &hb, &sa, &sb, &join_mode);
    if (rc) return rc;
    uint64_t total = 0;
    {
        Ws ws(c->ws, c->ws_bytes);
        hipError_t e = join_sorted(c->stream, sa->d_k, (const int64_t *)sa->d_v, sa->n,
                                   sb->d_k, (const int64_t *)sb->d_v, sb->n, join_mode,
                                   nullptr, nullptr, nullptr, 0, &total, ws);
        if (e != hipSuccess) {
            vega_gpu_free_rdd(c, ha); vega_gpu_free_rdd(c, hb);
            snprintf(c->err, sizeof c->err, "join count: %s", hipGetErrorString(e));
            return VEGA_ERR_HIP;
        }
    }
    if (total >= (1ULL << 32)) { /* u32 emit scan limit: refuse loudly */
        vega_gpu_free_rdd(c, ha); vega_gpu_free_rdd(c, hb);
        snprintf(c->err, sizeof c->err,
                 "join output %llu rows exceeds the 2^32-1 per-call limit",
                 (unsigned long long)total);
        return VEGA_ERR_UNSUPPORTED;
    }
    RddImpl *o;
    rc = new_rdd(c, total ? total : 1, 0, nparts, &o, out);
    if (rc) { vega_gpu_free_rdd(c, ha); vega_gpu_free_rdd(c, hb); return rc; }
    CTX_TRY(c, hipMalloc(&o->d_v2, (total ? total : 1) * 8));
    o->n = total;
    {
        Ws ws(c->ws, c->ws_bytes);
        uint64_t n2 = 0;
        hipError_t e = join_sorted(c->stream, sa->d_k, (const int64_t *)sa->d_v, sa->n,
                                   sb->d_k, (const int64_t *)sb->d_v, sb->n, join_mode,
                                   o->d_k, (int64_t *)o->d_v, (int64_t *)o->d_v2,
                                   total, &n2, ws);
        vega_gpu_free_rdd(c, ha);
        vega_gpu_free_rdd(c, hb);
        if (e != hipSuccess || n2 != total) {
            snprintf(c->err, sizeof c->err, "join emit: %s (n2=%llu total=%llu)",
                     hipGetErrorString(e), (unsigned long long)n2,
                     (unsigned long long)total);
            return VEGA_ERR_HIP;
        }
    }
    return VEGA_OK;
}

int vega_gpu_collect_join(vega_ctx_t *c, vega_rdd_t rdd, int64_t *keys,
                          int64_t *va, int64_t *vb, uint64_t *n) {
    RddImpl *r = get_rdd(c, rdd);
    if (!r) return VEGA_ERR_INVALID;
    if (!keys) { *n = r->n; return VEGA_OK; }
    if (*n < r->n || !r->d_v2) return VEGA_ERR_CAP;
    *n = r->n;
    if (r->n) {
        CTX_TRY(c, hipMemcpyAsync(keys, r->d_k, r->n * 8, hipMemcpyDeviceToHost, c->stream));
        CTX_TRY(c, hipMemcpyAsync(va, r->d_v, r->n * 8, hipMemcpyDeviceToHost, c->stream));
        CTX_TRY(c, hipMemcpyAsync(vb, r->d_v2, r->n * 8, hipMemcpyDeviceToHost, c->stream));
    }
    CTX_TRY(c, hipStreamSynchronize(c->stream));
    return VEGA_OK;
}

/* group_by_key (pair_rdd.rs:35-52, aggregator.rs:33-53): groups MATERIALIZED
 * in the engine — keys + u64 offsets + the values column in grouped order,
 * all device-resident (replaces the host-side numpy assembly). Value order
 * within each group is the row order (the grouping sort is stable), matching
 * the reference's per-partition append order. */
int vega_gpu_group_by_key(vega_ctx_t *c, vega_rdd_t rdd, uint32_t nparts,
                          vega_rdd_t *out) {
    if (!c) return VEGA_ERR_INVALID;
    if (c->ngpus > 1) return VEGA_ERR_UNSUPPORTED;
    RddImpl *r = get_rdd(c, rdd);
    if (!r) return VEGA_ERR_INVALID;
    int rc = ensure_ws(c, r->n);
    if (rc) return rc;
    RddImpl *o;
    rc = new_rdd(c, r->n ? r->n : 1, r->vtype, nparts, &o, out);
    if (rc) return rc;
    /* d_k: distinct keys; d_v: u64 offsets (reuse the n-row alloc, nk+1 <=
     * n+1 needs one extra slot) */
    CTX_TRY(c, hipFree(o->d_v));
    o->d_v = nullptr;
    CTX_TRY(c, hipMalloc(&o->d_v, (r->n + 2) * 8));
    CTX_TRY(c, hipMalloc(&o->d_v2, (r->n ? r->n : 1) * 8));
    Ws ws(c->ws, c->ws_bytes);
    const uint64_t *sk, *sv;
    CTX_TRY(c, group_sort_u64(c->stream, (const uint64_t *)r->d_k, (const uint64_t *)r->d_v,
                              r->n, 0, nullptr, 0, nullptr, ws, &sk, &sv));
    /* values in grouped order */
    if (r->n)
        CTX_TRY(c, hipMemcpyAsync(o->d_v2, sv, r->n * 8, hipMemcpyDeviceToDevice, c->stream));
    /* distinct keys + per-group counts (counts go to a transient) */
    int64_t *cnt = nullptr;
    CTX_TRY(c, hipMalloc(&cnt, (r->n ? r->n : 1) * 8));
    uint64_t nk = 0;
    {
        hipError_t e = seg_reduce(c->stream, sk, sv, r->n, VEGA_OP_COUNT,
                                  (uint64_t *)o->d_k, cnt, &nk, ws);
        if (e != hipSuccess) {
            (void)hipFree(cnt);
            snprintf(c->err, sizeof c->err, "group_by_key: %s", hipGetErrorString(e));
            return e == hipErrorNotSupported ? VEGA_ERR_UNSUPPORTED : VEGA_ERR_HIP;
        }
    }
    {
        hipError_t e = counts_to_offsets_u64(c->stream, cnt, nk, (uint64_t *)o->d_v, ws);
        (void)hipFree(cnt);
        if (e != hipSuccess) {
            snprintf(c->err, sizeof c->err, "group offsets: %s", hipGetErrorString(e));
            return VEGA_ERR_HIP;
        }
    }
    o->n = nk;
    o->n2 = r->n;
    o->grouped = true;
    return VEGA_OK;
}

/* collect a grouped rdd: keys[nk], offsets[nk+1] (u64), values[nvals].
 * Query sizes with keys == NULL. */
int vega_gpu_collect_groups(vega_ctx_t *c, vega_rdd_t rdd, int64_t *keys,
                            uint64_t *offsets, void *values, uint64_t *nk,
                            uint64_t *nvals) {
    RddImpl *r = get_rdd(c, rdd);
    if (!r || !r->grouped) return VEGA_ERR_INVALID;
    if (!keys) { *nk = r->n; *nvals = r->n2; return VEGA_OK; }
    if (*nk < r->n || *nvals < r->n2) return VEGA_ERR_CAP;
    *nk = r->n;
    *nvals = r->n2;
    if (r->n) {
        CTX_TRY(c, hipMemcpyAsync(keys, r->d_k, r->n * 8, hipMemcpyDeviceToHost, c->stream));
        CTX_TRY(c, hipMemcpyAsync(offsets, r->d_v, (r->n + 1) * 8, hipMemcpyDeviceToHost, c->stream));
    }
    if (r->n2)
        CTX_TRY(c, hipMemcpyAsync(values, r->d_v2, r->n2 * 8, hipMemcpyDeviceToHost, c->stream));
    CTX_TRY(c, hipStreamSynchronize(c->stream));
    return VEGA_OK;
}

/* cogroup (pair_rdd.rs:123-155 via co_grouped_rdd.rs:206-249): per key
 * present in EITHER side, the (Vec<V>, Vec<W>) ranges. Single call: caller
 * provides keys/offa/lena/offb/lenb sized cap >= |keys(a) U keys(b)|
 * (na+nb always suffices), vala[na], valb[nb]. Outputs: vala/valb are the
 * two value columns in grouped order; per key i, side A values are
 * vala[offa[i] .. offa[i]+lena[i]) and likewise for B. */
int vega_gpu_cogroup_collect(vega_ctx_t *c, vega_rdd_t a, vega_rdd_t b,
                             int64_t *keys, uint64_t *offa, uint64_t *lena,
                             uint64_t *offb, uint64_t *lenb, int64_t *vala,
                             int64_t *valb, uint64_t cap, uint64_t *h_nk) {
    if (!c) return VEGA_ERR_INVALID;
    if (c->ngpus > 1) return VEGA_ERR_UNSUPPORTED;
    RddImpl *ra = get_rdd(c, a), *rb = get_rdd(c, b);
    if (!ra || !rb || ra->vtype || rb->vtype) return VEGA_ERR_INVALID;
    vega_rdd_t ha = 0, hb = 0;
    RddImpl *sa, *sb;
    int mode = 2;
    int rc = group_two_sides(c, ra, rb, 1, &ha, &hb, &sa, &sb, &mode);
    if (rc) return rc;
    /* distinct keys + counts + offsets per side (device temporaries) */
    int64_t *ka_u = nullptr, *cnta = nullptr, *kb_u = nullptr, *cntb = nullptr;
    uint64_t *offa_d = nullptr, *offb_d = nullptr;
    int64_t *keys_d = nullptr;
    uint64_t *meta_d = nullptr; /* offa,lena,offb,lenb x cap */
    uint64_t nka = 0, nkb = 0, nk = 0;
    hipError_t e = hipSuccess;
    int ret = VEGA_ERR_HIP;
    do {
        if ((e = hipMalloc(&ka_u, (sa->n ? sa->n : 1) * 8)) != hipSuccess) break;
        if ((e = hipMalloc(&cnta, (sa->n ? sa->n : 1) * 8)) != hipSuccess) break;
        if ((e = hipMalloc(&kb_u, (sb->n ? sb->n : 1) * 8)) != hipSuccess) break;
        if ((e = hipMalloc(&cntb, (sb->n ? sb->n : 1) * 8)) != hipSuccess) break;
        if ((e = hipMalloc(&offa_d, (sa->n + 2) * 8)) != hipSuccess) break;
        if ((e = hipMalloc(&offb_d, (sb->n + 2) * 8)) != hipSuccess) break;
        {
            Ws ws(c->ws, c->ws_bytes);
            if ((e = seg_reduce(c->stream, (const uint64_t *)sa->d_k,
                                (const uint64_t *)sa->d_v, sa->n, VEGA_OP_COUNT,
                                (uint64_t *)ka_u, cnta, &nka, ws)) != hipSuccess) break;
        }
        {
            Ws ws(c->ws, c->ws_bytes);
            if ((e = seg_reduce(c->stream, (const uint64_t *)sb->d_k,
                                (const uint64_t *)sb->d_v, sb->n, VEGA_OP_COUNT,
                                (uint64_t *)kb_u, cntb, &nkb, ws)) != hipSuccess) break;
        }
        {
            Ws ws(c->ws, c->ws_bytes);
            if ((e = counts_to_offsets_u64(c->stream, cnta, nka, offa_d, ws)) != hipSuccess) break;
            if ((e = counts_to_offsets_u64(c->stream, cntb, nkb, offb_d, ws)) != hipSuccess) break;
        }
        uint64_t kcap = nka + nkb;
        if ((e = hipMalloc(&keys_d, (kcap ? kcap : 1) * 8)) != hipSuccess) break;
        if ((e = hipMalloc(&meta_d, (kcap ? kcap : 1) * 4 * 8)) != hipSuccess) break;
        {
            Ws ws(c->ws, c->ws_bytes);
            if ((e = cogroup_index(c->stream, ka_u, nka, offa_d, kb_u, nkb, offb_d,
                                   mode, keys_d, meta_d, meta_d + kcap,
                                   meta_d + 2 * kcap, meta_d + 3 * kcap,
                                   kcap, &nk, ws)) != hipSuccess) break;
        }
        *h_nk = nk; /* reported even on CAP so the caller can resize */
        if (nk > cap) { ret = VEGA_ERR_CAP; e = hipSuccess; break; }
        if (nk) {
            if ((e = hipMemcpyAsync(keys, keys_d, nk * 8, hipMemcpyDeviceToHost, c->stream)) != hipSuccess) break;
            if ((e = hipMemcpyAsync(offa, meta_d, nk * 8, hipMemcpyDeviceToHost, c->stream)) != hipSuccess) break;
            if ((e = hipMemcpyAsync(lena, meta_d + kcap, nk * 8, hipMemcpyDeviceToHost, c->stream)) != hipSuccess) break;
            if ((e = hipMemcpyAsync(offb, meta_d + 2 * kcap, nk * 8, hipMemcpyDeviceToHost, c->stream)) != hipSuccess) break;
            if ((e = hipMemcpyAsync(lenb, meta_d + 3 * kcap, nk * 8, hipMemcpyDeviceToHost, c->stream)) != hipSuccess) break;
        }
        if (sa->n && (e = hipMemcpyAsync(vala, sa->d_v, sa->n * 8, hipMemcpyDeviceToHost, c->stream)) != hipSuccess) break;
        if (sb->n && (e = hipMemcpyAsync(valb, sb->d_v, sb->n * 8, hipMemcpyDeviceToHost, c->stream)) != hipSuccess) break;
        if ((e = hipStreamSynchronize(c->stream)) != hipSuccess) break;
        ret = VEGA_OK;
    } while (0);
    if (e != hipSuccess) {
        snprintf(c->err, sizeof c->err, "cogroup: %s", hipGetErrorString(e));
        if (e == hipErrorNotSupported) ret = VEGA_ERR_UNSUPPORTED;
    }
    (void)hipStreamSynchronize(c->stream);
    if (ka_u) (void)hipFree(ka_u);
    if (cnta) (void)hipFree(cnta);
    if (kb_u) (void)hipFree(kb_u);
    if (cntb) (void)hipFree(cntb);
    if (offa_d) (void)hipFree(offa_d);
    if (offb_d) (void)hipFree(offb_d);
    if (keys_d) (void)hipFree(keys_d);
    if (meta_d) (void)hipFree(meta_d);
    vega_gpu_free_rdd(c, ha);
    vega_gpu_free_rdd(c, hb);
    return ret;
}

/* intersection / subtract (rdd.rs set compositions over CoGroupedRdd):
 * the element column is the key; results are key SETS (values zeroed). */
static int set_op_common(vega_ctx *c, vega_rdd_t a, vega_rdd_t b, int want,
                         uint32_t nparts, vega_rdd_t *out) {
    if (!c) return VEGA_ERR_INVALID;
    if (c->ngpus > 1) return VEGA_ERR_UNSUPPORTED;
    RddImpl *ra = get_rdd(c, a), *rb = get_rdd(c, b);
    if (!ra || !rb || ra->vtype || rb->vtype) return VEGA_ERR_INVALID;
    vega_rdd_t ha = 0, hb = 0;
    RddImpl *sa, *sb;
    int mode = 2;
    int rc = group_two_sides(c, ra, rb, nparts, &ha, &hb, &sa, &sb, &mode);
    if (rc) return rc;
    int64_t *ka_u = nullptr, *cnta = nullptr, *kb_u = nullptr, *cntb = nullptr;
    uint64_t nka = 0, nkb = 0, nsel = 0;
    hipError_t e = hipSuccess;
    int ret = VEGA_ERR_HIP;
    RddImpl *o = nullptr;
    do {
        if ((e = hipMalloc(&ka_u, (sa->n ? sa->n : 1) * 8)) != hipSuccess) break;
        if ((e = hipMalloc(&cnta, (sa->n ? sa->n : 1) * 8)) != hipSuccess) break;
        if ((e = hipMalloc(&kb_u, (sb->n ? sb->n : 1) * 8)) != hipSuccess) break;
        if ((e = hipMalloc(&cntb, (sb->n ? sb->n : 1) * 8)) != hipSuccess) break;
        {
            Ws ws(c->ws, c->ws_bytes);
            if ((e = seg_reduce(c->stream, (const uint64_t *)sa->d_k,
                                (const uint64_t *)sa->d_v, sa->n, VEGA_OP_COUNT,
                                (uint64_t *)ka_u, cnta, &nka, ws)) != hipSuccess) break;
        }
        {
            Ws ws(c->ws, c->ws_bytes);
            if ((e = seg_reduce(c->stream, (const uint64_t *)sb->d_k,
                                (const uint64_t *)sb->d_v, sb->n, VEGA_OP_COUNT,
                                (uint64_t *)kb_u, cntb, &nkb, ws)) != hipSuccess) break;
        }
        int rc2 = new_rdd(c, nka ? nka : 1, 0, nparts, &o, out);
        if (rc2) { ret = rc2; e = hipSuccess; break; }
        {
            Ws ws(c->ws, c->ws_bytes);
            if ((e = member_select(c->stream, ka_u, nka, kb_u, nkb, mode, want,
                                   o->d_k, &nsel, ws)) != hipSuccess) break;
        }
        o->n = nsel;
        if (nsel && (e = hipMemsetAsync(o->d_v, 0, nsel * 8, c->stream)) != hipSuccess) break;
        ret = VEGA_OK;
    } while (0);
    if (e != hipSuccess) {
        snprintf(c->err, sizeof c->err, "set op: %s", hipGetErrorString(e));
        if (e == hipErrorNotSupported) ret = VEGA_ERR_UNSUPPORTED;
    }
    (void)hipStreamSynchronize(c->stream);
    if (ka_u) (void)hipFree(ka_u);
    if (cnta) (void)hipFree(cnta);
    if (kb_u) (void)hipFree(kb_u);
    if (cntb) (void)hipFree(cntb);
    vega_gpu_free_rdd(c, ha);
    vega_gpu_free_rdd(c, hb);
    return ret;
}

int vega_gpu_intersection(vega_ctx_t *c, vega_rdd_t a, vega_rdd_t b,
                          uint32_t nparts, vega_rdd_t *out) {
    return set_op_common(c, a, b, 1, nparts, out);
}
int vega_gpu_subtract(vega_ctx_t *c, vega_rdd_t a, vega_rdd_t b,
                      uint32_t nparts, vega_rdd_t *out) {
    return set_op_common(c, a, b, 0, nparts, out);
}

int vega_gpu_map(vega_ctx_t *c, vega_rdd_t rdd, vega_map_op_t op, int64_t p0,
                 vega_rdd_t *out) {
    if (!c) return VEGA_ERR_INVALID;
    if (c->ngpus > 1) return VEGA_ERR_UNSUPPORTED; /* G>1: north-star ops only (this branch) */
    RddImpl *r = get_rdd(c, rdd);
    if (!r || r->vtype != 0) return VEGA_ERR_INVALID;
    RddImpl *o;
    int rc = new_rdd(c, r->n ? r->n : 1, 0, r->nparts, &o, out);
    if (rc) return rc;
    o->n = r->n;
    CTX_TRY(c, narrow_map(c->stream, r->d_k, (const int64_t *)r->d_v, r->n,
                          (int)op, p0, o->d_k, (int64_t *)o->d_v));
    return VEGA_OK;
}

int vega_gpu_filter(vega_ctx_t *c, vega_rdd_t rdd, vega_pred_t pred,
                    int64_t p0, int64_t p1, vega_rdd_t *out) {
    if (!c) return VEGA_ERR_INVALID;
    if (c->ngpus > 1) return VEGA_ERR_UNSUPPORTED; /* G>1: north-star ops only (this branch) */
    RddImpl *r = get_rdd(c, rdd);
    if (!r || r->vtype != 0) return VEGA_ERR_INVALID;
    int rc = ensure_ws(c, r->n);
    if (rc) return rc;
    RddImpl *o;
    rc = new_rdd(c, r->n ? r->n : 1, 0, r->nparts, &o, out);
    if (rc) return rc;
    Ws ws(c->ws, c->ws_bytes);
    uint64_t nout = 0;
    CTX_TRY(c, narrow_filter(c->stream, r->d_k, (const int64_t *)r->d_v, r->n,
                             (int)pred, p0, p1, o->d_k, (int64_t *)o->d_v, &nout, ws));
    o->n = nout;
    return VEGA_OK;
}

int vega_gpu_count(vega_ctx_t *c, vega_rdd_t rdd, uint64_t *n) {
    RddImpl *r = get_rdd(c, rdd);
    if (!r) return VEGA_ERR_INVALID;
    *n = r->n;
    return VEGA_OK;
}

int vega_gpu_collect(vega_ctx_t *c, vega_rdd_t rdd, int64_t *keys, void *vals, uint64_t *n) {
    RddImpl *r = get_rdd(c, rdd);
    if (!r) return VEGA_ERR_INVALID;
    if (!keys) {
        *n = r->n;
        return VEGA_OK;
    }
    if (*n < r->n) return VEGA_ERR_CAP;
    if (c->ngpus > 1 && !r->mk.empty()) { /* concat device shards */
        uint64_t off = 0;
        for (int g = 0; g < c->ngpus; ++g) {
            if (!r->mn[g]) continue;
            CTX_TRY(c, hipSetDevice(g));
            CTX_TRY(c, hipMemcpyAsync(keys + off, r->mk[g], r->mn[g] * 8,
                                      hipMemcpyDeviceToHost, c->mstreams[g]));
            if (vals)
                CTX_TRY(c, hipMemcpyAsync((char *)vals + off * 8, r->mv[g], r->mn[g] * 8,
                                          hipMemcpyDeviceToHost, c->mstreams[g]));
            off += r->mn[g];
        }
        for (int g = 0; g < c->ngpus; ++g) CTX_TRY(c, hipStreamSynchronize(c->mstreams[g]));
        (void)hipSetDevice(0);
        *n = off;
        return VEGA_OK;
    }
    *n = r->n;
    if (r->n) {
        CTX_TRY(c, hipMemcpyAsync(keys, r->d_k, r->n * 8, hipMemcpyDeviceToHost, c->stream));
        if (vals)
            CTX_TRY(c, hipMemcpyAsync(vals, r->d_v, r->n * 8, hipMemcpyDeviceToHost, c->stream));
    }
    CTX_TRY(c, hipStreamSynchronize(c->stream));
    return VEGA_OK;
}

int vega_gpu_free_rdd(vega_ctx_t *c, vega_rdd_t rdd) {
    auto it = c->rdds.find(rdd);
    if (it == c->rdds.end()) return VEGA_ERR_INVALID;
    CTX_TRY(c, hipStreamSynchronize(c->stream));
    free_rdd_buffers(c, it->second);
    delete it->second;
    c->rdds.erase(it);
    return VEGA_OK;
}

int vega_gpu_set_profiling(vega_ctx_t *c, int enabled) {
    (void)c;
    prof_enable(enabled != 0);
    return VEGA_OK;
}
int vega_gpu_kernel_stats(vega_ctx_t *c, char *buf, size_t buflen) {
    (void)c;
    return prof_stats_json(buf, buflen) < 0 ? VEGA_ERR_CAP : VEGA_OK;
}

/* =================== device-pointer API =================== */

size_t vega_dev_ws_bytes(uint64_t n) { return ws_bytes_for(n); }

/* separate global prof switch for the pointer API */
int vega_prof_enable(int on) { prof_enable(on != 0); return VEGA_OK; }
int vega_prof_stats(char *buf, size_t len) { return prof_stats_json(buf, len) < 0 ? VEGA_ERR_CAP : VEGA_OK; }

int vega_dev_gen_uniform_i64(void *stream, int64_t *keys, int64_t *vals, uint64_t n,
                             uint64_t seed, int key_bits, uint64_t start) {
    return gen_uniform((hipStream_t)stream, keys, vals, n, seed, key_bits, start, false)
               == hipSuccess ? VEGA_OK : VEGA_ERR_HIP;
}

/* f64-value variant of the generator (values exact dyadic uniform [0,1),
 * bit-identical to datagen.c's vega_gen_uniform_pairs_f64) */
int vega_dev_gen_uniform_f64(void *stream, int64_t *keys, double *vals, uint64_t n,
                             uint64_t seed, int key_bits, uint64_t start) {
    return gen_uniform((hipStream_t)stream, keys, (int64_t *)vals, n, seed, key_bits, start, true)
               == hipSuccess ? VEGA_OK : VEGA_ERR_HIP;
}

int vega_dev_partition_i64(void *stream, const int64_t *keys, const int64_t *vals,
                           uint64_t n, uint32_t nparts, int64_t *out_k, int64_t *out_v,
                           uint64_t *h_counts, void *d_ws, size_t ws_bytes) {
    Ws ws(d_ws, ws_bytes);
    hipError_t e = hash_partition((hipStream_t)stream, (const uint64_t *)keys,
                                  (const uint64_t *)vals, n, nparts,
                                  (uint64_t *)out_k, (uint64_t *)out_v, h_counts, ws);
    return e == hipSuccess ? VEGA_OK : (e == hipErrorOutOfMemory ? VEGA_ERR_NOMEM : VEGA_ERR_HIP);
}

int vega_dev_partition_range_i64(void *stream, const int64_t *keys, const int64_t *vals,
                                 uint64_t n, uint32_t nparts, const int64_t *d_splitters,
                                 int64_t *out_k, int64_t *out_v, uint64_t *h_counts,
                                 void *d_ws, size_t ws_bytes) {
    Ws ws(d_ws, ws_bytes);
    hipError_t e = range_partition((hipStream_t)stream, (const uint64_t *)keys,
                                   (const uint64_t *)vals, n, nparts, d_splitters,
                                   (uint64_t *)out_k, (uint64_t *)out_v, h_counts, ws);
    return e == hipSuccess ? VEGA_OK : (e == hipErrorOutOfMemory ? VEGA_ERR_NOMEM : VEGA_ERR_HIP);
}

int vega_dev_sort_reduce(void *stream, const int64_t *in_k, const void *in_v,
                         uint64_t n, int op, int64_t *out_k, void *out_v,
                         uint64_t *h_nout, void *d_ws, size_t ws_bytes) {
    Ws ws(d_ws, ws_bytes);
    hipError_t e = group_sort_reduce((hipStream_t)stream, (const uint64_t *)in_k,
                                     (const uint64_t *)in_v, n, op,
                                     (uint64_t *)out_k, out_v, h_nout, ws);
    if (e == hipErrorNotSupported) return VEGA_ERR_UNSUPPORTED;
    return e == hipSuccess ? VEGA_OK : (e == hipErrorOutOfMemory ? VEGA_ERR_NOMEM : VEGA_ERR_HIP);
}

int vega_dev_sort_pairs_i64(void *stream, int64_t *keys, int64_t *vals, uint64_t n,
                            void *d_ws, size_t ws_bytes) {
    Ws ws(d_ws, ws_bytes);
    const uint64_t *sk, *sv;
    hipError_t e = radix_sort_u64((hipStream_t)stream, (const uint64_t *)keys,
                                  (const uint64_t *)vals, n, true, true, ws, &sk, &sv);
    if (e != hipSuccess) return e == hipErrorOutOfMemory ? VEGA_ERR_NOMEM : VEGA_ERR_HIP;
    if ((const uint64_t *)keys != sk) {
        e = hipMemcpyAsync(keys, sk, n * 8, hipMemcpyDeviceToDevice, (hipStream_t)stream);
        if (e != hipSuccess) return VEGA_ERR_HIP;
        e = hipMemcpyAsync(vals, sv, n * 8, hipMemcpyDeviceToDevice, (hipStream_t)stream);
        if (e != hipSuccess) return VEGA_ERR_HIP;
    }
    return VEGA_OK;
}

int vega_dev_join_sorted(void *stream, const int64_t *ak, const int64_t *av, uint64_t na,
                         const int64_t *bk, const int64_t *bv, uint64_t nb,
                         int64_t *out_k, int64_t *out_va, int64_t *out_vb,
                         uint64_t cap, uint64_t *h_nout, void *d_ws, size_t ws_bytes) {
    Ws ws(d_ws, ws_bytes);
    hipError_t e = join_sorted((hipStream_t)stream, ak, av, na, bk, bv, nb, 0,
                               out_k, out_va, out_vb, cap, h_nout, ws);
    if (e == hipErrorNotSupported) return VEGA_ERR_UNSUPPORTED;
    return e == hipSuccess ? VEGA_OK : VEGA_ERR_HIP;
}

/* bring rows into the GROUPING order in place ((h32,key) lexicographic;
 * the cheap order vega_dev_join_grouped expects) */
int vega_dev_group_pairs_i64(void *stream, int64_t *keys, int64_t *vals, uint64_t n,
                             int *h_order_tag, void *d_ws, size_t ws_bytes) {
    Ws ws(d_ws, ws_bytes);
    hipError_t e = group_pairs_inplace((hipStream_t)stream, keys, vals, n, h_order_tag, ws);
    return e == hipSuccess ? VEGA_OK : (e == hipErrorOutOfMemory ? VEGA_ERR_NOMEM : VEGA_ERR_HIP);
}

/* sort-merge inner join; order_mode must match how BOTH sides are sorted:
 * 0 signed-key, 1 unsigned-key, 2 = (h32,key) lex (group tag 4) */
int vega_dev_join_grouped(void *stream, const int64_t *ak, const int64_t *av, uint64_t na,
                          const int64_t *bk, const int64_t *bv, uint64_t nb,
                          int order_mode,
                          int64_t *out_k, int64_t *out_va, int64_t *out_vb,
                          uint64_t cap, uint64_t *h_nout, void *d_ws, size_t ws_bytes) {
    Ws ws(d_ws, ws_bytes);
    hipError_t e = join_sorted((hipStream_t)stream, ak, av, na, bk, bv, nb, order_mode,
                               out_k, out_va, out_vb, cap, h_nout, ws);
    return e == hipSuccess ? VEGA_OK : VEGA_ERR_HIP;
}

/* diagnostic (VEGA_PHASE_PROF=1 builds/runs): per-phase shader-cycle sums of
 * the onesweep scatter — 0 prefetch, 1 rank, 2 publish+starts, 3 lookback,
 * 4 reorder, 5 writeout. */
int vega_phase_prof_read(unsigned long long out[8], int reset) {
    unsigned long long *b = vega::phase_prof_buf();
    if (!b) return VEGA_ERR_UNSUPPORTED;
    if (hipMemcpy(out, b, 8 * 8, hipMemcpyDeviceToHost) != hipSuccess) return VEGA_ERR_HIP;
    if (reset) (void)hipMemset(b, 0, 8 * 8);
    return VEGA_OK;
}

int vega_dev_checksum_pairs(void *stream, const int64_t *keys, const int64_t *vals,
                            uint64_t n, uint64_t *h_sum, void *d_ws, size_t ws_bytes) {
    Ws ws(d_ws, ws_bytes);
    return checksum_pairs((hipStream_t)stream, keys, vals, n, h_sum, ws) == hipSuccess
               ? VEGA_OK : VEGA_ERR_HIP;
}

} /* extern "C" */
