/* vega_internal.h — internal host-side interfaces between the kernel TU
 * (vega_kernels.hip) and the C-ABI TU (vega_api.hip). Not installed. */
#ifndef VEGA_INTERNAL_H
#define VEGA_INTERNAL_H

#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstddef>

namespace vega {

/* ---- profiling registry (enabled via vega_prof_enable) ---- */
void prof_enable(bool on);
bool prof_on();
/* record a closed event pair under `name`; takes ownership of events */
void prof_record(const char *name, hipEvent_t start, hipEvent_t stop);
/* drain into a JSON string: {"name":{"ms":x,"n":k},...} */
int prof_stats_json(char *buf, size_t len);

/* RAII-ish kernel timer: when profiling is on, brackets launches on `s` */
struct ProfScope {
    const char *name;
    hipStream_t s;
    hipEvent_t e0 = nullptr, e1 = nullptr;
    ProfScope(const char *n, hipStream_t stream);
    ~ProfScope();
};

/* ---- workspace carving ---- */
struct Ws {
    char *p;
    size_t left;
    Ws(void *ws, size_t bytes) : p((char *)ws), left(bytes) {}
    void *take(size_t bytes) {
        size_t a = (bytes + 255) & ~(size_t)255;
        if (a > left) return nullptr;
        void *r = p;
        p += a;
        left -= a;
        return r;
    }
};

/* ---- low-level ops (device pointers, async on stream) ---- */

/* exclusive scan of n uint32 in place; ws scratch */
hipError_t scan_u32_excl(hipStream_t s, uint32_t *a, uint64_t n, Ws &ws);

/* stable LSB radix sort of (key,val) u64 pairs by full 64-bit key with
 * degenerate-pass skipping. in_* are const; result pointers returned in
 * res_k / res_v (one of the two ws ping-pong buffers, or in_* when zero
 * passes were needed — in that case result aliases the input).
 * has_vals=false skips all value movement. */
hipError_t radix_sort_u64(hipStream_t s, const uint64_t *in_k, const uint64_t *in_v,
                          uint64_t n, bool has_vals, bool signed_order, Ws &ws,
                          const uint64_t **res_k, const uint64_t **res_v);

/* group equal keys adjacently for the reduce path (no total-order contract):
 * adaptive skipped key sort vs 40-bit hash sort + collision-run cleanup */
/* order_tag out: 0 = full-key unsigned order, 4 = (h32,key) lex order.
 * force_hbytes: 0 adaptive (reduce), 4 pinned order (joins). */
/* want_packed: hash path keeps the final pass's interleaved (k,v) layout
 * (cheaper writeout); *out_packed reports whether the result IS packed
 * (key i at res_k[2i]) — the narrow-key and fallback paths stay SoA. */
hipError_t group_sort_u64(hipStream_t s, const uint64_t *in_k, const uint64_t *in_v,
                          uint64_t n, int force_hbytes, int *order_tag,
                          int want_packed, int *out_packed, Ws &ws,
                          const uint64_t **res_k, const uint64_t **res_v);

/* segmented aggregate over key-sorted rows: one output row per equal-key
 * run. op: VEGA_OP_*. Returns #segments in *h_nout (after stream sync). */
hipError_t seg_reduce(hipStream_t s, const uint64_t *k, const void *v, uint64_t n,
                      int op, uint64_t *out_k, void *out_v, uint64_t *h_nout, Ws &ws,
                      bool v_prezeroed = false, bool packed = false);

/* grouping sort + segmented aggregate, accumulator init overlapped on a
 * side stream (the fast path reduce_by_key / group_count use) */
hipError_t group_sort_reduce(hipStream_t s, const uint64_t *in_k, const uint64_t *in_v,
                             uint64_t n, int op, uint64_t *out_k, void *out_v,
                             uint64_t *h_nout, Ws &ws);

/* hash-mod partition scatter; h_counts on host after sync */
hipError_t hash_partition(hipStream_t s, const uint64_t *in_k, const uint64_t *in_v,
                          uint64_t n, uint32_t nparts, uint64_t *out_k, uint64_t *out_v,
                          uint64_t *h_counts, Ws &ws);

/* range partition (sort exchange): bucket = #splitters <= key, signed */
hipError_t range_partition(hipStream_t s, const uint64_t *in_k, const uint64_t *in_v,
                           uint64_t n, uint32_t nparts, const int64_t *d_splitters,
                           uint64_t *out_k, uint64_t *out_v, uint64_t *h_counts, Ws &ws);

hipError_t gen_uniform(hipStream_t s, int64_t *keys, int64_t *vals, uint64_t n,
                       uint64_t seed, int key_bits, uint64_t start, bool f64_vals);

hipError_t narrow_map(hipStream_t s, const int64_t *in_k, const int64_t *in_v,
                      uint64_t n, int op, int64_t p0, int64_t *out_k, int64_t *out_v);
hipError_t narrow_filter(hipStream_t s, const int64_t *in_k, const int64_t *in_v,
                         uint64_t n, int pred, int64_t p0, int64_t p1,
                         int64_t *out_k, int64_t *out_v, uint64_t *h_nout, Ws &ws);

hipError_t checksum_pairs(hipStream_t s, const int64_t *k, const int64_t *v,
                          uint64_t n, uint64_t *h_sum, Ws &ws);

/* sort-merge inner join; hash_order=0: signed-key-sorted sides, 1: sides in
 * the grouping order ((h32,key) lexicographic, from group_pairs_inplace) */
/* mode: 0 signed-key order, 1 unsigned-key order, 2 (h32,key) lex order */
hipError_t join_sorted(hipStream_t s, const int64_t *ak, const int64_t *av, uint64_t na,
                       const int64_t *bk, const int64_t *bv, uint64_t nb,
                       int mode,
                       int64_t *out_k, int64_t *out_va, int64_t *out_vb,
                       uint64_t cap, uint64_t *h_nout, Ws &ws);

/* in-place grouping-order sort (the cheap 4-5 pass order joins use) */
hipError_t group_pairs_inplace(hipStream_t s, int64_t *keys, int64_t *vals,
                               uint64_t n, int *order_tag, Ws &ws);

/* u64 offsets (nk+1) from i64 group counts (cogroup / group_by_key) */
hipError_t counts_to_offsets_u64(hipStream_t s, const int64_t *counts, uint64_t nk,
                                 uint64_t *offsets, Ws &ws);

/* distinct keys of A present (want=1) / absent (want=0) in the sorted
 * distinct list kb_u — intersection / subtract (rdd.rs set ops) */
hipError_t member_select(hipStream_t s, const int64_t *ka_u, uint64_t nka,
                         const int64_t *kb_u, uint64_t nkb, int mode, int want,
                         int64_t *out_keys, uint64_t *h_nout, Ws &ws);

/* cogroup index: keys + per-key (offa,lena,offb,lenb) over the two distinct
 * lists and their u64 offsets; h_nk = nka + |B \ A| (co_grouped_rdd.rs
 * :206-249 output shape) */
hipError_t cogroup_index(hipStream_t s, const int64_t *ka_u, uint64_t nka,
                         const uint64_t *offa, const int64_t *kb_u, uint64_t nkb,
                         const uint64_t *offb, int mode, int64_t *keys,
                         uint64_t *o_offa, uint64_t *o_lena, uint64_t *o_offb,
                         uint64_t *o_lenb, uint64_t cap, uint64_t *h_nk, Ws &ws);

size_t ws_bytes_for(uint64_t n);

/* diagnostic phase-cycle buffer (VEGA_PHASE_PROF=1|2), else nullptr */
unsigned long long *phase_prof_buf();
int phase_prof_mode(); /* 0 off, 1 phases, 2 phases + walk counters */

} // namespace vega

#endif
