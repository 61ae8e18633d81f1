/* vega_common.h — constants and inline helpers shared by the product library
 * (vega_amd/csrc) and the CPU oracle (oracle/).
 *
 * The key hash replaces the reference's MetroHash64-based HashPartitioner
 * (/root/reference/src/partitioner.rs:21-25,54-57: `MetroHash64(key) as usize
 * % partitions`). fasthash/MetroHash sources are NOT vendored in the
 * reference and no reference test asserts concrete hash values or partition
 * assignments (partitioner.rs:60-82 only prints), so partition assignment is
 * UNPINNED; result-set parity of reduce_by_key/group_by_key/join is invariant
 * under any deterministic total partition function (each key is routed to
 * exactly one reducer). We use splitmix64 (public-domain finalizer constants)
 * on both the CPU oracle and the GPU path so the two agree bit-exactly.
 */
#ifndef VEGA_COMMON_H
#define VEGA_COMMON_H

#include <stdint.h>

#ifdef __cplusplus
#define VEGA_INLINE static inline
#else
#define VEGA_INLINE static inline
#endif

#ifdef __HIPCC__
#define VEGA_HD __host__ __device__
#else
#define VEGA_HD
#endif

/* splitmix64 finalizer: the partition/shuffle hash. */
VEGA_INLINE VEGA_HD uint64_t vega_hash_u64(uint64_t x) {
    x += 0x9E3779B97F4A7C15ULL;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ULL;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBULL;
    return x ^ (x >> 31);
}

/* partition id for key k (reference: partitioner.rs:54-57 `hash % partitions`) */
VEGA_INLINE VEGA_HD uint32_t vega_partition_of(int64_t k, uint32_t nparts) {
    return (uint32_t)(vega_hash_u64((uint64_t)k) % (uint64_t)nparts);
}

/* Counter-based deterministic RNG for synthetic inputs: draw j of stream
 * `seed` (benchmarks use seed = 0xC0FFEE + config index). */
VEGA_INLINE VEGA_HD uint64_t vega_rand_u64(uint64_t seed, uint64_t j) {
    return vega_hash_u64(seed ^ (0x9E3779B97F4A7C15ULL * (j + 1)));
}

/* ParallelCollection::slice contiguous chunking
 * (/root/reference/src/rdd/parallel_collection_rdd.rs:116-145):
 * partition p = rows [ p*n/P , (p+1)*n/P )  (integer division). */
VEGA_INLINE VEGA_HD uint64_t vega_slice_start(uint64_t n, uint32_t nparts, uint32_t p) {
    return ((__uint128_t)p * n) / nparts;
}

#endif /* VEGA_COMMON_H */
