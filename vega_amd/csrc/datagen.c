/* datagen.c — deterministic synthetic input generators (host side).
 *
 * Shared by bench.py, tests and the CPU-baseline leg. The GPU library
 * implements the SAME integer formulas as device kernels (vega_common.h
 * vega_rand_u64), so host- and device-generated inputs are bit-identical
 * for the uniform streams. Zipf uses double pow() and is host-only
 * (generated here, then H2D) to avoid libm cross-device drift.
 *
 * Row j of a stream is a pure function of (seed, j): generation is
 * order-free, so any rank can generate its own slice [start, start+n).
 *
 * Benchmark configs (BASELINE.json / SURVEY.md §8d):
 *   C0: 1e6 rows, keys uniform [0,2^20)          -> key_bits=20
 *   C1: 1e9 rows, keys uniform [0,2^63)          -> key_bits=63
 *   C2: 1e9 rows, Zipf s=1.1 over keyspace 1e8   -> zipf
 *   C3: 2e9 rows, keys uniform [0,2^63)          -> key_bits=63 (sort)
 *   C4: 5e8 x 5e8, keys uniform [0,5e8)          -> range
 */
#include <stdint.h>
#include <math.h>
#include "../../include/vega_common.h"

void vega_gen_uniform_pairs_i64(uint64_t seed, uint64_t start, uint64_t n,
                                int key_bits, int64_t *keys, int64_t *vals) {
    uint64_t mask = (key_bits >= 64) ? ~0ULL : ((1ULL << key_bits) - 1);
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (uint64_t i = 0; i < n; i++) {
        uint64_t j = start + i;
        keys[i] = (int64_t)(vega_rand_u64(seed, 2 * j) & mask);
        vals[i] = (int64_t)vega_rand_u64(seed, 2 * j + 1);
    }
}

void vega_gen_uniform_pairs_f64(uint64_t seed, uint64_t start, uint64_t n,
                                int key_bits, int64_t *keys, double *vals) {
    uint64_t mask = (key_bits >= 64) ? ~0ULL : ((1ULL << key_bits) - 1);
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (uint64_t i = 0; i < n; i++) {
        uint64_t j = start + i;
        keys[i] = (int64_t)(vega_rand_u64(seed, 2 * j) & mask);
        /* exact dyadic conversion: identical on CPU and GPU */
        vals[i] = (double)(vega_rand_u64(seed, 2 * j + 1) >> 11) * 0x1p-53;
    }
}

void vega_gen_uniform_range_pairs_i64(uint64_t seed, uint64_t start, uint64_t n,
                                      uint64_t range, int64_t *keys, int64_t *vals) {
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (uint64_t i = 0; i < n; i++) {
        uint64_t j = start + i;
        keys[i] = (int64_t)(vega_rand_u64(seed, 2 * j) % range);
        vals[i] = (int64_t)vega_rand_u64(seed, 2 * j + 1);
    }
}

/* Zipf-like keys via the continuous inverse CDF of density x^-s on
 * [1, M+1): x = ((pow(M+1,1-s)-1)*u + 1)^(1/(1-s)); key = floor(x)-1 in
 * [0, M). Deterministic; host-only (libm pow). */
void vega_gen_zipf_pairs_i64(uint64_t seed, uint64_t start, uint64_t n,
                             double s, uint64_t keyspace, int64_t *keys, int64_t *vals) {
    double a = 1.0 - s;
    double top = pow((double)keyspace + 1.0, a) - 1.0;
#ifdef _OPENMP
#pragma omp parallel for schedule(static)
#endif
    for (uint64_t i = 0; i < n; i++) {
        uint64_t j = start + i;
        double u = (double)(vega_rand_u64(seed, 2 * j) >> 11) * 0x1p-53;
        double x = pow(top * u + 1.0, 1.0 / a);
        int64_t k = (int64_t)x - 1;
        if (k < 0) k = 0;
        if (k >= (int64_t)keyspace) k = (int64_t)keyspace - 1;
        keys[i] = k;
        vals[i] = (int64_t)vega_rand_u64(seed, 2 * j + 1);
    }
}
