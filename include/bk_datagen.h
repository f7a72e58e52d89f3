// bk_datagen.h — deterministic synthetic table generator, shared verbatim by
// the CPU oracle (gcc) and the GPU kernels (hipcc device code), so that both
// sides see bit-identical inputs for any (seed, row, column).
//
// Replaces the data source below the drop-in boundary (the reference's
// RocksDB scan, src/engine/table_iterator.cpp:446-660, is OUT of scope —
// synthetic columnar generators stand in for the scan).
//
// Only integer ops and IEEE-754 double multiplies of exact constants are used
// (no libm), so CPU and gfx950 results are bit-identical by construction.
#ifndef BK_DATAGEN_H
#define BK_DATAGEN_H

#include <stdint.h>
#include "bk_common.h"

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define BK_HD __host__ __device__ static inline
#else
#define BK_HD static inline
#endif

/* splitmix64 (public-domain PRNG finalizer): stateless per (seed,row,col). */
BK_HD uint64_t bk_mix64(uint64_t x) {
    x += 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    return x ^ (x >> 31);
}

BK_HD uint64_t bk_cell_bits(uint64_t seed, uint64_t row, uint32_t col) {
    return bk_mix64(seed ^ bk_mix64(row ^ ((uint64_t)(col + 1) << 56)));
}

/* NULL decision: separate stream from the value stream. */
BK_HD int bk_cell_valid(uint64_t seed, uint64_t row, uint32_t col,
                        int32_t null_frac_x1e6) {
    if (null_frac_x1e6 <= 0) return 1;
    uint64_t u = bk_mix64(seed ^ 0xA5A5A5A5ull ^ bk_cell_bits(seed, row, col + 97));
    return (int)((u % 1000000ull) >= (uint64_t)null_frac_x1e6);
}

/* uniform integer in [lo, hi) */
BK_HD int64_t bk_gen_uniform_i64(uint64_t u, int64_t lo, int64_t hi) {
    uint64_t span = (uint64_t)(hi - lo);
    return lo + (int64_t)(u % span);
}

/* integer-only skewed distribution on [0, D): density concentrated near 0
 * ("Zipf-shaped" group keys for the GROUP BY configs; exact distribution is
 * a knob, identity CPU==GPU is the requirement). D must be <= 2^21. */
BK_HD int64_t bk_gen_cubeskew(uint64_t u, int64_t D) {
    uint64_t w = u >> 43;                 /* 21 bits */
    uint64_t t = (w * w) >> 21;           /* w^2 / 2^21, <= 2^21 */
    uint64_t c = (t * w) >> 21;           /* w^3 / 2^42, <= 2^21 */
    return (int64_t)(c % (uint64_t)D);
}

/* dict code uniform in [0, nwords) */
BK_HD int32_t bk_gen_dict(uint64_t u, int64_t nwords) {
    return (int32_t)(u % (uint64_t)nwords);
}

/* log-uniform "Zipf-1-like" integer on [0, D): pick an octave uniformly from
 * the ceil(log2(D)) octaves of [1, 2^ceil(log2 D)), then a value uniformly
 * inside it; values >= D wrap by modulo (slight distortion at the top
 * octave). Density ~ 1/(k+1). Integer-only => bit-identical CPU/GPU. */
BK_HD int64_t bk_gen_zipfoct(uint64_t u, int64_t D) {
    if (D <= 1) return 0;
    uint32_t noct = 1;
    while ((1ll << noct) < D && noct < 62) noct++;
    uint32_t e = (uint32_t)((u >> 40) % noct);            /* octave */
    uint64_t lo = bk_mix64(u ^ 0x0C7AE5ull);
    uint64_t k = ((uint64_t)1 << e) + (lo & (((uint64_t)1 << e) - 1)) - 1;
    return (int64_t)(k % (uint64_t)D);
}

/* approx N(0,1) double: (sum of four u16) centered and scaled. All steps are
 * exact int ops plus one exact int->double convert and one double multiply by
 * a constant, so CPU and GPU agree bitwise. Var(sum of 4 u16) = 4*(65536^2-1)/12;
 * scale = 1/sqrt(that) precomputed as a decimal literal (closest double). */
BK_HD double bk_gen_sumu16(uint64_t u) {
    int64_t s = (int64_t)(u & 0xFFFF) + (int64_t)((u >> 16) & 0xFFFF)
              + (int64_t)((u >> 32) & 0xFFFF) + (int64_t)((u >> 48) & 0xFFFF);
    /* mean = 4*65535/2 = 131070; sd = sqrt(4*(65536*65536-1)/12) = 37837.16... */
    return (double)(s - 131070) * 2.6429099261197387e-05;
}

/* apply a BkScalarFn to a packed-DATETIME int64 (identity for BK_FN_NONE).
 * Restates datetime.h:35-45 (year/month/day) and datetime.cpp:410-419 +
 * internal_functions.cpp hour/minute/second (time bit fields). */
BK_HD int64_t bk_scalar_fn(int32_t fn, int64_t v) {
    uint64_t dt = (uint64_t)v;
    switch (fn) {
        case 1 /*YEAR*/:   return (int64_t)(((dt >> 46) & 0x1FFFFull) / 13u);
        case 2 /*MONTH*/:  return (int64_t)(((dt >> 46) & 0x1FFFFull) % 13u);
        case 3 /*DAY*/:    return (int64_t)((dt >> 41) & 0x1Full);
        case 4 /*HOUR*/:   return (int64_t)((dt >> 36) & 0x1Full);
        case 5 /*MINUTE*/: return (int64_t)((dt >> 30) & 0x3Full);
        case 6 /*SECOND*/: return (int64_t)((dt >> 24) & 0x3Full);
        default:           return v;
    }
}

/* a valid packed DATETIME: year in [2019,2026), month 1-12, day 1-28,
 * h/m/s uniform — deterministic, integer-only (CPU == GPU) */
BK_HD int64_t bk_gen_datetime(uint64_t u) {
    uint64_t year  = 2019u + (u & 0xFF) % 7u;
    uint64_t month = 1u + ((u >> 8) & 0xFF) % 12u;
    uint64_t day   = 1u + ((u >> 16) & 0xFF) % 28u;
    uint64_t hour  = ((u >> 24) & 0xFF) % 24u;
    uint64_t minu  = ((u >> 33) & 0xFF) % 60u;
    uint64_t sec   = ((u >> 42) & 0xFF) % 60u;
    return (int64_t)(((year * 13u + month) << 46) | (day << 41) |
                     (hour << 36) | (minu << 30) | (sec << 24));
}

/* Generate one cell. Returns value through the matching out-param; the
 * caller dispatches storage by col_type. */
BK_HD int64_t bk_gen_i64(const BkColSpec* cs, uint64_t seed, uint64_t row, uint32_t col) {
    uint64_t u = bk_cell_bits(seed, row, col);
    switch ((BkDist)cs->dist) {
        case BK_DIST_UNIFORM_I64: return bk_gen_uniform_i64(u, cs->p0, cs->p1);
        case BK_DIST_CUBESKEW:    return bk_gen_cubeskew(u, cs->p0);
        case BK_DIST_DICT:        return (int64_t)bk_gen_dict(u, cs->p0);
        case BK_DIST_ZIPFOCT:     return bk_gen_zipfoct(u, cs->p0);
        case BK_DIST_DATETIME:    return bk_gen_datetime(u);
        default:                  return 0;
    }
}

BK_HD double bk_gen_f64(const BkColSpec* cs, uint64_t seed, uint64_t row, uint32_t col) {
    uint64_t u = bk_cell_bits(seed, row, col);
    (void)cs;
    return bk_gen_sumu16(u);
}

/* Deterministic dict word for a code (host-side; rows store codes, words are
 * materialized only when emitting VARCHAR values). Unique per code by
 * construction (the code is embedded in the word). */
#include <stdio.h>
static inline int bk_dict_word(uint64_t seed, int64_t code, char* out, int cap) {
    uint32_t h = (uint32_t)bk_mix64(seed ^ 0xD1C7ull ^ (uint64_t)code);
    return snprintf(out, (size_t)cap, "w%06lx_%08x", (unsigned long)code, h);
}

#endif /* BK_DATAGEN_H */
