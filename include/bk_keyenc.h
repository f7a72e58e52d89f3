// bk_keyenc.h — order-preserving (mem-comparable) value encoding, restated
// from baidu/BaikalDB include/common/key_encoder.h:104-173 (sign-flip for
// signed ints, sign-flip-or-complement for floats) and the big-endian byte
// order of include/common/mut_table_key.h:113-208 (append_i64/append_double).
//
// The GPU engine keeps encoded values as native uint64 (numeric order ==
// unsigned integer order); byte-swapped big-endian form is produced only when
// materializing the reference's MutTableKey group-key byte strings
// (src/exec/exec_node.cpp:555-571 encode_exprs_key).
//
// Shared by oracle (gcc), GPU kernels (hipcc) and the C++ host layer.
#ifndef BK_KEYENC_H
#define BK_KEYENC_H

#include <stdint.h>
#include <string.h>

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define BK_KHD __host__ __device__ static inline
#else
#define BK_KHD static inline
#endif

/* KeyEncoder::encode_i64 (key_encoder.h:128): flip sign bit. */
BK_KHD uint64_t bk_enc_i64(int64_t v) {
    return ((uint64_t)v) ^ 0x8000000000000000ull;
}
BK_KHD int64_t bk_dec_i64(uint64_t u) {
    return (int64_t)(u ^ 0x8000000000000000ull);
}

/* KeyEncoder::encode_f64 (key_encoder.h:157): >=0 -> set sign bit; <0 -> ~bits.
 * NOTE the reference tests `in >= 0.0`, so -0.0 encodes as a NEGATIVE branch
 * value? No: (-0.0 >= 0.0) is true in IEEE, so -0.0 takes the positive branch
 * (bits 0x8000.. | sign -> 0x8000000000000000|0x8000.. ) — restate exactly. */
BK_KHD uint64_t bk_enc_f64(double v) {
    uint64_t bits;
    memcpy(&bits, &v, 8);
    if (v >= 0.0) return bits | 0x8000000000000000ull;
    return ~bits;
}
BK_KHD double bk_dec_f64(uint64_t u) {
    if (u & 0x8000000000000000ull) u &= ~0x8000000000000000ull;
    else u = ~u;
    double v;
    memcpy(&v, &u, 8);
    return v;
}

/* KeyEncoder::encode_i32 / encode_i8 (key_encoder.h:104,120) */
BK_KHD uint32_t bk_enc_i32(int32_t v) { return ((uint32_t)v) ^ 0x80000000u; }
BK_KHD uint8_t  bk_enc_i8(int8_t v)   { return ((uint8_t)v) ^ 0x80u; }

/* byte swap to big-endian (KeyEncoder::to_endian_u64, key_encoder.h:54 — the
 * host is little-endian, asserted by the reference's own test
 * test/test_key_encoder.cpp:31-33). */
BK_KHD uint64_t bk_bswap64(uint64_t x) {
    return ((x & 0x00000000000000FFull) << 56) |
           ((x & 0x000000000000FF00ull) << 40) |
           ((x & 0x0000000000FF0000ull) << 24) |
           ((x & 0x00000000FF000000ull) << 8)  |
           ((x & 0x000000FF00000000ull) >> 8)  |
           ((x & 0x0000FF0000000000ull) >> 24) |
           ((x & 0x00FF000000000000ull) >> 40) |
           ((x & 0xFF00000000000000ull) >> 56);
}
BK_KHD uint32_t bk_bswap32(uint32_t x) {
    return ((x & 0x000000FFu) << 24) | ((x & 0x0000FF00u) << 8) |
           ((x & 0x00FF0000u) >> 8)  | ((x & 0xFF000000u) >> 24);
}

#endif /* BK_KEYENC_H */
