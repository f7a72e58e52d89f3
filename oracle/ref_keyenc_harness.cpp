// ref_keyenc_harness.cpp — *** TEST INFRASTRUCTURE ONLY ***
// Compiles the reference's own mem-comparable key encoder IN PLACE from
// /root/reference/include/common/key_encoder.h (self-contained header, no
// deps) and dumps golden vectors used to pin oracle/oracle.c's restatement
// (bk_keyenc.h). Built by `make ref` in this directory (only where
// /root/reference exists — this container, not the GPU box); the dumped
// vectors are committed as tests/golden/keyenc_golden.json and travel.
//
// No reference sources are copied into this repo; the header is included
// from its mounted location at build time only.
#include <cstdio>
#include <cstdint>
#include <cinttypes>
#include "key_encoder.h"   // resolved via -I /root/reference/include/common

using baikaldb::KeyEncoder;

int main() {
    const int64_t ivals[] = {
        0, 1, -1, 2, -2, 100, -100, 127, -128, 32767, -32768,
        2147483647LL, -2147483648LL, 9223372036854775807LL,
        (int64_t)(-9223372036854775807LL - 1), 123456789012345LL,
        -123456789012345LL, 42, -42, 1000000007LL
    };
    const double dvals[] = {
        0.0, -0.0, 1.0, -1.0, 0.5, -0.5, 120.5, 120.6, -120.5, -120.6,
        1e300, -1e300, 1e-300, -1e-300, 3.141592653589793,
        -3.141592653589793, 2.2250738585072014e-308, -2.2250738585072014e-308,
        1.7976931348623157e308, -1.7976931348623157e308
    };
    printf("{\n  \"encode_i64\": [\n");
    for (size_t i = 0; i < sizeof(ivals)/sizeof(ivals[0]); i++) {
        uint64_t enc = KeyEncoder::to_endian_u64(KeyEncoder::encode_i64(ivals[i]));
        uint64_t enc_native = KeyEncoder::encode_i64(ivals[i]);
        printf("    [%" PRId64 ", %" PRIu64 ", %" PRIu64 "]%s\n",
               ivals[i], enc_native, enc,
               i + 1 < sizeof(ivals)/sizeof(ivals[0]) ? "," : "");
    }
    printf("  ],\n  \"encode_f64\": [\n");
    for (size_t i = 0; i < sizeof(dvals)/sizeof(dvals[0]); i++) {
        uint64_t enc = KeyEncoder::to_endian_u64(KeyEncoder::encode_f64(dvals[i]));
        uint64_t enc_native = KeyEncoder::encode_f64(dvals[i]);
        printf("    [%.17g, %" PRIu64 ", %" PRIu64 "]%s\n",
               dvals[i], enc_native, enc,
               i + 1 < sizeof(dvals)/sizeof(dvals[0]) ? "," : "");
    }
    printf("  ],\n  \"encode_i32\": [\n");
    const int32_t i32vals[] = {0, 1, -1, 2147483647, (int32_t)(-2147483647 - 1), 12345, -12345};
    for (size_t i = 0; i < sizeof(i32vals)/sizeof(i32vals[0]); i++) {
        printf("    [%d, %u]%s\n", i32vals[i],
               KeyEncoder::encode_i32(i32vals[i]),
               i + 1 < sizeof(i32vals)/sizeof(i32vals[0]) ? "," : "");
    }
    printf("  ],\n  \"to_endian_u64\": [\n");
    const uint64_t uvals[] = {0ull, 0x1234567890123456ull, 0x295633CDFA778899ull,
                              0xFFFFFFFFFFFFFFFFull, 1ull};
    for (size_t i = 0; i < sizeof(uvals)/sizeof(uvals[0]); i++) {
        printf("    [%" PRIu64 ", %" PRIu64 "]%s\n", uvals[i],
               KeyEncoder::to_endian_u64(uvals[i]),
               i + 1 < sizeof(uvals)/sizeof(uvals[0]) ? "," : "");
    }
    printf("  ]\n}\n");
    return 0;
}
