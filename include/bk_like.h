/* bk_like.h — SQL LIKE ('%', '_', escape char) with charset-aware code
 * points. Restates, step for step, the reference's iterative
 * greedy-with-backtrack matcher LikePredicate::like
 * (/root/reference/include/expr/predicate.h:502-573) and its code-point
 * scanners Binary / UTF8Charset / GBKCharset
 * (/root/reference/src/expr/predicate.cpp:690-737), pinned against the
 * reference's own vectors in test/test_predicate.cpp:37-66
 * (tests/test_like_golden.py carries them verbatim).
 *
 * Return: 1 match, 0 no match, -1 invalid byte sequence — the reference's
 * boost::none; its like_one() then falls back from GBK to Binary
 * (src/expr/predicate.cpp:509-530), which bk_like_one mirrors. */
#ifndef BK_LIKE_H
#define BK_LIKE_H

#include <stddef.h>
#include <string.h>

#ifdef __cplusplus
extern "C" {
#endif

enum BkLikeCharset {
    BK_LIKE_BINARY = 0,
    BK_LIKE_UTF8 = 1,
    BK_LIKE_GBK = 2,
};

/* byte length of the code point at idx; 0 = invalid sequence */
static inline size_t bk_like_cp_(const char* s, size_t n, size_t idx,
                                 int cs) {
    if (idx >= n) return 0;
    unsigned char c = (unsigned char)s[idx];
    if (cs == BK_LIKE_BINARY) return 1;
    if (!(c & 0x80)) return 1;
    if (cs == BK_LIKE_GBK) {
        /* GBKCharset::next_code_point: lead 0x81..0xFE, trail 0x40..0x7E
         * or 0x80..0xFE */
        if (idx + 1 < n && c >= 0x81) {       /* c <= 0xFE implied (0xFF>0xFE
                                                 handled below) */
            unsigned char d = (unsigned char)s[idx + 1];
            if (c <= 0xFE &&
                ((d >= 0x40 && d <= 0x7E) || (d >= 0x80 && d <= 0xFE)))
                return 2;
        }
        return 0;
    }
    /* UTF8Charset: the contiguous continuation-byte count after the lead
     * must equal what the lead announces (get_char_size) */
    size_t num = 1, j = idx;
    while (++j < n && (((unsigned char)s[j]) & 0xC0) == 0x80) num++;
    if ((c & 0xE0) == 0xC0) return num == 2 ? 2 : 0;
    if ((c & 0xF0) == 0xE0) return num == 3 ? 3 : 0;
    if ((c & 0xF0) == 0xF0) return num == 4 ? 4 : 0;
    return 0;   /* stray continuation byte as lead */
}

static inline int bk_like_match(const char* t, size_t tn, const char* p,
                                size_t pn, int cs, char escape) {
    size_t tx = 0, px = 0, ntx = 0, npx = 0;
    while (tx < tn || px < pn) {
        if (px < pn) {
            size_t psz = bk_like_cp_(p, pn, px, cs);
            if (psz == 0) return -1;
            char pc = p[px];
            if (psz == 1 && pc == '_') {
                if (tx < tn) {
                    /* invalid target point consumes one byte here (the
                     * reference keeps t_offset = 1) */
                    size_t to = bk_like_cp_(t, tn, tx, cs);
                    if (to == 0) to = 1;
                    px++;
                    tx += to;
                    continue;
                }
            } else if (psz == 1 && pc == '%') {
                size_t to = 1;
                if (tx < tn) {
                    size_t z = bk_like_cp_(t, tn, tx, cs);
                    if (z > 0) to = z;
                }
                npx = px;          /* retry point: the '%' itself */
                ntx = tx + to;     /* ... with one more point consumed */
                px++;
                continue;
            } else {
                size_t ppx = px, ppsz = psz;
                if (psz == 1 && pc == escape && px + 1 < pn) {
                    ppx = px + 1;
                    ppsz = bk_like_cp_(p, pn, ppx, cs);
                    if (ppsz == 0) return -1;
                }
                if (tx < tn) {
                    size_t tsz = bk_like_cp_(t, tn, tx, cs);
                    if (tsz == 0) return -1;
                    if (tsz == ppsz && memcmp(t + tx, p + ppx, tsz) == 0) {
                        px = ppx + ppsz;
                        tx += tsz;
                        continue;
                    }
                }
            }
        }
        if (ntx > 0 && ntx <= tn) {   /* backtrack to the last '%' */
            px = npx;
            tx = ntx;
            continue;
        }
        return 0;
    }
    return 1;
}

/* like_one (src/expr/predicate.cpp:509-530): evaluate in the session
 * charset; an invalid sequence under GBK retries as Binary. */
static inline int bk_like_one(const char* t, size_t tn, const char* p,
                              size_t pn, int cs, char escape) {
    int r = bk_like_match(t, tn, p, pn, cs, escape);
    if (r < 0 && cs == BK_LIKE_GBK)
        r = bk_like_match(t, tn, p, pn, BK_LIKE_BINARY, escape);
    return r < 0 ? 0 : r;
}

#ifdef __cplusplus
}
#endif
#endif /* BK_LIKE_H */
