# like.py — SQL LIKE ('%', '_', escape) pattern matching + the dictionary
# bitmap compiler for the engine's BK_OP_IN_BITMAP pushdown.
#
# sql_like restates the reference's LikePredicate::like byte for byte
# (/root/reference/include/expr/predicate.h:502-573; charset scanners
# src/expr/predicate.cpp:690-737) and is pinned against the reference's own
# vectors (test/test_predicate.cpp:37-66) in tests/test_like_golden.py,
# together with the C++ restatement in include/bk_like.h (the two must
# agree bit for bit — the host layer compiles patterns with either).
BINARY, UTF8, GBK = 0, 1, 2


def _cp_len(s: bytes, idx: int, cs: int) -> int:
    """Byte length of the code point at idx; 0 = invalid sequence."""
    if idx >= len(s):
        return 0
    c = s[idx]
    if cs == BINARY:
        return 1
    if not (c & 0x80):
        return 1
    if cs == GBK:
        if idx + 1 < len(s) and 0x81 <= c <= 0xFE:
            d = s[idx + 1]
            if 0x40 <= d <= 0x7E or 0x80 <= d <= 0xFE:
                return 2
        return 0
    # UTF8: contiguous continuation count must equal what the lead announces
    num, j = 1, idx
    while j + 1 < len(s) and (s[j + 1] & 0xC0) == 0x80:
        num += 1
        j += 1
    if (c & 0xE0) == 0xC0:
        return 2 if num == 2 else 0
    if (c & 0xF0) == 0xE0:
        return 3 if num == 3 else 0
    if (c & 0xF0) == 0xF0:
        return 4 if num == 4 else 0
    return 0


def like_match(target, pattern, charset=BINARY, escape=b"\\"):
    """1 match / 0 no match / -1 invalid sequence (the reference's
    boost::none)."""
    t = target.encode() if isinstance(target, str) else bytes(target)
    p = pattern.encode() if isinstance(pattern, str) else bytes(pattern)
    esc = escape[0] if isinstance(escape, (bytes, bytearray)) else ord(escape)
    tn, pn = len(t), len(p)
    tx = px = ntx = npx = 0
    while tx < tn or px < pn:
        if px < pn:
            psz = _cp_len(p, px, charset)
            if psz == 0:
                return -1
            pc = p[px]
            if psz == 1 and pc == 0x5F:                    # '_'
                if tx < tn:
                    to = _cp_len(t, tx, charset) or 1
                    px += 1
                    tx += to
                    continue
            elif psz == 1 and pc == 0x25:                  # '%'
                to = 1
                if tx < tn:
                    z = _cp_len(t, tx, charset)
                    if z > 0:
                        to = z
                npx, ntx = px, tx + to
                px += 1
                continue
            else:
                ppx, ppsz = px, psz
                if psz == 1 and pc == esc and px + 1 < pn:
                    ppx = px + 1
                    ppsz = _cp_len(p, ppx, charset)
                    if ppsz == 0:
                        return -1
                if tx < tn:
                    tsz = _cp_len(t, tx, charset)
                    if tsz == 0:
                        return -1
                    if tsz == ppsz and t[tx:tx + tsz] == p[ppx:ppx + ppsz]:
                        px = ppx + ppsz
                        tx += tsz
                        continue
        if 0 < ntx <= tn:                                  # backtrack to '%'
            px, tx = npx, ntx
            continue
        return 0
    return 1


def sql_like(target, pattern, charset=BINARY, escape=b"\\"):
    """LikePredicate::like_one semantics: a GBK-invalid sequence retries as
    Binary (src/expr/predicate.cpp:509-530); invalid otherwise = no match."""
    r = like_match(target, pattern, charset, escape)
    if r < 0 and charset == GBK:
        r = like_match(target, pattern, BINARY, escape)
    return r == 1


def like_accept_codes(words, pattern, charset=BINARY, escape=b"\\",
                      negate=False):
    """Compile a LIKE pattern against a dictionary word list into the
    accept-code list for the engine's BK_OP_IN_BITMAP pushdown (the
    cstore-dict trick: the host matches once per distinct word, the GPU
    filters by code membership)."""
    out = [c for c, w in enumerate(words)
           if sql_like(w, pattern, charset, escape) != negate]
    return out
