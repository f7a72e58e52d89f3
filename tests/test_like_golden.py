# SQL LIKE golden tests: the reference's OWN vectors
# (/root/reference/test/test_predicate.cpp:37-66, byte-exact — the GBK
# strings are raw GBK bytes) against BOTH restatements:
#   - Python  baikaldb_amd/like.py (binding-level pattern->bitmap compiler)
#   - C++     include/bk_like.h via the bkgpu_like_match export
# plus a randomized cross-check that the two stay bit-identical.
import ctypes as C
import os
import random

import pytest

from baikaldb_amd.like import (BINARY, UTF8, GBK, like_match, sql_like,
                               like_accept_codes)

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(REPO, "baikaldb_amd", "libbkgpu.so")

# (expect, charset, target, pattern) — test_predicate.cpp:37-66 verbatim.
# b"\\\\" in the C++ source is one backslash in the string = the escape char.
VECTORS = [
    (True, BINARY, b"www.bad/aca?bd_vid", b"www.bad/aca?bd_vid"),
    (True, BINARY, b"abc", b"a_c"),
    (True, BINARY, b"abc", b"%"),
    (True, BINARY, b"axxx", b"a%x%x"),
    (True, GBK, b"axxx", b"a%x%x"),
    (True, GBK, b"\xd6\xd0\xce\xc4testbd_vid\xd6\xd0\xce\xc4test",
     b"\xd6\xd0\xce\xc4testbd_vid\xd6\xd0\xce\xc4test"),
    (True, GBK, b"\xd6\xd0\xce\xc4testbd_vid\xd6\xd0\xce\xc4test",
     b"%testbd_vid\xd6\xd0\xce\xc4tes%"),
    # GBK point boundaries differ from bytes: "\xbb\xbc" + "\xc1\xcb" are the
    # two GBK points; "%\xbc\xc1%" straddles them -> no GBK match, but a
    # Binary match (test_predicate.cpp:45-47)
    (False, GBK, b"\xbb\xbc\xc1\xcb", b"%\xbc\xc1%"),
    (True, BINARY, b"\xbb\xbc\xc1\xcb", b"%\xbc\xc1%"),
    (True, GBK, b"\xd6\xd0%\xce\xc4", b"\xd6\xd0\\%\xce\xc4"),
    (False, GBK, b"\xd6\xd0\xb2\xe2\xca\xd4\xce\xc4", b"\xd6\xd0\\%\xce\xc4"),
    (False, GBK, b"\xd6\xd0f\xce\xc4", b"\xd6\xd0\\_\xce\xc4"),
    (True, GBK, b"\xd6\xd0f\xce\xc4", b"\xd6\xd0_\xce\xc4"),
    (False, GBK, b"\xd6\xd0%\xce\xc4", b"\xd6\xd0\xb2\xe2\xca\xd4\xce\xc4"),
    (True, GBK, b"\xd6\xd0aaa\xce\xc4", b"\xd6\xd0%\xce\xc4"),
    (True, GBK, b"", b""),
    (True, GBK, b"test", b"te%st"),
    (True, BINARY, b"test", b"te%st"),
    (True, GBK, b"test", b"te%%st"),
    (True, BINARY, b"test", b"te%%st"),
    (True, GBK, b"test", b"%test%"),
    (True, BINARY, b"test", b"%test%"),
    (True, GBK, b"test", b"_%_%_%_"),
    (True, GBK, b"test", b"_%_%st"),
    (True, GBK, b"3hello", b"3%hello"),
    (True, BINARY, b"3hello", b"3%hello"),
    (False, GBK, b"a" * 27, b"a%a%a%a%a%a%a%a%b"),
    (False, BINARY, b"a" * 27, b"a%a%a%a%a%a%a%a%b"),
]


@pytest.fixture(scope="module")
def clib():
    if not os.path.exists(LIB):
        pytest.skip("libbkgpu.so not built")
    lib = C.CDLL(LIB)
    lib.bkgpu_like_match.restype = C.c_int
    lib.bkgpu_like_match.argtypes = [C.c_char_p, C.c_int64, C.c_char_p,
                                     C.c_int64, C.c_int, C.c_char]
    return lib


def test_reference_vectors_python():
    for expect, cs, target, pattern in VECTORS:
        got = like_match(target, pattern, cs)
        assert got == (1 if expect else 0), (target, pattern, cs, got)


def test_reference_vectors_c(clib):
    for expect, cs, target, pattern in VECTORS:
        got = clib.bkgpu_like_match(target, len(target), pattern,
                                    len(pattern), cs, b"\\")
        assert got == (1 if expect else 0), (target, pattern, cs, got)


def test_utf8_code_points():
    # '_' consumes one CODE POINT under utf8mb4, one BYTE under binary
    s = "中文".encode("utf-8")        # 6 bytes, 2 points
    assert sql_like(s, b"__", UTF8)
    assert not sql_like(s, b"__", BINARY)
    assert sql_like(s, b"______", BINARY)
    assert sql_like(s, "中_".encode("utf-8"), UTF8)
    assert sql_like(s, b"%" + "文".encode("utf-8"), UTF8)
    # invalid utf8 in the target under a LITERAL pattern -> boost::none
    # (-1); under '_' the reference consumes one byte and keeps going
    # (predicate.h:519-524 keeps t_offset = 1)
    assert like_match(b"\xff\xfe", b"ab", UTF8) == -1
    assert like_match(b"\xff\xfe", b"__", UTF8) == 1


def test_escape_char_variants():
    assert sql_like(b"50%", b"50\\%")
    assert not sql_like(b"505", b"50\\%")
    assert sql_like(b"a_b", b"a#_b", escape=b"#")
    assert not sql_like(b"axb", b"a#_b", escape=b"#")
    # trailing escape matches itself (px+1 < pn fails -> literal)
    assert sql_like(b"a\\", b"a\\")


def test_python_c_agree_fuzz(clib):
    rng = random.Random(20260916)
    alpha = b"ab%_\\\xd6\xc4\xe4"
    for _ in range(3000):
        t = bytes(rng.choice(alpha) for _ in range(rng.randrange(0, 8)))
        p = bytes(rng.choice(alpha) for _ in range(rng.randrange(0, 8)))
        cs = rng.choice([BINARY, UTF8, GBK])
        py = like_match(t, p, cs)
        cc = clib.bkgpu_like_match(t, len(t), p, len(p), cs, b"\\")
        assert py == cc, (t, p, cs, py, cc)


def test_accept_codes_bitmap_compiler():
    words = ["apple", "apricot", "banana", "grape", "a%b", "axb"]
    assert like_accept_codes(words, b"ap%") == [0, 1]
    assert like_accept_codes(words, b"%an%") == [2]
    assert like_accept_codes(words, b"a\\%b") == [4]
    assert like_accept_codes(words, b"a_b") == [4, 5]
    assert like_accept_codes(words, b"ap%", negate=True) == [2, 3, 4, 5]
