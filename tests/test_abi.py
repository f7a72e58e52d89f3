# CPU-side checks of the C-ABI shared library: it loads and exports every
# symbol include/bkgpu.h declares (no compute calls — no GPU here).
import ctypes as C
import os
import re
import subprocess

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(REPO, "baikaldb_amd", "libbkgpu.so")
HDR = os.path.join(REPO, "include", "bkgpu.h")


def _build_if_needed():
    csrc = os.path.join(REPO, "baikaldb_amd", "csrc")
    srcs = [os.path.join(csrc, f) for f in
            ("bkgpu.hip", "bkexec.cpp", "bkparquet.cpp", "bkarrow.cpp",
             "bkcstore.cpp")]   # must match __graft_entry__.build()
    if not os.path.exists(LIB) or \
            os.path.getmtime(LIB) < max(os.path.getmtime(s) for s in srcs):
        subprocess.run(["hipcc", "--offload-arch=gfx950", "-O3", "-std=c++17",
                        "-munsafe-fp-atomics", "-fPIC", "-shared", *srcs,
                        "-o", LIB],
                       check=True, capture_output=True)


def _declared_symbols():
    with open(HDR) as f:
        text = f.read()
    return sorted(set(re.findall(r"\b(bkgpu_\w+)\s*\(", text)))


def test_library_loads_and_exports_header_symbols():
    _build_if_needed()
    lib = C.CDLL(LIB)
    syms = _declared_symbols()
    assert len(syms) >= 15
    for s in syms:
        assert hasattr(lib, s), f"missing export: {s}"


def test_engine_refuses_without_gpu():
    """On a machine with no GPU the engine must fail loudly, not fall back."""
    _build_if_needed()
    import sys
    sys.path.insert(0, REPO)
    import torch
    if torch.cuda.is_available():
        pytest.skip("GPU present")
    from baikaldb_amd import GpuEngine, NativeEngineMissing
    with pytest.raises((NativeEngineMissing, RuntimeError)):
        GpuEngine()


def test_substr_reference_semantics():
    """substr_ref restates internal_functions.cpp substr() exactly: 1-based
    positions, negative start counts from the end (pos = size+start), start
    0 or past-end -> empty, len<=0 -> empty, 2-arg form runs to the end.
    (The old remap lambda mishandled start<=0 — real parity fix.)"""
    from baikaldb_amd.engine import substr_ref

    s = "abcdef"
    # (start, len, expected) — MySQL SUBSTRING semantics as the reference
    # implements them
    cases = [
        (1, 3, "abc"), (2, 2, "bc"), (6, 1, "f"), (6, 10, "f"),
        (7, 1, ""), (100, 5, ""),
        (0, 3, ""),                      # pos 0: --pos -> -1 -> empty
        (-1, 1, "f"), (-3, 2, "de"), (-6, 2, "ab"),
        (-7, 2, ""),                     # size+start < 0 -> empty
        (1, 0, ""), (1, -2, ""),         # len <= 0 -> empty
        (3, None, "cdef"), (-2, None, "ef"), (0, None, ""),
    ]
    for start, ln, exp in cases:
        got = substr_ref(s, start, ln)
        assert got == exp, (start, ln, got, exp)
    assert substr_ref("", 1, 1) == ""
    assert substr_ref("", -1, None) == ""


def test_string_fn_byte_semantics():
    """upper/lower/reverse match internal_functions.cpp byte-wise behavior:
    ::toupper/::tolower touch only ASCII letters (non-ASCII UTF-8 bytes pass
    through), reverse reverses BYTES (not codepoints)."""
    from baikaldb_amd.engine import GpuEngine

    up = GpuEngine.STRING_FNS["upper"]
    lo = GpuEngine.STRING_FNS["lower"]
    rv = GpuEngine.STRING_FNS["reverse"]
    assert up("aBc9_z") == "ABC9_Z"
    assert lo("AbC9_Z") == "abc9_z"
    # non-ASCII letters are NOT case-mapped (C locale tolower on bytes)
    assert up("café") == "CAFé"
    assert lo("CAFÉ") == "cafÉ"
    assert rv("abc") == "cba"
    # byte reversal of multibyte input mirrors std::reverse on the raw
    # bytes: the result's bytes are exactly the reversed input bytes
    got = rv("aé").encode("utf-8", "surrogateescape")
    assert got == "aé".encode("utf-8")[::-1]


def test_substr_left_right_byte_positions():
    """substr/left/right index BYTES like std::string (multibyte chars can
    split; surrogateescape keeps the raw bytes) — via the same
    resolve_string_fn the dict remap uses."""
    from baikaldb_amd.engine import substr_ref, resolve_string_fn

    # "a\u00e9" = bytes 61 C3 A9: substr(2,2) -> C3 A9 = "\u00e9"
    assert substr_ref("a\u00e9", 2, 2) == "\u00e9"
    # split the multibyte char: substr(2,1) -> the lone C3 byte
    assert substr_ref("a\u00e9", 2, 1).encode("utf-8", "surrogateescape") \
        == b"\xc3"
    assert substr_ref("a\u00e9", -2, None) == "\u00e9"

    for fn, w, exp_bytes in [
        (("left", 2), "a\u00e9", b"a\xc3"),
        (("right", 2), "\u00e9a", b"\xa9a"),
        (("left", 0), "abc", b""),
        (("right", 99), "abc", b"abc"),
        (("substr", 2, 2), "abcd", b"bc"),
    ]:
        f = resolve_string_fn(fn)
        assert f(w).encode("utf-8", "surrogateescape") == exp_bytes, (fn, w)


def test_exports_cover_all_capi_headers():
    """Every C symbol include/*.h declares must export from the product .so
    (bk_exec.h ExecNode view, bk_arrow.h IPC writer, bk_cstore.h decoder —
    bkgpu.h is covered by the dedicated test above)."""
    _build_if_needed()
    lib = C.CDLL(LIB)
    missing = []
    for hdr, pat in [("bk_exec.h", r"\b(bkexec_\w+)\s*\("),
                     ("bk_arrow.h", r"\b(bk_arrow_\w+)\s*\("),
                     ("bk_cstore.h", r"\b(bk_cstore_\w+)\s*\(")]:
        path = os.path.join(REPO, "include", hdr)
        if not os.path.exists(path):
            continue
        with open(path) as f:
            syms = sorted(set(re.findall(pat, f.read())))
        for s in syms:
            if not hasattr(lib, s):
                missing.append(f"{hdr}:{s}")
    assert not missing, missing


def test_headers_compile_standalone_as_c():
    """Every include/*.h is a self-contained C header (the FFI boundary a
    cgo/JNI/ctypes consumer includes with no C++ toolchain)."""
    import glob
    import tempfile
    for h in sorted(glob.glob(os.path.join(REPO, "include", "*.h"))):
        with tempfile.NamedTemporaryFile("w", suffix=".c", delete=False) as f:
            f.write(f'#include "{h}"\n')
            path = f.name
        try:
            r = subprocess.run(["gcc", "-fsyntax-only", "-I", REPO, path],
                               capture_output=True, text=True)
            assert r.returncode == 0, (h, r.stderr[:400])
        finally:
            os.unlink(path)
