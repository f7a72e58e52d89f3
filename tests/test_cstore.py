# CSTORE per-column KV decode (bk_cstore.h / bkcstore.cpp): synthetic KV
# buffers written in the reference's exact on-disk byte format
# (mut_table_key.h:113-208 key encodings + table_record.cpp:362-470 values;
# the generator below restates them in Python independently), decoded by
# the C++ reader, compared against the source columns — including the
# merge-join semantics of table_iterator.cpp:525-597 (missing column key
# -> field default / NULL).
import ctypes as C
import os
import struct

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(REPO, "baikaldb_amd", "libbkgpu.so")

TYPE_INT64, TYPE_DOUBLE, TYPE_STRING = 6, 12, 13


class Field(C.Structure):
    _fields_ = [("field_id", C.c_int32), ("col_type", C.c_int32),
                ("kv", C.c_void_p), ("kv_len", C.c_int64),
                ("has_default", C.c_int32), ("def_i", C.c_int64),
                ("def_d", C.c_double), ("def_s", C.c_char_p)]


@pytest.fixture(scope="module")
def lib():
    if not os.path.exists(LIB):
        pytest.skip("libbkgpu.so not built")
    lib = C.CDLL(LIB)
    lib.bk_cstore_decode.restype = C.c_void_p
    lib.bk_cstore_decode.argtypes = [C.c_int64, C.c_int64, C.c_void_p,
                                     C.c_int64, C.c_int, C.POINTER(Field)]
    lib.bk_cstore_last_error.restype = C.c_char_p
    lib.bk_cstore_nrows.restype = C.c_int64
    lib.bk_cstore_nrows.argtypes = [C.c_void_p]
    lib.bk_cstore_col.restype = C.c_void_p
    lib.bk_cstore_col.argtypes = [C.c_void_p, C.c_int]
    lib.bk_cstore_valid.restype = C.POINTER(C.c_uint8)
    lib.bk_cstore_valid.argtypes = [C.c_void_p, C.c_int]
    lib.bk_cstore_nwords.restype = C.c_int64
    lib.bk_cstore_nwords.argtypes = [C.c_void_p, C.c_int]
    lib.bk_cstore_word.restype = C.c_char_p
    lib.bk_cstore_word.argtypes = [C.c_void_p, C.c_int, C.c_int64]
    lib.bk_cstore_free.argtypes = [C.c_void_p]
    return lib


# ---- independent Python restatement of the on-disk byte format ----
def enc_i64(v):
    """KeyEncoder::encode_i64 + to big-endian (key_encoder.h:128,54)."""
    return struct.pack(">Q", (int(v) ^ (1 << 63)) & (2 ** 64 - 1))


def enc_i32(v):
    return struct.pack(">I", (int(v) ^ (1 << 31)) & (2 ** 32 - 1))


def kv_stream(pairs):
    out = bytearray()
    for k, v in pairs:
        out += struct.pack("<I", len(k)) + k + struct.pack("<I", len(v)) + v
    return bytes(out)


def make_streams(region, table_id, pks, fields):
    """fields: list of (field_id, type, values, valid_or_None, default).
    Returns (primary_bytes, [column_bytes...]). Rows whose value is NULL or
    equals the default are NOT written (table_record.cpp:362-470)."""
    order = np.argsort(pks, kind="stable")
    prim = []
    row_prefix = enc_i64(region) + enc_i64(table_id)
    for i in order:
        prim.append((row_prefix + enc_i64(pks[i]), b""))
    cols = []
    for fid, typ, vals, valid, default in fields:
        prefix = enc_i64(region) + enc_i32(table_id) + enc_i32(fid)
        pairs = []
        for i in order:
            if valid is not None and not valid[i]:
                continue                       # NULL: not stored
            v = vals[i]
            if default is not None and v == default:
                continue                       # default: not stored
            if typ == TYPE_DOUBLE:
                val = struct.pack("<d", float(v))
            elif typ == TYPE_STRING:
                val = v.encode() if isinstance(v, str) else bytes(v)
            else:
                val = struct.pack("<q", int(v))
            pairs.append((prefix + enc_i64(pks[i]), val))
        cols.append(kv_stream(pairs))
    return kv_stream(prim), cols


def decode(lib, region, table_id, prim, cols, fdefs):
    farr = (Field * len(fdefs))()
    keep = []
    for i, (fid, typ, default) in enumerate(fdefs):
        buf = C.create_string_buffer(cols[i], len(cols[i]))
        keep.append(buf)
        farr[i].field_id = fid
        farr[i].col_type = typ
        farr[i].kv = C.cast(buf, C.c_void_p)
        farr[i].kv_len = len(cols[i])
        if default is None:
            farr[i].has_default = 0
        else:
            farr[i].has_default = 1
            if typ == TYPE_DOUBLE:
                farr[i].def_d = float(default)
            elif typ == TYPE_STRING:
                farr[i].def_s = default.encode()
            else:
                farr[i].def_i = int(default)
    pbuf = C.create_string_buffer(prim, len(prim))
    h = lib.bk_cstore_decode(region, table_id, C.cast(pbuf, C.c_void_p),
                             len(prim), len(fdefs), farr)
    assert h, lib.bk_cstore_last_error().decode()
    return h


def col_np(lib, h, col, typ, n):
    p = lib.bk_cstore_col(h, col)
    dt = {TYPE_INT64: np.int64, TYPE_DOUBLE: np.float64,
          TYPE_STRING: np.int32}[typ]
    return np.ctypeslib.as_array(
        C.cast(p, C.POINTER({TYPE_INT64: C.c_int64, TYPE_DOUBLE: C.c_double,
                             TYPE_STRING: C.c_int32}[typ])), (n,)).copy()


def test_decode_int64_double_roundtrip(lib):
    rng = np.random.default_rng(5)
    n = 5000
    pks = rng.permutation(n).astype(np.int64) * 3 - n
    a = rng.integers(-(2 ** 62), 2 ** 62, n)
    d = rng.standard_normal(n)
    prim, cols = make_streams(7, 42, pks, [
        (10, TYPE_INT64, a, None, None),
        (11, TYPE_DOUBLE, d, None, None)])
    h = decode(lib, 7, 42, prim, cols,
               [(10, TYPE_INT64, None), (11, TYPE_DOUBLE, None)])
    assert lib.bk_cstore_nrows(h) == n
    order = np.argsort(pks, kind="stable")
    assert np.array_equal(col_np(lib, h, 0, TYPE_INT64, n), pks[order])
    assert np.array_equal(col_np(lib, h, 1, TYPE_INT64, n), a[order])
    assert np.array_equal(col_np(lib, h, 2, TYPE_DOUBLE, n), d[order])
    assert not lib.bk_cstore_valid(h, 1)
    lib.bk_cstore_free(h)


def test_missing_keys_null_and_default(lib):
    """The write side skips NULL and default-valued fields; the merge-join
    must synthesize them back (get_column's default_expr_value branch)."""
    n = 1000
    rng = np.random.default_rng(6)
    pks = np.arange(n, dtype=np.int64)
    a = rng.integers(0, 50, n)            # many rows == default 7
    av = (rng.random(n) > 0.3).astype(np.uint8)
    d = rng.standard_normal(n)
    dv = (rng.random(n) > 0.5).astype(np.uint8)
    prim, cols = make_streams(1, 2, pks, [
        (3, TYPE_INT64, a, av, 7),
        (4, TYPE_DOUBLE, d, dv, None)])
    h = decode(lib, 1, 2, prim, cols,
               [(3, TYPE_INT64, 7), (4, TYPE_DOUBLE, None)])
    got_a = col_np(lib, h, 1, TYPE_INT64, n)
    va = lib.bk_cstore_valid(h, 1)
    # field 3 has a default: NULL rows were not written, so they read back
    # as the DEFAULT (the reference cannot distinguish them — exactly its
    # semantics when a default exists)
    for r in range(n):
        expect = a[r] if av[r] else 7
        assert got_a[r] == expect, r
    assert not va     # every row lands on a value -> no validity array
    got_d = col_np(lib, h, 2, TYPE_DOUBLE, n)
    vd = lib.bk_cstore_valid(h, 4 - 3 + 1)
    assert vd
    for r in range(n):
        if dv[r]:
            assert vd[r] == 1 and got_d[r] == d[r]
        else:
            assert vd[r] == 0
    lib.bk_cstore_free(h)


def test_string_fields_dict(lib):
    n = 500
    rng = np.random.default_rng(8)
    pks = np.arange(n, dtype=np.int64) * 2
    words = ["cherry", "apple", "banana", "fig"]
    vals = [words[i] for i in rng.integers(0, 4, n)]
    sv = (rng.random(n) > 0.2).astype(np.uint8)
    prim, cols = make_streams(9, 33, pks, [(5, TYPE_STRING, vals, sv, None)])
    h = decode(lib, 9, 33, prim, cols, [(5, TYPE_STRING, None)])
    nw = lib.bk_cstore_nwords(h, 1)
    got_words = [lib.bk_cstore_word(h, 1, c).decode() for c in range(nw)]
    assert got_words == sorted(set(w for w, v in zip(vals, sv) if v))
    codes = col_np(lib, h, 1, TYPE_STRING, n)
    va = lib.bk_cstore_valid(h, 1)
    for r in range(n):
        if sv[r]:
            assert got_words[codes[r]] == vals[r]
        else:
            assert va[r] == 0
    lib.bk_cstore_free(h)


def test_empty_and_malformed(lib):
    h = decode(lib, 1, 1, b"", [b""], [(2, TYPE_INT64, None)])
    assert lib.bk_cstore_nrows(h) == 0
    lib.bk_cstore_free(h)
    bad = C.create_string_buffer(b"\x10\x00\x00\x00oops", 8)
    assert not lib.bk_cstore_decode(1, 1, C.cast(bad, C.c_void_p), 8, 0, None)


@pytest.mark.gpu
def test_cstore_to_hbm_group_by(lib):
    """End-to-end: KV-format buffers -> bkgpu_table_from_cstore (HBM) ->
    GROUP BY, identical to the same query on a directly-uploaded table
    (VERDICT round-1 'done' criterion for §8f.1's second half)."""
    import torch
    if torch.cuda.is_available():
        torch.cuda.init()
    from baikaldb_amd import GpuEngine, QueryPlan
    from baikaldb_amd.engine import GpuTable
    lib.bkgpu_table_from_cstore.restype = C.c_void_p
    lib.bkgpu_table_from_cstore.argtypes = [C.c_void_p]
    rng = np.random.default_rng(11)
    n = 200_000
    pks = np.arange(n, dtype=np.int64)
    g = rng.integers(0, 500, n)
    v = rng.integers(0, 1000, n)
    d = rng.standard_normal(n)
    prim, cols = make_streams(3, 77, pks, [
        (10, TYPE_INT64, g, None, None),
        (11, TYPE_INT64, v, None, None),
        (12, TYPE_DOUBLE, d, None, None)])
    h = decode(lib, 3, 77, prim, cols,
               [(10, TYPE_INT64, None), (11, TYPE_INT64, None),
                (12, TYPE_DOUBLE, None)])
    eng = GpuEngine()
    th = lib.bkgpu_table_from_cstore(h)
    assert th
    types = [TYPE_INT64, TYPE_INT64, TYPE_INT64, TYPE_DOUBLE]
    t1 = GpuTable(eng, th, types, n)
    # reference table: direct upload of the same columns
    t2 = eng.create_table([(TYPE_INT64, 0, 0, 0, 0)] * 3
                          + [(TYPE_DOUBLE, 0, 0, 0, 0)], n)
    for i, arr in enumerate([pks, g.astype(np.int64), v.astype(np.int64),
                             d]):
        eng.upload(t2, i, np.ascontiguousarray(arr))
    plan = QueryPlan(types, conjuncts=[(2, "<", 800)], group=[1],
                     aggs=[("count_star", -1), ("sum", 2), ("avg", 3),
                           ("min", 0)])
    r1 = eng.filter_agg(t1, plan, expected_groups=1 << 10)
    r2 = eng.filter_agg(t2, plan, expected_groups=1 << 10)
    g1, g2 = r1.fetch(sorted=True), r2.fetch(sorted=True)
    r1.free()
    r2.free()
    t1.free()
    t2.free()
    lib.bk_cstore_free(h)
    assert g1["ngroups"] == g2["ngroups"]
    assert np.array_equal(g1["enc"], g2["enc"])
    assert np.array_equal(g1["agg_i"], g2["agg_i"])
    # decode is bit-exact; the AVG(double) residual is only the f64 atomic
    # reduction ORDER differing between the two GPU runs
    np.testing.assert_allclose(g1["agg_d"], g2["agg_d"], rtol=1e-10,
                               atol=1e-12)
