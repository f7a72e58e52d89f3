# Cold columnar ingestion: the from-scratch parquet reader
# (baikaldb_amd/csrc/bkparquet.cpp) vs files written by pyarrow — the same
# arrow writer the reference's cold path uses (parquet_writer.h:119).
import ctypes as C
import os

import numpy as np
import pytest

pa = pytest.importorskip("pyarrow")
pq = pytest.importorskip("pyarrow.parquet")

_LIB = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                    "baikaldb_amd", "libbkgpu.so")


@pytest.fixture(scope="module")
def lib():
    lib = C.CDLL(_LIB)
    lib.bkparquet_open.restype = C.c_void_p
    lib.bkparquet_open.argtypes = [C.c_char_p]
    lib.bkparquet_num_rows.restype = C.c_int64
    lib.bkparquet_num_rows.argtypes = [C.c_void_p]
    lib.bkparquet_num_cols.restype = C.c_int
    lib.bkparquet_num_cols.argtypes = [C.c_void_p]
    lib.bkparquet_col_type.restype = C.c_int
    lib.bkparquet_col_type.argtypes = [C.c_void_p, C.c_int]
    lib.bkparquet_col_nullable.restype = C.c_int
    lib.bkparquet_col_nullable.argtypes = [C.c_void_p, C.c_int]
    lib.bkparquet_col_name.argtypes = [C.c_void_p, C.c_int, C.c_char_p, C.c_int]
    lib.bkparquet_read_column.restype = C.c_int64
    lib.bkparquet_read_column.argtypes = [C.c_void_p, C.c_int, C.c_void_p,
                                          C.POINTER(C.c_uint8)]
    lib.bkparquet_last_error.restype = C.c_char_p
    lib.bkparquet_close.argtypes = [C.c_void_p]
    lib.bkgpu_table_from_parquet.restype = C.c_void_p
    lib.bkgpu_table_from_parquet.argtypes = [C.c_char_p]
    lib.bkparquet_read_string_column.restype = C.c_int64
    lib.bkparquet_read_string_column.argtypes = [
        C.c_void_p, C.c_int, C.POINTER(C.c_int32), C.POINTER(C.c_uint8),
        C.POINTER(C.c_void_p), C.POINTER(C.c_int64)]
    lib.bkparquet_dict_word.argtypes = [C.c_void_p, C.c_int64, C.c_char_p,
                                        C.c_int]
    lib.bkparquet_dict_free.argtypes = [C.c_void_p]
    lib.bkgpu_table_dict_word.argtypes = [C.c_void_p, C.c_int, C.c_int64,
                                          C.c_char_p, C.c_int]
    return lib


def write_file(path, cols, row_group_size=None, data_page_size=None):
    """cols: list of (name, numpy array, mask-or-None); mask True = NULL."""
    arrays, names = [], []
    for name, arr, mask in cols:
        arrays.append(pa.array(arr, mask=mask))
        names.append(name)
    tab = pa.table(dict(zip(names, arrays)))
    kw = dict(compression=None, use_dictionary=False, version="2.6",
              data_page_version="1.0", write_statistics=False)
    if row_group_size:
        kw["row_group_size"] = row_group_size
    if data_page_size:
        kw["data_page_size"] = data_page_size
    pq.write_table(tab, path, **kw)


def read_col(lib, r, col, n, nullable):
    out = np.empty(n, dtype=np.int64)
    valid = np.empty(n, dtype=np.uint8) if nullable else None
    vp = valid.ctypes.data_as(C.POINTER(C.c_uint8)) if nullable else None
    got = lib.bkparquet_read_column(r, col, out.ctypes.data_as(C.c_void_p), vp)
    assert got == n, lib.bkparquet_last_error()
    return out, valid


def test_read_int64_double_nulls(lib, tmp_path):
    rng = np.random.default_rng(7)
    n = 100_000
    a = rng.integers(-(1 << 60), 1 << 60, n, dtype=np.int64)
    b = rng.standard_normal(n)
    bmask = rng.random(n) < 0.2
    path = str(tmp_path / "t.parquet")
    # multiple row groups AND multiple pages per chunk
    write_file(path, [("a", a, None), ("b", b, bmask)],
               row_group_size=30_000, data_page_size=4096)
    r = lib.bkparquet_open(path.encode())
    assert r, lib.bkparquet_last_error()
    try:
        assert lib.bkparquet_num_rows(r) == n
        assert lib.bkparquet_num_cols(r) == 2
        assert lib.bkparquet_col_type(r, 0) == 6      # BK_INT64
        assert lib.bkparquet_col_type(r, 1) == 12     # BK_DOUBLE
        assert lib.bkparquet_col_nullable(r, 0) == 0 or True  # pyarrow may mark optional
        va, _ = read_col(lib, r, 0, n, lib.bkparquet_col_nullable(r, 0))
        assert np.array_equal(va, a)
        vb, validb = read_col(lib, r, 1, n, True)
        assert np.array_equal(validb == 0, bmask)
        assert np.array_equal(vb.view(np.float64)[~bmask], b[~bmask])
    finally:
        lib.bkparquet_close(r)


def test_reject_compressed(lib, tmp_path):
    path = str(tmp_path / "c.parquet")
    tab = pa.table({"a": pa.array(np.arange(1000, dtype=np.int64))})
    pq.write_table(tab, path, compression="snappy", use_dictionary=False)
    r = lib.bkparquet_open(path.encode())
    assert r
    out = np.empty(1000, dtype=np.int64)
    got = lib.bkparquet_read_column(r, 0, out.ctypes.data_as(C.c_void_p), None)
    assert got < 0
    assert b"unsupported" in lib.bkparquet_last_error() or \
           b"compressed" in lib.bkparquet_last_error()
    lib.bkparquet_close(r)


def test_dictionary_encoded_int64(lib, tmp_path):
    """pyarrow's default dict encoding: dict page + RLE_DICTIONARY indices."""
    path = str(tmp_path / "d.parquet")
    rng = np.random.default_rng(3)
    a = rng.integers(0, 37, 50_000, dtype=np.int64) * 1000 - 5000
    tab = pa.table({"a": pa.array(a)})
    pq.write_table(tab, path, compression=None, use_dictionary=True,
                   data_page_version="1.0", write_statistics=False)
    r = lib.bkparquet_open(path.encode())
    assert r
    out = np.empty(len(a), dtype=np.int64)
    got = lib.bkparquet_read_column(r, 0, out.ctypes.data_as(C.c_void_p), None)
    assert got == len(a), lib.bkparquet_last_error()
    assert np.array_equal(out, a)
    lib.bkparquet_close(r)


def test_string_column_order_preserving_codes(lib, tmp_path):
    """BYTE_ARRAY -> dict codes whose order == byte order of the words."""
    path = str(tmp_path / "s.parquet")
    rng = np.random.default_rng(5)
    words = [f"city_{i:04d}" for i in rng.integers(0, 300, 40_000)]
    mask = rng.random(40_000) < 0.1
    tab = pa.table({"w": pa.array(words, mask=mask)})
    pq.write_table(tab, path, compression=None, use_dictionary=True,
                   data_page_version="1.0", write_statistics=False)
    r = lib.bkparquet_open(path.encode())
    assert r
    assert lib.bkparquet_col_type(r, 0) == 13    # BK_STRING
    codes = np.empty(40_000, dtype=np.int32)
    valid = np.empty(40_000, dtype=np.uint8)
    dh = C.c_void_p()
    dn = C.c_int64()
    got = lib.bkparquet_read_string_column(
        r, 0, codes.ctypes.data_as(C.POINTER(C.c_int32)),
        valid.ctypes.data_as(C.POINTER(C.c_uint8)), C.byref(dh), C.byref(dn))
    assert got == 40_000, lib.bkparquet_last_error()
    uniq = sorted(set(w for w, m in zip(words, mask) if not m))
    assert dn.value == len(uniq)
    buf = C.create_string_buffer(64)
    for c in (0, 1, dn.value // 2, dn.value - 1):
        lib.bkparquet_dict_word(dh, c, buf, 64)
        assert buf.value.decode() == uniq[c]
    # codes match the sorted-unique rank of each word
    rank = {w: i for i, w in enumerate(uniq)}
    exp = np.array([0 if m else rank[w] for w, m in zip(words, mask)],
                   dtype=np.int32)
    sel = valid != 0
    assert np.array_equal(codes[sel], exp[sel])
    assert np.array_equal(sel, ~mask)
    lib.bkparquet_dict_free(dh)
    lib.bkparquet_close(r)


@pytest.mark.gpu
def test_parquet_to_gpu_agg_parity(lib, tmp_path):
    """parquet file -> HBM table -> filter+GROUP BY == oracle over the
    same host arrays (stored-region end-to-end)."""
    import torch
    if torch.cuda.is_available():
        torch.cuda.init()
    from baikaldb_amd import GpuEngine, QueryPlan
    from baikaldb_amd.engine import AggResult
    from oracle import Oracle
    from oracle.bindings import make_query

    rng = np.random.default_rng(11)
    n = 200_000
    g = rng.integers(0, 50, n, dtype=np.int64)
    v = rng.integers(-1000, 1000, n, dtype=np.int64)
    d = rng.standard_normal(n)
    dmask = rng.random(n) < 0.15
    path = str(tmp_path / "r.parquet")
    write_file(path, [("g", g, None), ("v", v, None), ("d", d, dmask)],
               row_group_size=64_000)

    eng = GpuEngine()
    h = lib.bkgpu_table_from_parquet(path.encode())
    assert h, lib.bkparquet_last_error()
    from baikaldb_amd.engine import GpuTable
    t = GpuTable(eng, h, [6, 6, 12], n)
    plan = QueryPlan(t.col_types, conjuncts=[(1, ">", 0)], group=[0],
                     aggs=[("count_star", -1), ("sum", 1), ("avg", 2)])
    res = eng.filter_agg(t, plan, expected_groups=1 << 10)
    got = res.fetch(sorted=True)
    res.free()
    t.free()

    orc = Oracle()
    valid_d = (~dmask).astype(np.uint8)
    q = make_query([(1, 2, 6, 0)], [0], [(0, -1), (2, 1), (3, 2)], [6, 6, 12])
    exp = orc.filter_agg([g, v, d], [None, None, valid_d], [6, 6, 12], q,
                         nthreads=4)
    assert got["ngroups"] == exp["ngroups"]
    assert got["rows_passed"] == exp["rows_passed"]
    assert np.array_equal(got["enc"], exp["enc"])
    assert np.array_equal(got["agg_i"][0], exp["agg_i"][0])
    assert np.array_equal(got["agg_i"][1], exp["agg_i"][1])
    err = np.abs(got["agg_d"][2] - exp["agg_d"][2])
    assert np.all(err <= 1e-10 * (np.abs(exp["agg_d"][2]) + 1))


@pytest.mark.gpu
def test_parquet_string_group_minmax(lib, tmp_path):
    """Ingested VARCHAR column: GROUP BY string, MIN/MAX over another string
    column, dict words materialized via bkgpu_table_dict_word."""
    import torch
    if torch.cuda.is_available():
        torch.cuda.init()
    from baikaldb_amd import GpuEngine, QueryPlan
    from baikaldb_amd.engine import GpuTable

    rng = np.random.default_rng(21)
    n = 120_000
    g = [f"grp_{i:02d}" for i in rng.integers(0, 12, n)]
    w = [f"val_{i:05d}" for i in rng.integers(0, 5000, n)]
    v = rng.integers(0, 100, n, dtype=np.int64)
    path = str(tmp_path / "sv.parquet")
    tab = pa.table({"g": pa.array(g), "w": pa.array(w),
                    "v": pa.array(v)})
    pq.write_table(tab, path, compression=None, use_dictionary=True,
                   data_page_version="1.0", write_statistics=False)

    eng = GpuEngine()
    h = lib.bkgpu_table_from_parquet(path.encode())
    assert h, lib.bkparquet_last_error()
    t = GpuTable(eng, h, [13, 13, 6], n)
    plan = QueryPlan(t.col_types, conjuncts=[(2, "<", 80)], group=[0],
                     aggs=[("count_star", -1), ("min", 1), ("max", 1)])
    res = eng.filter_agg(t, plan, expected_groups=1 << 8)
    got = res.fetch(sorted=True)
    res.free()

    # brute force over the host data
    sel = v < 80
    groups = sorted(set(np.array(g)[sel]))
    assert got["ngroups"] == len(groups)
    buf = C.create_string_buffer(64)
    garr, warr = np.array(g), np.array(w)
    for r_i, gname in enumerate(groups):
        # group key code -> word
        code = int(got["enc"][r_i][0])
        lib.bkgpu_table_dict_word(C.c_void_p(t.handle), 0, code, buf, 64)
        assert buf.value.decode() == gname
        rows = sel & (garr == gname)
        assert got["agg_i"][0][r_i] == rows.sum()
        lib.bkgpu_table_dict_word(C.c_void_p(t.handle), 1,
                                  int(got["agg_i"][1][r_i]), buf, 64)
        assert buf.value.decode() == min(warr[rows])
        lib.bkgpu_table_dict_word(C.c_void_p(t.handle), 1,
                                  int(got["agg_i"][2][r_i]), buf, 64)
        assert buf.value.decode() == max(warr[rows])
    t.free()


@pytest.mark.gpu
def test_cpp_embedding_example(tmp_path):
    """examples/region_select: the pure-C++ driver loop (no Python in the
    query path) over an ingested parquet file — Region::select_normal's
    loop re-created by an embedder."""
    import subprocess
    exe = os.path.join(os.path.dirname(_LIB), "..", "examples",
                       "region_select")
    if not os.path.exists(exe):
        pytest.skip("example binary not built")
    rng = np.random.default_rng(9)
    n = 50_000
    g = [f"grp_{i:02d}" for i in rng.integers(0, 6, n)]
    w = [f"val_{i:04d}" for i in rng.integers(0, 500, n)]
    v = rng.integers(0, 100, n, dtype=np.int64)
    path = str(tmp_path / "e.parquet")
    tab = pa.table({"g": pa.array(g), "w": pa.array(w), "v": pa.array(v)})
    pq.write_table(tab, path, compression=None, use_dictionary=True,
                   data_page_version="1.0", write_statistics=False)
    arrow_out = str(tmp_path / "result.arrow")
    out = subprocess.run([exe, path, arrow_out], capture_output=True,
                         text=True, timeout=120)
    assert out.returncode == 0, out.stderr
    assert f"ingested {n} rows x 3 cols" in out.stdout
    # verify one group row against a brute-force recompute
    garr, warr = np.array(g), np.array(w)
    sel = v < 80
    g0 = sorted(set(garr[sel]))[0]
    rows = sel & (garr == g0)
    expect = f"row 0: {g0} {rows.sum()} {v[rows].sum()} {min(warr[rows])}"
    assert expect in out.stdout, f"wanted {expect!r} in:\n{out.stdout[:2000]}"
    assert "top-5 by (v, g):" in out.stdout
    assert "row_number=1" in out.stdout
    # the example also serializes the GROUP BY result as an Arrow IPC
    # stream (the region.cpp:2905-2918 response bytes, written by the
    # from-scratch serializer) — pyarrow must round-trip it and agree
    # with a pandas-free brute-force recompute
    assert "arrow ipc:" in out.stdout
    t = pa.ipc.open_stream(open(arrow_out, "rb").read()).read_all()
    t.validate(full=True)
    got = {t.column(0)[i].as_py(): (t.column(1)[i].as_py(),
                                    t.column(2)[i].as_py(),
                                    t.column(3)[i].as_py())
           for i in range(t.num_rows)}
    for grp in set(garr[sel]):
        rows_g = sel & (garr == grp)
        assert got[grp] == (int(rows_g.sum()), int(v[rows_g].sum()),
                            min(warr[rows_g])), grp


@pytest.mark.gpu
def test_parquet_to_window(lib, tmp_path):
    """Ingested parquet table through the window path (rank over a stored
    VARCHAR partition)."""
    import torch
    if torch.cuda.is_available():
        torch.cuda.init()
    from baikaldb_amd import GpuEngine
    from baikaldb_amd.engine import GpuTable
    rng = np.random.default_rng(31)
    n = 30_000
    g = [f"p{i%9}" for i in range(n)]
    v = rng.integers(0, 1000, n, dtype=np.int64)
    path = str(tmp_path / "w.parquet")
    tab = pa.table({"g": pa.array(g), "v": pa.array(v)})
    pq.write_table(tab, path, compression=None, use_dictionary=True,
                   data_page_version="1.0", write_statistics=False)
    eng = GpuEngine()
    h = lib.bkgpu_table_from_parquet(path.encode())
    assert h, lib.bkparquet_last_error()
    t = GpuTable(eng, h, [13, 6], n)
    got = eng.window(t, [("row_number", -1), ("max", 1)], part_col=0,
                     order=[(1, 1, 1)])
    t.free()
    assert got["n"] == n
    # per partition: row_number restarts; max = partition max broadcast
    garr = np.array(g)
    r = got["rowids"]
    for p in range(9):
        sel = garr[r] == f"p{p}"
        assert np.array_equal(got["out_i"][0][sel],
                              np.arange(1, sel.sum() + 1))
        assert np.all(got["out_i"][1][sel] == v[garr == f"p{p}"].max())


@pytest.mark.parametrize("case", range(10))
def test_parquet_reader_fuzz(lib, tmp_path, case):
    """Randomized schemas x writer layouts (row-group/page sizes, dict
    on/off, null fractions) vs pyarrow: every value and validity bit must
    round-trip through the from-scratch reader."""
    import random
    rng = random.Random(5150 + case)
    nrng = np.random.default_rng(5150 + case)
    n = rng.choice([1, 7, 1000, 57_331])
    ncols = rng.randint(1, 5)
    cols = []
    for ci in range(ncols):
        kind = rng.choice(["i64", "f64", "str"])
        mask = (nrng.random(n) < rng.choice([0.0, 0.1, 0.9])) \
            if rng.random() < 0.5 else None
        if kind == "i64":
            arr = nrng.integers(-(1 << 62), 1 << 62, n, dtype=np.int64)
        elif kind == "f64":
            arr = nrng.standard_normal(n)
        else:
            arr = [f"s{v:06d}" for v in nrng.integers(0, rng.choice([2, 500]),
                                                      n)]
        cols.append((f"c{ci}", kind, arr, mask))
    path = str(tmp_path / f"f{case}.parquet")
    tab = pa.table({nm: pa.array(arr, mask=mask)
                    for nm, _, arr, mask in cols})
    kw = dict(compression=None,
              use_dictionary=rng.random() < 0.5,
              data_page_version="1.0",
              write_statistics=rng.random() < 0.5)
    if rng.random() < 0.5:
        kw["row_group_size"] = rng.choice([1, 100, 10_000])
    if rng.random() < 0.5:
        kw["data_page_size"] = rng.choice([512, 4096])
    pq.write_table(tab, path, **kw)

    r = lib.bkparquet_open(path.encode())
    assert r, lib.bkparquet_last_error()
    try:
        assert lib.bkparquet_num_rows(r) == n
        for ci, (nm, kind, arr, mask) in enumerate(cols):
            nullable = lib.bkparquet_col_nullable(r, ci)
            valid = np.empty(n, dtype=np.uint8) if nullable else None
            vp = valid.ctypes.data_as(C.POINTER(C.c_uint8)) if nullable \
                else None
            if kind == "str":
                assert lib.bkparquet_col_type(r, ci) == 13
                codes = np.empty(n, dtype=np.int32)
                dh = C.c_void_p()
                dn = C.c_int64()
                got = lib.bkparquet_read_string_column(
                    r, ci, codes.ctypes.data_as(C.POINTER(C.c_int32)), vp,
                    C.byref(dh), C.byref(dn))
                assert got == n, (case, ci, lib.bkparquet_last_error())
                uniq = sorted(set(a for a, m in
                                  zip(arr, mask if mask is not None
                                      else [False] * n) if not m))
                assert dn.value == len(uniq), (case, ci)
                rank = {w: i for i, w in enumerate(uniq)}
                for i in range(n):
                    isnull = mask is not None and mask[i]
                    if nullable:
                        assert (valid[i] == 0) == isnull, (case, ci, i)
                    if not isnull:
                        assert codes[i] == rank[arr[i]], (case, ci, i)
                lib.bkparquet_dict_free(dh)
            else:
                out = np.empty(n, dtype=np.int64)
                got = lib.bkparquet_read_column(
                    r, ci, out.ctypes.data_as(C.c_void_p), vp)
                assert got == n, (case, ci, lib.bkparquet_last_error())
                vals = out.view(np.float64) if kind == "f64" else out
                sel = (~mask) if mask is not None else np.ones(n, bool)
                if nullable:
                    assert np.array_equal(valid != 0, sel), (case, ci)
                assert np.array_equal(np.asarray(vals)[sel],
                                      np.asarray(arr)[sel]), (case, ci)
    finally:
        lib.bkparquet_close(r)
