# Derived expression columns (bkgpu_table_derive_prog): the engine-side
# projection of an arbitrary-depth expression tree — the reference evaluates
# ScalarFnCall trees per row wherever an expr slot appears
# (scalar_fn_call.cpp:194-225); here the compiled postfix program runs once
# over the table and the result is a first-class column, so WINDOW fn
# inputs, ORDER BY keys and SELECT out_cols become expression-valued with
# zero kernel changes. Parity: the derived column must equal a host-side
# numpy evaluation uploaded as a plain column (bit-exact; int64 wraps,
# mixed domains in f64, either-NULL => NULL).
import numpy as np
import pytest

pytestmark = pytest.mark.gpu

TYPE_INT64, TYPE_DOUBLE = 6, 12
SEED = 0xD54E


@pytest.fixture(scope="module")
def eng():
    import torch
    if torch.cuda.is_available():
        torch.cuda.init()
    from baikaldb_amd import GpuEngine
    return GpuEngine()


def _upload_table(eng, cols, valids):
    """cols: list of np arrays (int64 or f64); valids: list of uint8|None."""
    specs = []
    for c, v in zip(cols, valids):
        t = TYPE_DOUBLE if c.dtype == np.float64 else TYPE_INT64
        specs.append((t, 0, 0, 1 << 31, 1 if v is not None else 0))
    t = eng.create_table(specs, len(cols[0]))
    for i, (c, v) in enumerate(zip(cols, valids)):
        eng.upload(t, i, c, v)
    return t


def test_derived_equals_host_eval_int64_wrap(eng):
    """(a*b + 1000003) in the int64 domain: wraparound must match the
    reference's int64 arithmetic (operators.cpp multiplies wrap)."""
    from baikaldb_amd import QueryPlan
    rng = np.random.default_rng(SEED)
    n = 150_000
    a = rng.integers(-(1 << 62), 1 << 62, n, dtype=np.int64)
    b = rng.integers(-(1 << 20), 1 << 20, n, dtype=np.int64)
    host = ((a.astype(np.uint64) * b.astype(np.uint64))
            + np.uint64(1000003)).astype(np.int64)
    t = _upload_table(eng, [a, b, host], [None, None, None])
    try:
        nc = eng.derive_expr(t, ("add", ("mul", 0, 1), ("liti", 1000003)))
        assert nc == 3
        # derived col 3 must aggregate exactly like the uploaded host col 2
        plan = QueryPlan(t.col_types,
                         aggs=[("min", 2), ("max", 2), ("sum", 2),
                               ("min", 3), ("max", 3), ("sum", 3)])
        res = eng.filter_agg(t, plan)
        got = res.fetch()
        res.free()
        for k in range(3):
            assert got["agg_i"][k][0] == got["agg_i"][k + 3][0]
        assert got["agg_i"][0][0] == host.min()
        assert got["agg_i"][2][0] == int(host.astype(np.uint64).sum()
                                         .astype(np.int64))
    finally:
        t.free()


def test_derived_double_domain_and_nulls(eng):
    """Mixed int/double promotes to f64 (arg-cast rule); NULL in either
    input nulls the result — COUNT over the derived column must see it."""
    from baikaldb_amd import QueryPlan
    rng = np.random.default_rng(SEED + 1)
    n = 120_000
    a = rng.integers(0, 1000, n, dtype=np.int64)
    d = rng.standard_normal(n)
    va = (rng.random(n) > 0.2).astype(np.uint8)
    t = _upload_table(eng, [a, d], [va, None])
    try:
        nc = eng.derive_expr(t, ("mul", ("add", 0, 1.5), 1))
        assert t.col_types[nc] == TYPE_DOUBLE
        plan = QueryPlan(t.col_types,
                         aggs=[("count", nc), ("sum", nc), ("min", nc)])
        res = eng.filter_agg(t, plan)
        got = res.fetch()
        res.free()
    finally:
        t.free()
    host = (a.astype(np.float64) + 1.5) * d
    m = va.astype(bool)
    assert got["agg_i"][0][0] == int(m.sum())
    assert abs(got["agg_d"][1][0] - host[m].sum()) <= 1e-9 * (
        np.abs(host[m]).sum() + 1)
    assert got["agg_d"][2][0] == host[m].min()


def test_order_by_expression(eng):
    """ORDER BY (a-b)*(a+b) LIMIT k through sort_topk on the derived
    column: rowids equal numpy's stable argsort of the host expression."""
    rng = np.random.default_rng(SEED + 2)
    n = 90_000
    a = rng.integers(-(1 << 20), 1 << 20, n, dtype=np.int64)
    b = rng.integers(-(1 << 20), 1 << 20, n, dtype=np.int64)
    t = _upload_table(eng, [a, b], [None, None])
    try:
        nc = eng.derive_expr(t, ("mul", ("sub", 0, 1), ("add", 0, 1)))
        rowids = eng.sort_topk(t, [(nc, 1, 1)], 40)
    finally:
        t.free()
    host = (a - b) * (a + b)
    expect = np.lexsort((np.arange(n), host))[:40]
    assert np.array_equal(np.asarray(rowids), expect)


def test_window_expression_input(eng):
    """SUM(a*b+c) OVER (PARTITION BY p ORDER BY o): the window over the
    derived column equals the window over the identical host-evaluated
    uploaded column — expression inputs for WINDOW fns (the §6 open item)."""
    rng = np.random.default_rng(SEED + 3)
    n = 100_000
    p = rng.integers(0, 97, n, dtype=np.int64)
    o = rng.integers(0, 50, n, dtype=np.int64)
    a = rng.integers(-500, 500, n, dtype=np.int64)
    b = rng.integers(-40, 40, n, dtype=np.int64)
    c = rng.integers(0, 10_000, n, dtype=np.int64)
    host = a * b + c
    t = _upload_table(eng, [p, o, a, b, c, host], [None] * 6)
    try:
        nc = eng.derive_expr(t, ("add", ("mul", 2, 3), 4))
        fns = [("sum", nc), ("min", nc), ("lag", nc, 2), ("max", 5)]
        got = eng.window(t, fns, part_col=0, order=[(1, 1, 1)])
        ref = eng.window(t, [("sum", 5), ("min", 5), ("lag", 5, 2),
                             ("max", nc)], part_col=0, order=[(1, 1, 1)])
    finally:
        t.free()
    assert got["n"] == ref["n"] == n
    assert np.array_equal(got["rowids"], ref["rowids"])
    assert np.array_equal(got["out_i"], ref["out_i"])
    assert np.array_equal(got["out_null"], ref["out_null"])


def test_derive_prog_rejects_bad_programs(eng):
    rng = np.random.default_rng(SEED + 4)
    a = rng.integers(0, 10, 1000, dtype=np.int64)
    t = _upload_table(eng, [a], [None])
    try:
        with pytest.raises(Exception):
            eng.derive_expr(t, ("add", 0, 7))  # col 7 out of range
        with pytest.raises(ValueError):
            eng.derive_expr(t, ("bogus", 0, 1))
    finally:
        t.free()
