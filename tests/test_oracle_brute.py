# Oracle-vs-numpy-brute randomized fuzz over the FULL query surface the
# fuzz generator produces (plain/arith/OR/fn conjuncts, fn group keys,
# plain + expression aggregate inputs, nullable columns). The brute here is
# an INDEPENDENT restatement in numpy — it shares only the deterministic
# input generator with the oracle, so a bug common to oracle and GPU cannot
# hide behind their mutual parity (SURVEY §8c pinning).
import ctypes as C

import numpy as np
import pytest

import tests.test_gpu_fuzz as fz
from oracle import BkColSpec

TYPE_INT64, TYPE_DOUBLE, TYPE_STRING, TYPE_DATETIME = 6, 12, 13, 14
OPS = {"=": 0, "!=": 1, ">": 2, ">=": 3, "<": 4, "<=": 5}
AGGMAP = {"count_star": 0, "count": 1, "sum": 2, "avg": 3, "min": 4, "max": 5}


def np_scalar_fn(name, v):
    dt = v.astype(np.uint64)
    if name == "year":
        return ((dt >> np.uint64(46)) & np.uint64(0x1FFFF)) // np.uint64(13)
    if name == "month":
        return ((dt >> np.uint64(46)) & np.uint64(0x1FFFF)) % np.uint64(13)
    if name == "day":
        return (dt >> np.uint64(41)) & np.uint64(0x1F)
    if name == "hour":
        return (dt >> np.uint64(36)) & np.uint64(0x1F)
    if name == "minute":
        return (dt >> np.uint64(30)) & np.uint64(0x3F)
    return (dt >> np.uint64(24)) & np.uint64(0x3F)


def brute(cols, valids, types, conjuncts, group, aggs):
    n = len(cols[0])
    ok = np.ones(n, dtype=bool)
    or_terms = {}
    with np.errstate(over="ignore"):
        for cjt in conjuncts:
            col, op, lit = cjt[0], cjt[1], cjt[2]
            og = cjt[3] if len(cjt) > 3 else 0
            if isinstance(col, tuple) and col[0] in ("add", "sub", "mul"):
                _, a, b = col
                va = (valids[a] != 0) if valids[a] is not None else np.ones(n, bool)
                vb = (valids[b] != 0) if valids[b] is not None else np.ones(n, bool)
                valid = va & vb
                if types[a] == TYPE_DOUBLE or types[b] == TYPE_DOUBLE:
                    x = cols[a].astype(np.float64)
                    y = cols[b].astype(np.float64)
                    lit = float(lit)
                else:
                    x = cols[a].astype(np.int64)
                    y = cols[b].astype(np.int64)
                v = {"add": x + y, "sub": x - y, "mul": x * y}[col[0]]
            elif isinstance(col, tuple):
                fn, c = col
                valid = (valids[c] != 0) if valids[c] is not None else np.ones(n, bool)
                v = np_scalar_fn(fn, cols[c]).astype(np.int64)
            else:
                valid = (valids[col] != 0) if valids[col] is not None \
                    else np.ones(n, bool)
                v = cols[col]
            if op in ("in", "not_in"):
                hit = np.isin(v, np.asarray(lit))
                term = valid & (hit if op == "in" else ~hit)
            else:
                cmpf = {"=": v == lit, "!=": v != lit, "<": v < lit,
                        "<=": v <= lit, ">": v > lit, ">=": v >= lit}[op]
                term = valid & cmpf
            if og:
                or_terms.setdefault(og, np.zeros(n, bool))
                or_terms[og] |= term
            else:
                ok &= term
    for t in or_terms.values():
        ok &= t
    idx = np.nonzero(ok)[0]

    # group keys (fn keys extract first)
    kvals, knull = [], []
    for g in group:
        if isinstance(g, tuple):
            fn, c = g
            kvals.append(np_scalar_fn(fn, cols[c][idx]).astype(np.int64))
            knull.append(np.zeros(len(idx), bool) if valids[c] is None
                         else (valids[c][idx] == 0))
        else:
            kv = cols[g][idx]
            kvals.append(kv if types[g] == TYPE_DOUBLE
                         else kv.astype(np.int64))
            knull.append(np.zeros(len(idx), bool) if valids[g] is None
                         else (valids[g][idx] == 0))
    if group:
        key = np.zeros(len(idx), dtype=object)
        for r in range(len(idx)):
            key[r] = tuple(
                (None if knull[k][r] else
                 (float(kvals[k][r]) if kvals[k].dtype == np.float64
                  else int(kvals[k][r])))
                for k in range(len(group)))
        uniq = sorted(set(key.tolist()),
                      key=lambda t: [(x is None, x) for x in t])
    else:
        key = np.zeros(len(idx), dtype=object)
        key[:] = [()] * len(idx)
    out = {}
    with np.errstate(over="ignore"):
        for u in (uniq if group else [()]):
            sel = np.array([k == u for k in key], dtype=bool) if group else \
                np.ones(len(idx), bool)
            rows = idx[sel]
            row = []
            for name, colspec in aggs:
                if name == "count_star":
                    row.append(("i", len(rows)))
                    continue
                if isinstance(colspec, tuple):
                    _, a, b = colspec
                    va = np.ones(len(rows), bool) if valids[a] is None \
                        else (valids[a][rows] != 0)
                    vb = np.ones(len(rows), bool) if valids[b] is None \
                        else (valids[b][rows] != 0)
                    vmask = va & vb
                    dbl = (types[a] == TYPE_DOUBLE or types[b] == TYPE_DOUBLE)
                    if dbl:
                        x = cols[a][rows].astype(np.float64)
                        y = cols[b][rows].astype(np.float64)
                        vv = {"add": x + y, "sub": x - y,
                              "mul": x * y}[colspec[0]]
                    else:
                        x = cols[a][rows].astype(np.int64)
                        y = cols[b][rows].astype(np.int64)
                        vv = {"add": x + y, "sub": x - y,
                              "mul": x * y}[colspec[0]]
                    vv = vv[vmask]
                    is_dbl = dbl
                else:
                    c = colspec
                    vmask = np.ones(len(rows), bool) if valids[c] is None \
                        else (valids[c][rows] != 0)
                    vv = cols[c][rows][vmask]
                    is_dbl = types[c] == TYPE_DOUBLE
                if name == "count":
                    row.append(("i", int(vmask.sum())))
                elif len(vv) == 0:
                    row.append(("null", None))
                elif name == "sum":
                    row.append(("d", float(vv.sum())) if is_dbl
                               else ("i", int(vv.astype(np.int64).sum())))
                elif name == "avg":
                    row.append(("d", float(vv.astype(np.float64).mean())))
                elif name == "min":
                    row.append(("d", float(vv.min())) if is_dbl
                               else ("i", int(vv.min())))
                else:
                    row.append(("d", float(vv.max())) if is_dbl
                               else ("i", int(vv.max())))
            out[u] = row
    return len(idx), out


@pytest.mark.parametrize("cs", range(20))
def test_oracle_vs_numpy_brute(oracle, cs):
    import random
    from oracle.bindings import make_query
    from baikaldb_amd.plan import _FNS, _ARITH
    rng = random.Random(777_000 + cs)
    specs, conjuncts, group, aggs = fz.random_case(rng)
    n = rng.choice([1000, 6000])
    seed = rng.randrange(1 << 40)
    arr = (BkColSpec * len(specs))()
    for i, s in enumerate(specs):
        (arr[i].col_type, arr[i].dist, arr[i].p0, arr[i].p1,
         arr[i].null_frac_x1e6) = s
    cols, valids = oracle.generate_table(list(arr), n, seed)
    types = [s[0] for s in specs]

    # oracle side (same conversion as run_both)
    oconj = []
    for cjt in conjuncts:
        col, op, lit = cjt[0], cjt[1], cjt[2]
        og = cjt[3] if len(cjt) > 3 else 0
        fn, col2, arith = 0, -1, 0
        if isinstance(col, tuple) and col[0] in _ARITH:
            arith, col2, col = _ARITH[col[0]], col[2], col[1]
            if types[col] == TYPE_DOUBLE or types[col2] == TYPE_DOUBLE:
                lit = float(lit)
        elif isinstance(col, tuple):
            fn = _FNS[col[0]]
            col = col[1]
        ops = {"=": 0, "!=": 1, ">": 2, ">=": 3, "<": 4, "<=": 5,
               "in": 6, "not_in": 7}
        ct = TYPE_DOUBLE if (types[col] == TYPE_DOUBLE and
                             not isinstance(lit, (list, tuple))) or \
            isinstance(lit, float) else TYPE_INT64
        oconj.append((col, ops[op], ct, lit, fn, og, col2, arith))
    ogroup = [(_FNS[g[0]], g[1]) if isinstance(g, tuple) else g for g in group]
    oaggs = []
    for a, c in aggs:
        if isinstance(c, tuple):
            c = (_ARITH[c[0]], c[1], c[2])
        oaggs.append((AGGMAP[a], c))
    q = make_query(oconj, ogroup, oaggs, types)
    exp = oracle.filter_agg(cols, valids, types, q, nthreads=3, dict_seed=seed)

    rp, bout = brute(cols, valids, types, conjuncts, group, aggs)
    assert exp["rows_passed"] == rp, f"case {cs}"
    if group:
        assert exp["ngroups"] == len(bout), f"case {cs}"
    # spot-check aggregate values of up to 10 groups via the oracle's
    # canonical (sorted) emission order versus the brute's sorted keys
    ng = exp["ngroups"]
    take = min(ng, 10)
    bkeys = sorted(bout.keys(), key=lambda t: [(x is None and False, x if x is not None else -(1 << 62)) for x in t]) if group else [()]
    # order differences for nullable keys: compare as MULTISETS of rows
    def rowrepr(vals):
        return tuple(("%s:%.9g" % (k, v)) if v is not None else "null"
                     for k, v in vals)
    brows = sorted(rowrepr(v) for v in bout.values())
    erows = []
    for r in range(ng):
        vals = []
        for a, (name, cspec) in enumerate(aggs):
            has = exp["agg_has"][a][r]
            if not has:
                vals.append(("null", None))
                continue
            if isinstance(cspec, tuple):
                dbl = (types[cspec[1]] == TYPE_DOUBLE or
                       types[cspec[2]] == TYPE_DOUBLE)
            else:
                dbl = cspec >= 0 and types[cspec] == TYPE_DOUBLE
            if name in ("count_star", "count"):
                vals.append(("i", int(exp["agg_i"][a][r])))
            elif name == "avg":
                vals.append(("d", float(exp["agg_d"][a][r])))
            elif dbl:
                vals.append(("d", float(exp["agg_d"][a][r])))
            else:
                vals.append(("i", int(exp["agg_i"][a][r])))
        erows.append(rowrepr(vals))
    erows.sort()
    assert len(erows) == len(brows), f"case {cs}"
    # float formatting to 9 significant digits absorbs reduction-order noise
    assert erows == brows, f"case {cs}"
