import sys, random
sys.path.insert(0, "/root/repo")
import numpy as np
from oracle import Oracle
from oracle.bindings import make_query, BkColSpec, BkWindowFn, BkOrderSpec
import os
orc = Oracle(os.environ.get("ORACLE_ASAN_LIB", "/tmp/liboracle_asan.so"))
import tests.test_gpu_fuzz as fz

for cs in range(25):
    rng = random.Random(777 + cs)
    specs, conjuncts, group, aggs, = fz.random_case(rng)
    n = 4000
    arr = (BkColSpec * len(specs))()
    for i, s in enumerate(specs):
        (arr[i].col_type, arr[i].dist, arr[i].p0, arr[i].p1,
         arr[i].null_frac_x1e6) = s
    cols, valids = orc.generate_table(list(arr), n, rng.getrandbits(40))
    types = [s[0] for s in specs]
    ops = {"=":0,"!=":1,">":2,">=":3,"<":4,"<=":5,"in":6,"not_in":7}
    am = {"count_star":0,"count":1,"sum":2,"avg":3,"min":4,"max":5}
    oc = []
    for col, op, lit, *og in conjuncts:
        fn = 0
        if isinstance(col, tuple):
            from baikaldb_amd.plan import _FNS
            fn = _FNS[col[0]]; col = col[1]
        ct = 12 if (types[col]==12 and not isinstance(lit,(list,tuple))) or isinstance(lit,float) else 6
        oc.append((col, ops[op], ct, lit, fn, og[0] if og else 0))
    from baikaldb_amd.plan import _FNS as _GF
    group = [(_GF[g[0]], g[1]) if isinstance(g, tuple) and isinstance(g[0], str)
             else g for g in group]
    q = make_query(oc, group, [(am[a], c) for a, c in aggs], types)
    orc.filter_agg(cols, valids, types, q, nthreads=3, dict_seed=7)
    # sort + window paths
    icols = [c for c in range(len(types)) if types[c] != 13][:2]
    if icols:
        order = [(icols[0], 1, 1)]
        orc.sort_topk(cols, valids, types, order, 50)
        orc.window(cols, valids, types,
                   [(10, -1, 0), (2, icols[0], 0), (19, -1, 0), (20, -1, 3)],
                   part_col=group[0] if group and isinstance(group[0], int) else -1,
                   order=order)
        orc.window(cols, valids, types, [(2, icols[0], 0), (1, icols[0], 0)],
                   part_col=-1, order=order, frame=(2, 1))
print("ASAN DRIVE OK")
