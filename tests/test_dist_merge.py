# CPU (gloo, world_size=2) coverage of the N>1 path: region-sharded partial
# aggregation + the MERGE_AGG exchange (agg_node.cpp:539-543) that bench.py
# performs over RCCL/xGMI on GPUs. Here each rank runs the oracle on its
# region set, partials are exchanged with torch.distributed (gloo), and the
# merged result must equal the whole-range single-pass result.
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

SEED = 777
TYPE_INT64, TYPE_DOUBLE, TYPE_STRING = 6, 12, 13


def merge_partials(parts, naggs, agg_types, agg_is_double):
    """Merge partial group maps with AggFnCall::merge semantics
    (src/expr/agg_fn_call.cpp:781-830). parts: list of dicts
    key(flag, e0, e1) -> list of (val_i, val_d, cnt, has) per agg."""
    out = {}
    for p in parts:
        for key, states in p.items():
            if key not in out:
                out[key] = [list(s) for s in states]
                continue
            dst = out[key]
            for a in range(naggs):
                vi, vd, cnt, has = states[a]
                at = agg_types[a]
                if not has:
                    continue
                if at in ("count_star", "count"):
                    dst[a][0] += vi
                    dst[a][3] = True
                elif at in ("sum", "avg"):
                    if agg_is_double[a] or at == "avg":
                        dst[a][1] = dst[a][1] + vd if dst[a][3] else vd
                    else:
                        dst[a][0] = int(np.int64(np.uint64(
                            (dst[a][0] if dst[a][3] else 0) & (2**64 - 1))
                            + np.uint64(vi & (2**64 - 1))))
                    dst[a][2] += cnt
                    dst[a][3] = True
                elif at == "min":
                    if not dst[a][3] or (vd if agg_is_double[a] else vi) < \
                            (dst[a][1] if agg_is_double[a] else dst[a][0]):
                        dst[a][0], dst[a][1] = vi, vd
                    dst[a][2] += cnt
                    dst[a][3] = True
                elif at == "max":
                    if not dst[a][3] or (vd if agg_is_double[a] else vi) > \
                            (dst[a][1] if agg_is_double[a] else dst[a][0]):
                        dst[a][0], dst[a][1] = vi, vd
                    dst[a][2] += cnt
                    dst[a][3] = True
    return out


def oracle_partial(rank, nshards, nrows):
    import sys
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from oracle import Oracle, BkColSpec
    from oracle.bindings import make_query
    orc = Oracle()
    specs = (BkColSpec * 4)()
    for i, s in enumerate([(TYPE_INT64, 0, 0, 1 << 31, 0),
                           (TYPE_INT64, 1, 500, 0, 0),
                           (TYPE_INT64, 0, 0, 100, 0),
                           (TYPE_DOUBLE, 3, 0, 0, 0)]):
        specs[i].col_type, specs[i].dist, specs[i].p0, specs[i].p1, \
            specs[i].null_frac_x1e6 = s
    col_types = [TYPE_INT64, TYPE_INT64, TYPE_INT64, TYPE_DOUBLE]
    per = nrows // nshards
    begin = rank * per if rank >= 0 else 0
    count = per if rank >= 0 else nrows
    cols, valids = orc.generate_table(list(specs), count, SEED, row_begin=begin)
    q = make_query([(0, 4, TYPE_INT64, int((1 << 31) * 0.6))], [1],
                   [(0, -1), (2, 2), (2, 3), (3, 3), (4, 2), (5, 3)], col_types)
    r = orc.filter_agg(cols, valids, col_types, q, nthreads=2, dict_seed=SEED)
    # partial map: key -> per-agg states (vi, vd, cnt(unknown->0), has)
    out = {}
    for g in range(r["ngroups"]):
        key = (int(r["flags"][g]), int(r["enc"][g][0]), int(r["enc"][g][1]))
        states = []
        for a in range(6):
            has = bool(r["agg_has"][a][g])
            states.append((int(r["agg_i"][a][g]), float(r["agg_d"][a][g]),
                           int(r["agg_i"][0][g]), has))  # cnt ~ count* (unused for exactness here)
        out[key] = states
    return out, r


def _worker(rank, world, rendezvous, results):
    torch.distributed.init_process_group(
        "gloo", init_method=rendezvous, rank=rank, world_size=world)
    part, _ = oracle_partial(rank, world, 40_000)
    gathered = [None] * world
    torch.distributed.all_gather_object(gathered, part)
    if rank == 0:
        agg_types = ["count_star", "sum", "sum", "avg", "min", "max"]
        is_dbl = [False, False, True, True, False, True]
        merged = merge_partials(gathered, 6, agg_types, is_dbl)
        results.put(merged)
    torch.distributed.destroy_process_group()


def test_gloo_shard_merge_matches_whole():
    import tempfile
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    rdvfile = tempfile.NamedTemporaryFile(delete=False)
    rendezvous = f"file://{rdvfile.name}"
    procs = [ctx.Process(target=_worker, args=(r, 2, rendezvous, results))
             for r in range(2)]
    for p in procs:
        p.start()
    merged = results.get(timeout=180)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    whole, r = oracle_partial(-1, 1, 40_000)
    assert set(merged.keys()) == set(whole.keys())
    for key in whole:
        w, m = whole[key], merged[key]
        # COUNT(*) and integer SUM bit-exact
        assert m[0][0] == w[0][0]
        assert m[1][0] == w[1][0]
        # double sums/avg inputs within tolerance
        assert m[2][1] == pytest.approx(w[2][1], rel=1e-9, abs=1e-9)
        # min/max exact
        assert m[4][0] == w[4][0]
        assert m[5][1] == w[5][1]


def _bk_mix64(x):
    # bk_datagen.h bk_mix64 (splitmix64 finalizer) — restated exactly, so
    # the CPU test partitions with the SAME hash the engine's key_hash uses
    # on device (a key must land on the same part on every rank)
    M = 2**64 - 1
    x = (int(x) + 0x9E3779B97F4A7C15) & M
    x = (((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & M)
    x = (((x ^ (x >> 27)) * 0x94D049BB133111EB) & M)
    return np.uint64(x ^ (x >> 31))


def _key_part(key, nparts):
    flag, e0, e1 = key
    h = _bk_mix64(np.uint64(e0) ^ np.uint64(0x9E3779B97F4A7C15))
    h = _bk_mix64(h ^ np.uint64(e1))
    h = _bk_mix64(h ^ np.uint64(flag))
    return int(h) % nparts


def _worker_partitioned(rank, world, rendezvous, results):
    """Hash-partitioned exchange protocol (the all-to-all bench.py drives
    over RCCL; gloo has no all_to_all, so parts travel via
    all_gather_object): every rank keeps only part `rank`, merges the
    peers' part-`rank` blobs, and the SHARDS must be disjoint and union to
    the whole result."""
    torch.distributed.init_process_group(
        "gloo", init_method=rendezvous, rank=rank, world_size=world)
    part, _ = oracle_partial(rank, world, 40_000)
    split = [dict() for _ in range(world)]
    for key, states in part.items():
        split[_key_part(key, world)][key] = states
    gathered = [None] * world
    torch.distributed.all_gather_object(gathered, split)
    agg_types = ["count_star", "sum", "sum", "avg", "min", "max"]
    is_dbl = [False, False, True, True, False, True]
    mine = merge_partials([g[rank] for g in gathered], 6, agg_types, is_dbl)
    shards = [None] * world
    torch.distributed.all_gather_object(shards, mine)
    if rank == 0:
        results.put(shards)
    torch.distributed.destroy_process_group()


def test_gloo_partitioned_exchange_matches_whole():
    import tempfile
    ctx = mp.get_context("spawn")
    results = ctx.Queue()
    rdvfile = tempfile.NamedTemporaryFile(delete=False)
    rendezvous = f"file://{rdvfile.name}"
    procs = [ctx.Process(target=_worker_partitioned,
                         args=(r, 2, rendezvous, results))
             for r in range(2)]
    for p in procs:
        p.start()
    shards = results.get(timeout=180)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0

    whole, _ = oracle_partial(-1, 1, 40_000)
    # shards are disjoint and union to the whole key set
    assert not (set(shards[0]) & set(shards[1]))
    assert set(shards[0]) | set(shards[1]) == set(whole.keys())
    # every key landed on the part its hash names
    for p, shard in enumerate(shards):
        for key in shard:
            assert _key_part(key, 2) == p
    merged = {**shards[0], **shards[1]}
    for key in whole:
        w, m = whole[key], merged[key]
        assert m[0][0] == w[0][0]                       # COUNT(*)
        assert m[1][0] == w[1][0]                       # SUM int64
        assert m[2][1] == pytest.approx(w[2][1], rel=1e-9, abs=1e-9)
        assert m[4][0] == w[4][0]                       # MIN
        assert m[5][1] == w[5][1]                       # MAX
