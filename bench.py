#!/usr/bin/env python3
"""bench.py — measures BASELINE.json's metric: scanned rows/sec (whole node)
plus achieved HBM GB/s on the 1e9-row filter+GROUP BY workload (config 3 of
BASELINE.json, the configuration the metric is quoted on; it fits one GPU).

One "step" = one pass of the fused scan+filter+aggregate pipeline over
the full synthetic table resident in HBM (data generated on device before the
timed region; `data: synthetic`). With N>1 ranks each rank owns its region
set (weak scaling: per-GPU rows fixed) and the step includes the RCCL
merge-aggregate: a hash-partitioned all-to-all of part blobs (the
repartition the reference's ExchangeSenderNode does over brpc,
exchange_sender_node.h:228-235), every rank merging its key-hash shard in
parallel.

Usage: python bench.py [--gpus N] [--steps K] [--warmup W] [--config NAME]
The driver launches N>1 via torch.distributed.run; ranks read RANK/WORLD_SIZE.
"""
import argparse
import json
import os
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

TYPE_INT64, TYPE_DOUBLE, TYPE_STRING = 6, 12, 13
D_UNI, D_SKEW, D_DICT, D_SUM16 = 0, 1, 2, 3
SEED = 20260915

# ---------------- workload definitions (BASELINE.json configs) -------------
# config3: 1e9 rows mixed INT64/DOUBLE/VARCHAR(32)-dict; 3 predicates;
# GROUP BY (c2 skew-1e5, c7 dict-65536); aggs COUNT(*), SUM(c3:I64),
# SUM(c4:DBL), AVG(c5:DBL).
CONFIGS = {
    # BASELINE config 2: "1e8-row 8xINT64, WHERE c1<K AND c2=K2 GROUP BY c3
    # SUM(c4)" — c2 is a low-cardinality column (sysbench-style), K2 one of
    # its values; c3 ~ Zipf over 1e5 distinct.
    "config2_1e8_8int64": dict(
        nrows=100_000_000,
        specs=[(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
               (TYPE_INT64, D_UNI, 0, 20, 0)]
            + [(TYPE_INT64, 4, 100_000, 0, 0)]      # group key ~ Zipf/1e5
            + [(TYPE_INT64, D_UNI, 0, 1000, 0)]
            + [(TYPE_INT64, D_UNI, 0, 1 << 31, 0)] * 4,
        conjuncts=[(0, "<", 1 << 30), (1, "=", 7)],
        group=[2], aggs=[("sum", 3)],
        expected_groups=1 << 18,
        # algorithmic bytes/row: c0,c1 for every row (16 B); group c2 + agg
        # c3 only for the 0.5 * 0.05 surviving fraction
        bytes_per_row=16 + 16 * 0.5 * 0.05),
    # the north_star target shape: "1e9-row 8-INT64-col scan+filter+GROUP BY
    # at 1 GPU, >= 40% of HBM3E peak read bandwidth, >= 10x CPU" — config2's
    # query at 1e9 rows
    "config2_1e9_8int64": dict(
        nrows=1_000_000_000,
        specs=[(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
               (TYPE_INT64, D_UNI, 0, 20, 0)]
            + [(TYPE_INT64, 4, 100_000, 0, 0)]
            + [(TYPE_INT64, D_UNI, 0, 1000, 0)]
            + [(TYPE_INT64, D_UNI, 0, 1 << 31, 0)] * 4,
        conjuncts=[(0, "<", 1 << 30), (1, "=", 7)],
        group=[2], aggs=[("sum", 3)],
        expected_groups=1 << 18,
        bytes_per_row=16 + 16 * 0.5 * 0.05),
    "config3_1e9_mixed": dict(
        nrows=1_000_000_000,
        specs=[(TYPE_INT64, D_UNI, 0, 1 << 31, 0),      # c0 predicate
               (TYPE_INT64, D_UNI, 0, 1 << 31, 0),      # c1 predicate
               (TYPE_INT64, 4, 16384, 0, 0),            # c2 group key 1 (Zipf-like)
               (TYPE_INT64, D_UNI, 0, 1000, 0),         # c3 SUM int64
               (TYPE_DOUBLE, D_SUM16, 0, 0, 0),         # c4 SUM double
               (TYPE_DOUBLE, D_SUM16, 0, 0, 0),         # c5 AVG double
               (TYPE_INT64, D_UNI, 0, 1 << 31, 0),      # c6 untouched
               # c7: VARCHAR via dict codes; group-key cardinality is
               # generator-controlled (BASELINE: target 1e4-1e7 groups):
               # 64 distinct codes x 16384 c2 values ~> ~1e6 group pairs
               (TYPE_STRING, D_DICT, 64, 0, 0)],
        conjuncts=[(0, "<", 1 << 30),                    # sel 0.5
                   (1, "<", int((1 << 31) * 0.9)),       # sel 0.9
                   (7, "!=", 123)],                      # sel ~1
        group=[2, 7],
        aggs=[("count_star", -1), ("sum", 3), ("sum", 4), ("avg", 5)],
        expected_groups=1 << 21,
        # c0,c1 for all rows (16 B); dict c7 for the 0.45 surviving c0&c1
        # fraction (4 B); group c2 + aggs c3,c4,c5 for the 0.45*(65535/65536)
        # survivors (32 B)
        bytes_per_row=16 + 4 * 0.45 + 32 * 0.45 * (65535 / 65536)),
    # BASELINE config 5: 1e9-row ORDER BY c0,c1 LIMIT 1e6 (GPU radix top-N
    # selection; with N>1 each rank owns a region set and rank 0 merges the
    # gathered per-rank top-Ks — SelectManagerNode's heap-merge role,
    # select_manager_node.cpp:304-344).
    "config5_1e9_sort": dict(
        nrows=1_000_000_000,
        specs=[(TYPE_INT64, D_UNI, 0, 1 << 31, 0),
               (TYPE_INT64, D_UNI, 0, 1 << 31, 0),
               (TYPE_INT64, D_UNI, 0, 1 << 31, 0)],
        order=[(0, 1, 1), (1, 1, 1)],
        limit=1_000_000,
        conjuncts=[], group=[], aggs=[],
        expected_groups=0,
        bytes_per_row=16.0),  # the two 8-B key columns, read once
}


def log(msg):
    if int(os.environ.get("RANK", "0")) == 0:
        print(msg, file=sys.stderr, flush=True)


def cpu_baseline_leg(cfg, sample_rows):
    """Time the oracle (the reference row engine restated; kind='port') on the
    host cores, on a bounded sample of the same workload."""
    from concurrent.futures import ThreadPoolExecutor
    import numpy as np
    from oracle import Oracle, BkColSpec
    from oracle.bindings import make_query
    orc = Oracle()
    specs = (BkColSpec * len(cfg["specs"]))()
    for i, s in enumerate(cfg["specs"]):
        (specs[i].col_type, specs[i].dist, specs[i].p0, specs[i].p1,
         specs[i].null_frac_x1e6) = s
    col_types = [s[0] for s in cfg["specs"]]
    # generate columns in parallel (untimed setup; ctypes releases the GIL)
    import ctypes as Ct
    cols = []
    for ci, sp in enumerate(specs):
        dt = {6: np.int64, 12: np.float64, 13: np.int32}[sp.col_type]
        cols.append(np.empty(sample_rows, dtype=dt))
    def gen(ci):
        orc.lib.orc_generate_column(Ct.byref(specs[ci]), SEED, ci, 0,
                                    sample_rows,
                                    cols[ci].ctypes.data_as(Ct.c_void_p), None)
    with ThreadPoolExecutor(max_workers=len(cols)) as ex:
        list(ex.map(gen, range(len(cols))))
    valids = [None] * len(cols)
    ops = {"=": 0, "!=": 1, ">": 2, ">=": 3, "<": 4, "<=": 5}
    aggmap = {"count_star": 0, "count": 1, "sum": 2, "avg": 3, "min": 4, "max": 5}
    conj = []
    for col, op, lit in cfg["conjuncts"]:
        ct = TYPE_DOUBLE if (col_types[col] == TYPE_DOUBLE or isinstance(lit, float)) \
            else TYPE_INT64
        conj.append((col, ops[op], ct, lit))
    q = make_query(conj, cfg["group"], [(aggmap[a], c) for a, c in cfg["aggs"]],
                   col_types)
    cores = min(os.cpu_count() or 1, 128)  # oracle caps at 128 worker threads
    t0 = time.perf_counter()
    if "order" in cfg:
        # top-N selection (single-threaded heap walk — the reference's
        # TopNSorter is per-region single-threaded too, topn_sorter.h:32)
        cores = 1
        lim = min(cfg["limit"], sample_rows)
        orc.sort_topk(cols, valids, col_types, cfg["order"], lim, q=q)
    else:
        orc.filter_agg(cols, valids, col_types, q, nthreads=cores,
                       dict_seed=SEED, sort_keys=False)
    dt = time.perf_counter() - t0
    return {"value": sample_rows / dt, "unit": "rows/s", "cores": cores,
            "kind": "port",
            "sample": f"{sample_rows} rows of the same workload, {dt:.1f}s"}


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--config", default="config3_1e9_mixed")
    ap.add_argument("--rows", type=int, default=0, help="override rows per GPU")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    n_gpus = max(world, 1)

    import torch
    dist = None
    if world > 1 or (world == 1 and os.environ.get("BK_FORCE_DIST")):
        # BK_FORCE_DIST exercises the full RCCL exchange path on a single
        # rank (validation on 1-GPU boxes; the driver launches real N>1)
        import torch.distributed as tdist
        tdist.init_process_group(backend="nccl")
        dist = tdist
        torch.cuda.set_device(local_rank)

    # pin clocks: the box's DVFS governor drops to ~100 MHz between steps and
    # ramps slowly, making otherwise-identical steps bimodal (measured in
    # profiles/; junction temps ~46C, nowhere near thermal limits)
    if local_rank == 0:
        import subprocess
        subprocess.run(["rocm-smi", "--setperflevel", "high"],
                       capture_output=True, check=False)

    from baikaldb_amd import GpuEngine, QueryPlan
    eng = GpuEngine(device=local_rank)

    cfg = CONFIGS[args.config]
    nrows = args.rows or cfg["nrows"]
    # weak scaling: each rank owns its own region set of `nrows` rows
    row_begin = rank * nrows

    log(f"[bench] generating {nrows} rows on rank {rank} (device {local_rank})")
    t = eng.create_table(cfg["specs"], nrows)
    eng.generate(t, SEED, row_begin=row_begin)
    eng.sync()

    plan = QueryPlan(t.col_types, conjuncts=cfg["conjuncts"], group=cfg["group"],
                     aggs=cfg["aggs"])

    is_sort = "order" in cfg

    def one_sort_step(timed_kernels):
        rowids = eng.sort_topk(t, cfg["order"], cfg["limit"], plan=plan)
        if timed_kernels is not None:
            timed_kernels.append(eng.topk_kernel_ms())
        if dist is not None:
            # gather per-rank top-Ks; rank 0 merges (stable LSD argsorts)
            import numpy as np
            k0 = np.zeros(len(rowids), dtype=np.int64)
            k1 = np.zeros(len(rowids), dtype=np.int64)
            dscr = np.zeros(len(rowids), dtype=np.float64)
            nul = np.zeros(len(rowids), dtype=np.uint8)
            import ctypes as Ct
            for col, dst in ((cfg["order"][0][0], k0), (cfg["order"][1][0], k1)):
                eng.lib.bkgpu_gather(t.handle, col,
                                     rowids.ctypes.data_as(Ct.POINTER(Ct.c_int64)),
                                     len(rowids),
                                     dst.ctypes.data_as(Ct.POINTER(Ct.c_int64)),
                                     dscr.ctypes.data_as(Ct.POINTER(Ct.c_double)),
                                     nul.ctypes.data_as(Ct.POINTER(Ct.c_uint8)))
            pack = torch.stack([torch.from_numpy(k0), torch.from_numpy(k1),
                                torch.from_numpy(rowids)]).cuda()
            buf = torch.zeros(3, cfg["limit"], dtype=torch.int64, device="cuda")
            buf[:, :pack.shape[1]] = pack
            gath = [torch.zeros_like(buf) for _ in range(world)]
            dist.all_gather(gath, buf)
            if rank == 0:
                allk = torch.cat(gath, dim=1)
                idx = torch.argsort(allk[2], stable=True)
                for row in (1, 0):
                    idx = idx[torch.argsort(allk[row][idx], stable=True)]
                _final = allk[:, idx[:cfg["limit"]]]
            torch.cuda.synchronize()
        return len(rowids), 0

    def one_step(timed_kernels):
        if is_sort:
            return one_sort_step(timed_kernels)
        res = eng.filter_agg(t, plan, expected_groups=cfg["expected_groups"])
        if timed_kernels is not None:
            timed_kernels.append(res.kernel_ms)
        if os.environ.get("BK_BENCH_DEBUG"):
            log(f"[dbg] breakdown { {k: round(v, 1) for k, v in res.breakdown().items()} }")
        if dist is not None:
            eng.sync()
            # hash-partitioned all-to-all exchange — the repartition the
            # reference's ExchangeSenderNode does over brpc
            # (exchange_sender_node.h:228-235), here over RCCL/xGMI: rank r
            # keeps hash-part r and merges every peer's part-r blob, so the
            # merge work spreads across ranks and the traffic across all 7
            # xGMI links (replaces the serialized gather-to-rank-0 merge).
            per_group = 20 + 16 * len(cfg["aggs"])
            t_p0 = time.perf_counter()
            counts = res.part_counts(world)
            if os.environ.get("BK_BENCH_DEBUG"):
                log(f"[dbg] part_counts {(time.perf_counter()-t_p0)*1e3:.1f} ms")
            scounts = torch.tensor(counts, dtype=torch.int64, device="cuda")
            rcounts = torch.zeros_like(scounts)
            dist.all_to_all_single(rcounts, scounts)
            in_splits = [c * per_group for c in counts]
            sbuf = torch.zeros(max(sum(in_splits), 1), dtype=torch.uint8,
                               device="cuda")
            off = 0
            for p in range(world):
                if counts[p] > 0:
                    res.export_part(world, p, sbuf.data_ptr() + off,
                                    counts[p])
                off += in_splits[p]
            rlist = [int(x) for x in rcounts.cpu()]
            out_splits = [c * per_group for c in rlist]
            rbuf = torch.zeros(max(sum(out_splits), 1), dtype=torch.uint8,
                               device="cuda")
            dist.all_to_all_single(rbuf, sbuf, out_splits, in_splits)
            torch.cuda.synchronize()
            # local merge of the received part-`rank` blobs: the final
            # result is SHARDED by key hash (disjoint across ranks); the
            # target is sized exactly (sum of received counts is an upper
            # bound: keys dedup across peers)
            t_x0 = time.perf_counter()
            merged = eng.agg_empty(plan,
                                   expected_groups=max(sum(rlist), 1))
            off = 0
            for p in range(world):
                if rlist[p] > 0:
                    merged.merge_blob(rbuf.data_ptr() + off, rlist[p])
                off += out_splits[p]
            eng.sync()
            # group count from the fill counter (insert-only target: every
            # distinct key claimed exactly once) — no compact needed
            tot = torch.tensor([merged.nfilled], dtype=torch.int64,
                               device="cuda")
            dist.all_reduce(tot)
            merged.free()
            if os.environ.get("BK_BENCH_DEBUG"):
                log(f"[dbg] exchange merge leg {(time.perf_counter()-t_x0)*1e3:.1f} ms")
            rp = res.rows_passed
            ng = int(tot.item())
            res.free()
            return rp, ng
        rp = res.rows_passed
        ng = res.ngroups
        res.free()
        return rp, ng

    log(f"[bench] warmup x{args.warmup}")
    for _ in range(args.warmup):
        one_step(None)
    eng.sync()
    if dist is not None:
        dist.barrier()
    torch.cuda.synchronize() if torch.cuda.is_available() else None

    dbg = os.environ.get("BK_BENCH_DEBUG")
    kms = []
    t0 = time.perf_counter()
    for _ in range(args.steps):
        s0 = time.perf_counter()
        rp, ng = one_step(kms)
        if dbg:
            log(f"[dbg] step wall {(time.perf_counter()-s0)*1e3:.1f} ms, "
                f"kernels {kms[-1] if kms else 0:.1f} ms")
    eng.sync()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    if dist is not None:
        dist.barrier()
    t1 = time.perf_counter()
    elapsed = t1 - t0
    if dist is not None:  # max over ranks
        e = torch.tensor([elapsed], dtype=torch.float64, device="cuda")
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = float(e.item())

    ms_per_step = elapsed / args.steps * 1000.0
    total_rows = nrows * n_gpus  # whole-job rows scanned per step
    rows_per_sec = total_rows / (elapsed / args.steps)

    if rank != 0:
        return

    # roofline of the dominant kernel (fused filter+agg), HIP-event timed on
    # its launch stream inside libbkgpu (bkgpu_agg_kernel_ms)
    avg_kernel_ms = sum(kms) / len(kms) if kms else 0.0
    algo_bytes = cfg["bytes_per_row"] * nrows  # per launch (this rank)
    achieved_gbs = (algo_bytes / (avg_kernel_ms / 1000.0)) / 1e9 if avg_kernel_ms else 0.0
    peak_gbs = 8000.0  # HBM3E spec peak (MI355X_MICROARCH.md)
    # traffic: HBM bytes per launch from separate rocprofv3 --pmc passes
    # (FETCH_SIZE x2 wide-read calibration + WRITE_SIZE; see profiles/).
    # Regenerated by tools/pmc_traffic.py; env BK_TRAFFIC_BYTES overrides.
    traffic = os.environ.get("BK_TRAFFIC_BYTES")
    if traffic is None:
        tf = os.path.join(REPO, "profiles", f"traffic_{args.config}.json")
        if os.path.exists(tf) and nrows == cfg["nrows"]:
            with open(tf) as f:
                traffic = json.load(f).get("bytes_per_launch")
    roofline = {"bound": "hbm", "achieved": round(achieved_gbs, 1),
                "peak": peak_gbs, "unit": "GB/s",
                "frac": round(achieved_gbs / peak_gbs, 4),
                "traffic": float(traffic) if traffic else None}

    cpu_baseline = None
    if n_gpus == 1 and not args.no_cpu_baseline:
        sample = min(nrows, 100_000_000)
        log(f"[bench] cpu baseline leg on {sample} rows")
        cpu_baseline = cpu_baseline_leg(cfg, sample)

    line = {
        "metric": "scanned rows/sec (whole node), 1e9-row filter+GROUP BY",
        "value": round(rows_per_sec, 1),
        "unit": "rows/s",
        "n_gpus": n_gpus,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(ms_per_step, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,  # reference publishes no number
        "dtype": "int64/f64",
        "data": "synthetic",
        "config": {"workload": args.config, "rows_per_gpu": nrows,
                   "groups": ng, "rows_passed_rank0": rp,
                   "predicates": len(cfg["conjuncts"]),
                   "group_keys": len(cfg["group"]), "aggs": len(cfg["aggs"]),
                   **({"order_keys": len(cfg["order"]), "limit": cfg["limit"]}
                      if "order" in cfg else {})},
        "roofline": roofline,
        "cpu_baseline": cpu_baseline,
    }
    print(json.dumps(line), flush=True)


if __name__ == "__main__":
    main()
    # clean rendezvous shutdown (all ranks reach here: rank!=0 returns
    # early from main after the timed loop's final barrier)
    try:
        import torch.distributed as _td
        if _td.is_initialized():
            _td.destroy_process_group()
    except Exception:
        pass
