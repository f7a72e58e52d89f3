# Guards the bench.py driver contract: config shapes stay self-consistent
# (column references valid, analytic byte counts sane, wire-blob size math
# matches the engine's export layout).
import json
import os

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_configs_self_consistent():
    import bench
    for name, cfg in bench.CONFIGS.items():
        ncols = len(cfg["specs"])
        for col, _op, _lit in cfg["conjuncts"]:
            assert 0 <= col < ncols, (name, col)
        for col in cfg["group"]:
            assert 0 <= col < ncols, (name, col)
        for _fn, col in cfg["aggs"]:
            assert col == -1 or 0 <= col < ncols, (name, col)
        for col, _asc, _nf in cfg.get("order", ()):
            assert 0 <= col < ncols, (name, col)
        assert 0 < cfg["bytes_per_row"] <= 8 * ncols + 1, name
        assert len(cfg["group"]) <= 4 and len(cfg["aggs"]) <= 8, name


def test_wire_blob_per_group_math():
    """bench.one_step sizes peer blobs analytically as 20 + 16*naggs; the
    header documents [flags u32][k0 u64][k1 u64][states u64*2*naggs] — keep
    the two in sync (include/bkgpu.h wire format)."""
    hdr = open(os.path.join(REPO, "include", "bkgpu.h")).read()
    assert "[ flags: u32 * n ][ k0: u64 * n ][ k1: u64 * n ]" in hdr
    # 4 + 8 + 8 + 16*naggs == 20 + 16*naggs
    src = open(os.path.join(REPO, "bench.py")).read()
    assert "per_group = 20 + 16 * len(cfg[\"aggs\"])" in src


def test_baseline_metric_matches_bench():
    base = json.load(open(os.path.join(REPO, "BASELINE.json")))
    src = open(os.path.join(REPO, "bench.py")).read()
    assert "config3_1e9_mixed" in src
    assert base.get("metric") is None or "rows" in str(base.get("metric", ""))
