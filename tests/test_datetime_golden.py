# Parity pinning: the packed-DATETIME bit layout of our bk_scalar_fn /
# bk_gen_datetime vs vectors produced by the REFERENCE's own inline
# extraction functions compiled in place (oracle/_ref/datetime_ref; see
# oracle/ref_datetime_harness.cpp). Fixtures committed in
# tests/golden/datetime_golden.json.
import json
import os

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "golden", "datetime_golden.json")


def test_datetime_extractions_match_reference(oracle):
    import ctypes as C
    vecs = json.load(open(GOLDEN))
    assert len(vecs) == 256
    lib = oracle.lib
    lib.orc_scalar_fn.restype = C.c_int64
    lib.orc_scalar_fn.argtypes = [C.c_int32, C.c_int64]
    FN_YEAR, FN_MONTH, FN_DAY = 1, 2, 3
    for v in vecs:
        dt = C.c_int64(v["dt"])
        assert lib.orc_scalar_fn(FN_YEAR, dt) == v["year"], v
        assert lib.orc_scalar_fn(FN_MONTH, dt) == v["month"], v
        assert lib.orc_scalar_fn(FN_DAY, dt) == v["day"], v
        # the generator packs valid calendar dates; the reference's own
        # date_str round-trip confirms the y/m/d fields line up
        assert v["date_str"] == f"{v['year']:04d}-{v['month']:02d}-{v['day']:02d}"
