# Pins the oracle's (and thus the GPU path's) mem-comparable encoding against
# golden vectors produced by the REFERENCE's own key_encoder.h compiled
# in-place (oracle/ref_keyenc_harness.cpp; fixtures committed in
# tests/golden/keyenc_golden.json). Reference: include/common/key_encoder.h:104-173.
import json
import os
import struct

GOLDEN = os.path.join(os.path.dirname(__file__), "golden", "keyenc_golden.json")


def _load():
    with open(GOLDEN) as f:
        return json.load(f)


def test_encode_i64_matches_reference(oracle):
    g = _load()
    for val, enc_native, enc_be in g["encode_i64"]:
        got = oracle.lib.orc_encode_i64(val)
        assert got == enc_native, (val, got, enc_native)
        # big-endian byte-string form (MutTableKey::append_i64, mut_table_key.h:113)
        got_be = struct.unpack("<Q", struct.pack(">Q", got))[0]
        assert got_be == enc_be, (val, got_be, enc_be)
        # round trip
        assert oracle.lib.orc_decode_i64(got) == val


def test_encode_f64_matches_reference(oracle):
    g = _load()
    for val, enc_native, enc_be in g["encode_f64"]:
        got = oracle.lib.orc_encode_f64(val)
        assert got == enc_native, (val, got, enc_native)
        got_be = struct.unpack("<Q", struct.pack(">Q", got))[0]
        assert got_be == enc_be, (val, got_be, enc_be)
        back = oracle.lib.orc_decode_f64(got)
        assert back == val or (val != val and back != back)


def test_encoding_is_order_preserving(oracle):
    # reference test semantics: test/test_key_encoder.cpp:72-112
    import random
    rng = random.Random(7)
    ivals = [rng.randint(-2**63, 2**63 - 1) for _ in range(2000)] + [0, 1, -1, 2**63 - 1, -2**63]
    encs = [(v, oracle.lib.orc_encode_i64(v)) for v in ivals]
    for (v1, e1), (v2, e2) in zip(encs, encs[1:]):
        assert (v1 < v2) == (e1 < e2)
    dvals = [rng.uniform(-1e300, 1e300) for _ in range(2000)] + [0.0, -0.0, 1e-300, -1e-300]
    dencs = [(v, oracle.lib.orc_encode_f64(v)) for v in dvals]
    for (v1, e1), (v2, e2) in zip(dencs, dencs[1:]):
        assert (v1 < v2) == (e1 < e2), (v1, v2)
