# Dictionary order-preservation invariant: bk_dict_word produces words whose
# byte order (ExprValue STRING compare = byte compare, expr_value.h:895-945)
# equals dict-code order, for ANY seed — the property that makes integer
# MIN/MAX over dict codes a correct string MIN/MAX (DESIGN.md "string agg").
import ctypes as C
import random


def test_dict_word_order_preserving(oracle):
    lib = oracle.lib
    rng = random.Random(42)
    buf_a = C.create_string_buffer(64)
    buf_b = C.create_string_buffer(64)
    for _ in range(2000):
        seed = rng.getrandbits(64)
        a = rng.randrange(0, 1 << 22)
        b = rng.randrange(0, 1 << 22)
        lib.orc_dict_word(C.c_uint64(seed), C.c_int64(a), buf_a, 64)
        lib.orc_dict_word(C.c_uint64(seed), C.c_int64(b), buf_b, 64)
        wa, wb = buf_a.value, buf_b.value
        if a < b:
            assert wa < wb, (seed, a, b, wa, wb)
        elif a > b:
            assert wa > wb, (seed, a, b, wa, wb)
        else:
            assert wa == wb
