# oracle — *** TEST INFRASTRUCTURE ONLY ***
# ctypes bindings for the CPU restatement of BaikalDB's row-engine hot path
# (oracle/oracle.c). Only tests/, __graft_entry__.smoke() and bench.py's
# cpu_baseline leg may import this package; the product GPU path must never
# route through it.
from .bindings import (  # noqa: F401
    Oracle,
    BkColSpec,
    BkConjunct,
    BkAggSpec,
    BkOrderSpec,
    BkQuerySpec,
    TYPE_INT64,
    TYPE_DOUBLE,
    TYPE_STRING,
    DIST_UNIFORM,
    DIST_CUBESKEW,
    DIST_DICT,
    DIST_SUMU16,
    OP_EQ, OP_NE, OP_GT, OP_GE, OP_LT, OP_LE,
    AGG_COUNT_STAR, AGG_COUNT, AGG_SUM, AGG_AVG, AGG_MIN, AGG_MAX,
)
