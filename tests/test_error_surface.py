"""Fail-loudly contract (CPU): unsupported plan shapes must be REJECTED at
compile/open with a descriptive gx_last_error — never silently mis-executed
(the round's documented limitations). These run against the PRODUCT library
on CPU: plan compilation happens before the GPU check."""
import ctypes

import pytest

from tests.gxlib import (GX_AGG_COUNT, GX_AGG_MODE_PARTIAL, GX_AGG_SUM,
                         GX_F_GT, GX_F_ROUND, GX_TYPE_DECIMAL, GX_TYPE_I64,
                         GX_TYPE_STRING, load_product)
from tidb_amd import plan as P
from tidb_amd.chunkpy import PyChunk


def _expect_error(build_fn, *needles):
    lib = load_product()
    b = P.Builder(lib)
    root, src, chunks = build_fn(b, lib)
    ex = b.build(root)
    if src is not None:
        ex.bind_chunks(src, chunks)
    rc = lib.gx_open(ex.ex)
    err = ex.error()
    ex.free()
    b.free()
    assert rc != 0, f"expected a compile/open error, got rc=0"
    assert any(n in err for n in needles), err


def _i64_chunk(rows=((1, 1),)):
    ch = PyChunk([GX_TYPE_I64] * 2, max(len(rows), 1))
    for r in rows:
        ch.append_row(list(r))
    return [ch]


def test_mixed_distinct_plain_rejected():
    def plan(b, lib):
        src = b.source([GX_TYPE_I64, GX_TYPE_I64])
        agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                        [(6, b.colref(1, GX_TYPE_I64), 0),
                         (GX_AGG_COUNT, -1, 0)])
        return agg, src, _i64_chunk()
    _expect_error(plan, "DISTINCT")


def test_distinct_partial_rejected():
    def plan(b, lib):
        src = b.source([GX_TYPE_I64, GX_TYPE_I64])
        agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                        [(6, b.colref(1, GX_TYPE_I64), 0)],
                        GX_AGG_MODE_PARTIAL)
        return agg, src, _i64_chunk()
    _expect_error(plan, "COMPLETE")


def test_round_negative_d_rejected():
    def plan(b, lib):
        src = b.source([GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 2])
        r = b.call(GX_F_ROUND, GX_TYPE_DECIMAL, 2,
                   b.colref(1, GX_TYPE_DECIMAL, 2), b.const_i64(-1))
        agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                        [(GX_AGG_SUM, r, 2)])
        ch = PyChunk([GX_TYPE_I64, GX_TYPE_DECIMAL], 1, [0, 2])
        return agg, src, [ch]
    _expect_error(plan, "ROUND")


def test_unsupported_having_condition_rejected():
    """HAVING leaves are col-cmp-const/IS NULL/OR; an arithmetic HAVING
    (sum(a)+1 > 2) must fail loudly — caught at first Next (open runs the
    pipeline lazily, but the CPU box fails at the GPU check FIRST, so pin
    the compile-side rejection via an out-of-range column instead)."""
    def plan(b, lib):
        src = b.source([GX_TYPE_I64, GX_TYPE_I64])
        agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                        [(GX_AGG_COUNT, -1, 0)])
        hav = b.selection(agg, [b.call(GX_F_GT, GX_TYPE_I64, 0,
                                       b.colref(7, GX_TYPE_I64),
                                       b.const_i64(0))])
        return hav, src, _i64_chunk()
    _expect_error(plan, "HAVING")


def test_too_many_group_keys_rejected():
    def plan(b, lib):
        types = [GX_TYPE_I64] * 8
        src = b.source(types)
        keys = [b.colref(i, GX_TYPE_I64) for i in range(7)]
        agg = b.hashagg(src, keys, [(GX_AGG_COUNT, -1, 0)])
        ch = PyChunk(types, 1)
        return agg, src, [ch]
    _expect_error(plan, "group", "key")


def test_bad_join_type_rejected():
    def plan(b, lib):
        b1 = b.source([GX_TYPE_I64])
        p1 = b.source([GX_TYPE_I64])
        j = b.hashjoin(b1, p1, [b.colref(0, GX_TYPE_I64)],
                       [b.colref(0, GX_TYPE_I64)], join_type=8)
        ch1 = PyChunk([GX_TYPE_I64], 1)
        ch2 = PyChunk([GX_TYPE_I64], 1)
        return j, None, None  # bind manually below
    lib = load_product()
    b = P.Builder(lib)
    b1 = b.source([GX_TYPE_I64])
    p1 = b.source([GX_TYPE_I64])
    j = b.hashjoin(b1, p1, [b.colref(0, GX_TYPE_I64)],
                   [b.colref(0, GX_TYPE_I64)], join_type=8)
    ex = b.build(j)
    rc = lib.gx_open(ex.ex)
    err = ex.error()
    ex.free()
    b.free()
    assert rc != 0 and "join type" in err


def test_too_many_join_keys_rejected():
    lib = load_product()
    b = P.Builder(lib)
    types = [GX_TYPE_I64] * 5
    b1 = b.source(types)
    p1 = b.source(types)
    keys_b = [b.colref(i, GX_TYPE_I64) for i in range(5)]
    keys_p = [b.colref(i, GX_TYPE_I64) for i in range(5)]
    j = b.hashjoin(b1, p1, keys_b, keys_p)
    ex = b.build(j)
    rc = lib.gx_open(ex.ex)
    err = ex.error()
    ex.free()
    b.free()
    assert rc != 0 and "key" in err


def test_long_string_const_rejected():
    def plan(b, lib):
        from tests.gxlib import GX_F_EQ
        src = b.source([GX_TYPE_STRING, GX_TYPE_I64])
        cond = b.call(GX_F_EQ, GX_TYPE_I64, 0,
                      b.colref(0, GX_TYPE_STRING),
                      lib.gx_pb_const_str(b.pb, b"x" * 20, 20))
        sel = b.selection(src, [cond])
        ch = PyChunk([GX_TYPE_STRING, GX_TYPE_I64], 1, None, [64, None])
        return sel, src, [ch]
    _expect_error(plan, "string const", "16")


def test_tuple_outside_distinct_rejected():
    def plan(b, lib):
        from tests.gxlib import GX_F_TUPLE
        src = b.source([GX_TYPE_I64, GX_TYPE_I64])
        tup = b.call(GX_F_TUPLE, GX_TYPE_I64, 0,
                     b.colref(0, GX_TYPE_I64), b.colref(1, GX_TYPE_I64))
        agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                        [(GX_AGG_SUM, tup, 0)])
        return agg, src, _i64_chunk()
    _expect_error(plan, "tuple", "DISTINCT", "decimal",
                  "unsupported function")
