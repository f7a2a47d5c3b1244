"""Standalone Projection operator (ProjectionExec, projection.go:77):
computed expressions materialize as device columns — decimals encode to the
canonical 40-byte MyDecimal struct ON DEVICE — while passthrough column
references alias the input buffers. Parity bar: decimal DISPLAY equality
(same digits, same frac) with the oracle's per-row builtin results.
"""
from fractions import Fraction

import pytest

from tests.gxlib import (GX_F_GT, GX_F_MINUS, GX_F_MUL, GX_TPCH_LINEITEM,
                         GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_STRING,
                         GX_TYPE_TIME, load_oracle, load_product)
from tidb_amd import plan as P


def _proj_plan(lib, with_selection=False, with_cast=False):
    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    child = src
    if with_selection:
        cond = b.call(GX_F_GT, GX_TYPE_I64, 0,
                      b.colref(P.L_SHIPDATE, GX_TYPE_TIME),
                      b.const_time(lib.gx_time_from_date(1995, 3, 15)))
        child = b.selection(src, [cond])
    price = b.colref(P.L_EXTPRICE, GX_TYPE_DECIMAL, 2)
    disc = b.colref(P.L_DISCOUNT, GX_TYPE_DECIMAL, 2)
    one = P._const_dec_one(lib, b)
    om_d = b.call(GX_F_MINUS, GX_TYPE_DECIMAL, 2, one, disc)
    rev = b.call(GX_F_MUL, GX_TYPE_DECIMAL, 4, price, om_d)
    exprs = [b.colref(P.L_ORDERKEY, GX_TYPE_I64), rev, price,
             b.colref(P.L_RETFLAG, GX_TYPE_STRING)]
    out_types = [GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_DECIMAL,
                 GX_TYPE_STRING]
    out_fracs = [0, 4, 2, 0]
    if with_cast:
        from tests.gxlib import GX_F_CAST_INT
        exprs.append(b.call(GX_F_CAST_INT, GX_TYPE_I64, 0, rev))
        out_types.append(GX_TYPE_I64)
        out_fracs.append(0)
    proj = b.projection(child, exprs)
    return b, src, proj, out_types, out_fracs


def run_proj(lib, n_rows, with_selection=False, with_cast=False):
    b, src, proj, out_types, out_fracs = _proj_plan(lib, with_selection,
                                                    with_cast)
    ex = b.build(proj)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n_rows)
    ex.open()
    rows = ex.pull_all(out_types, out_fracs,
                       data_caps=[None, None, None, 2048] + [None])
    ex.close()
    ex.free()
    b.free()
    return rows


def test_oracle_projection():
    from tests.test_oracle_q1 import pull_lineitem
    lib = load_oracle()
    raw = pull_lineitem(lib, 4000)
    got = run_proj(lib, 4000)
    assert len(got) == 4000
    for r, g in zip(raw, got):
        assert g[0] == r[0]
        assert Fraction(g[1]) == Fraction(r[2]) * (1 - Fraction(r[3]))
        assert g[2] == r[2]
        assert g[3] == r[5]
        # scale of the product is f1+f2 = 4 digits
        assert len(g[1].split(".")[1]) == 4


@pytest.mark.gpu
def test_projection_parity():
    """Row-for-row equality incl. decimal DISPLAY (device MyDecimal encode
    vs the oracle's DecimalMul/Sub outputs)."""
    a = run_proj(load_oracle(), 50000)
    b = run_proj(load_product(), 50000)
    assert len(a) == len(b) == 50000
    assert a == b


@pytest.mark.gpu
def test_projection_over_selection_parity():
    a = run_proj(load_oracle(), 30000, with_selection=True)
    b = run_proj(load_product(), 30000, with_selection=True)
    assert len(a) == len(b) > 1000
    assert a == b


@pytest.mark.gpu
def test_projection_cast_parity():
    """cast(rev as signed) — ROUND_SCALE to 0 + int64 output column."""
    a = run_proj(load_oracle(), 20000, with_cast=True)
    b = run_proj(load_product(), 20000, with_cast=True)
    assert len(a) == len(b) == 20000
    assert a == b


def _null_proj_run(lib):
    """NULL propagation through computed expressions (NULL operand -> NULL
    result; builtin_arithmetic_vec.go null merge) + NULL passthrough."""
    import ctypes
    from tidb_amd.chunkpy import PyChunk

    def decb(s):
        out = (ctypes.c_uint8 * 40)()
        assert lib.gx_dec_from_string(s.encode(), len(s.encode()), out) == 0
        return bytes(out)

    b = P.Builder(lib)
    src = b.source([GX_TYPE_DECIMAL, GX_TYPE_DECIMAL, GX_TYPE_I64], [2, 2, 0])
    x = b.colref(0, GX_TYPE_DECIMAL, 2)
    y = b.colref(1, GX_TYPE_DECIMAL, 2)
    prod = b.call(GX_F_MUL, GX_TYPE_DECIMAL, 4, x, y)
    proj = b.projection(src, [prod, b.colref(2, GX_TYPE_I64), x])
    ex = b.build(proj)
    ch = PyChunk([GX_TYPE_DECIMAL, GX_TYPE_DECIMAL, GX_TYPE_I64], 8, [2, 2, 0])
    rows = [("1.50", "2.00", 7), (None, "3.00", None), ("-0.25", None, 1),
            (None, None, 0), ("10.00", "0.10", -5)]
    for r in rows:
        ch.append_row([None if r[0] is None else decb(r[0]),
                       None if r[1] is None else decb(r[1]), r[2]])
    ex.bind_chunks(src, [ch])
    ex.open()
    got = ex.pull_all([GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_DECIMAL],
                      [4, 0, 2])
    ex.close()
    ex.free()
    b.free()
    return got


def test_oracle_projection_nulls():
    lib = load_oracle()
    got = _null_proj_run(lib)
    assert got == [("3.0000", 7, "1.50"), (None, None, None),
                   (None, 1, "-0.25"), (None, 0, None),
                   ("1.0000", -5, "10.00")]


@pytest.mark.gpu
def test_projection_nulls_parity():
    assert _null_proj_run(load_product()) == _null_proj_run(load_oracle())
