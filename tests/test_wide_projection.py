"""Wide projection (BASELINE config 5's shape, scaled to round-1 VM
capacity): ~20 chained decimal expressions over 4 loaded columns feeding 12
aggregates — exceeds the VM's 12 physical registers unless the compiler
recycles dead subexpression registers."""
from fractions import Fraction

import pytest

from tests.gxlib import (GX_AGG_COUNT, GX_AGG_SUM, GX_F_CAST_DEC, GX_F_MINUS,
                         GX_F_MUL, GX_F_PLUS, GX_TPCH_LINEITEM,
                         GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_STRING,
                         load_oracle)
from tidb_amd import plan as P


def wide_plan(lib):
    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    qty = b.colref(P.L_QUANTITY, GX_TYPE_DECIMAL, 2)
    price = b.colref(P.L_EXTPRICE, GX_TYPE_DECIMAL, 2)
    disc = b.colref(P.L_DISCOUNT, GX_TYPE_DECIMAL, 2)
    tax = b.colref(P.L_TAX, GX_TYPE_DECIMAL, 2)
    rf = b.colref(P.L_RETFLAG, GX_TYPE_STRING)
    ls = b.colref(P.L_LINESTATUS, GX_TYPE_STRING)
    one = P._const_dec_one(lib, b)

    def mul(a, bb, fr):
        return b.call(GX_F_MUL, GX_TYPE_DECIMAL, fr, a, bb)

    def add(a, bb, fr):
        return b.call(GX_F_PLUS, GX_TYPE_DECIMAL, fr, a, bb)

    def sub(a, bb, fr):
        return b.call(GX_F_MINUS, GX_TYPE_DECIMAL, fr, a, bb)

    # a chain of derived expressions, each consumed once (dead after use)
    e1 = sub(one, disc, 2)            # 1-d
    e2 = add(one, tax, 2)             # 1+t
    e3 = mul(price, e1, 4)            # p(1-d)
    e4 = mul(e3, e2, 6)               # charge
    e5 = add(qty, price, 2)
    e6 = sub(price, qty, 2)
    e7 = mul(e5, e6, 4)               # (q+p)(p-q)
    e8 = mul(qty, qty, 4)
    e9 = add(e8, e7, 4)
    e10 = b.call(GX_F_CAST_DEC, GX_TYPE_DECIMAL, 2, e9)
    e11 = mul(disc, tax, 4)
    e12 = add(e11, e11, 4)
    e13 = mul(qty, tax, 4)
    e14 = sub(e13, e11, 4)
    e15 = mul(e10, e2, 4)  # cast result consumed, not pinned as a root
    # 15 expression nodes over 4 loads + 1 const; peak liveness stays within
    # the VM's 12 physical registers only because dead subexpressions
    # (e1, e2, e5, e6, e8, e10, e11, e13) release their registers
    outs = [rf, ls, e4, e9, e12, e14, e15, qty, price]
    proj = b.projection(src, outs)
    fr_of = [None, None, 6, 4, 4, 4, 4, 2, 2]
    aggs = []
    for i in range(2, 9):
        aggs.append((GX_AGG_SUM, b.colref(i, GX_TYPE_DECIMAL, fr_of[i]),
                     fr_of[i]))
    aggs.append((GX_AGG_COUNT, -1, 0))
    agg = b.hashagg(proj, [b.colref(0, GX_TYPE_STRING),
                           b.colref(1, GX_TYPE_STRING)], aggs)
    out_types = ([GX_TYPE_STRING, GX_TYPE_STRING] + [GX_TYPE_DECIMAL] * 7 +
                 [GX_TYPE_I64])
    out_fracs = [0, 0] + [fr_of[i] for i in range(2, 9)] + [0]
    return b, src, agg, out_types, out_fracs


def run_wide(lib, n_rows):
    b, src, agg, out_types, out_fracs = wide_plan(lib)
    ex = b.build(agg)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n_rows)
    ex.open()
    caps = [2048 if t == 4 else None for t in out_types]
    rows = ex.pull_all(out_types, out_fracs, data_caps=caps)
    ex.close()
    ex.free()
    b.free()
    return sorted(rows)


def test_wide_oracle_spotcheck(oracle_lib):
    from tests.test_oracle_q1 import pull_lineitem
    rows = run_wide(oracle_lib, 2000)
    raw = pull_lineitem(oracle_lib, 2000)
    # spot-check agg sum(qty^2 + (q+p)(p-q)) per group (output col 3)
    want = {}
    for r in raw:
        q, p = Fraction(r[1]), Fraction(r[2])
        k = (r[5], r[6])
        want[k] = want.get(k, 0) + q * q + (q + p) * (p - q)
    for row in rows:
        assert Fraction(row[3]) == want[(row[0], row[1])]


@pytest.mark.gpu
def test_wide_parity():
    from tests.gxlib import load_product
    assert run_wide(load_oracle(), 50000) == run_wide(load_product(), 50000)


@pytest.mark.gpu
def test_wide_parity_interpreted(monkeypatch):
    from tests.gxlib import load_product
    monkeypatch.setenv("GX_NO_JIT", "1")
    got = run_wide(load_product(), 20000)
    monkeypatch.delenv("GX_NO_JIT")
    assert got == run_wide(load_oracle(), 20000)
