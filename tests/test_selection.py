"""Standalone Selection operator (SelectionExec, select.go:750-785): CNF
filter over one source, survivors emitted in row order. Device shape: no
row-at-a-time AppendRow copy — survivor indices compact wave-aggregated,
columns (incl. varlen + null bitmaps) gather through the index.
"""
import pytest

from tests.gxlib import (GX_F_EQ, GX_F_GT, GX_F_LT, GX_TPCH_CUSTOMER,
                         GX_TPCH_LINEITEM, GX_TYPE_DECIMAL, GX_TYPE_I64,
                         GX_TYPE_STRING, GX_TYPE_TIME, load_oracle,
                         load_product)
from tidb_amd import plan as P


def _li_selection_plan(lib, topn=None):
    """l_shipdate > 1995-03-15 AND l_orderkey < l_quantity-scaled bound AND
    l_returnflag = 'A' — mixed const/col conjuncts over the lineitem shape."""
    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    c1 = b.call(GX_F_GT, GX_TYPE_I64, 0, b.colref(P.L_SHIPDATE, GX_TYPE_TIME),
                b.const_time(lib.gx_time_from_date(1995, 3, 15)))
    c2 = b.call(GX_F_EQ, GX_TYPE_I64, 0, b.colref(P.L_RETFLAG, GX_TYPE_STRING),
                lib.gx_pb_const_str(b.pb, b"A", 1))
    root = b.selection(src, [c1, c2])
    if topn is not None:
        root = b.topn(root, [b.colref(P.L_ORDERKEY, GX_TYPE_I64),
                             b.colref(P.L_SHIPDATE, GX_TYPE_TIME)],
                      [0, 0], topn)
    return b, src, root


def run_li_selection(lib, n_rows, topn=None):
    b, src, root = _li_selection_plan(lib, topn)
    ex = b.build(root)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n_rows)
    ex.open()
    rows = ex.pull_all(P.LINEITEM_TYPES, P.LINEITEM_FRACS,
                       data_caps=[None] * 5 + [2048, 2048] + [None])
    ex.close()
    ex.free()
    b.free()
    return rows


def test_oracle_selection():
    from tests.test_oracle_q1 import pull_lineitem
    lib = load_oracle()
    raw = pull_lineitem(lib, 8000)
    cut = (1995 << 50) | (3 << 46) | (15 << 41)
    want = [r for r in raw if (r[7] & ~0xF) > cut and r[5] == "A"]
    got = run_li_selection(lib, 8000)
    assert got == want
    assert 0 < len(got) < 8000


@pytest.mark.gpu
def test_selection_parity():
    """Survivors preserve ROW ORDER (SelectionExec appends in input order) —
    exact list compare, not a multiset."""
    a = run_li_selection(load_oracle(), 60000)
    b = run_li_selection(load_product(), 60000)
    assert len(a) == len(b) > 1000
    assert a == b


@pytest.mark.gpu
def test_selection_topn_parity():
    a = run_li_selection(load_oracle(), 30000, topn=100)
    b = run_li_selection(load_product(), 30000, topn=100)
    assert len(a) == len(b) == 100
    assert a == b


@pytest.mark.gpu
def test_selection_varlen_parity():
    """Varlen output (c_mktsegment) + col-cmp-const on the key column."""

    def run(lib):
        b = P.Builder(lib)
        src = b.source(P.CUSTOMER_TYPES)
        cond = b.call(GX_F_LT, GX_TYPE_I64, 0,
                      b.colref(P.C_CUSTKEY, GX_TYPE_I64), b.const_i64(400))
        root = b.selection(src, [cond])
        ex = b.build(root)
        ex.bind_tpch(src, GX_TPCH_CUSTOMER, 1000)
        ex.open()
        rows = ex.pull_all(P.CUSTOMER_TYPES, [0, 0],
                           data_caps=[None, 65536])
        ex.close()
        ex.free()
        b.free()
        return rows

    a = run(load_oracle())
    b = run(load_product())
    assert len(a) == len(b) > 100
    assert a == b


def _run_or_selection(lib, n=3000):
    """Disjunctive conjuncts (LogicOr inside the CNF): WHERE (qty < 10.00
    OR retflag = 'R' OR qty > 45.00) AND shipdate > 1995-03-15."""
    import ctypes

    from tests.gxlib import GX_F_OR

    def dec(s):
        out = (ctypes.c_uint8 * 40)()
        assert lib.gx_dec_from_string(s.encode(), len(s.encode()), out) == 0
        return bytes(out)

    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    qty = b.colref(P.L_QUANTITY, GX_TYPE_DECIMAL, 2)
    c_or = b.call(GX_F_OR, GX_TYPE_I64, 0,
                  b.call(GX_F_OR, GX_TYPE_I64, 0,
                         b.call(GX_F_LT, GX_TYPE_I64, 0, qty,
                                b.const_dec(dec("10.00"))),
                         b.call(GX_F_EQ, GX_TYPE_I64, 0,
                                b.colref(P.L_RETFLAG, GX_TYPE_STRING),
                                lib.gx_pb_const_str(b.pb, b"R", 1))),
                  b.call(GX_F_GT, GX_TYPE_I64, 0, qty,
                         b.const_dec(dec("45.00"))))
    c_date = b.call(GX_F_GT, GX_TYPE_I64, 0,
                    b.colref(P.L_SHIPDATE, GX_TYPE_TIME),
                    b.const_time(lib.gx_time_from_date(1995, 3, 15)))
    root = b.selection(src, [c_or, c_date])
    ex = b.build(root)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n)
    ex.open()
    rows = ex.pull_all(P.LINEITEM_TYPES, P.LINEITEM_FRACS,
                       data_caps=[None] * 5 + [2048, 2048] + [None])
    ex.close()
    ex.free()
    b.free()
    return rows


def test_oracle_or_selection():
    from decimal import Decimal
    lib = load_oracle()
    rows = _run_or_selection(lib)
    from tests.test_oracle_q1 import pull_lineitem
    raw = pull_lineitem(lib, 3000)
    cut = (1995 << 50) | (3 << 46) | (15 << 41)
    want = [r for r in raw
            if (Decimal(r[1]) < 10 or r[5] == "R" or Decimal(r[1]) > 45)
            and (r[7] & ~0xF) > cut]
    assert rows == want
    assert 0 < len(rows) < len(raw)


@pytest.mark.gpu
def test_or_selection_parity():
    want = _run_or_selection(load_oracle())
    got = _run_or_selection(load_product())
    assert got == want


def _run_or_agg(lib, n=3000):
    """OR inside the FUSED CNF: count(*) group by retflag where
    (qty < 10.00 OR qty > 45.00) AND shipdate > 1995-03-15 — the hipRTC
    generator declines disjunctions, the interpreted kernel groups them."""
    import ctypes

    from tests.gxlib import GX_AGG_COUNT, GX_F_OR

    def dec(s):
        out = (ctypes.c_uint8 * 40)()
        assert lib.gx_dec_from_string(s.encode(), len(s.encode()), out) == 0
        return bytes(out)

    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    qty = b.colref(P.L_QUANTITY, GX_TYPE_DECIMAL, 2)
    c_or = b.call(GX_F_OR, GX_TYPE_I64, 0,
                  b.call(GX_F_LT, GX_TYPE_I64, 0, qty,
                         b.const_dec(dec("10.00"))),
                  b.call(GX_F_GT, GX_TYPE_I64, 0, qty,
                         b.const_dec(dec("45.00"))))
    c_date = b.call(GX_F_GT, GX_TYPE_I64, 0,
                    b.colref(P.L_SHIPDATE, GX_TYPE_TIME),
                    b.const_time(lib.gx_time_from_date(1995, 3, 15)))
    sel = b.selection(src, [c_or, c_date])
    agg = b.hashagg(sel, [b.colref(P.L_RETFLAG, GX_TYPE_STRING)],
                    [(GX_AGG_COUNT, -1, 0)])
    ex = b.build(agg)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n)
    ex.open()
    rows = sorted(ex.pull_all([GX_TYPE_STRING, GX_TYPE_I64], [0, 0],
                              data_caps=[1 << 12, None]))
    ex.close()
    ex.free()
    b.free()
    return rows


def test_oracle_or_agg():
    from decimal import Decimal
    lib = load_oracle()
    rows = _run_or_agg(lib)
    from tests.test_oracle_q1 import pull_lineitem
    raw = pull_lineitem(lib, 3000)
    cut = (1995 << 50) | (3 << 46) | (15 << 41)
    want = {}
    for r in raw:
        if (Decimal(r[1]) < 10 or Decimal(r[1]) > 45) and (r[7] & ~0xF) > cut:
            want[r[5]] = want.get(r[5], 0) + 1
    assert rows == sorted(want.items())
    assert len(rows) >= 2


@pytest.mark.gpu
def test_or_agg_parity():
    want = _run_or_agg(load_oracle())
    got = _run_or_agg(load_product())
    assert got == want
