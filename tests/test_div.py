"""Decimal division builtin (builtin_arithmetic_vec.go:67 divide sig;
types.DecimalDiv mydecimal.go:1311 / doDiv:1168): quotient truncated at the
word-granular result scale (scale-2 / scale-2 -> 18 frac digits), division
by zero -> NULL. Exercised through sum(price/qty) so parity is value-level."""
from fractions import Fraction

import pytest

from tests.gxlib import (GX_AGG_COUNT, GX_AGG_SUM, GX_F_DIV,
                         GX_TPCH_LINEITEM, GX_TYPE_DECIMAL, GX_TYPE_I64,
                         GX_TYPE_STRING, load_oracle)
from tidb_amd import plan as P

SR = 18  # word-granular result scale for scale-2 / scale-2


def div_plan(lib):
    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    price = b.colref(P.L_EXTPRICE, GX_TYPE_DECIMAL, 2)
    qty = b.colref(P.L_QUANTITY, GX_TYPE_DECIMAL, 2)
    rf = b.colref(P.L_RETFLAG, GX_TYPE_STRING)
    ls = b.colref(P.L_LINESTATUS, GX_TYPE_STRING)
    ratio = b.call(GX_F_DIV, GX_TYPE_DECIMAL, SR, price, qty)
    proj = b.projection(src, [rf, ls, ratio])
    prf = b.colref(0, GX_TYPE_STRING)
    pls = b.colref(1, GX_TYPE_STRING)
    pr = b.colref(2, GX_TYPE_DECIMAL, SR)
    agg = b.hashagg(proj, [prf, pls],
                    [(GX_AGG_SUM, pr, SR), (GX_AGG_COUNT, pr, 0)])
    out_types = [GX_TYPE_STRING, GX_TYPE_STRING, GX_TYPE_DECIMAL, GX_TYPE_I64]
    out_fracs = [0, 0, SR, 0]
    return b, src, agg, out_types, out_fracs


def run_div(lib, n_rows):
    b, src, agg, out_types, out_fracs = div_plan(lib)
    ex = b.build(agg)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n_rows)
    ex.open()
    caps = [2048 if t == 4 else None for t in out_types]
    rows = ex.pull_all(out_types, out_fracs, data_caps=caps)
    ex.close()
    ex.free()
    b.free()
    return sorted(rows)


def test_div_oracle_vs_python(oracle_lib):
    """Independent check: truncated quotient at scale 18, summed exactly."""
    from tests.test_oracle_q1 import pull_lineitem
    raw = pull_lineitem(oracle_lib, 3000)
    groups = {}
    for r in raw:
        price = Fraction(r[2]) * 10**2
        qty = Fraction(r[1]) * 10**2
        q_units = (int(price) * 10**SR) // int(qty)  # both positive
        key = (r[5], r[6])
        s, c = groups.get(key, (0, 0))
        groups[key] = (s + q_units, c + 1)
    got = run_div(oracle_lib, 3000)
    assert len(got) == len(groups)
    for rf, ls, s, c in got:
        ws, wc = groups[(rf, ls)]
        assert c == wc
        assert Fraction(s) == Fraction(ws, 10**SR), (rf, ls)


def test_div_by_zero_null(oracle_lib):
    """qty = 0 divisor -> NULL; NULL never enters the sum or its count."""
    from tidb_amd.chunkpy import PyChunk
    from tidb_amd.decimals import str_to_decimal_bytes
    lib = oracle_lib
    d = lambda s: str_to_decimal_bytes(lib, s)
    t = lib.gx_time_from_date(1995, 1, 1)
    rows = [
        (1, d("2.00"), d("10.00"), d("0.00"), d("0.00"), "A", "F", t),
        (2, d("0.00"), d("30.00"), d("0.00"), d("0.00"), "A", "F", t),
        (3, d("4.00"), d("10.00"), d("0.00"), d("0.00"), "A", "F", t),
    ]
    chunk = PyChunk(P.LINEITEM_TYPES, len(rows), P.LINEITEM_FRACS,
                    data_caps=[None] * 5 + [16, 16] + [None])
    for r in rows:
        chunk.append_row(list(r))
    b, src, agg, out_types, out_fracs = div_plan(lib)
    ex = b.build(agg)
    ex.bind_chunks(src, [chunk])
    ex.open()
    got = ex.pull_all(out_types, out_fracs,
                      data_caps=[2048, 2048, None, None])
    ex.close()
    ex.free()
    b.free()
    assert len(got) == 1
    rf, ls, s, c = got[0]
    assert c == 2  # the div-by-zero row is NULL, not counted
    assert Fraction(s) == Fraction(10**SR * 5 + 10**SR * 5 // 2, 10**SR)


@pytest.mark.gpu
def test_div_parity():
    from tests.gxlib import load_product
    assert run_div(load_oracle(), 50000) == run_div(load_product(), 50000)


@pytest.mark.gpu
def test_q1_wide_vm_parity(monkeypatch):
    """Force the int128 (wide) VM for plain Q1: covers the retry-on-overflow
    kernel that narrow-friendly data otherwise never exercises."""
    import os
    from tests.gxlib import load_product
    from tests.test_gpu_parity import _run_q1, _as_map
    monkeypatch.setenv("GX_FORCE_WIDE", "1")
    got = _as_map(_run_q1(load_product(), 65536))
    monkeypatch.delenv("GX_FORCE_WIDE")
    want = _as_map(_run_q1(load_oracle(), 65536))
    assert got == want
