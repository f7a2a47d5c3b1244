"""MIN/MAX aggregates (aggfuncs/func_max_min.go): device accumulation uses
an order-preserving biased-u64 encoding so both extremes run as unsigned
max from a zero-initialized table; NULL args never participate; a group
with no non-null arg yields NULL."""
from fractions import Fraction

import pytest

from tests.gxlib import (GX_AGG_COUNT, GX_AGG_MAX, GX_AGG_MIN,
                         GX_TPCH_LINEITEM, GX_TYPE_DECIMAL, GX_TYPE_I64,
                         GX_TYPE_STRING, load_oracle)
from tidb_amd import plan as P


def mm_plan(lib):
    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    price = b.colref(P.L_EXTPRICE, GX_TYPE_DECIMAL, 2)
    qty = b.colref(P.L_QUANTITY, GX_TYPE_DECIMAL, 2)
    rf = b.colref(P.L_RETFLAG, GX_TYPE_STRING)
    ls = b.colref(P.L_LINESTATUS, GX_TYPE_STRING)
    proj = b.projection(src, [rf, ls, price, qty])
    agg = b.hashagg(proj, [b.colref(0, GX_TYPE_STRING),
                           b.colref(1, GX_TYPE_STRING)],
                    [(GX_AGG_MIN, b.colref(2, GX_TYPE_DECIMAL, 2), 2),
                     (GX_AGG_MAX, b.colref(2, GX_TYPE_DECIMAL, 2), 2),
                     (GX_AGG_MIN, b.colref(3, GX_TYPE_DECIMAL, 2), 2),
                     (GX_AGG_MAX, b.colref(3, GX_TYPE_DECIMAL, 2), 2),
                     (GX_AGG_COUNT, -1, 0)])
    out_types = ([GX_TYPE_STRING, GX_TYPE_STRING] + [GX_TYPE_DECIMAL] * 4 +
                 [GX_TYPE_I64])
    out_fracs = [0, 0, 2, 2, 2, 2, 0]
    return b, src, agg, out_types, out_fracs


def run_mm(lib, n_rows):
    b, src, agg, out_types, out_fracs = mm_plan(lib)
    ex = b.build(agg)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n_rows)
    ex.open()
    caps = [2048 if t == 4 else None for t in out_types]
    rows = ex.pull_all(out_types, out_fracs, data_caps=caps)
    ex.close()
    ex.free()
    b.free()
    return sorted(rows)


def test_minmax_oracle_vs_python(oracle_lib):
    from tests.test_oracle_q1 import pull_lineitem
    raw = pull_lineitem(oracle_lib, 3000)
    want = {}
    for r in raw:
        k = (r[5], r[6])
        p, q = Fraction(r[2]), Fraction(r[1])
        lo_p, hi_p, lo_q, hi_q, c = want.get(k, (p, p, q, q, 0))
        want[k] = (min(lo_p, p), max(hi_p, p), min(lo_q, q), max(hi_q, q),
                   c + 1)
    got = run_mm(oracle_lib, 3000)
    assert len(got) == len(want)
    for rf, ls, mnp, mxp, mnq, mxq, c in got:
        w = want[(rf, ls)]
        assert (Fraction(mnp), Fraction(mxp), Fraction(mnq), Fraction(mxq),
                c) == w


def test_minmax_nulls(oracle_lib):
    """A group whose arg rows are all NULL yields NULL min/max but counts
    its rows in count(*)."""
    from tidb_amd.chunkpy import PyChunk
    from tidb_amd.decimals import str_to_decimal_bytes
    lib = oracle_lib
    d = lambda s: str_to_decimal_bytes(lib, s)
    t = lib.gx_time_from_date(1995, 1, 1)
    rows = [
        (1, d("2.00"), None, d("0.00"), d("0.00"), "A", "F", t),
        (2, d("4.00"), None, d("0.00"), d("0.00"), "A", "F", t),
        (3, d("6.00"), d("-12.34"), d("0.00"), d("0.00"), "N", "O", t),
        (4, d("8.00"), d("5.00"), d("0.00"), d("0.00"), "N", "O", t),
    ]
    chunk = PyChunk(P.LINEITEM_TYPES, len(rows), P.LINEITEM_FRACS,
                    data_caps=[None] * 5 + [16, 16] + [None])
    for r in rows:
        chunk.append_row(list(r))
    b, src, agg, out_types, out_fracs = mm_plan(lib)
    ex = b.build(agg)
    ex.bind_chunks(src, [chunk])
    ex.open()
    got = sorted(ex.pull_all(out_types, out_fracs,
                             data_caps=[2048, 2048, None, None, None, None,
                                        None]))
    ex.close()
    ex.free()
    b.free()
    af = [r for r in got if r[0] == "A"][0]
    no = [r for r in got if r[0] == "N"][0]
    assert af[2] is None and af[3] is None and af[6] == 2
    assert Fraction(no[2]) == Fraction("-12.34")
    assert Fraction(no[3]) == Fraction("5.00")


@pytest.mark.gpu
def test_minmax_parity():
    from tests.gxlib import load_product
    assert run_mm(load_oracle(), 50000) == run_mm(load_product(), 50000)


@pytest.mark.gpu
def test_minmax_parity_interpreted(monkeypatch):
    from tests.gxlib import load_product
    monkeypatch.setenv("GX_NO_JIT", "1")
    got = run_mm(load_product(), 20000)
    monkeypatch.delenv("GX_NO_JIT")
    assert got == run_mm(load_oracle(), 20000)


@pytest.mark.gpu
def test_minmax_nulls_parity():
    from tests.gxlib import load_product
    a = _run_nulls(load_oracle())
    b = _run_nulls(load_product())
    assert a == b


def _run_nulls(lib):
    from tidb_amd.chunkpy import PyChunk
    from tidb_amd.decimals import str_to_decimal_bytes
    d = lambda s: str_to_decimal_bytes(lib, s)
    t = lib.gx_time_from_date(1995, 1, 1)
    rows = [
        (1, d("2.00"), None, d("0.00"), d("0.00"), "A", "F", t),
        (2, d("4.00"), None, d("0.00"), d("0.00"), "A", "F", t),
        (3, d("6.00"), d("-12.34"), d("0.00"), d("0.00"), "N", "O", t),
        (4, d("8.00"), d("5.00"), d("0.00"), d("0.00"), "N", "O", t),
    ]
    chunk = PyChunk(P.LINEITEM_TYPES, len(rows), P.LINEITEM_FRACS,
                    data_caps=[None] * 5 + [16, 16] + [None])
    for r in rows:
        chunk.append_row(list(r))
    b, src, agg, out_types, out_fracs = mm_plan(lib)
    ex = b.build(agg)
    ex.bind_chunks(src, [chunk])
    ex.open()
    got = sorted(ex.pull_all(out_types, out_fracs,
                             data_caps=[2048, 2048, None, None, None, None,
                                        None]))
    ex.close()
    ex.free()
    b.free()
    return got
