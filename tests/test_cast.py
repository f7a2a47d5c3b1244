"""Cast builtins (builtin_cast_vec.go; ProduceDecWithSpecifiedTp
datum.go:1629): cast(decimal as decimal(s)) rounds HalfUp to the target
frac; cast(decimal as signed) = Round(0, HalfUp) + ToInt; cast(int as
decimal) pads. Exercised through sums so GPU parity is value-level."""
from fractions import Fraction

import pytest

from tests.gxlib import (GX_AGG_COUNT, GX_AGG_SUM, GX_F_CAST_DEC,
                         GX_F_CAST_INT, GX_TPCH_LINEITEM, GX_TYPE_DECIMAL,
                         GX_TYPE_I64, GX_TYPE_STRING, load_oracle)
from tidb_amd import plan as P


def cast_plan(lib, to_int=False):
    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    price = b.colref(P.L_EXTPRICE, GX_TYPE_DECIMAL, 2)
    rf = b.colref(P.L_RETFLAG, GX_TYPE_STRING)
    ls = b.colref(P.L_LINESTATUS, GX_TYPE_STRING)
    if to_int:
        # decimal -> signed int -> decimal(1): both casts on the path
        as_int = b.call(GX_F_CAST_INT, GX_TYPE_I64, 0, price)
        val = b.call(GX_F_CAST_DEC, GX_TYPE_DECIMAL, 1, as_int)
        sr = 1
    else:
        # 90145.00 -> cast to frac 1 -> half-up rounding at the cent digit
        val = b.call(GX_F_CAST_DEC, GX_TYPE_DECIMAL, 1, price)
        sr = 1
    proj = b.projection(src, [rf, ls, val])
    agg = b.hashagg(proj, [b.colref(0, GX_TYPE_STRING),
                           b.colref(1, GX_TYPE_STRING)],
                    [(GX_AGG_SUM, b.colref(2, GX_TYPE_DECIMAL, sr), sr),
                     (GX_AGG_COUNT, -1, 0)])
    out_types = [GX_TYPE_STRING, GX_TYPE_STRING, GX_TYPE_DECIMAL, GX_TYPE_I64]
    out_fracs = [0, 0, sr, 0]
    return b, src, agg, out_types, out_fracs


def run_cast(lib, n_rows, to_int=False):
    b, src, agg, out_types, out_fracs = cast_plan(lib, to_int)
    ex = b.build(agg)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n_rows)
    ex.open()
    caps = [2048 if t == 4 else None for t in out_types]
    rows = ex.pull_all(out_types, out_fracs, data_caps=caps)
    ex.close()
    ex.free()
    b.free()
    return sorted(rows)


def half_up(units, down):
    """round_half_up(units / 10^down) in integer arithmetic."""
    q, r = divmod(abs(units), 10 ** down)
    if 2 * r >= 10 ** down:
        q += 1
    return q if units >= 0 else -q


def test_cast_dec_oracle_vs_python(oracle_lib):
    from tests.test_oracle_q1 import pull_lineitem
    raw = pull_lineitem(oracle_lib, 3000)
    groups = {}
    for r in raw:
        cents = int(Fraction(r[2]) * 100)
        tenths = half_up(cents, 1)  # frac 2 -> frac 1
        key = (r[5], r[6])
        s, c = groups.get(key, (0, 0))
        groups[key] = (s + tenths, c + 1)
    got = run_cast(oracle_lib, 3000)
    assert len(got) == len(groups)
    for rf, ls, s, c in got:
        ws, wc = groups[(rf, ls)]
        assert c == wc and Fraction(s) == Fraction(ws, 10), (rf, ls)


def test_cast_int_oracle_vs_python(oracle_lib):
    from tests.test_oracle_q1 import pull_lineitem
    raw = pull_lineitem(oracle_lib, 3000)
    groups = {}
    for r in raw:
        cents = int(Fraction(r[2]) * 100)
        whole = half_up(cents, 2)  # Round(0, HalfUp) + ToInt
        key = (r[5], r[6])
        s, c = groups.get(key, (0, 0))
        groups[key] = (s + whole * 10, c + 1)  # re-cast to frac 1 pads a zero
    got = run_cast(oracle_lib, 3000, to_int=True)
    for rf, ls, s, c in got:
        ws, wc = groups[(rf, ls)]
        assert c == wc and Fraction(s) == Fraction(ws, 10), (rf, ls)


@pytest.mark.gpu
@pytest.mark.parametrize("to_int", [False, True])
def test_cast_parity(to_int):
    from tests.gxlib import load_product
    assert run_cast(load_oracle(), 50000, to_int) == \
        run_cast(load_product(), 50000, to_int)
