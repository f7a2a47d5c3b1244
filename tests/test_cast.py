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


def _run_round_abs(lib):
    """ROUND(x, d) (builtinRoundWithFracDecSig: scale min(d, ret frac),
    half away from zero) and ABS (builtinAbsDecSig/IntSig), standalone
    projection + as a sum arg in the fused pipeline."""
    import ctypes

    import numpy as np

    from tests.gxlib import (GX_AGG_SUM, GX_F_ABS, GX_F_ROUND,
                             GX_TYPE_DECIMAL, GX_TYPE_I64)
    from tidb_amd import plan as P
    from tidb_amd.chunkpy import PyChunk

    def dec(s):
        out = (ctypes.c_uint8 * 40)()
        assert lib.gx_dec_from_string(s.encode(), len(s.encode()), out) == 0
        return bytes(out)

    rng = np.random.default_rng(31)
    rows = []
    for i in range(3000):
        sign = "-" if rng.random() < 0.5 else ""
        rows.append((int(rng.integers(0, 9)),
                     f"{sign}{int(rng.integers(0, 999))}.{int(rng.integers(0, 1000)):03d}",
                     int(rng.integers(-(10 ** 6), 10 ** 6))))
    chunks = []
    types = [GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64]
    for base in range(0, len(rows), 1000):
        part = rows[base:base + 1000]
        ch = PyChunk(types, len(part), [0, 3, 0])
        for k, d, v in part:
            ch.append_row([k, dec(d), v])
        chunks.append(ch)

    b = P.Builder(lib)
    src = b.source(types, [0, 3, 0])
    x = b.colref(1, GX_TYPE_DECIMAL, 3)
    proj = b.projection(src, [
        b.call(GX_F_ROUND, GX_TYPE_DECIMAL, 2, x, b.const_i64(2)),
        b.call(GX_F_ROUND, GX_TYPE_DECIMAL, 3, x, b.const_i64(0)),
        b.call(GX_F_ABS, GX_TYPE_DECIMAL, 3, x),
        b.call(GX_F_ABS, GX_TYPE_I64, 0, b.colref(2, GX_TYPE_I64)),
    ])
    ex = b.build(proj)
    ex.bind_chunks(src, chunks)
    ex.open()
    out1 = ex.pull_all([GX_TYPE_DECIMAL, GX_TYPE_DECIMAL, GX_TYPE_DECIMAL,
                        GX_TYPE_I64], [2, 3, 3, 0])
    ex.close()
    ex.free()

    agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                    [(GX_AGG_SUM, b.call(GX_F_ABS, GX_TYPE_DECIMAL, 3, x),
                      3)])
    ex = b.build(agg)
    ex.bind_chunks(src, chunks)
    ex.open()
    out2 = sorted(ex.pull_all([GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 3]))
    ex.close()
    ex.free()
    b.free()
    return rows, out1, out2


def test_oracle_round_abs():
    from decimal import ROUND_HALF_UP, Decimal
    lib = load_oracle()
    rows, out1, out2 = _run_round_abs(lib)
    for (r2, r0, ad, ai), (k, d, v) in zip(out1, rows):
        D = Decimal(d)
        # mydecimal ModeHalfUp = half AWAY from zero (python ROUND_HALF_UP)
        assert Decimal(r2) == D.quantize(Decimal("0.01"), ROUND_HALF_UP)
        # ROUND(x, 0) at ret frac 3 displays 3 decimals of the integer value
        assert Decimal(r0) == D.quantize(Decimal("1"), ROUND_HALF_UP)
        assert Decimal(ad) == abs(D)
        assert ai == abs(v)
    want = {}
    for k, d, v in rows:
        want[k] = want.get(k, Decimal(0)) + abs(Decimal(d))
    assert out2 == sorted((k, str(s.quantize(Decimal("0.001"))))
                          for k, s in want.items())


@pytest.mark.gpu
def test_round_abs_parity():
    from tests.gxlib import load_product
    _, w1, w2 = _run_round_abs(load_oracle())
    _, g1, g2 = _run_round_abs(load_product())
    assert g1 == w1
    assert g2 == w2
