"""Randomized decimal expression fuzz (CPU): random +,-,* trees over
decimal columns with mixed scales, oracle projection vs exact Fraction
arithmetic (these ops are exact in MyDecimal; division's word-granular
truncation is pinned separately in test_div/test_golden_sql)."""
import ctypes
from fractions import Fraction

import numpy as np
import pytest

from tests.gxlib import GX_TYPE_DECIMAL, GX_TYPE_I64, load_oracle

GX_F_PLUS, GX_F_MINUS, GX_F_MUL = 16, 17, 18

from tidb_amd import plan as P
from tidb_amd.chunkpy import PyChunk


def _dec(lib, s):
    out = (ctypes.c_uint8 * 40)()
    assert lib.gx_dec_from_string(s.encode(), len(s.encode()), out) == 0
    return bytes(out)


def _rand_expr(rng, b, cols, scales, depth):
    """Returns (expr_id, python_fn, scale)."""
    if depth == 0 or rng.random() < 0.3:
        if rng.random() < 0.25:
            sc = int(rng.integers(0, 4))
            val = f"{int(rng.integers(-999, 1000))}.{int(rng.integers(0, 10**sc)):0{sc}d}" if sc else str(int(rng.integers(-999, 1000)))
            lib = b.lib
            e = b.const_dec(_dec_cached(lib, val))
            f = Fraction(val)
            return e, (lambda row, f=f: f), sc
        i = int(rng.integers(0, len(cols)))
        return (b.colref(cols[i], GX_TYPE_DECIMAL, scales[i]),
                (lambda row, i=i: row[i]), scales[i])
    op = int(rng.integers(0, 3))
    la, lf, ls = _rand_expr(rng, b, cols, scales, depth - 1)
    ra, rf, rs = _rand_expr(rng, b, cols, scales, depth - 1)
    func = [GX_F_PLUS, GX_F_MINUS, GX_F_MUL][op]
    sc = ls + rs if func == GX_F_MUL else max(ls, rs)
    e = b.call(func, GX_TYPE_DECIMAL, sc, la, ra)
    if func == GX_F_PLUS:
        fn = lambda row: None if (lf(row) is None or rf(row) is None) \
            else lf(row) + rf(row)
    elif func == GX_F_MINUS:
        fn = lambda row: None if (lf(row) is None or rf(row) is None) \
            else lf(row) - rf(row)
    else:
        fn = lambda row: None if (lf(row) is None or rf(row) is None) \
            else lf(row) * rf(row)
    return e, fn, sc


_dc = {}


def _dec_cached(lib, s):
    if s not in _dc:
        _dc[s] = _dec(lib, s)
    return _dc[s]


@pytest.mark.parametrize("seed", [5, 23, 77, 131])
def test_fuzz_decimal_exprs(seed):
    lib = load_oracle()
    rng = np.random.default_rng(seed)
    scales = [int(rng.integers(0, 5)) for _ in range(3)]
    types = [GX_TYPE_DECIMAL] * 3
    n = 400
    data = []
    for i in range(n):
        row = []
        for sc in scales:
            if rng.random() < 0.1:
                row.append(None)
            else:
                iv = int(rng.integers(-10**5, 10**5))
                row.append(Fraction(iv, 10**sc))
        data.append(row)

    b = P.Builder(lib)
    src = b.source(types, scales)
    exprs, fns, out_scales = [], [], []
    for _ in range(4):
        e, f, sc = _rand_expr(rng, b, [0, 1, 2], scales, 3)
        exprs.append(e)
        fns.append(f)
        out_scales.append(sc)
    proj = b.projection(src, exprs)
    ex = b.build(proj)
    ch = PyChunk(types, n, scales)
    for row in data:
        vals = []
        for v, sc in zip(row, scales):
            if v is None:
                vals.append(None)
            else:
                q = v * 10**sc
                s = f"{'-' if q < 0 else ''}{abs(q.numerator)//10**sc}"
                if sc:
                    s += f".{abs(q.numerator) % 10**sc:0{sc}d}"
                vals.append(_dec_cached(lib, s))
        ch.append_row(vals)
    ex.bind_chunks(src, [ch])
    ex.open()
    got = ex.pull_all([GX_TYPE_DECIMAL] * 4, out_scales)
    ex.close()
    ex.free()
    b.free()
    assert len(got) == n
    for i, out in enumerate(got):
        for j, v in enumerate(out):
            want = fns[j](data[i])
            if want is None:
                assert v is None, (i, j, v)
            else:
                assert Fraction(v) == want, (i, j, v, want)
