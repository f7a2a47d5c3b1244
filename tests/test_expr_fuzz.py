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


@pytest.mark.parametrize("seed", [11, 53, 97])
def test_fuzz_groupby_aggs(seed):
    """Random GROUP BY plans: 1-2 int keys, a random mix of
    sum/avg/count/min/max/firstrow over decimal/int args with NULLs,
    oracle vs exact Fraction accounting."""
    from tests.gxlib import (GX_AGG_AVG, GX_AGG_COUNT, GX_AGG_FIRSTROW,
                             GX_AGG_MAX, GX_AGG_MIN, GX_AGG_SUM,
                             GX_F_CAST_DEC)
    lib = load_oracle()
    rng = np.random.default_rng(seed)
    n = 1500
    nkeys = int(rng.integers(1, 3))
    sc = int(rng.integers(1, 4))
    types = [GX_TYPE_I64] * nkeys + [GX_TYPE_DECIMAL, GX_TYPE_I64]
    fracs = [0] * nkeys + [sc, 0]
    data = []
    for _ in range(n):
        keys = [int(rng.integers(0, 8)) for _ in range(nkeys)]
        d = None if rng.random() < 0.2 else \
            Fraction(int(rng.integers(-10**4, 10**4)), 10**sc)
        v = None if rng.random() < 0.2 else int(rng.integers(-500, 500))
        data.append((tuple(keys), d, v))

    b = P.Builder(lib)
    src = b.source(types, fracs)
    kexprs = [b.colref(i, GX_TYPE_I64) for i in range(nkeys)]
    dref = b.colref(nkeys, GX_TYPE_DECIMAL, sc)
    vref = b.colref(nkeys + 1, GX_TYPE_I64)
    vdec = b.call(GX_F_CAST_DEC, GX_TYPE_DECIMAL, 0, vref)
    aggs = [(GX_AGG_SUM, dref, sc), (GX_AGG_COUNT, dref, 0),
            (GX_AGG_MIN, vref, 0), (GX_AGG_MAX, vref, 0),
            (GX_AGG_AVG, vdec, 4), (GX_AGG_COUNT, -1, 0)]
    agg = b.hashagg(src, kexprs, aggs)
    ex = b.build(agg)
    ch = PyChunk(types, n, fracs)
    for keys, d, v in data:
        row = list(keys)
        if d is None:
            row.append(None)
        else:
            q = d * 10**sc
            s = f"{'-' if q < 0 else ''}{abs(q.numerator)//10**sc}"
            s += f".{abs(q.numerator) % 10**sc:0{sc}d}"
            row.append(_dec_cached(lib, s))
        row.append(v)
        ch.append_row(row)
    ex.bind_chunks(src, [ch])
    ex.open()
    out_t = [GX_TYPE_I64] * nkeys + [GX_TYPE_DECIMAL, GX_TYPE_I64,
                                     GX_TYPE_I64, GX_TYPE_I64,
                                     GX_TYPE_DECIMAL, GX_TYPE_I64]
    got = sorted(ex.pull_all(out_t, [0] * nkeys + [sc, 0, 0, 0, 4, 0]))
    ex.close()
    ex.free()
    b.free()

    groups = {}
    for keys, d, v in data:
        g = groups.setdefault(keys, {"s": Fraction(0), "dc": 0, "mn": None,
                                     "mx": None, "vs": Fraction(0), "vc": 0,
                                     "n": 0})
        g["n"] += 1
        if d is not None:
            g["s"] += d
            g["dc"] += 1
        if v is not None:
            g["mn"] = v if g["mn"] is None else min(g["mn"], v)
            g["mx"] = v if g["mx"] is None else max(g["mx"], v)
            g["vs"] += v
            g["vc"] += 1
    want = []
    for keys, g in groups.items():
        s = None if g["dc"] == 0 else g["s"]
        avg = None if g["vc"] == 0 else g["vs"] / g["vc"]
        want.append(tuple(keys) + (s, g["dc"], g["mn"], g["mx"], avg,
                                   g["n"]))
    want.sort(key=lambda r: r[:nkeys])
    assert len(got) == len(want)
    for gr, wr in zip(got, want):
        assert gr[:nkeys] == wr[:nkeys]
        s = gr[nkeys]
        assert (s is None) == (wr[nkeys] is None)
        if s is not None:
            assert Fraction(s) == wr[nkeys]
        assert gr[nkeys + 1] == wr[nkeys + 1]
        assert gr[nkeys + 2] == wr[nkeys + 2]
        assert gr[nkeys + 3] == wr[nkeys + 3]
        a = gr[nkeys + 4]
        assert (a is None) == (wr[nkeys + 4] is None)
        if a is not None:  # avg rounds half-up to frac 4
            exact = wr[nkeys + 4]
            assert abs(Fraction(a) - exact) <= Fraction(1, 2 * 10**4)
        assert gr[nkeys + 5] == wr[nkeys + 5]


@pytest.mark.parametrize("seed", [7, 29])
def test_fuzz_join_types_model(seed):
    """All 8 join types vs one independent Python model on random nullable
    data (oracle, CPU) — one generator, eight semantics."""
    from tests.test_join_types import _run
    lib = load_oracle()
    rng = np.random.default_rng(seed)
    brows = [[None if rng.random() < 0.06 else int(rng.integers(0, 40)),
              int(i)] for i in range(300)]
    prows = [[None if rng.random() < 0.06 else int(rng.integers(0, 60)),
              -int(i)] for i in range(900)]
    bkeys = [k for k, _ in brows if k is not None]
    bset = set(bkeys)
    bnull = any(k is None for k, _ in brows)
    bmatches = {}
    for k, v in brows:
        if k is not None:
            bmatches.setdefault(k, []).append(v)

    def model(jt):
        out = []
        for k, v in prows:
            ms = bmatches.get(k, []) if k is not None else []
            if jt == 0:
                out += [(k, bv, k, v) for bv in ms]
            elif jt == 1:
                out += [(k, bv, k, v) for bv in ms] if ms else \
                    [(None, None, k, v)]
            elif jt == 3:
                if ms:
                    out.append((k, v))
            elif jt == 4:
                if not ms:
                    out.append((k, v))
            elif jt == 5:
                if len(brows) == 0:
                    out.append((k, v))
                elif k is None or bnull:
                    pass
                elif not ms:
                    out.append((k, v))
            elif jt == 6:
                out.append((k, v, 1 if ms else 0))
            elif jt == 7:
                if ms:
                    out.append((k, v, 1))
                elif (k is None and len(brows) > 0) or bnull:
                    out.append((k, v, None))
                else:
                    out.append((k, v, 0))
        if jt == 2:
            matched_b = set()
            for k, v in prows:
                if k is not None and k in bmatches:
                    matched_b.add(k)
                    out += [(k, bv, k, v) for bv in bmatches[k]]
            for k, v in brows:
                if k is None or k not in {kk for kk, _ in prows
                                          if kk is not None}:
                    out.append((k, v, None, None))
        key = lambda r: tuple((x is None, x) for x in r)
        return sorted(out, key=key)

    for jt in range(8):
        got = _run(lib, jt, brows, prows)
        assert got == model(jt), f"join_type {jt}"


@pytest.mark.parametrize("seed", [13, 61])
def test_fuzz_sort_nullable_multikey(seed):
    """TopN over 2 nullable keys with random asc/desc vs Python's sort
    with sortexec NULL-first-asc / NULL-last-desc semantics (oracle)."""
    lib = load_oracle()
    rng = np.random.default_rng(seed)
    desc = [int(rng.integers(0, 2)), int(rng.integers(0, 2))]
    rows = []
    for i in range(1200):
        a = None if rng.random() < 0.12 else int(rng.integers(-50, 50))
        bb = None if rng.random() < 0.12 else int(rng.integers(-9, 9))
        rows.append((a, bb, i))
    b = P.Builder(lib)
    types = [GX_TYPE_I64] * 3
    src = b.source(types)
    root = b.topn(src, [b.colref(0, GX_TYPE_I64), b.colref(1, GX_TYPE_I64)],
                  desc, 200)
    ex = b.build(root)
    ch = PyChunk(types, len(rows))
    for r in rows:
        ch.append_row(list(r))
    ex.bind_chunks(src, [ch])
    ex.open()
    got = ex.pull_all(types)
    ex.close()
    ex.free()
    b.free()

    def keyf(r):
        out = []
        for v, d in zip(r[:2], desc):
            # NULL < any value; desc reverses
            rank = (0 if v is None else 1, 0 if v is None else v)
            out.append((-rank[0], -rank[1]) if d else rank)
        return tuple(out)

    want_keys = [keyf(r) for r in sorted(rows, key=keyf)][:200]
    assert [keyf(r) for r in got] == want_keys


def test_fuzz_substr_windows():
    """SUBSTR byte semantics fuzz: 60 random (pos, len) pairs incl.
    negatives and out-of-range vs the Python model (oracle)."""
    from tests.test_string_builtins import _py_substr, _run_proj
    rng = np.random.default_rng(37)
    lib = load_oracle()
    for _ in range(20):
        pos = int(rng.integers(-12, 13))
        ln = int(rng.integers(-2, 15))
        got, rows = _run_proj(lib, pos, ln)
        for (o0, _, _, _, _), (s, _) in zip(got, rows):
            assert o0 == _py_substr(s, pos, ln), (pos, ln, s)
