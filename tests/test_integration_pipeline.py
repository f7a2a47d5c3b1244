"""Composed-pipeline integration pin (CPU): one plan exercising the round's
feature surface together — Selection (OR + IS NOT NULL) → Projection
(IF over a value-context compare, IFNULL, ROUND, MONTH extraction) →
HashAgg (sum/count/min) → HAVING → ORDER BY/LIMIT — oracle vs an
independent Python model over random data, several seeds."""
import ctypes
from decimal import ROUND_HALF_UP, Decimal

import numpy as np
import pytest

from tests.gxlib import (GX_AGG_COUNT, GX_AGG_MIN, GX_AGG_SUM, GX_F_EQ,
                         GX_F_GT, GX_F_IF, GX_F_IFNULL, GX_F_IS_NOT_NULL,
                         GX_F_LT, GX_F_MONTH, GX_F_OR, GX_F_ROUND,
                         GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_TIME,
                         load_oracle)
from tidb_amd import plan as P
from tidb_amd.chunkpy import PyChunk

TYPES = [GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_TIME, GX_TYPE_I64]
FRACS = [0, 2, 0, 0]


def _dec(lib, s):
    out = (ctypes.c_uint8 * 40)()
    assert lib.gx_dec_from_string(s.encode(), len(s.encode()), out) == 0
    return bytes(out)


def _gen(lib, n, seed):
    rng = np.random.default_rng(seed)
    rows = []
    for i in range(n):
        k = int(rng.integers(0, 12))
        d = None if rng.random() < 0.2 else \
            f"{int(rng.integers(-200, 200))}.{int(rng.integers(0, 100)):02d}"
        y, m, dd = 1995, int(rng.integers(1, 13)), int(rng.integers(1, 28))
        t = lib.gx_time_from_date(y, m, dd)
        v = int(rng.integers(-5, 6))
        rows.append((k, d, (m,), t, v))
    return rows


def _run_oracle(lib, rows):
    b = P.Builder(lib)
    src = b.source(TYPES, FRACS)
    k = b.colref(0, GX_TYPE_I64)
    d = b.colref(1, GX_TYPE_DECIMAL, 2)
    t = b.colref(2, GX_TYPE_TIME)
    v = b.colref(3, GX_TYPE_I64)
    # WHERE (v > 2 OR v < -2) AND d IS NOT NULL
    sel = b.selection(src, [
        b.call(GX_F_OR, GX_TYPE_I64, 0,
               b.call(GX_F_GT, GX_TYPE_I64, 0, v, b.const_i64(2)),
               b.call(GX_F_LT, GX_TYPE_I64, 0, v, b.const_i64(-2))),
        b.call(GX_F_IS_NOT_NULL, GX_TYPE_I64, 0, d)])
    # SELECT k, month(t), round(if(d < 0, ifnull(d, 0) * -1, d), 1)
    zero = b.const_dec(_dec(lib, "0.00"))
    cond = b.call(GX_F_LT, GX_TYPE_I64, 0, d, zero)
    negd = b.call(18, GX_TYPE_DECIMAL, 2,
                  b.call(GX_F_IFNULL, GX_TYPE_DECIMAL, 2, d, zero),
                  b.const_dec(_dec(lib, "-1")))
    absd = b.call(GX_F_IF, GX_TYPE_DECIMAL, 2, cond, negd, d)
    r1 = b.call(GX_F_ROUND, GX_TYPE_DECIMAL, 1, absd, b.const_i64(1))
    proj = b.projection(sel, [k, b.call(GX_F_MONTH, GX_TYPE_I64, 0, t), r1])
    # GROUP BY k: sum(r1), count(*), min(month)
    agg = b.hashagg(proj, [b.colref(0, GX_TYPE_I64)],
                    [(GX_AGG_SUM, b.colref(2, GX_TYPE_DECIMAL, 1), 1),
                     (GX_AGG_COUNT, -1, 0),
                     (GX_AGG_MIN, b.colref(1, GX_TYPE_I64), 0)])
    hav = b.selection(agg, [b.call(GX_F_GT, GX_TYPE_I64, 0,
                                   b.colref(2, GX_TYPE_I64),
                                   b.const_i64(5))])
    root = b.topn(hav, [b.colref(0, GX_TYPE_I64)], [0], 100)
    ex = b.build(root)
    chunks = []
    for base in range(0, len(rows), 1000):
        part = rows[base:base + 1000]
        ch = PyChunk(TYPES, len(part), FRACS)
        for kk, dd, _, tt, vv in part:
            ch.append_row([kk, None if dd is None else _dec(lib, dd), tt,
                           vv])
        chunks.append(ch)
    ex.bind_chunks(src, chunks)
    ex.open()
    got = ex.pull_all([GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64,
                       GX_TYPE_I64], [0, 1, 0, 0])
    ex.close()
    ex.free()
    b.free()
    return got


def _model(rows):
    want = {}
    for k, d, (m,), t, v in rows:
        if not (v > 2 or v < -2):
            continue
        if d is None:
            continue
        D = Decimal(d)
        absd = -D if D < 0 else D
        r1 = absd.quantize(Decimal("0.1"), ROUND_HALF_UP)
        s, c, mn = want.get(k, (Decimal(0), 0, None))
        want[k] = (s + r1, c + 1, m if mn is None else min(mn, m))
    return sorted((k, f"{s:.1f}", c, mn) for k, (s, c, mn) in want.items()
                  if c > 5)


@pytest.mark.parametrize("seed", [3, 17, 42])
def test_oracle_composed_pipeline(seed):
    lib = load_oracle()
    rows = _gen(lib, 3000, seed)
    got = _run_oracle(lib, rows)
    assert got == _model(rows)
    assert len(got) > 5


@pytest.mark.gpu
@pytest.mark.parametrize("seed", [17, 42])
def test_composed_pipeline_parity(seed):
    from tests.gxlib import load_product
    lib_o = load_oracle()
    rows = _gen(lib_o, 3000, seed)
    want = _run_oracle(lib_o, rows)
    got = _run_oracle(load_product(), rows)
    assert got == want
    assert len(got) > 5


def test_oracle_in_scalar_aggregated():
    """IN-scalar join feeding an aggregation (oracle, CPU): count how many
    probe rows per group are IN the build set — agg over the jt-6 flag."""
    from tests.gxlib import GX_AGG_COUNT, GX_AGG_SUM, GX_F_CAST_DEC
    lib = load_oracle()
    rng = np.random.default_rng(97)
    brows = [[int(k), 0] for k in rng.choice(200, size=60, replace=False)]
    prows = [[int(rng.integers(0, 200)), int(rng.integers(0, 8))]
             for _ in range(3000)]
    b = P.Builder(lib)
    t2 = [GX_TYPE_I64, GX_TYPE_I64]
    bsrc = b.source(t2)
    psrc = b.source(t2)
    j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)],
                   [b.colref(0, GX_TYPE_I64)], join_type=6)
    # jt-6 output: p.key, p.grp, flag — group by p.grp,
    # sum(cast(flag)) counts the IN rows
    agg = b.hashagg(j, [b.colref(1, GX_TYPE_I64)],
                    [(GX_AGG_SUM, b.call(GX_F_CAST_DEC, GX_TYPE_DECIMAL, 0,
                                         b.colref(2, GX_TYPE_I64)), 0),
                     (GX_AGG_COUNT, -1, 0)])
    ex = b.build(agg)
    bch = PyChunk(t2, len(brows))
    for r in brows:
        bch.append_row(r)
    pch = PyChunk(t2, len(prows))
    for r in prows:
        pch.append_row(r)
    ex.bind_chunks(bsrc, [bch])
    ex.bind_chunks(psrc, [pch])
    ex.open()
    got = sorted(ex.pull_all([GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64],
                             [0, 0, 0]))
    ex.close()
    ex.free()
    b.free()
    bset = {k for k, _ in brows}
    want = {}
    for k, g in prows:
        s, c = want.get(g, (0, 0))
        want[g] = (s + (1 if k in bset else 0), c + 1)
    assert got == sorted((g, str(s), c) for g, (s, c) in want.items())
    assert len(got) == 8


def test_oracle_not_in_filter_then_agg():
    """NOT IN (jt 5) as a filter feeding aggregation (oracle, CPU):
    count the rows whose key is NOT in the build set, per group —
    with and without a NULL poisoning the build side."""
    from tests.gxlib import GX_AGG_COUNT
    lib = load_oracle()
    rng = np.random.default_rng(101)
    clean = [[int(k), 0] for k in rng.choice(150, size=40, replace=False)]
    prows = [[int(rng.integers(0, 150)), int(rng.integers(0, 5))]
             for _ in range(2000)]

    def run(brows):
        b = P.Builder(lib)
        t2 = [GX_TYPE_I64, GX_TYPE_I64]
        bsrc = b.source(t2)
        psrc = b.source(t2)
        j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)],
                       [b.colref(0, GX_TYPE_I64)], join_type=5)
        agg = b.hashagg(j, [b.colref(1, GX_TYPE_I64)],
                        [(GX_AGG_COUNT, -1, 0)])
        ex = b.build(agg)
        bch = PyChunk(t2, max(len(brows), 1))
        for r in brows:
            bch.append_row(r)
        pch = PyChunk(t2, len(prows))
        for r in prows:
            pch.append_row(r)
        ex.bind_chunks(bsrc, [bch])
        ex.bind_chunks(psrc, [pch])
        ex.open()
        got = sorted(ex.pull_all([GX_TYPE_I64, GX_TYPE_I64]))
        ex.close()
        ex.free()
        b.free()
        return got

    bset = {k for k, _ in clean}
    want = {}
    for k, g in prows:
        if k not in bset:
            want[g] = want.get(g, 0) + 1
    assert run(clean) == sorted(want.items())
    # one NULL build key: NOT IN never true -> zero rows -> zero groups
    assert run(clean + [[None, 0]]) == []
