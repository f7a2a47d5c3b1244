"""DISTINCT aggregates (AggFuncDesc.HasDistinct,
/root/reference/pkg/expression/aggregation/descriptor.go; the executor's
distinct checker dedups update rows per group): COUNT/SUM/AVG over the
DISTINCT non-NULL values of one arg column.

Oracle computes them directly with per-group value sets (the reference's
shape). The product rewrites the plan: the device kernel groups by
(orig keys..., arg) — the group table IS the distinct check — and the host
decode folds back to the original keys. Results must be identical.

Semantics pinned: NULLs are excluded from distinct counts and sums; a group
whose values are all NULL still EXISTS (count 0, sum/avg NULL); scalar
(no group-by) distinct over an empty table yields the single default row;
avg(distinct) = sum(distinct)/count(distinct) with DivPrecisionIncrement
and the display-frac round.
"""
import ctypes

import numpy as np
import pytest

from tests.gxlib import (GX_AGG_MODE_PARTIAL, GX_F_GT, GX_TYPE_DECIMAL,
                         GX_TYPE_I64, GX_TYPE_STRING, load_oracle,
                         load_product)
from tidb_amd import plan as P
from tests.gxlib import (GX_AGG_AVG_DISTINCT, GX_AGG_COUNT_DISTINCT,
                         GX_AGG_SUM_DISTINCT)
from tidb_amd.chunkpy import PyChunk

TYPES = [GX_TYPE_I64, GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_STRING]
FRACS = [0, 0, 2, 0]


def _dec(lib, s):
    out = (ctypes.c_uint8 * 40)()
    assert lib.gx_dec_from_string(s.encode(), len(s.encode()), out) == 0
    return bytes(out)


def _data(n=4000, seed=17):
    rng = np.random.default_rng(seed)
    rows = []
    for i in range(n):
        k = int(rng.integers(0, 40))
        v = None if rng.random() < 0.2 else int(rng.integers(0, 12))
        d = None if rng.random() < 0.2 else f"{int(rng.integers(0, 9))}.75"
        s = None if rng.random() < 0.2 else f"s{int(rng.integers(0, 5))}"
        rows.append([k, v, d, s])
    rows.append([777, None, None, None])  # an all-NULL group
    return rows


def _chunks(lib, rows):
    out = []
    for base in range(0, len(rows), 1000):
        part = rows[base:base + 1000]
        ch = PyChunk(TYPES, len(part), FRACS, [None, None, None, 8192])
        for r in part:
            ch.append_row([r[0], r[1],
                           None if r[2] is None else _dec(lib, r[2]), r[3]])
        out.append(ch)
    return out


def _run(lib, rows, aggs, out_types, out_fracs, group=True, sel=False,
         topn=None):
    b = P.Builder(lib)
    src = b.source(TYPES, FRACS)
    node = src
    if sel:
        node = b.selection(src, [b.call(GX_F_GT, GX_TYPE_I64, 0,
                                        b.colref(0, GX_TYPE_I64),
                                        b.const_i64(10))])
    keys = [b.colref(0, GX_TYPE_I64)] if group else []
    agg = b.hashagg(node, keys,
                    [(f, b.colref(c, t, fr), of) for f, c, t, fr, of in aggs])
    root = agg
    if topn is not None:
        root = b.topn(agg, [b.colref(topn, GX_TYPE_I64),
                            b.colref(0, GX_TYPE_I64)], [1, 0], 5)
    ex = b.build(root)
    ex.bind_chunks(src, _chunks(lib, rows))
    ex.open()
    got = ex.pull_all(out_types, out_fracs)
    ex.close()
    ex.free()
    b.free()
    return got if topn is not None else sorted(got)


CD_I64 = (GX_AGG_COUNT_DISTINCT, 1, GX_TYPE_I64, 0, 0)
CD_STR = (GX_AGG_COUNT_DISTINCT, 3, GX_TYPE_STRING, 0, 0)
SD_DEC = (GX_AGG_SUM_DISTINCT, 2, GX_TYPE_DECIMAL, 2, 2)
AD_DEC = (GX_AGG_AVG_DISTINCT, 2, GX_TYPE_DECIMAL, 2, 6)


def test_oracle_count_distinct_grouped():
    lib = load_oracle()
    rows = _data()
    got = _run(lib, rows, [CD_I64], [GX_TYPE_I64, GX_TYPE_I64], [0, 0])
    want = {}
    for k, v, _, _ in rows:
        want.setdefault(k, set())
        if v is not None:
            want[k].add(v)
    assert got == sorted((k, len(s)) for k, s in want.items())
    assert (777, 0) in got  # all-NULL group exists with count 0


def test_oracle_sum_avg_distinct():
    from fractions import Fraction
    lib = load_oracle()
    rows = _data()
    got = _run(lib, rows, [SD_DEC, AD_DEC],
               [GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_DECIMAL], [0, 2, 6])
    want = {}
    for k, _, d, _ in rows:
        want.setdefault(k, set())
        if d is not None:
            want[k].add(Fraction(d))
    for k, s, a in got:
        vals = want[k]
        if not vals:
            assert s is None and a is None
        else:
            assert Fraction(s) == sum(vals)
            exact = sum(vals) / len(vals)
            assert abs(Fraction(a) - exact) <= Fraction(1, 10 ** 6)


def test_oracle_scalar_distinct_empty():
    lib = load_oracle()
    assert _run(lib, [], [CD_I64], [GX_TYPE_I64], [0], group=False) == [(0,)]
    assert _run(lib, [], [SD_DEC], [GX_TYPE_DECIMAL], [2],
                group=False) == [(None,)]


def test_partial_distinct_rejected_both_libs():
    for lib in (load_oracle(), load_product()):
        b = P.Builder(lib)
        src = b.source(TYPES, FRACS)
        agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                        [(GX_AGG_COUNT_DISTINCT, b.colref(1, GX_TYPE_I64),
                          0)], GX_AGG_MODE_PARTIAL)
        ex = b.build(agg)
        ch = PyChunk(TYPES, 1, FRACS, [None, None, None, 64])
        ch.append_row([1, 1, None, None])
        ex.bind_chunks(src, [ch])
        rc = lib.gx_open(ex.ex)
        assert rc != 0  # COMPLETE only
        assert "COMPLETE" in ex.error() or "DISTINCT" in ex.error()
        ex.free()
        b.free()


@pytest.mark.gpu
@pytest.mark.parametrize("aggs,out_types,out_fracs", [
    ([CD_I64], [GX_TYPE_I64, GX_TYPE_I64], [0, 0]),
    ([CD_STR], [GX_TYPE_I64, GX_TYPE_I64], [0, 0]),
    ([SD_DEC, AD_DEC],
     [GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_DECIMAL], [0, 2, 6]),
])
def test_distinct_grouped_parity(aggs, out_types, out_fracs):
    rows = _data()
    want = _run(load_oracle(), rows, aggs, out_types, out_fracs)
    got = _run(load_product(), rows, aggs, out_types, out_fracs)
    assert got == want
    assert len(got) > 30


@pytest.mark.gpu
def test_distinct_with_selection_parity():
    rows = _data()
    args = ([CD_I64], [GX_TYPE_I64, GX_TYPE_I64], [0, 0])
    want = _run(load_oracle(), rows, *args, sel=True)
    got = _run(load_product(), rows, *args, sel=True)
    assert got == want
    assert all(k > 10 for k, _ in got)


@pytest.mark.gpu
def test_distinct_scalar_parity():
    for rows in ([], _data(500)):
        want = _run(load_oracle(), rows, [CD_I64, SD_DEC],
                    [GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 2], group=False)
        got = _run(load_product(), rows, [CD_I64, SD_DEC],
                   [GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 2], group=False)
        assert got == want
        assert len(got) == 1


@pytest.mark.gpu
def test_distinct_topn_parity():
    """ORDER BY count(distinct v) DESC, k LIMIT 5 over the distinct agg —
    the host post-sort must see the FOLDED (user-schema) rows."""
    rows = _data()
    args = ([CD_I64], [GX_TYPE_I64, GX_TYPE_I64], [0, 0])
    want = _run(load_oracle(), rows, *args, topn=1)
    got = _run(load_product(), rows, *args, topn=1)
    assert got == want
    assert len(got) == 5


def _run_join_distinct(lib):
    """count(distinct li.val) group by orders.prio over orders ⋈ lineitem —
    the distinct rewrite riding the agg-over-join pipeline (the fused
    kernel runs over the materialized joined table)."""
    rng = np.random.default_rng(23)
    brows = [[i, int(rng.integers(0, 5))] for i in range(200)]
    prows = []
    for i in range(5000):
        k = int(rng.integers(0, 260))  # some probe keys miss the build
        v = None if rng.random() < 0.15 else int(rng.integers(0, 9))
        prows.append([k, v])
    b = P.Builder(lib)
    t2 = [GX_TYPE_I64, GX_TYPE_I64]
    bsrc = b.source(t2)
    psrc = b.source(t2)
    j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)],
                   [b.colref(0, GX_TYPE_I64)])
    # joined schema: b.key, b.prio, p.key, p.val
    agg = b.hashagg(j, [b.colref(1, GX_TYPE_I64)],
                    [(GX_AGG_COUNT_DISTINCT, b.colref(3, GX_TYPE_I64), 0)])
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
    got = ex.pull_all(t2, [0, 0])
    ex.close()
    ex.free()
    b.free()
    # independent expectation
    want = {}
    bmap = {k: p for k, p in brows}
    for k, v in prows:
        if k in bmap:
            want.setdefault(bmap[k], set())
            if v is not None:
                want[bmap[k]].add(v)
    assert sorted(got) == sorted((p, len(s)) for p, s in want.items())
    return sorted(got)


def test_oracle_join_distinct():
    _run_join_distinct(load_oracle())


@pytest.mark.gpu
def test_join_distinct_parity():
    want = _run_join_distinct(load_oracle())
    got = _run_join_distinct(load_product())
    assert got == want


def _run_multicol(lib, rows, group=True):
    """count(distinct v, d, s) — the multi-column distinct: rows with ANY
    NULL element are excluded; identity is the element tuple."""
    from tests.gxlib import GX_F_TUPLE
    b = P.Builder(lib)
    src = b.source(TYPES, FRACS)
    tup = b.call(GX_F_TUPLE, GX_TYPE_I64, 0,
                 b.colref(1, GX_TYPE_I64), b.colref(2, GX_TYPE_DECIMAL, 2),
                 b.colref(3, GX_TYPE_STRING))
    keys = [b.colref(0, GX_TYPE_I64)] if group else []
    agg = b.hashagg(src, keys, [(GX_AGG_COUNT_DISTINCT, tup, 0)])
    ex = b.build(agg)
    ex.bind_chunks(src, _chunks(lib, rows))
    ex.open()
    out_t = [GX_TYPE_I64, GX_TYPE_I64] if group else [GX_TYPE_I64]
    got = sorted(ex.pull_all(out_t, [0] * len(out_t)))
    ex.close()
    ex.free()
    b.free()
    return got


def test_oracle_multicol_distinct():
    """Pins aggregate.result's count(distinct b,c,d) group-by-id pattern
    (adapted to i64/decimal/string columns): every row with ANY NULL
    element contributes 0; full rows count distinct tuples."""
    lib = load_oracle()
    rows = _data()
    got = _run_multicol(lib, rows)
    want = {}
    for k, v, d, s in rows:
        want.setdefault(k, set())
        if v is not None and d is not None and s is not None:
            want[k].add((v, d, s))
    assert got == sorted((k, len(t)) for k, t in want.items())
    # and the aggregate.result golden itself: one row per id; rows with any
    # NULL -> 0, the single full row -> 1
    g_rows = [[1, 1, "3.00", None], [2, 1, None, "6"], [3, None, "1.00", "2"],
              [4, None, None, "1"], [5, None, "2.00", None],
              [6, 3, None, None], [7, None, None, None],
              [8, 1, "2.00", "3"]]
    got = _run_multicol(lib, g_rows)
    assert got == [(1, 0), (2, 0), (3, 0), (4, 0), (5, 0), (6, 0), (7, 0),
                   (8, 1)]


@pytest.mark.gpu
def test_multicol_distinct_parity():
    rows = _data()
    want = _run_multicol(load_oracle(), rows)
    got = _run_multicol(load_product(), rows)
    assert got == want
    want = _run_multicol(load_oracle(), rows, group=False)
    got = _run_multicol(load_product(), rows, group=False)
    assert got == want
