"""HAVING — a Selection over the aggregate's output (the reference
evaluates HAVING above the agg the same way; aggregate.result HAVING
blocks). Product: the fused aggregation runs on device and the (small)
decoded group rows filter on host. Leaves: col cmp const, IS [NOT] NULL,
OR trees of those."""
import ctypes

import numpy as np
import pytest

from tests.gxlib import (GX_AGG_COUNT, GX_AGG_SUM, GX_F_GT, GX_F_IS_NULL,
                         GX_F_LT, GX_F_OR, GX_TYPE_DECIMAL, GX_TYPE_I64,
                         load_oracle, load_product)
from tidb_amd import plan as P
from tidb_amd.chunkpy import PyChunk


def _dec(lib, s):
    out = (ctypes.c_uint8 * 40)()
    assert lib.gx_dec_from_string(s.encode(), len(s.encode()), out) == 0
    return bytes(out)


def _data(n=4000, seed=61):
    rng = np.random.default_rng(seed)
    rows = []
    for i in range(n):
        k = int(rng.integers(0, 30))
        d = None if rng.random() < 0.3 else f"{int(rng.integers(0, 50))}.25"
        rows.append((k, d))
    rows += [(777, None)] * 3  # an all-NULL group (sum NULL)
    return rows


def _run(lib, rows, having):
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 2])
    agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                    [(GX_AGG_SUM, b.colref(1, GX_TYPE_DECIMAL, 2), 2),
                     (GX_AGG_COUNT, -1, 0)])
    hav = b.selection(agg, having(b))
    ex = b.build(hav)
    chunks = []
    for base in range(0, len(rows), 1000):
        part = rows[base:base + 1000]
        ch = PyChunk([GX_TYPE_I64, GX_TYPE_DECIMAL], len(part), [0, 2])
        for k, d in part:
            ch.append_row([k, None if d is None else _dec(lib, d)])
        chunks.append(ch)
    ex.bind_chunks(src, chunks)
    ex.open()
    got = sorted(ex.pull_all([GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64],
                             [0, 2, 0]))
    ex.close()
    ex.free()
    b.free()
    return got


def _cases(lib):
    return {
        "count_gt": lambda b: [b.call(GX_F_GT, GX_TYPE_I64, 0,
                                      b.colref(2, GX_TYPE_I64),
                                      b.const_i64(140))],
        "sum_lt": lambda b: [b.call(GX_F_LT, GX_TYPE_I64, 0,
                                    b.colref(1, GX_TYPE_DECIMAL, 2),
                                    b.const_dec(_dec(lib, "3000.00")))],
        "sum_null": lambda b: [b.call(GX_F_IS_NULL, GX_TYPE_I64, 0,
                                      b.colref(1, GX_TYPE_DECIMAL, 2))],
        "or": lambda b: [b.call(GX_F_OR, GX_TYPE_I64, 0,
                                b.call(GX_F_GT, GX_TYPE_I64, 0,
                                       b.colref(2, GX_TYPE_I64),
                                       b.const_i64(150)),
                                b.call(GX_F_LT, GX_TYPE_I64, 0,
                                       b.colref(0, GX_TYPE_I64),
                                       b.const_i64(3)))],
    }


def test_oracle_having():
    from fractions import Fraction
    lib = load_oracle()
    rows = _data()
    got = _run(lib, rows, _cases(lib)["count_gt"])
    want = {}
    for k, d in rows:
        s, c = want.get(k, (Fraction(0), 0))
        want[k] = (s + (Fraction(d) if d else 0), c + 1)
    exp = sorted((k, None if all(d is None for kk, d in rows if kk == k)
                  else f"{float(s):.2f}", c)
                 for k, (s, c) in want.items() if c > 140)
    assert [(k, c) for k, _, c in got] == [(k, c) for k, _, c in exp]
    # the NULL-sum group filters correctly too
    got_null = _run(lib, rows, _cases(lib)["sum_null"])
    assert [k for k, _, _ in got_null] == [777]


@pytest.mark.gpu
@pytest.mark.parametrize("case", ["count_gt", "sum_lt", "sum_null", "or"])
def test_having_parity(case):
    rows = _data()
    want = _run(load_oracle(), rows, _cases(load_oracle())[case])
    got = _run(load_product(), rows, _cases(load_product())[case])
    assert got == want
    assert 0 < len(got)


def _run_topn_having(lib, rows):
    """ORDER BY count DESC, k LIMIT 5 over a HAVING filter over the agg."""
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 2])
    agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                    [(GX_AGG_SUM, b.colref(1, GX_TYPE_DECIMAL, 2), 2),
                     (GX_AGG_COUNT, -1, 0)])
    hav = b.selection(agg, [b.call(GX_F_GT, GX_TYPE_I64, 0,
                                   b.colref(2, GX_TYPE_I64),
                                   b.const_i64(120))])
    root = b.topn(hav, [b.colref(2, GX_TYPE_I64),
                        b.colref(0, GX_TYPE_I64)], [1, 0], 5)
    ex = b.build(root)
    chunks = []
    for base in range(0, len(rows), 1000):
        part = rows[base:base + 1000]
        ch = PyChunk([GX_TYPE_I64, GX_TYPE_DECIMAL], len(part), [0, 2])
        for k, d in part:
            ch.append_row([k, None if d is None else _dec(lib, d)])
        chunks.append(ch)
    ex.bind_chunks(src, chunks)
    ex.open()
    got = ex.pull_all([GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64], [0, 2, 0])
    ex.close()
    ex.free()
    b.free()
    cnts = [c for _, _, c in got]
    assert cnts == sorted(cnts, reverse=True) and all(c > 120 for c in cnts)
    return got


def test_oracle_topn_having():
    rows = _data()
    got = _run_topn_having(load_oracle(), rows)
    assert 0 < len(got) <= 5


@pytest.mark.gpu
def test_topn_having_parity():
    rows = _data()
    want = _run_topn_having(load_oracle(), rows)
    got = _run_topn_having(load_product(), rows)
    assert got == want


def _run_having_distinct(lib, rows):
    """HAVING over a DISTINCT aggregate: the decode folds the distinct
    rewrite FIRST, then filters the user-schema rows."""
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 2])
    agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                    [(6, b.colref(1, GX_TYPE_DECIMAL, 2), 0)])
    hav = b.selection(agg, [b.call(GX_F_GT, GX_TYPE_I64, 0,
                                   b.colref(1, GX_TYPE_I64),
                                   b.const_i64(20))])
    ex = b.build(hav)
    chunks = []
    for base in range(0, len(rows), 1000):
        part = rows[base:base + 1000]
        ch = PyChunk([GX_TYPE_I64, GX_TYPE_DECIMAL], len(part), [0, 2])
        for k, d in part:
            ch.append_row([k, None if d is None else _dec(lib, d)])
        chunks.append(ch)
    ex.bind_chunks(src, chunks)
    ex.open()
    got = sorted(ex.pull_all([GX_TYPE_I64, GX_TYPE_I64]))
    ex.close()
    ex.free()
    b.free()
    assert all(c > 20 for _, c in got)
    return got


def test_oracle_having_distinct():
    rows = _data()
    got = _run_having_distinct(load_oracle(), rows)
    want = {}
    for k, d in rows:
        want.setdefault(k, set())
        if d is not None:
            want[k].add(d)
    assert got == sorted((k, len(s)) for k, s in want.items() if len(s) > 20)
    assert len(got) > 3


@pytest.mark.gpu
def test_having_distinct_parity():
    rows = _data()
    want = _run_having_distinct(load_oracle(), rows)
    got = _run_having_distinct(load_product(), rows)
    assert got == want


def _run_having_over_join(lib):
    """HAVING above an aggregation over a JOIN: the agg-over-join pipeline
    materializes the joined table, the fused agg runs over it, and the
    decode filters groups."""
    rng = np.random.default_rng(71)
    brows = [[i, int(rng.integers(0, 6))] for i in range(150)]
    prows = [[int(rng.integers(0, 200)), int(rng.integers(1, 9))]
             for _ in range(4000)]
    b = P.Builder(lib)
    t2 = [GX_TYPE_I64, GX_TYPE_I64]
    bsrc = b.source(t2)
    psrc = b.source(t2)
    j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)],
                   [b.colref(0, GX_TYPE_I64)])
    # joined: b.key, b.grp, p.key, p.val — group by b.grp, count(*)
    agg = b.hashagg(j, [b.colref(1, GX_TYPE_I64)],
                    [(GX_AGG_COUNT, -1, 0)])
    hav = b.selection(agg, [b.call(GX_F_GT, GX_TYPE_I64, 0,
                                   b.colref(1, GX_TYPE_I64),
                                   b.const_i64(450))])
    ex = b.build(hav)
    bch = PyChunk(t2, len(brows))
    for r in brows:
        bch.append_row(r)
    pch = PyChunk(t2, len(prows))
    for r in prows:
        pch.append_row(r)
    ex.bind_chunks(bsrc, [bch])
    ex.bind_chunks(psrc, [pch])
    ex.open()
    got = sorted(ex.pull_all(t2))
    ex.close()
    ex.free()
    b.free()
    bmap = {k: g for k, g in brows}
    want = {}
    for k, _ in prows:
        if k in bmap:
            want[bmap[k]] = want.get(bmap[k], 0) + 1
    assert got == sorted((g, c) for g, c in want.items() if c > 450)
    assert 0 < len(got) < 6  # the cut removes some groups (seeded data)
    return got


def test_oracle_having_over_join():
    _run_having_over_join(load_oracle())


@pytest.mark.gpu
def test_having_over_join_parity():
    want = _run_having_over_join(load_oracle())
    got = _run_having_over_join(load_product())
    assert got == want


def test_oracle_topn_having_distinct():
    """Full stack: ORDER BY count-distinct DESC LIMIT 3 over HAVING over
    COUNT(DISTINCT) (oracle, CPU)."""
    from tests.gxlib import GX_AGG_COUNT_DISTINCT
    lib = load_oracle()
    rows = _data()
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 2])
    agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                    [(GX_AGG_COUNT_DISTINCT, b.colref(1, GX_TYPE_DECIMAL, 2),
                      0)])
    hav = b.selection(agg, [b.call(GX_F_GT, GX_TYPE_I64, 0,
                                   b.colref(1, GX_TYPE_I64),
                                   b.const_i64(10))])
    root = b.topn(hav, [b.colref(1, GX_TYPE_I64),
                        b.colref(0, GX_TYPE_I64)], [1, 0], 3)
    ex = b.build(root)
    chunks = []
    for base in range(0, len(rows), 1000):
        part = rows[base:base + 1000]
        ch = PyChunk([GX_TYPE_I64, GX_TYPE_DECIMAL], len(part), [0, 2])
        for k, d in part:
            ch.append_row([k, None if d is None else _dec(lib, d)])
        chunks.append(ch)
    ex.bind_chunks(src, chunks)
    ex.open()
    got = ex.pull_all([GX_TYPE_I64, GX_TYPE_I64])
    ex.close()
    ex.free()
    b.free()
    want = {}
    for k, d in rows:
        want.setdefault(k, set())
        if d is not None:
            want[k].add(d)
    exp = sorted(((k, len(s)) for k, s in want.items() if len(s) > 10),
                 key=lambda r: (-r[1], r[0]))[:3]
    assert got == exp
    assert len(got) == 3
