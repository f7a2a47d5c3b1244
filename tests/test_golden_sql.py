"""Known-answer tests lifted from the reference's SQL golden files
(tests/integrationtest/r/executor/aggregate.result): literal expected output
strings pin the oracle's aggregate semantics end to end (avg display frac,
div+avg chains, NULL exclusion).
"""
import ctypes

from tests.gxlib import (GX_AGG_AVG, GX_F_CAST_DEC, GX_F_DIV, GX_TYPE_DECIMAL,
                         GX_TYPE_I64, load_oracle)
from tidb_amd import plan as P
from tidb_amd.chunkpy import PyChunk


def _run(lib, rows, build_plan, out_types, out_fracs, src_types, src_fracs):
    b = P.Builder(lib)
    src = b.source(src_types, src_fracs)
    root = build_plan(b, src)
    ex = b.build(root)
    ch = PyChunk(src_types, max(len(rows), 1), src_fracs)
    for r in rows:
        ch.append_row(list(r))
    ex.bind_chunks(src, [ch])
    ex.open()
    got = ex.pull_all(out_types, out_fracs)
    ex.close()
    ex.free()
    b.free()
    return got


def test_avg_group_by_int():
    """aggregate.result:428-433 — select avg(a) from t group by a over
    (-120),(127): avg of an int column displays at frac 4."""
    lib = load_oracle()

    def plan(b, src):
        a = b.colref(0, GX_TYPE_I64)
        # the planner wraps int avg args in a decimal cast
        # (expression.WrapWithCastAsDecimal)
        ad = b.call(GX_F_CAST_DEC, GX_TYPE_DECIMAL, 0, a)
        return b.hashagg(src, [a], [(GX_AGG_AVG, ad, 4)])

    got = sorted(_run(lib, [(-120,), (127,)], plan,
                      [GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 4],
                      [GX_TYPE_I64], [0]))
    assert got == [(-120, "-120.0000"), (127, "127.0000")]


def test_avg_of_int_division():
    """aggregate.result:463-476 — SELECT AVG(col_bigint / col_smallint) FROM
    td over 10 rows (2 NULL bigints): expected 25769363061037.62077260
    (div scale 4 + avg increment 4 = display frac 8; NULL rows excluded)."""
    lib = load_oracle()
    rows = [(None, 22876), (9220557287087669248, 32767), (28030, 32767),
            (-3309864251140603904, 32767), (4, 0), (None, 0), (4, -23828),
            (54720, 32767), (0, 29815), (10017, -32661)]

    def plan(b, src):
        c1 = b.colref(0, GX_TYPE_I64)
        c2 = b.colref(1, GX_TYPE_I64)
        q = b.call(GX_F_DIV, GX_TYPE_DECIMAL, 4, c1, c2)
        proj = b.projection(src, [q])
        return b.hashagg(proj, [], [(GX_AGG_AVG,
                                     b.colref(0, GX_TYPE_DECIMAL, 4), 8)])

    got = _run(lib, rows, plan, [GX_TYPE_DECIMAL], [8],
               [GX_TYPE_I64, GX_TYPE_I64], [0, 0])
    assert got == [("25769363061037.62077260",)]


def test_count_nullable_column_group_by():
    """aggregate.result:11-14 — count(c) per group over (1,NULL),(2,1):
    NULL rows don't count."""
    from tests.gxlib import GX_AGG_COUNT
    lib = load_oracle()

    def plan(b, src):
        gid = b.colref(0, GX_TYPE_I64)
        c = b.colref(1, GX_TYPE_I64)
        return b.hashagg(src, [gid], [(GX_AGG_COUNT, c, 0)])

    got = sorted(_run(lib, [(1, None), (2, 1)], plan,
                      [GX_TYPE_I64, GX_TYPE_I64], [0, 0],
                      [GX_TYPE_I64, GX_TYPE_I64], [0, 0]))
    assert got == [(1, 0), (2, 1)]


def test_minmax_string_null_group():
    """aggregate.result:46-51 — MIN(b), MAX(b) over a varchar with one group
    all-NULL: that group's extremes are NULL."""
    from tests.gxlib import GX_AGG_MAX, GX_AGG_MIN, GX_TYPE_STRING
    lib = load_oracle()

    def plan(b, src):
        a = b.colref(0, GX_TYPE_I64)
        s = b.colref(1, GX_TYPE_STRING)
        return b.hashagg(src, [a], [(GX_AGG_MIN, s, 0), (GX_AGG_MAX, s, 0)])

    got = sorted(_run(lib, [(1, "11"), (3, None)], plan,
                      [GX_TYPE_I64, GX_TYPE_STRING, GX_TYPE_STRING],
                      [0, 0, 0], [GX_TYPE_I64, GX_TYPE_STRING], [0, 0]))
    assert got == [(1, "11", "11"), (3, None, None)]
