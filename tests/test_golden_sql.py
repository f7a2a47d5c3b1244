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


def test_count_empty_table_grouped_vs_scalar():
    """aggregate.result:55-64 — over an EMPTY t(a,b,c):
    `select count(a) from t group by a` returns ZERO rows, while
    `select count(a) from t` (no group-by) returns the single default
    row `0`. Then after `insert t values(0,0,0)`:
    `select count(b) from t group by a` returns one row `1`."""
    from tests.gxlib import GX_AGG_COUNT
    lib = load_oracle()

    def grouped(b, src):
        a = b.colref(0, GX_TYPE_I64)
        return b.hashagg(src, [a], [(GX_AGG_COUNT,
                                     b.colref(1, GX_TYPE_I64), 0)])

    def scalar(b, src):
        return b.hashagg(src, [], [(GX_AGG_COUNT,
                                    b.colref(0, GX_TYPE_I64), 0)])

    t3 = [GX_TYPE_I64] * 3
    assert _run(lib, [], grouped, [GX_TYPE_I64, GX_TYPE_I64], [0, 0],
                t3, [0, 0, 0]) == []
    assert _run(lib, [], scalar, [GX_TYPE_I64], [0], t3, [0, 0, 0]) == [(0,)]
    assert _run(lib, [(0, 0, 0)], grouped, [GX_TYPE_I64, GX_TYPE_I64],
                [0, 0], t3, [0, 0, 0]) == [(0, 1)]


def test_count_filtered_two_key_group_order_limit():
    """aggregate.result:66-85 — t(a,b,c) with rows (0,0,0),(1,1,1),(3,3,6),
    (3,2,5),(2,1,4),(1,1,3),(1,1,2):
    `select count(a) from t where b>0 group by a, b` gives counts
    {1,1,1,3} (in any order), and with `order by a limit 1` the first
    group (a=1,b=1) has count 3."""
    from tests.gxlib import GX_AGG_COUNT, GX_F_GT
    lib = load_oracle()
    rows = [(0, 0, 0), (1, 1, 1), (3, 3, 6), (3, 2, 5), (2, 1, 4),
            (1, 1, 3), (1, 1, 2)]
    t3 = [GX_TYPE_I64] * 3

    def plan(b, src):
        a = b.colref(0, GX_TYPE_I64)
        bb = b.colref(1, GX_TYPE_I64)
        sel = b.selection(src, [b.call(GX_F_GT, GX_TYPE_I64, 0, bb,
                                       b.const_i64(0))])
        return b.hashagg(sel, [a, bb], [(GX_AGG_COUNT, a, 0)])

    got = _run(lib, rows, plan, [GX_TYPE_I64] * 3, [0] * 3, t3, [0] * 3)
    assert sorted(c for _, _, c in got) == [1, 1, 1, 3]

    def plan_limit(b, src):
        a = b.colref(0, GX_TYPE_I64)
        bb = b.colref(1, GX_TYPE_I64)
        sel = b.selection(src, [b.call(GX_F_GT, GX_TYPE_I64, 0, bb,
                                       b.const_i64(0))])
        agg = b.hashagg(sel, [a, bb], [(GX_AGG_COUNT, a, 0)])
        return b.topn(agg, [b.colref(0, GX_TYPE_I64)], [0], 1)

    got = _run(lib, rows, plan_limit, [GX_TYPE_I64] * 3, [0] * 3, t3, [0] * 3)
    assert [c for _, _, c in got] == [3]  # group (a=1,b=1)


def test_max_both_sides_over_join():
    """aggregate.result:86-91 — t(1,1,1),(2,1,1); tt(1,2,1);
    `select max(a.b), max(b.b) from t a join tt b on a.a = b.a group by
    a.c` returns one row `1 2` (only a=1 joins; group a.c=1)."""
    from tests.gxlib import GX_AGG_MAX
    lib = load_oracle()
    t3 = [GX_TYPE_I64] * 3
    b = P.Builder(lib)
    tsrc = b.source(t3, [0] * 3)
    ttsrc = b.source(t3, [0] * 3)
    j = b.hashjoin(tsrc, ttsrc, [b.colref(0, GX_TYPE_I64)],
                   [b.colref(0, GX_TYPE_I64)])
    # joined schema: t.a,t.b,t.c,tt.a,tt.b,tt.c
    agg = b.hashagg(j, [b.colref(2, GX_TYPE_I64)],
                    [(GX_AGG_MAX, b.colref(1, GX_TYPE_I64), 0),
                     (GX_AGG_MAX, b.colref(4, GX_TYPE_I64), 0)])
    ex = b.build(agg)
    ch_t = PyChunk(t3, 2)
    for r in [(1, 1, 1), (2, 1, 1)]:
        ch_t.append_row(list(r))
    ch_tt = PyChunk(t3, 1)
    ch_tt.append_row([1, 2, 1])
    ex.bind_chunks(tsrc, [ch_t])
    ex.bind_chunks(ttsrc, [ch_tt])
    ex.open()
    got = ex.pull_all([GX_TYPE_I64] * 3, [0] * 3)
    ex.close()
    ex.free()
    b.free()
    assert got == [(1, 1, 2)]


def test_count_distinct_with_nulls():
    """aggregate.result (distinct agg block) — t(a,b) with rows
    (NULL,NULL),(1,NULL),(NULL,1),(1,2),(3,4):
    `select count(distinct a) from t` = 2 (NULLs excluded, dup 1 merged),
    with and without distinct-agg push-down (one execution shape here)."""
    from tests.gxlib import GX_TYPE_I64
    lib = load_oracle()
    rows = [(None, None), (1, None), (None, 1), (1, 2), (3, 4)]

    def plan(b, src):
        return b.hashagg(src, [], [(6, b.colref(0, GX_TYPE_I64), 0)])

    got = _run(lib, rows, plan, [GX_TYPE_I64], [0],
               [GX_TYPE_I64, GX_TYPE_I64], [0, 0])
    assert got == [(2,)]


def test_sum_avg_distinct_group_by():
    """aggregate.result — t1(a,b) = (1,1),(2,2),(3,3),(1,4),(1,1),(3,5),
    (2,2),(3,5),(3,3): `select avg(distinct b) ... group by a order by a`
    = 2.5000, 2.0000, 4.0000 and `sum(distinct b)` = 5, 2, 8 (the planner
    wraps int distinct args in a decimal cast; avg displays at frac 4)."""
    from tests.gxlib import GX_TYPE_I64
    lib = load_oracle()
    rows = [(1, 1), (2, 2), (3, 3), (1, 4), (1, 1), (3, 5), (2, 2), (3, 5),
            (3, 3)]

    def plan(b, src):
        bd = b.call(GX_F_CAST_DEC, GX_TYPE_DECIMAL, 0,
                    b.colref(1, GX_TYPE_I64))
        return b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                         [(8, bd, 4), (7, bd, 0)])

    got = _run(lib, rows, plan, [GX_TYPE_I64, GX_TYPE_DECIMAL,
                                 GX_TYPE_DECIMAL], [0, 4, 0],
               [GX_TYPE_I64, GX_TYPE_I64], [0, 0])
    assert sorted(got) == [(1, "2.5000", "5"), (2, "2.0000", "2"),
                           (3, "4.0000", "8")]


def test_not_in_null_aware_join_goldens():
    """jointest/join.result:1460-1500 — t1={1}, t2(b)={1,NULL}:
    `where 1 not in (select b from t2)` and `where 2 not in ...` both
    return NO rows (a NULL y defeats NOT IN); `select 1 in (select b from
    t2)` = 1; after deleting b=1 (t2={NULL}): `1 in (...)` = NULL; with
    t2 empty: NOT IN accepts and the IN scalar is 0."""
    from tests.gxlib import GX_TYPE_I64
    from tidb_amd.chunkpy import PyChunk
    lib = load_oracle()

    def run(jt, build_vals, probe_vals, out_types):
        b = P.Builder(lib)
        bsrc = b.source([GX_TYPE_I64])
        psrc = b.source([GX_TYPE_I64])
        j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)],
                       [b.colref(0, GX_TYPE_I64)], join_type=jt)
        ex = b.build(j)
        bch = PyChunk([GX_TYPE_I64], max(len(build_vals), 1))
        for v in build_vals:
            bch.append_row([v])
        pch = PyChunk([GX_TYPE_I64], max(len(probe_vals), 1))
        for v in probe_vals:
            pch.append_row([v])
        ex.bind_chunks(bsrc, [bch])
        ex.bind_chunks(psrc, [pch])
        ex.open()
        rows = ex.pull_all(out_types)
        ex.close()
        ex.free()
        b.free()
        return rows

    t2 = [1, None]
    # where 1 not in (select b from t2)  -> empty ; where 2 not in -> empty
    assert run(5, t2, [1], [GX_TYPE_I64]) == []
    assert run(5, t2, [2], [GX_TYPE_I64]) == []
    # select 1 in (select b from t2) -> 1
    assert run(7, t2, [1], [GX_TYPE_I64] * 2) == [(1, 1)]
    # select 2 in (select b from t2) -> NULL (no match, NULL evidence)
    assert run(7, t2, [2], [GX_TYPE_I64] * 2) == [(2, None)]
    # delete b=1: t2 = {NULL}
    assert run(7, [None], [1], [GX_TYPE_I64] * 2) == [(1, None)]
    assert run(5, [None], [1], [GX_TYPE_I64]) == []
    # t2 empty: NOT IN accepts; IN scalar is plain 0
    assert run(5, [], [1], [GX_TYPE_I64]) == [(1,)]
    assert run(7, [], [1], [GX_TYPE_I64] * 2) == [(1, 0)]
    # select 1 in (select 1 from t2={NULL}) -> 1 (the subquery projects 1)
    assert run(7, [1], [1], [GX_TYPE_I64] * 2) == [(1, 1)]


def test_ifnull_with_div_by_zero_golden():
    """expression/builtin.result:1342-1369 — t(b int) rows (0),(NULL),(4):
    `select ifnull(b, b/0) from t` = 0.0000, NULL, 4.0000 — division by
    zero yields NULL (never an error) and IFNULL falls back per row; the
    display frac comes from the unified decimal type."""
    from tests.gxlib import GX_F_IFNULL, GX_TYPE_I64
    from tidb_amd.chunkpy import PyChunk
    lib = load_oracle()
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64])
    col = b.colref(0, GX_TYPE_I64)
    bd = b.call(GX_F_CAST_DEC, GX_TYPE_DECIMAL, 4, col)
    div = b.call(GX_F_DIV, GX_TYPE_DECIMAL, 4, bd, b.const_i64(0))
    proj = b.projection(src, [b.call(GX_F_IFNULL, GX_TYPE_DECIMAL, 4, bd,
                                     div)])
    ex = b.build(proj)
    ch = PyChunk([GX_TYPE_I64], 3)
    for v in (0, None, 4):
        ch.append_row([v])
    ex.bind_chunks(src, [ch])
    ex.open()
    got = ex.pull_all([GX_TYPE_DECIMAL], [4])
    ex.close()
    ex.free()
    b.free()
    assert got == [("0.0000",), (None,), ("4.0000",)]


def test_if_golden():
    """expression/builtin.result:1337-1341 — t(b int) rows (0),(NULL),(4):
    `select if(b=0, 1, 1/b) from t` = 1.0000, NULL, 0.2500 — a NULL
    condition picks the else branch, and 1/NULL is NULL; display frac 4
    from the division type."""
    from tests.gxlib import GX_F_EQ, GX_F_IF, GX_TYPE_I64
    from tidb_amd.chunkpy import PyChunk
    lib = load_oracle()
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64])
    col = b.colref(0, GX_TYPE_I64)
    bd = b.call(GX_F_CAST_DEC, GX_TYPE_DECIMAL, 4, col)
    cond = b.call(GX_F_EQ, GX_TYPE_I64, 0, col, b.const_i64(0))
    one = b.call(GX_F_CAST_DEC, GX_TYPE_DECIMAL, 4, b.const_i64(1))
    div = b.call(GX_F_DIV, GX_TYPE_DECIMAL, 4, one, bd)
    # the planner rounds the unified IF type to the display frac
    # (ProduceDecWithSpecifiedTp) — expressed as the cast wrapper here
    ife = b.call(GX_F_IF, GX_TYPE_DECIMAL, 4, cond, one, div)
    proj = b.projection(src, [b.call(GX_F_CAST_DEC, GX_TYPE_DECIMAL, 4,
                                     ife)])
    ex = b.build(proj)
    ch = PyChunk([GX_TYPE_I64], 3)
    for v in (0, None, 4):
        ch.append_row([v])
    ex.bind_chunks(src, [ch])
    ex.open()
    got = ex.pull_all([GX_TYPE_DECIMAL], [4])
    ex.close()
    ex.free()
    b.free()
    assert got == [("1.0000",), (None,), ("0.2500",)]


def test_order_by_limit_offset_goldens():
    """executor.result:388-410 — t(a,b) = (1,1),(2,2),(3,30),(4,40),(5,5),
    (6,6): `order by a limit 1, k` (offset 1) for k=1..4, and the keyless
    `where a > 0 limit 1, 1` = row a=2 (child order preserved)."""
    from tests.gxlib import GX_F_GT, GX_TYPE_I64
    from tidb_amd.chunkpy import PyChunk
    lib = load_oracle()
    rows = [(1, 1), (2, 2), (3, 30), (4, 40), (5, 5), (6, 6)]

    def run(limit, offset, keyless=False):
        b = P.Builder(lib)
        src = b.source([GX_TYPE_I64] * 2)
        node = src
        if keyless:
            node = b.selection(src, [b.call(GX_F_GT, GX_TYPE_I64, 0,
                                            b.colref(0, GX_TYPE_I64),
                                            b.const_i64(0))])
            root = b.topn(node, [], [], limit, offset)
        else:
            root = b.topn(src, [b.colref(0, GX_TYPE_I64)], [0], limit,
                          offset)
        ex = b.build(root)
        ch = PyChunk([GX_TYPE_I64] * 2, len(rows))
        for r in rows:
            ch.append_row(list(r))
        ex.bind_chunks(src, [ch])
        ex.open()
        out = ex.pull_all([GX_TYPE_I64] * 2)
        ex.close()
        ex.free()
        b.free()
        return out

    assert run(1, 1) == [(2, 2)]
    assert run(2, 1) == [(2, 2), (3, 30)]
    assert run(3, 1) == [(2, 2), (3, 30), (4, 40)]
    assert run(4, 1) == [(2, 2), (3, 30), (4, 40), (5, 5)]
    assert run(1, 1, keyless=True) == [(2, 2)]
