"""Standalone inner HashJoin operator — the general HashJoinV2 equivalent
(reference pkg/executor/join/hash_join_v2.go): duplicate build keys via a
chained hash table (hash_table_v2.go:22-53 heads + join_row_table.go
next_row_ptr chains), NULL join keys never match, output = build cols ++
probe cols, one row per matching (build,probe) pair
(inner_join_probe.go:27-86).

Join output ORDER is unspecified (the reference's concurrent probe workers
make its output order nondeterministic too), so parity compares sorted
multisets. Values must be bit/digit-exact.
"""
import ctypes

import numpy as np
import pytest

from tests.gxlib import (GX_F_GT, GX_F_LT, GX_TPCH_LINEITEM, GX_TPCH_ORDERS,
                         GX_TYPE_DECIMAL, GX_TYPE_F64, GX_TYPE_I64,
                         GX_TYPE_STRING, GX_TYPE_TIME, load_oracle,
                         load_product)
from tidb_amd import plan as P
from tidb_amd.chunkpy import PyChunk

BUILD_TYPES = [GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_STRING]
BUILD_FRACS = [0, 2, 0]
PROBE_TYPES = [GX_TYPE_I64, GX_TYPE_I64, GX_TYPE_TIME]
PROBE_FRACS = [0, 0, 0]
OUT_TYPES = BUILD_TYPES + PROBE_TYPES
OUT_FRACS = BUILD_FRACS + PROBE_FRACS


def _dec_bytes(lib, s):
    out = (ctypes.c_uint8 * 40)()
    assert lib.gx_dec_from_string(s.encode(), len(s.encode()), out) == 0
    return bytes(out)


def _to_chunk(lib, types, fracs, rows):
    """rows hold decoded values (int for i64/time, str for string, decimal as
    str, None for NULL); decimals are encoded through the library under test."""
    ch = PyChunk(types, max(len(rows), 1), fracs)
    for r in rows:
        vals = []
        for v, t in zip(r, types):
            if v is None or t != GX_TYPE_DECIMAL:
                vals.append(v)
            else:
                vals.append(_dec_bytes(lib, v))
        ch.append_row(vals)
    return ch


def _canon(rows):
    return sorted(rows, key=lambda r: tuple((x is None, 0 if x is None else x)
                                            for x in r))


def _join_plan(lib, with_filters):
    b = P.Builder(lib)
    bsrc = b.source(BUILD_TYPES, BUILD_FRACS)
    psrc = b.source(PROBE_TYPES, PROBE_FRACS)
    bchild, pchild = bsrc, psrc
    if with_filters:
        # build: key > 1; probe: payload < 100
        cond_b = b.call(GX_F_GT, GX_TYPE_I64, 0, b.colref(0, GX_TYPE_I64),
                        b.const_i64(1))
        bchild = b.selection(bsrc, [cond_b])
        cond_p = b.call(GX_F_LT, GX_TYPE_I64, 0, b.colref(1, GX_TYPE_I64),
                        b.const_i64(100))
        pchild = b.selection(psrc, [cond_p])
    j = b.hashjoin(bchild, pchild, [b.colref(0, GX_TYPE_I64)],
                   [b.colref(0, GX_TYPE_I64)])
    return b, bsrc, psrc, j


def run_join(lib, build_rows, probe_rows, with_filters=False):
    b, bsrc, psrc, j = _join_plan(lib, with_filters)
    ex = b.build(j)
    ex.bind_chunks(bsrc, [_to_chunk(lib, BUILD_TYPES, BUILD_FRACS, build_rows)])
    ex.bind_chunks(psrc, [_to_chunk(lib, PROBE_TYPES, PROBE_FRACS, probe_rows)])
    ex.open()
    rows = ex.pull_all(OUT_TYPES, OUT_FRACS)
    ex.close()
    ex.free()
    b.free()
    return _canon(rows)


BUILD_ROWS = [
    (1, "1.50", "A"),
    (2, "2.25", "B"),
    (2, "-0.10", "C"),     # duplicate build key
    (2, None, "D"),        # duplicate with NULL payload
    (3, "9.99", "E"),
    (None, "7.00", "F"),   # NULL key: never matches
    (7, "0.01", "G"),      # never probed
]
DATE = (1995 << 50) | (3 << 46) | (15 << 41)
PROBE_ROWS = [
    (2, 10, DATE),
    (2, 200, DATE + (1 << 41)),   # filtered out when with_filters
    (1, 50, DATE),
    (None, 60, DATE),             # NULL key: never matches
    (5, 70, DATE),                # no build match
    (3, 80, DATE),
    (1, 90, DATE),
]


def expected_join(with_filters=False):
    out = []
    for pr in PROBE_ROWS:
        if pr[0] is None:
            continue
        if with_filters and not pr[1] < 100:
            continue
        for br in BUILD_ROWS:
            if br[0] is None or br[0] != pr[0]:
                continue
            if with_filters and not br[0] > 1:
                continue
            # decimals display through Round-free ToString: trailing zeros kept
            out.append(br + pr)
    return _canon(out)


# ---------------- CPU: oracle vs independent computation ----------------

def test_oracle_join_duplicates():
    lib = load_oracle()
    assert run_join(lib, BUILD_ROWS, PROBE_ROWS) == expected_join()


def test_oracle_join_filters():
    lib = load_oracle()
    assert run_join(lib, BUILD_ROWS, PROBE_ROWS, with_filters=True) == \
        expected_join(with_filters=True)


# ---------------- GPU: product vs oracle ----------------

@pytest.fixture(scope="module")
def libs():
    return load_oracle(), load_product()


@pytest.mark.gpu
def test_join_parity_dup_keys(libs):
    oracle, product = libs
    got = run_join(product, BUILD_ROWS, PROBE_ROWS)
    assert got == run_join(oracle, BUILD_ROWS, PROBE_ROWS)
    assert len(got) == len(expected_join())


@pytest.mark.gpu
def test_join_parity_filters(libs):
    oracle, product = libs
    assert run_join(product, BUILD_ROWS, PROBE_ROWS, with_filters=True) == \
        run_join(oracle, BUILD_ROWS, PROBE_ROWS, with_filters=True)


@pytest.mark.gpu
def test_join_parity_empty_sides(libs):
    oracle, product = libs
    assert run_join(product, [], PROBE_ROWS) == run_join(oracle, [], PROBE_ROWS) == []
    assert run_join(product, BUILD_ROWS, []) == run_join(oracle, BUILD_ROWS, []) == []


def _run_join_random(lib, bkeys, bpay, pkeys, ppay):
    """i64-only columns filled straight from numpy (bulk upload)."""
    b = P.Builder(lib)
    bsrc = b.source([GX_TYPE_I64, GX_TYPE_I64])
    psrc = b.source([GX_TYPE_I64, GX_TYPE_I64])
    j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)],
                   [b.colref(0, GX_TYPE_I64)])
    ex = b.build(j)

    def chunk_of(k, p):
        ch = PyChunk([GX_TYPE_I64, GX_TYPE_I64], len(k))
        for col, arr in zip(ch.columns, (k, p)):
            col.data[:len(arr) * 8] = arr.astype("<i8").view(np.uint8)
            col.length = len(arr)
        return ch

    ex.bind_chunks(bsrc, [chunk_of(bkeys, bpay)])
    ex.bind_chunks(psrc, [chunk_of(pkeys, ppay)])
    ex.open()
    rows = ex.pull_all([GX_TYPE_I64] * 4)
    ex.close()
    ex.free()
    b.free()
    return sorted(rows)


def _run_join_filtered(lib, bkeys, bpay, pkeys, ppay, topn_limit=None):
    """Selection over the join = the join's other conditions
    (inner_join_probe.go:75): build.payload < probe.payload AND
    probe.payload < 9000."""
    b = P.Builder(lib)
    bsrc = b.source([GX_TYPE_I64, GX_TYPE_I64])
    psrc = b.source([GX_TYPE_I64, GX_TYPE_I64])
    j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)],
                   [b.colref(0, GX_TYPE_I64)])
    colcol = b.call(GX_F_LT, GX_TYPE_I64, 0, b.colref(1, GX_TYPE_I64),
                    b.colref(3, GX_TYPE_I64))
    colconst = b.call(GX_F_LT, GX_TYPE_I64, 0, b.colref(3, GX_TYPE_I64),
                      b.const_i64(9000))
    root = b.selection(j, [colcol, colconst])
    if topn_limit is not None:
        root = b.topn(root, [b.colref(3, GX_TYPE_I64),
                             b.colref(1, GX_TYPE_I64)], [0, 0], topn_limit)

    def chunk_of(k, p):
        ch = PyChunk([GX_TYPE_I64, GX_TYPE_I64], len(k))
        for col, arr in zip(ch.columns, (k, p)):
            col.data[:len(arr) * 8] = arr.astype("<i8").view(np.uint8)
            col.length = len(arr)
        return ch

    ex = b.build(root)
    ex.bind_chunks(bsrc, [chunk_of(bkeys, bpay)])
    ex.bind_chunks(psrc, [chunk_of(pkeys, ppay)])
    ex.open()
    rows = ex.pull_all([GX_TYPE_I64] * 4)
    ex.close()
    ex.free()
    b.free()
    return rows if topn_limit is not None else sorted(rows)


def test_oracle_join_other_condition():
    lib = load_oracle()
    bkeys, bpay, pkeys, ppay = _topn_data()
    got = _run_join_filtered(lib, bkeys, bpay, pkeys, ppay)
    want = sorted((int(bk), int(bp), int(pk), int(pp))
                  for pk, pp in zip(pkeys, ppay)
                  for bk, bp in zip(bkeys, bpay)
                  if bk == pk and bp < pp and pp < 9000)
    assert got == want
    assert 0 < len(got)


@pytest.mark.gpu
def test_join_other_condition_parity(libs):
    oracle, product = libs
    data = _topn_data()
    assert _run_join_filtered(product, *data) == _run_join_filtered(oracle, *data)


@pytest.mark.gpu
def test_join_other_condition_topn_parity(libs):
    """TopN over Selection over HashJoin — the full composed pipeline."""
    oracle, product = libs
    data = _topn_data()
    want = _run_join_filtered(oracle, *data, topn_limit=40)
    got = _run_join_filtered(product, *data, topn_limit=40)
    assert len(want) == 40
    assert got == want


def _run_join_topn(lib, bkeys, bpay, pkeys, ppay, limit, offset=0):
    """ORDER BY (probe payload, build payload) + limit/offset over the joined
    rows (TopNExec/SortExec over the join child)."""
    b = P.Builder(lib)
    bsrc = b.source([GX_TYPE_I64, GX_TYPE_I64])
    psrc = b.source([GX_TYPE_I64, GX_TYPE_I64])
    j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)],
                   [b.colref(0, GX_TYPE_I64)])
    root = b.topn(j, [b.colref(3, GX_TYPE_I64), b.colref(1, GX_TYPE_I64)],
                  [0, 0], limit, offset)
    ex = b.build(root)

    def chunk_of(k, p):
        ch = PyChunk([GX_TYPE_I64, GX_TYPE_I64], len(k))
        for col, arr in zip(ch.columns, (k, p)):
            col.data[:len(arr) * 8] = arr.astype("<i8").view(np.uint8)
            col.length = len(arr)
        return ch

    ex.bind_chunks(bsrc, [chunk_of(bkeys, bpay)])
    ex.bind_chunks(psrc, [chunk_of(pkeys, ppay)])
    ex.open()
    rows = ex.pull_all([GX_TYPE_I64] * 4)
    ex.close()
    ex.free()
    b.free()
    return rows


def _topn_data():
    rng = np.random.default_rng(7)
    bkeys = rng.integers(0, 200, 1500)
    bpay = np.arange(1500) * 7 - 5000          # unique tiebreaker
    pkeys = rng.integers(0, 400, 10000)
    ppay = np.arange(10000) - 3000             # unique primary key
    return bkeys, bpay, pkeys, ppay


def _topn_expected(limit, offset):
    bkeys, bpay, pkeys, ppay = _topn_data()
    rows = [(int(bk), int(bp), int(pk), int(pp))
            for pk, pp in zip(pkeys, ppay)
            for bk, bp in zip(bkeys, bpay) if bk == pk]
    rows.sort(key=lambda r: (r[3], r[1]))
    return rows[offset:offset + limit]


def test_oracle_join_topn():
    lib = load_oracle()
    got = _run_join_topn(lib, *_topn_data(), limit=50, offset=5)
    assert got == _topn_expected(50, 5)


@pytest.mark.gpu
def test_join_topn_parity(libs):
    oracle, product = libs
    data = _topn_data()
    want = _run_join_topn(oracle, *data, limit=50, offset=5)
    got = _run_join_topn(product, *data, limit=50, offset=5)
    assert len(want) == 50
    assert got == want


@pytest.mark.gpu
def test_join_full_orderby_parity(libs):
    """Full ORDER BY (limit -1) over the joined rows."""
    oracle, product = libs
    rng = np.random.default_rng(9)
    bkeys = rng.integers(0, 50, 300)
    bpay = np.arange(300)
    pkeys = rng.integers(0, 100, 2000)
    ppay = np.arange(2000)

    def run(lib):
        b = P.Builder(lib)
        bsrc = b.source([GX_TYPE_I64, GX_TYPE_I64])
        psrc = b.source([GX_TYPE_I64, GX_TYPE_I64])
        j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)],
                       [b.colref(0, GX_TYPE_I64)])
        root = b.sort(j, [b.colref(3, GX_TYPE_I64), b.colref(1, GX_TYPE_I64)],
                      [1, 0])  # ppay desc, bpay asc

        def chunk_of(k, p):
            ch = PyChunk([GX_TYPE_I64, GX_TYPE_I64], len(k))
            for col, arr in zip(ch.columns, (k, p)):
                col.data[:len(arr) * 8] = arr.astype("<i8").view(np.uint8)
                col.length = len(arr)
            return ch

        ex = b.build(root)
        ex.bind_chunks(bsrc, [chunk_of(bkeys, bpay)])
        ex.bind_chunks(psrc, [chunk_of(pkeys, ppay)])
        ex.open()
        rows = ex.pull_all([GX_TYPE_I64] * 4)
        ex.close()
        ex.free()
        b.free()
        return rows

    want = run(oracle)
    got = run(product)
    assert len(want) > 5000
    assert got == want


@pytest.mark.gpu
def test_join_parity_random_larger(libs):
    """2k build rows over 300 distinct keys (~7 duplicates per key) probed by
    40k rows — exercises long chains and the match-pair reservation path."""
    oracle, product = libs
    rng = np.random.default_rng(42)
    bkeys = rng.integers(0, 300, 2000)
    bpay = rng.integers(-10**9, 10**9, 2000)
    pkeys = rng.integers(0, 600, 40000)
    ppay = np.arange(40000)
    got = _run_join_random(product, bkeys, bpay, pkeys, ppay)
    want = _run_join_random(oracle, bkeys, bpay, pkeys, ppay)
    assert len(got) == len(want) > 100000
    assert got == want


def _two_key_plan(lib):
    """2-column join key (int64, time) — SerializeKeys concatenation
    (codec.go:852-910) specialized to fixed 8-byte keys."""
    b = P.Builder(lib)
    bsrc = b.source([GX_TYPE_I64, GX_TYPE_TIME, GX_TYPE_I64])
    psrc = b.source([GX_TYPE_I64, GX_TYPE_TIME, GX_TYPE_I64])
    j = b.hashjoin(bsrc, psrc,
                   [b.colref(0, GX_TYPE_I64), b.colref(1, GX_TYPE_TIME)],
                   [b.colref(0, GX_TYPE_I64), b.colref(1, GX_TYPE_TIME)])
    return b, bsrc, psrc, j


TK_BUILD = [
    (1, DATE, 100),
    (1, DATE, 101),                 # duplicate (key0,key1)
    (1, DATE + (1 << 41), 102),     # same key0, different date
    (2, DATE, 103),
    (None, DATE, 104),              # NULL first key
    (2, None, 105),                 # NULL second key
]
TK_PROBE = [
    (1, DATE, 1),
    (1, DATE + (1 << 41), 2),
    (2, DATE, 3),
    (2, None, 4),                   # NULL key: no match
    (3, DATE, 5),
]


def run_two_key(lib):
    b, bsrc, psrc, j = _two_key_plan(lib)
    ex = b.build(j)
    t3 = [GX_TYPE_I64, GX_TYPE_TIME, GX_TYPE_I64]
    ex.bind_chunks(bsrc, [_to_chunk(lib, t3, [0, 0, 0], TK_BUILD)])
    ex.bind_chunks(psrc, [_to_chunk(lib, t3, [0, 0, 0], TK_PROBE)])
    ex.open()
    rows = ex.pull_all(t3 + t3)
    ex.close()
    ex.free()
    b.free()
    return _canon(rows)


def test_oracle_join_two_keys():
    lib = load_oracle()
    want = _canon([br + pr for pr in TK_PROBE for br in TK_BUILD
                   if None not in (br[0], br[1], pr[0], pr[1])
                   and br[0] == pr[0] and br[1] == pr[1]])
    got = run_two_key(lib)
    assert got == want
    assert len(got) == 4  # (1,DATE)x2 matches, (1,DATE+1d), (2,DATE)


@pytest.mark.gpu
def test_join_parity_two_keys(libs):
    oracle, product = libs
    assert run_two_key(product) == run_two_key(oracle)


@pytest.mark.gpu
def test_join_parity_generator(libs):
    """orders ⋈ lineitem on orderkey over the synthetic generator tables,
    order-date filter on the build side."""
    oracle, product = libs

    def run(lib):
        b = P.Builder(lib)
        orders = b.source(P.ORDERS_TYPES)
        cond = b.call(GX_F_LT, GX_TYPE_I64, 0,
                      b.colref(P.O_ORDERDATE, GX_TYPE_TIME),
                      b.const_time(lib.gx_time_from_date(1995, 3, 15)))
        sel_o = b.selection(orders, [cond])
        li = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
        j = b.hashjoin(sel_o, li, [b.colref(P.O_ORDERKEY, GX_TYPE_I64)],
                       [b.colref(P.L_ORDERKEY, GX_TYPE_I64)])
        ex = b.build(j)
        ex.bind_tpch(orders, GX_TPCH_ORDERS, 5000)
        ex.bind_tpch(li, GX_TPCH_LINEITEM, 20000)
        ex.open()
        out_types = P.ORDERS_TYPES + P.LINEITEM_TYPES
        out_fracs = [0] * 4 + P.LINEITEM_FRACS
        rows = ex.pull_all(out_types, out_fracs)
        ex.close()
        ex.free()
        b.free()
        return sorted(rows)

    got = run(product)
    want = run(oracle)
    assert len(got) == len(want) > 1000
    assert got == want


# ---------------- aggregation over joined rows ----------------

def _agg_over_join_plan(lib, topn_limit=None):
    """sum(probe value dec(s2)), count(*), group by build.grp — an aggregate
    shape the Q3-class fused pipeline does not cover (group key from the
    BUILD side): exercises HashAgg over materialized joined rows."""
    from tests.gxlib import GX_AGG_SUM, GX_AGG_COUNT
    b = P.Builder(lib)
    bsrc = b.source([GX_TYPE_I64, GX_TYPE_I64])           # key, grp
    psrc = b.source([GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 2])  # key, val
    j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)],
                   [b.colref(0, GX_TYPE_I64)])
    grp = b.colref(1, GX_TYPE_I64)
    val = b.colref(3, GX_TYPE_DECIMAL, 2)
    agg = b.hashagg(j, [grp], [(GX_AGG_SUM, val, 2), (GX_AGG_COUNT, -1, 0)])
    root = agg
    if topn_limit is not None:
        root = b.topn(agg, [b.colref(0, GX_TYPE_I64)], [0], topn_limit)
    return b, bsrc, psrc, root


def _agg_over_join_data():
    rng = np.random.default_rng(11)
    n_b, n_p = 800, 6000
    build = [(int(k), int(g)) for k, g in zip(rng.integers(0, 150, n_b),
                                              rng.integers(0, 40, n_b))]
    probe = [(int(k), "%d.%02d" % (v // 100, v % 100))
             for k, v in zip(rng.integers(0, 300, n_p),
                             rng.integers(0, 100000, n_p))]
    return build, probe


def run_agg_over_join(lib, topn_limit=None):
    build, probe = _agg_over_join_data()
    b, bsrc, psrc, root = _agg_over_join_plan(lib, topn_limit)
    ex = b.build(root)
    ex.bind_chunks(bsrc, [_to_chunk(lib, [GX_TYPE_I64, GX_TYPE_I64], [0, 0], build)])
    ex.bind_chunks(psrc, [_to_chunk(lib, [GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 2], probe)])
    ex.open()
    rows = ex.pull_all([GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64], [0, 2, 0])
    ex.close()
    ex.free()
    b.free()
    return rows if topn_limit is not None else sorted(rows)


def test_oracle_agg_over_join():
    from fractions import Fraction
    lib = load_oracle()
    build, probe = _agg_over_join_data()
    got = run_agg_over_join(lib)
    groups = {}
    for pk, pv in probe:
        for bk, bg in build:
            if bk == pk:
                s, c = groups.get(bg, (Fraction(0), 0))
                groups[bg] = (s + Fraction(pv), c + 1)
    want = sorted((g, s, c) for g, (s, c) in groups.items())
    assert [(r[0], Fraction(r[1]), r[2]) for r in got] == want
    assert len(got) == len(want) > 10


@pytest.mark.gpu
def test_agg_over_join_parity(libs):
    oracle, product = libs
    assert run_agg_over_join(product) == run_agg_over_join(oracle)


@pytest.mark.gpu
def test_agg_over_join_topn_parity(libs):
    """TopN over HashAgg over HashJoin (host post-sort of the agg output)."""
    oracle, product = libs
    want = run_agg_over_join(oracle, topn_limit=15)
    got = run_agg_over_join(product, topn_limit=15)
    assert len(want) == 15
    assert got == want


@pytest.mark.gpu
def test_join_parity_varlen_output(libs):
    """General varlen output column (c_mktsegment, 8-10 byte strings):
    two-pass varlen gather (lengths -> scan -> bytes) through the match
    index — the chunk.Column offsets+data layout end to end."""
    from tests.gxlib import GX_TPCH_CUSTOMER, GX_TPCH_ORDERS
    oracle, product = libs

    def run(lib):
        b = P.Builder(lib)
        cust = b.source(P.CUSTOMER_TYPES)
        orders = b.source(P.ORDERS_TYPES)
        j = b.hashjoin(cust, orders,
                       [b.colref(P.C_CUSTKEY, GX_TYPE_I64)],
                       [b.colref(P.O_CUSTKEY, GX_TYPE_I64)])
        ex = b.build(j)
        ex.bind_tpch(cust, GX_TPCH_CUSTOMER, 500)
        ex.bind_tpch(orders, GX_TPCH_ORDERS, 5000)
        ex.open()
        out_types = P.CUSTOMER_TYPES + P.ORDERS_TYPES
        rows = ex.pull_all(out_types, [0] * 6,
                           data_caps=[None, 65536] + [None] * 4)
        ex.close()
        ex.free()
        b.free()
        return sorted(rows)

    got = run(product)
    want = run(oracle)
    assert len(got) == len(want) > 1000
    assert got == want


def _full_agg_over_join(lib):
    """avg/min/max/firstrow + sum/count over joined rows, group by a build
    column — the full aggfuncs family (func_avg.go, func_max_min.go,
    func_sum.go) over a join child."""
    from tests.gxlib import (GX_AGG_AVG, GX_AGG_COUNT, GX_AGG_FIRSTROW,
                            GX_AGG_MAX, GX_AGG_MIN, GX_AGG_SUM)
    build, probe = _agg_over_join_data()
    b = P.Builder(lib)
    bsrc = b.source([GX_TYPE_I64, GX_TYPE_I64])
    psrc = b.source([GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 2])
    j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)],
                   [b.colref(0, GX_TYPE_I64)])
    grp = b.colref(1, GX_TYPE_I64)
    val = b.colref(3, GX_TYPE_DECIMAL, 2)
    key = b.colref(2, GX_TYPE_I64)
    agg = b.hashagg(j, [grp], [
        (GX_AGG_SUM, val, 2), (GX_AGG_AVG, val, 6), (GX_AGG_MIN, key, 0),
        (GX_AGG_MAX, key, 0), (GX_AGG_COUNT, -1, 0), (GX_AGG_FIRSTROW, grp, 0),
    ])
    ex = b.build(agg)
    ex.bind_chunks(bsrc, [_to_chunk(lib, [GX_TYPE_I64, GX_TYPE_I64], [0, 0], build)])
    ex.bind_chunks(psrc, [_to_chunk(lib, [GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 2], probe)])
    ex.open()
    out_t = [GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_DECIMAL, GX_TYPE_I64,
             GX_TYPE_I64, GX_TYPE_I64, GX_TYPE_I64]
    rows = ex.pull_all(out_t, [0, 2, 6, 0, 0, 0, 0])
    ex.close()
    ex.free()
    b.free()
    return sorted(rows)


@pytest.mark.gpu
def test_full_agg_family_over_join_parity(libs):
    oracle, product = libs
    got = _full_agg_over_join(product)
    want = _full_agg_over_join(oracle)
    assert len(got) == len(want) > 10
    assert got == want


def test_oracle_join_varlen_output():
    """CPU pin for the varlen-output join: oracle hashjoin of generator
    customer ⋈ orders equals a pure-Python join of the pulled tables."""
    from tests.gxlib import GX_TPCH_CUSTOMER, GX_TPCH_ORDERS
    lib = load_oracle()

    def pull(types, table, n, caps):
        b = P.Builder(lib)
        src = b.source(types)
        ex = b.build(src)
        ex.bind_tpch(src, table, n)
        ex.open()
        rows = ex.pull_all(types, [0] * len(types), data_caps=caps)
        ex.close()
        ex.free()
        b.free()
        return rows

    cust = pull(P.CUSTOMER_TYPES, GX_TPCH_CUSTOMER, 300, [None, 65536])
    orders = pull(P.ORDERS_TYPES, GX_TPCH_ORDERS, 3000, [None] * 4)
    want = sorted(tuple(c) + tuple(o) for o in orders for c in cust
                  if c[0] == o[1])

    b = P.Builder(lib)
    csrc = b.source(P.CUSTOMER_TYPES)
    osrc = b.source(P.ORDERS_TYPES)
    j = b.hashjoin(csrc, osrc, [b.colref(P.C_CUSTKEY, GX_TYPE_I64)],
                   [b.colref(P.O_CUSTKEY, GX_TYPE_I64)])
    ex = b.build(j)
    ex.bind_tpch(csrc, GX_TPCH_CUSTOMER, 300)
    ex.bind_tpch(osrc, GX_TPCH_ORDERS, 3000)
    ex.open()
    got = sorted(ex.pull_all(P.CUSTOMER_TYPES + P.ORDERS_TYPES, [0] * 6,
                             data_caps=[None, 65536] + [None] * 4))
    ex.close()
    ex.free()
    b.free()
    assert got == want
    assert len(got) > 500


# ---------------- nested joins (stage compilation) ----------------

def _nested_data():
    rng = np.random.default_rng(21)
    t0 = [(int(k), int(v)) for k, v in zip(rng.integers(0, 40, 200),
                                           np.arange(200))]        # inner build
    t1 = [(int(k), int(k2), int(v)) for k, k2, v in zip(
        rng.integers(0, 80, 1000), rng.integers(0, 60, 1000),
        np.arange(1000))]                                          # inner probe
    t2 = [(int(k2), int(v)) for k2, v in zip(rng.integers(0, 120, 4000),
                                             np.arange(4000))]     # outer probe
    return t0, t1, t2


def _run_nested(lib):
    """(t0 ⋈ t1 on k) ⋈ t2 on k2 — duplicate keys at every level; the outer
    build side is the inner join's materialized output."""
    t0, t1, t2 = _nested_data()
    b = P.Builder(lib)
    s0 = b.source([GX_TYPE_I64, GX_TYPE_I64])
    s1 = b.source([GX_TYPE_I64, GX_TYPE_I64, GX_TYPE_I64])
    s2 = b.source([GX_TYPE_I64, GX_TYPE_I64])
    j1 = b.hashjoin(s0, s1, [b.colref(0, GX_TYPE_I64)],
                    [b.colref(0, GX_TYPE_I64)])
    # j1 out: t0.k, t0.v, t1.k, t1.k2, t1.v  -> join t2 on k2 (col 3)
    j2 = b.hashjoin(j1, s2, [b.colref(3, GX_TYPE_I64)],
                    [b.colref(0, GX_TYPE_I64)])
    ex = b.build(j2)
    for src, rows, tps in ((s0, t0, 2), (s1, t1, 3), (s2, t2, 2)):
        ex.bind_chunks(src, [_to_chunk(lib, [GX_TYPE_I64] * tps, [0] * tps,
                                       rows)])
    ex.open()
    rows = ex.pull_all([GX_TYPE_I64] * 7)
    ex.close()
    ex.free()
    b.free()
    return sorted(rows)


def test_oracle_nested_join():
    t0, t1, t2 = _nested_data()
    want = sorted(a + bb + c
                  for bb in t1 for a in t0 if a[0] == bb[0]
                  for c in t2 if c[0] == bb[1])
    got = _run_nested(load_oracle())
    assert got == want
    assert len(got) > 3000


@pytest.mark.gpu
def test_nested_join_parity(libs):
    oracle, product = libs
    got = _run_nested(product)
    want = _run_nested(oracle)
    assert len(got) == len(want) > 3000
    assert got == want


@pytest.mark.gpu
def test_q3_class_agg_over_nested_join_parity(libs):
    """customer⋈orders⋈lineitem with a CUSTOMER-side group key — a Q3-class
    query the fused join-aggregate pipeline rejects: runs as nested join
    stages + fused aggregation over the materialized output."""
    from tests.gxlib import (GX_AGG_COUNT, GX_AGG_SUM, GX_F_EQ, GX_F_GT,
                             GX_F_MINUS, GX_F_MUL, GX_TPCH_CUSTOMER,
                             GX_TPCH_LINEITEM, GX_TPCH_ORDERS)
    oracle, product = libs

    def run(lib):
        b = P.Builder(lib)
        cust = b.source(P.CUSTOMER_TYPES)
        seg = b.colref(P.C_MKTSEGMENT, GX_TYPE_STRING)
        sel_c = b.selection(cust, [b.call(GX_F_EQ, GX_TYPE_I64, 0, seg,
                                          lib.gx_pb_const_str(b.pb, b"BUILDING", 8))])
        orders = b.source(P.ORDERS_TYPES)
        li = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
        sdate = b.colref(P.L_SHIPDATE, GX_TYPE_TIME)
        sel_l = b.selection(li, [b.call(GX_F_GT, GX_TYPE_I64, 0, sdate,
                                        b.const_time(lib.gx_time_from_date(1995, 3, 15)))])
        j1 = b.hashjoin(sel_c, orders, [b.colref(P.C_CUSTKEY, GX_TYPE_I64)],
                        [b.colref(P.O_CUSTKEY, GX_TYPE_I64)])
        # j1 out: c_custkey, c_mktsegment, o_orderkey, o_custkey, o_odate, o_prio
        j2 = b.hashjoin(j1, sel_l, [b.colref(2, GX_TYPE_I64)],
                        [b.colref(P.L_ORDERKEY, GX_TYPE_I64)])
        ckey = b.colref(0, GX_TYPE_I64)
        price = b.colref(6 + P.L_EXTPRICE, GX_TYPE_DECIMAL, 2)
        disc = b.colref(6 + P.L_DISCOUNT, GX_TYPE_DECIMAL, 2)
        one = P._const_dec_one(lib, b)
        rev = b.call(GX_F_MUL, GX_TYPE_DECIMAL, 4, price,
                     b.call(GX_F_MINUS, GX_TYPE_DECIMAL, 2, one, disc))
        proj = b.projection(j2, [ckey, rev])
        agg = b.hashagg(proj, [b.colref(0, GX_TYPE_I64)],
                        [(GX_AGG_SUM, b.colref(1, GX_TYPE_DECIMAL, 4), 4),
                         (GX_AGG_COUNT, -1, 0)])
        ex = b.build(agg)
        ex.bind_tpch(cust, GX_TPCH_CUSTOMER, 1000)
        ex.bind_tpch(orders, GX_TPCH_ORDERS, 10000)
        ex.bind_tpch(li, GX_TPCH_LINEITEM, 40000)
        ex.open()
        rows = ex.pull_all([GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64],
                           [0, 4, 0])
        ex.close()
        ex.free()
        b.free()
        return sorted(rows)

    got = run(product)
    want = run(oracle)
    assert len(got) == len(want) > 50
    assert got == want


def test_oracle_full_agg_family_over_join():
    """Independent (non-oracle) pin of avg/min/max/firstrow/sum/count over
    joined rows — guards the oracle itself (a firstrow aux-flag bug was
    caught by exactly this kind of check)."""
    from fractions import Fraction
    lib = load_oracle()
    build, probe = _agg_over_join_data()
    got = _full_agg_over_join(lib)
    groups = {}
    for pk, pv in probe:
        for bk, bg in build:
            if bk != pk:
                continue
            g = groups.setdefault(bg, {"sum": Fraction(0), "cnt": 0,
                                       "min": None, "max": None})
            g["sum"] += Fraction(pv)
            g["cnt"] += 1
            g["min"] = pk if g["min"] is None else min(g["min"], pk)
            g["max"] = pk if g["max"] is None else max(g["max"], pk)
    assert len(got) == len(groups) > 10
    for r in got:
        g = groups[r[0]]
        assert Fraction(r[1]) == g["sum"]
        # avg = Round(sum / count, 6, HalfUp) — check via scaled integers
        num, den = (g["sum"] / g["cnt"]).numerator, (g["sum"] / g["cnt"]).denominator
        scaled2 = abs(num) * 10**6 * 2
        q, rem = divmod(scaled2, den * 2)
        avg = q + (1 if 2 * rem >= den * 2 else 0)
        if num < 0:
            avg = -avg
        assert round(Fraction(r[2]) * 10**6) == avg
        assert r[3] == g["min"] and r[4] == g["max"]
        assert r[5] == g["cnt"]
        assert r[6] == r[0]  # firstrow(group col) == the group value


def test_oracle_join_multi_chunk_sources():
    """Sources bound as MULTIPLE chunks (the Next-pulls-many-chunks shape):
    same result as a single concatenated chunk."""
    lib = load_oracle()
    b1 = BUILD_ROWS[:3]
    b2 = BUILD_ROWS[3:]
    p1 = PROBE_ROWS[:4]
    p2 = PROBE_ROWS[4:]
    b, bsrc, psrc, j = _join_plan(lib, False)
    ex = b.build(j)
    ex.bind_chunks(bsrc, [_to_chunk(lib, BUILD_TYPES, BUILD_FRACS, b1),
                          _to_chunk(lib, BUILD_TYPES, BUILD_FRACS, b2)])
    ex.bind_chunks(psrc, [_to_chunk(lib, PROBE_TYPES, PROBE_FRACS, p1),
                          _to_chunk(lib, PROBE_TYPES, PROBE_FRACS, p2)])
    ex.open()
    rows = ex.pull_all(OUT_TYPES, OUT_FRACS)
    ex.close()
    ex.free()
    b.free()
    assert _canon(rows) == expected_join()


def _run_join_sort_nullable(lib, limit, offset):
    """ORDER BY unique probe payload + limit/OFFSET over joined rows whose
    BUILD payload column carries NULLs. Regression for the null-bitmap
    bit-shift at unaligned sort offsets (emitTableChunk srcPos%8 != 0):
    with offset=5 the first emitted chunk starts mid-byte in the bitmap."""
    b = P.Builder(lib)
    bsrc = b.source([GX_TYPE_I64, GX_TYPE_I64])
    psrc = b.source([GX_TYPE_I64, GX_TYPE_I64])
    j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)],
                   [b.colref(0, GX_TYPE_I64)])
    root = b.topn(j, [b.colref(3, GX_TYPE_I64)], [0], limit, offset)
    ex = b.build(root)
    bch = PyChunk([GX_TYPE_I64, GX_TYPE_I64], 256)
    for i in range(200):
        bch.append_row([i, None if i % 3 == 0 else i * 11])
    pch = PyChunk([GX_TYPE_I64, GX_TYPE_I64], 1024)
    rng = np.random.default_rng(31)
    pkeys = rng.integers(0, 200, 600)
    for i, k in enumerate(pkeys):
        pch.append_row([int(k), i])
    ex.bind_chunks(bsrc, [bch])
    ex.bind_chunks(psrc, [pch])
    ex.open()
    rows = ex.pull_all([GX_TYPE_I64] * 4)
    ex.close()
    ex.free()
    b.free()
    want = sorted(((int(k), None if k % 3 == 0 else int(k) * 11, int(k), i)
                   for i, k in enumerate(pkeys)), key=lambda r: r[3])
    return rows, want[offset:offset + limit]


def test_oracle_join_sort_nullable_offset():
    lib = load_oracle()
    got, want = _run_join_sort_nullable(lib, 80, 5)
    assert got == want


@pytest.mark.gpu
@pytest.mark.parametrize("offset", [5, 13, 0])
def test_join_sort_nullable_offset_parity(libs, offset):
    oracle, product = libs
    got, want = _run_join_sort_nullable(product, 80, offset)
    assert got == want
    got_o, _ = _run_join_sort_nullable(oracle, 80, offset)
    assert got == got_o


@pytest.mark.gpu
def test_join_reopen_stable(libs):
    """Re-open/Next cycles on one executor: identical results and no
    unbounded device-buffer growth (per-run buffers are freed at re-run)."""
    _, product = libs
    b = P.Builder(product)
    bsrc = b.source([GX_TYPE_I64, GX_TYPE_I64])
    psrc = b.source([GX_TYPE_I64, GX_TYPE_I64])
    j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)],
                   [b.colref(0, GX_TYPE_I64)])
    root = b.topn(j, [b.colref(3, GX_TYPE_I64)], [0], 40, 3)
    ex = b.build(root)
    bch = PyChunk([GX_TYPE_I64, GX_TYPE_I64], 128)
    for i in range(100):
        bch.append_row([i, i * 7])
    pch = PyChunk([GX_TYPE_I64, GX_TYPE_I64], 512)
    rng = np.random.default_rng(5)
    for i, k in enumerate(rng.integers(0, 100, 400)):
        pch.append_row([int(k), i])
    ex.bind_chunks(bsrc, [bch])
    ex.bind_chunks(psrc, [pch])
    first = None
    for _ in range(4):
        ex.open()
        rows = ex.pull_all([GX_TYPE_I64] * 4)
        ex.close()
        if first is None:
            first = rows
        assert rows == first
    ex.free()
    b.free()
