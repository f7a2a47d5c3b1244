"""Full ORDER BY over a large source stream (sortexec/sort.go:50,546 analog):
oracle semantics on CPU, device radix sort parity on GPU (-m gpu).

Ties: TiDB's sort is not stable across equal keys, so parity compares
(a) the key-column sequences exactly and (b) the full row multiset.
"""
import pytest

from tests.gxlib import (GX_TPCH_LINEITEM, GX_TYPE_I64, GX_TYPE_TIME,
                         GX_TYPE_DECIMAL, load_oracle)
from tidb_amd import plan as P

N_ROWS = 20000


def sort_plan(lib, keys, desc, limit=-1, offset=0):
    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    key_exprs = [b.colref(c, t, f) for (c, t, f) in keys]
    if limit >= 0:
        root = b.topn(src, key_exprs, desc, limit, offset)
    else:
        root = b.sort(src, key_exprs, desc)
    return b, src, root


def run_sort(lib, keys, desc, n_rows=N_ROWS, limit=-1, offset=0):
    b, src, root = sort_plan(lib, keys, desc, limit, offset)
    ex = b.build(root)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n_rows)
    ex.open()
    rows = ex.pull_all(P.LINEITEM_TYPES, P.LINEITEM_FRACS,
                       data_caps=[None] * 5 + [2048, 2048] + [None])
    ex.close()
    ex.free()
    b.free()
    return rows


KEYS_2 = [(P.L_SHIPDATE, GX_TYPE_TIME, 0), (P.L_ORDERKEY, GX_TYPE_I64, 0)]


def test_sort_oracle_sorted(oracle_lib):
    rows = run_sort(oracle_lib, KEYS_2, [0, 1])  # shipdate asc, orderkey desc
    assert len(rows) == N_ROWS
    keys = [(r[7] & ~0xF, -r[0]) for r in rows]
    assert keys == sorted(keys)


def test_sort_oracle_permutation(oracle_lib):
    from tests.test_oracle_q1 import pull_lineitem
    rows = run_sort(oracle_lib, KEYS_2, [0, 0], n_rows=5000)
    raw = pull_lineitem(oracle_lib, 5000)
    assert sorted(map(tuple, rows)) == sorted(map(tuple, raw))


@pytest.mark.gpu
@pytest.mark.parametrize("n", [5000, 300000])
def test_sort_parity(n):
    from tests.gxlib import load_product
    a = run_sort(load_oracle(), KEYS_2, [0, 1], n_rows=n)
    b = run_sort(load_product(), KEYS_2, [0, 1], n_rows=n)
    assert len(a) == len(b) == n
    # key columns identical row-for-row; full rows equal as multisets
    assert [(r[7], r[0]) for r in a] == [(r[7], r[0]) for r in b]
    assert sorted(map(tuple, a)) == sorted(map(tuple, b))


@pytest.mark.gpu
def test_sort_parity_decimal_key():
    """decimal sort key: parsed to int64 units on device; oracle compares
    MyDecimal (types/mydecimal.go Compare semantics on aligned fracs)."""
    from tests.gxlib import load_product
    keys = [(P.L_EXTPRICE, GX_TYPE_DECIMAL, 2),
            (P.L_ORDERKEY, GX_TYPE_I64, 0)]
    a = run_sort(load_oracle(), keys, [1, 0], n_rows=20000)
    b = run_sort(load_product(), keys, [1, 0], n_rows=20000)
    assert [(r[2], r[0]) for r in a] == [(r[2], r[0]) for r in b]
    assert sorted(map(tuple, a)) == sorted(map(tuple, b))


@pytest.mark.gpu
def test_topn_source_limit_offset_parity():
    from tests.gxlib import load_product
    a = run_sort(load_oracle(), KEYS_2, [0, 1], n_rows=8000, limit=100,
                 offset=7)
    b = run_sort(load_product(), KEYS_2, [0, 1], n_rows=8000, limit=100,
                 offset=7)
    assert len(a) == len(b) == 100
    assert [(r[7], r[0]) for r in a] == [(r[7], r[0]) for r in b]


@pytest.mark.gpu
def test_spill_sort_parity(monkeypatch):
    """Out-of-core sort: GX_SORT_RUN_ROWS forces 7 device-sorted runs spilled
    to host + k-way merge on emission (sort_spill.go / multi_way_merge.go
    analog). Same key order and row multiset as the in-HBM sort / oracle."""
    from tests.gxlib import load_product
    a = run_sort(load_oracle(), KEYS_2, [0, 1], n_rows=100000)
    monkeypatch.setenv("GX_SORT_RUN_ROWS", "15000")
    b = run_sort(load_product(), KEYS_2, [0, 1], n_rows=100000)
    monkeypatch.delenv("GX_SORT_RUN_ROWS")
    assert len(a) == len(b) == 100000
    assert [(r[7], r[0]) for r in a] == [(r[7], r[0]) for r in b]
    assert sorted(map(tuple, a)) == sorted(map(tuple, b))


@pytest.mark.gpu
def test_spill_topn_limit_offset_parity(monkeypatch):
    """limit/offset served from the spill merge (TopN over a spilled sort)."""
    from tests.gxlib import load_product
    a = run_sort(load_oracle(), KEYS_2, [0, 1], n_rows=50000, limit=200,
                 offset=13)
    monkeypatch.setenv("GX_SORT_RUN_ROWS", "9000")
    b = run_sort(load_product(), KEYS_2, [0, 1], n_rows=50000, limit=200,
                 offset=13)
    monkeypatch.delenv("GX_SORT_RUN_ROWS")
    assert len(a) == len(b) == 200
    assert [(r[7], r[0]) for r in a] == [(r[7], r[0]) for r in b]


@pytest.mark.gpu
def test_spill_sort_decimal_key_parity(monkeypatch):
    """Spilled runs with a decimal sort key (composed int64-unit keys per
    run; merge compares the same composed keys)."""
    from tests.gxlib import load_product
    keys = [(P.L_EXTPRICE, GX_TYPE_DECIMAL, 2), (P.L_ORDERKEY, GX_TYPE_I64, 0)]
    a = run_sort(load_oracle(), keys, [1, 0], n_rows=40000)
    monkeypatch.setenv("GX_SORT_RUN_ROWS", "7000")
    b = run_sort(load_product(), keys, [1, 0], n_rows=40000)
    monkeypatch.delenv("GX_SORT_RUN_ROWS")
    assert [(r[2], r[0]) for r in a] == [(r[2], r[0]) for r in b]
    assert sorted(map(tuple, a)) == sorted(map(tuple, b))
