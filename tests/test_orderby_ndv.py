"""ORDER BY over the aggregate (full Q1 statement shape) and mid/high-NDV
grouping — oracle semantics on CPU, product parity on GPU (-m gpu)."""
import pytest

from tests.gxlib import (GX_AGG_COUNT, GX_AGG_SUM, GX_TPCH_LINEITEM,
                         GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_STRING,
                         load_oracle)
from tidb_amd import plan as P


def q1_orderby_plan(lib):
    """Q1 + its final Sort (ORDER BY l_returnflag, l_linestatus) — the
    complete statement (tpch golden plan's trailing Sort)."""
    b, src, agg, out_types, out_fracs = P.q1_plan(lib)
    rf = b.colref(0, GX_TYPE_STRING)
    ls = b.colref(1, GX_TYPE_STRING)
    root = b.sort(agg, [rf, ls], [0, 0])
    return b, src, root, out_types, out_fracs


def run_plan(lib, build_fn, n_rows, **kw):
    b, src, root, out_types, out_fracs = build_fn(lib)
    ex = b.build(root)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n_rows, **kw)
    ex.open()
    caps = [2048 if t == 4 else None for t in out_types]
    rows = ex.pull_all(out_types, out_fracs, data_caps=caps)
    ex.close()
    ex.free()
    b.free()
    return rows


def orderkey_agg_plan(lib):
    """Mid-NDV grouping: sum(quantity), count(*) group by l_orderkey —
    thousands of groups (exceeds the device LDS table; exercises the
    global-direct path), ordered by orderkey."""
    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    okey = b.colref(P.L_ORDERKEY, GX_TYPE_I64)
    qty = b.colref(P.L_QUANTITY, GX_TYPE_DECIMAL, 2)
    agg = b.hashagg(src, [okey], [(GX_AGG_SUM, qty, 2), (GX_AGG_COUNT, -1, 0)])
    root = b.sort(agg, [b.colref(0, GX_TYPE_I64)], [0])
    out_types = [GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64]
    out_fracs = [0, 2, 0]
    return b, src, root, out_types, out_fracs


def test_q1_orderby_oracle(oracle_lib):
    rows = run_plan(oracle_lib, q1_orderby_plan, 20000)
    keys = [(r[0], r[1]) for r in rows]
    assert keys == sorted(keys)
    assert 4 <= len(rows) <= 6


def test_orderkey_agg_oracle(oracle_lib):
    from fractions import Fraction
    rows = run_plan(oracle_lib, orderkey_agg_plan, 8000)
    assert len(rows) > 1000  # ~2000 distinct orderkeys
    keys = [r[0] for r in rows]
    assert keys == sorted(keys)
    # independent check of one group
    from tests.test_oracle_q1 import pull_lineitem
    raw = pull_lineitem(oracle_lib, 8000)
    k0 = rows[0][0]
    want_cnt = sum(1 for r in raw if r[0] == k0)
    want_qty = sum(Fraction(r[1]) for r in raw if r[0] == k0)
    assert rows[0][2] == want_cnt
    assert Fraction(rows[0][1]) == want_qty


@pytest.mark.gpu
def test_q1_orderby_parity():
    from tests.gxlib import load_product
    a = run_plan(load_oracle(), q1_orderby_plan, 50000)
    b = run_plan(load_product(), q1_orderby_plan, 50000)
    assert a == b


@pytest.mark.gpu
def test_orderkey_agg_parity():
    """NDV ~2000 > kLdsGroups: exercises the global-direct aggregation."""
    from tests.gxlib import load_product
    a = run_plan(load_oracle(), orderkey_agg_plan, 8000)
    b = run_plan(load_product(), orderkey_agg_plan, 8000)
    assert a == b


@pytest.mark.gpu
def test_orderkey_agg_high_ndv_parity():
    """NDV ~60k > the default 8192-slot global table: exercises the
    kErrGlobalFull grow-and-rerun retry (globalGroupsLog2 13 -> 16)."""
    from tests.gxlib import load_product
    a = run_plan(load_oracle(), orderkey_agg_plan, 240000)
    b = run_plan(load_product(), orderkey_agg_plan, 240000)
    assert len(a) > 8192
    assert a == b
