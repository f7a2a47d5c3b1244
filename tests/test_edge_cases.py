"""Edge cases the reference's own tests cover: empty inputs,
single rows, fully-filtered scans, and TopN larger than the input."""
import pytest

from tests.gxlib import (GX_TPCH_CUSTOMER, GX_TPCH_LINEITEM, GX_TPCH_ORDERS,
                         GX_TYPE_I64, GX_TYPE_TIME, load_oracle)
from tidb_amd import plan as P


def run_q1_n(lib, n_rows):
    from tests.test_gpu_parity import _run_q1
    return _run_q1(lib, n_rows)


def test_q1_empty_oracle(oracle_lib):
    assert run_q1_n(oracle_lib, 0) == []


def test_q1_one_row_oracle(oracle_lib):
    rows = run_q1_n(oracle_lib, 1)
    assert len(rows) == 1 and rows[0][9] == 1  # count(*) == 1


def test_sort_empty_oracle(oracle_lib):
    from tests.test_full_sort import run_sort, KEYS_2
    assert run_sort(oracle_lib, KEYS_2, [0, 0], n_rows=0) == []


def test_topn_larger_than_input_oracle(oracle_lib):
    from tests.test_full_sort import run_sort, KEYS_2
    rows = run_sort(oracle_lib, KEYS_2, [0, 1], n_rows=7, limit=100)
    assert len(rows) == 7


def q3_no_match(lib):
    """Customer table too small for any BUILDING match at n=1 is flaky;
    instead use 0 lineitem rows -> empty join output -> 0 TopN rows."""
    b, (cust, orders, li), topn, out_types, out_fracs = P.q3_plan(lib)
    ex = b.build(topn)
    ex.bind_tpch(cust, GX_TPCH_CUSTOMER, 100)
    ex.bind_tpch(orders, GX_TPCH_ORDERS, 400)
    ex.bind_tpch(li, GX_TPCH_LINEITEM, 0)
    ex.open()
    rows = ex.pull_all(out_types, out_fracs, data_caps=[None] * 4)
    ex.close()
    ex.free()
    b.free()
    return rows


def test_q3_empty_probe_oracle(oracle_lib):
    assert q3_no_match(oracle_lib) == []


def _scalar_agg_rows(lib, nrows):
    """SELECT count(c0), sum(c1) with NO group-by: zero input rows must
    still produce ONE row (count=0, sum NULL) — HashAggExec empty-input
    semantics (executor/aggregate tests; aggregate.result goldens)."""
    import ctypes

    from tests.gxlib import (GX_AGG_COUNT, GX_AGG_SUM, GX_TYPE_DECIMAL)
    from tidb_amd.chunkpy import PyChunk
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 2])
    agg = b.hashagg(src, [], [(GX_AGG_COUNT, b.colref(0, GX_TYPE_I64), 0),
                              (GX_AGG_SUM, b.colref(1, GX_TYPE_DECIMAL, 2),
                               2)])
    ex = b.build(agg)
    ch = PyChunk([GX_TYPE_I64, GX_TYPE_DECIMAL], max(nrows, 1), [0, 2])
    out = (ctypes.c_uint8 * 40)()
    assert lib.gx_dec_from_string(b"1.50", 4, out) == 0
    for i in range(nrows):
        ch.append_row([i, bytes(out)])
    ex.bind_chunks(src, [ch])
    ex.open()
    rows = ex.pull_all([GX_TYPE_I64, GX_TYPE_DECIMAL], [0, 2])
    ex.close()
    ex.free()
    b.free()
    return rows


def test_scalar_agg_empty_oracle(oracle_lib):
    assert _scalar_agg_rows(oracle_lib, 0) == [(0, None)]
    assert _scalar_agg_rows(oracle_lib, 3) == [(3, "4.50")]


@pytest.mark.gpu
@pytest.mark.parametrize("n", [0, 3])
def test_scalar_agg_empty_parity(n):
    from tests.gxlib import load_product
    assert _scalar_agg_rows(load_product(), n) == \
        _scalar_agg_rows(load_oracle(), n)


@pytest.mark.gpu
@pytest.mark.parametrize("n", [0, 1, 63, 64, 65, 255, 1023])
def test_q1_tiny_parity(n):
    from tests.gxlib import load_product
    from tests.test_gpu_parity import _as_map
    assert _as_map(run_q1_n(load_product(), n)) == \
        _as_map(run_q1_n(load_oracle(), n))


@pytest.mark.gpu
def test_q3_empty_probe_parity():
    from tests.gxlib import load_product
    assert q3_no_match(load_product()) == q3_no_match(load_oracle()) == []


@pytest.mark.gpu
def test_sort_tiny_parity():
    from tests.gxlib import load_product
    from tests.test_full_sort import run_sort, KEYS_2
    for n in (0, 1, 65):
        a = run_sort(load_oracle(), KEYS_2, [0, 1], n_rows=n)
        b = run_sort(load_product(), KEYS_2, [0, 1], n_rows=n)
        assert [(r[7], r[0]) for r in a] == [(r[7], r[0]) for r in b]
        assert sorted(map(tuple, a)) == sorted(map(tuple, b))


def _run_limit(lib, limit, offset, with_sel):
    """Keyless TOPN == plain LIMIT/OFFSET (LimitExec, executor/limit.go):
    child row order preserved, no sort."""
    from tidb_amd.chunkpy import PyChunk
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64, GX_TYPE_I64])
    node = src
    if with_sel:
        from tests.gxlib import GX_F_LT
        node = b.selection(src, [b.call(GX_F_LT, GX_TYPE_I64, 0,
                                        b.colref(1, GX_TYPE_I64),
                                        b.const_i64(500))])
    root = b.topn(node, [], [], limit, offset)
    ex = b.build(root)
    ch = PyChunk([GX_TYPE_I64] * 2, 1000)
    for i in range(1000):
        ch.append_row([i, (i * 7) % 1000])
    ex.bind_chunks(src, [ch])
    ex.open()
    rows = ex.pull_all([GX_TYPE_I64] * 2)
    ex.close()
    ex.free()
    b.free()
    return rows


def test_oracle_keyless_limit():
    rows = _run_limit(load_oracle(), 7, 0, False)
    assert rows == [(i, (i * 7) % 1000) for i in range(7)]
    rows = _run_limit(load_oracle(), 5, 3, False)
    assert [r[0] for r in rows] == [3, 4, 5, 6, 7]
    rows = _run_limit(load_oracle(), 4, 2, True)
    want = [(i, (i * 7) % 1000) for i in range(1000) if (i * 7) % 1000 < 500]
    assert rows == want[2:6]


@pytest.mark.gpu
@pytest.mark.parametrize("limit,offset,with_sel", [
    (7, 0, False), (5, 3, False), (4, 2, True), (2000, 0, True)])
def test_keyless_limit_parity(limit, offset, with_sel):
    from tests.gxlib import load_product
    want = _run_limit(load_oracle(), limit, offset, with_sel)
    got = _run_limit(load_product(), limit, offset, with_sel)
    assert got == want


def test_oracle_limit_offset_beyond_input():
    """OFFSET past the input: empty result, no error (LimitExec)."""
    rows = _run_limit(load_oracle(), 5, 5000, False)
    assert rows == []
    rows = _run_limit(load_oracle(), 0, 0, False)
    assert rows == []  # LIMIT 0


def test_oracle_topn_offset_beyond_groups():
    from tests.test_full_sort import KEYS_2, run_sort
    rows = run_sort(load_oracle(), KEYS_2, [0, 0], n_rows=7, limit=10,
                    offset=20)
    assert rows == []
