"""The reference's golden Q1 plan carries firstrow(group col) aggregates
(4 sums + 3 avgs + count(*) + 2 firstrow of the group cols;
aggfuncs/builder.go). The device engine decodes them from the group key."""
import pytest

from tests.gxlib import GX_TPCH_LINEITEM, load_oracle
from tidb_amd import plan as P


def run_golden(lib, n_rows):
    b, src, agg, out_types, out_fracs = P.q1_plan(lib, firstrow=True)
    ex = b.build(agg)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n_rows)
    ex.open()
    caps = [2048 if t == 4 else None for t in out_types]
    rows = ex.pull_all(out_types, out_fracs, data_caps=caps)
    ex.close()
    ex.free()
    b.free()
    return sorted(rows)


def test_firstrow_oracle(oracle_lib):
    rows = run_golden(oracle_lib, 20000)
    assert 4 <= len(rows) <= 6
    for r in rows:
        assert r[10] == r[0] and r[11] == r[1]  # firstrow == its group col


@pytest.mark.gpu
def test_firstrow_parity():
    from tests.gxlib import load_product
    assert run_golden(load_oracle(), 50000) == \
        run_golden(load_product(), 50000)


def test_firstrow_i64_value_one():
    """firstrow over an int64 column whose value is 1 must NOT come back NULL
    (regression: the oracle's aux field doubled as null-flag and i64 value)."""
    import numpy as np
    from tests.gxlib import (GX_AGG_FIRSTROW, GX_AGG_COUNT, GX_TYPE_I64,
                             load_oracle)
    from tidb_amd import plan as P
    from tidb_amd.chunkpy import PyChunk
    lib = load_oracle()
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64])
    g = b.colref(0, GX_TYPE_I64)
    agg = b.hashagg(src, [g], [(GX_AGG_FIRSTROW, g, 0), (GX_AGG_COUNT, -1, 0)])
    ex = b.build(agg)
    ch = PyChunk([GX_TYPE_I64], 8)
    for v in [1, 0, 1, 2, 1]:
        ch.append_row([v])
    ex.bind_chunks(src, [ch])
    ex.open()
    rows = sorted(ex.pull_all([GX_TYPE_I64] * 3))
    ex.close()
    ex.free()
    b.free()
    assert rows == [(0, 0, 1), (1, 1, 3), (2, 2, 1)]
