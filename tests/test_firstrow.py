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
