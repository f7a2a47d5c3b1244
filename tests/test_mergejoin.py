"""Merge join (join/merge_join.go): inner join over inputs sorted on the
join keys. The oracle enforces the sorted-children contract statically and
executes the inner-join semantics; the device engine maps the plan onto its
join pipeline (sorts below joins cannot change the TopN'd result)."""
import pytest

from tests.gxlib import (GX_TPCH_CUSTOMER, GX_TPCH_LINEITEM, GX_TPCH_ORDERS,
                         load_oracle)
from tidb_amd import plan as P

N_LI = 40000
N_ORD = N_LI // 4
N_CUST = max(N_ORD // 10, 1)


def run_q3_merge(lib, limit=10):
    b, (cust, orders, li), topn, out_types, out_fracs = P.q3_merge_plan(lib, limit)
    ex = b.build(topn)
    ex.bind_tpch(cust, GX_TPCH_CUSTOMER, N_CUST)
    ex.bind_tpch(orders, GX_TPCH_ORDERS, N_ORD)
    ex.bind_tpch(li, GX_TPCH_LINEITEM, N_LI)
    ex.open()
    rows = ex.pull_all(out_types, out_fracs, data_caps=[None] * 4)
    ex.close()
    ex.free()
    b.free()
    return rows


def test_mergejoin_oracle_matches_hashjoin(oracle_lib):
    from tests.test_oracle_q3 import run_q3
    assert run_q3_merge(oracle_lib) == run_q3(oracle_lib)


def test_mergejoin_contract(oracle_lib):
    """Children not sorted on the join key -> build error."""
    from tests.gxlib import GX_TYPE_I64
    b = P.Builder(oracle_lib)
    cust = b.source(P.CUSTOMER_TYPES)
    orders = b.source(P.ORDERS_TYPES)
    j = b.mergejoin(cust, orders,
                    [b.colref(P.C_CUSTKEY, GX_TYPE_I64)],
                    [b.colref(P.O_CUSTKEY, GX_TYPE_I64)])
    ex = b.build(j)
    ex.bind_tpch(cust, GX_TPCH_CUSTOMER, 100)
    ex.bind_tpch(orders, GX_TPCH_ORDERS, 400)
    import pytest as _pt
    with _pt.raises(AssertionError):
        ex.open()
    ex.free()
    b.free()


@pytest.mark.gpu
def test_mergejoin_parity():
    from tests.gxlib import load_product
    assert run_q3_merge(load_product()) == run_q3_merge(load_oracle())
