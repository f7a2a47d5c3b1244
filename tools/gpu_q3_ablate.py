"""Q3 probe ablation: vary the lineitem filter selectivity / bloom."""
import os, sys, time, ctypes
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tests.gxlib import (GX_TPCH_CUSTOMER, GX_TPCH_LINEITEM, GX_TPCH_ORDERS,
                         GX_F_GT, GX_F_LT, GX_F_EQ, GX_TYPE_I64, GX_TYPE_TIME,
                         GX_TYPE_DECIMAL, GX_TYPE_STRING, GX_AGG_SUM, load_product)
from tidb_amd import plan as P

lib = load_product()
lib.gx_last_kernel_ms.restype = ctypes.c_double
lib.gx_last_kernel_ms.argtypes = [ctypes.c_void_p]
SF = 100
n_li, n_ord, n_cust = 6_000_000*SF, 1_500_000*SF, 150_000*SF

def run(cutoff_ymd, label):
    import tidb_amd.plan as PP
    b = PP.Builder(lib)
    cust = b.source(PP.CUSTOMER_TYPES)
    seg = b.colref(1, GX_TYPE_STRING)
    cond_c = b.call(GX_F_EQ, GX_TYPE_I64, 0, seg, lib.gx_pb_const_str(b.pb, b"BUILDING", 8))
    sel_c = b.selection(cust, [cond_c])
    orders = b.source(PP.ORDERS_TYPES)
    cond_o = b.call(GX_F_LT, GX_TYPE_I64, 0, b.colref(2, GX_TYPE_TIME),
                    b.const_time(lib.gx_time_from_date(1995, 3, 15)))
    sel_o = b.selection(orders, [cond_o])
    j1 = b.hashjoin(sel_c, sel_o, [b.colref(0, GX_TYPE_I64)], [b.colref(1, GX_TYPE_I64)])
    li = b.source(PP.LINEITEM_TYPES, PP.LINEITEM_FRACS)
    cond_l = b.call(GX_F_GT, GX_TYPE_I64, 0, b.colref(7, GX_TYPE_TIME),
                    b.const_time(lib.gx_time_from_date(*cutoff_ymd)))
    sel_l = b.selection(li, [cond_l])
    j2 = b.hashjoin(j1, sel_l, [b.colref(2, GX_TYPE_I64)], [b.colref(0, GX_TYPE_I64)])
    okey = b.colref(6 + 0, GX_TYPE_I64); odate = b.colref(4, GX_TYPE_TIME); prio = b.colref(5, GX_TYPE_I64)
    price = b.colref(6 + 2, GX_TYPE_DECIMAL, 2); disc = b.colref(6 + 3, GX_TYPE_DECIMAL, 2)
    one = P._const_dec_one(lib, b)
    rev = b.call(18, GX_TYPE_DECIMAL, 4, price, b.call(17, GX_TYPE_DECIMAL, 2, one, disc))
    proj = b.projection(j2, [okey, odate, prio, rev])
    agg = b.hashagg(proj, [b.colref(0, GX_TYPE_I64), b.colref(1, GX_TYPE_TIME), b.colref(2, GX_TYPE_I64)],
                    [(GX_AGG_SUM, b.colref(3, GX_TYPE_DECIMAL, 4), 4)])
    topn = b.topn(agg, [b.colref(3, GX_TYPE_DECIMAL, 4), b.colref(1, GX_TYPE_TIME)], [1, 0], 10)
    ex = b.build(topn)
    ex.bind_tpch(cust, GX_TPCH_CUSTOMER, n_cust)
    ex.bind_tpch(orders, GX_TPCH_ORDERS, n_ord)
    ex.bind_tpch(li, GX_TPCH_LINEITEM, n_li)
    for i in range(3):
        ex.open()
        rows = ex.pull_all([GX_TYPE_I64, GX_TYPE_TIME, GX_TYPE_I64, GX_TYPE_DECIMAL], [0,0,0,4], data_caps=[None]*4)
        ex.close()
    print(f"{label}: probe {lib.gx_last_kernel_ms(ex.ex):.2f} ms, {len(rows)} rows", flush=True)
    ex.free(); b.free()

run((1999, 1, 1), "pred rejects all (scan+filter only)")
run((1998, 6, 1), "pred ~7% pass")
run((1995, 3, 15), "normal 54% pass")
os.environ; 
