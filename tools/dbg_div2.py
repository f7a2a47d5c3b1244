"""Same plan shape with different top ops to isolate the DIV bug."""
from tests.gxlib import (load_product, load_oracle, GX_AGG_SUM, GX_AGG_COUNT,
                         GX_TYPE_DECIMAL, GX_TYPE_STRING, GX_TYPE_I64,
                         GX_F_DIV, GX_F_MUL, GX_F_MINUS)
from tidb_amd import plan as P
from tidb_amd.chunkpy import PyChunk
from tidb_amd.decimals import str_to_decimal_bytes

def make_plan(lib, func, sr):
    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    price = b.colref(P.L_EXTPRICE, GX_TYPE_DECIMAL, 2)
    qty = b.colref(P.L_QUANTITY, GX_TYPE_DECIMAL, 2)
    rf = b.colref(P.L_RETFLAG, GX_TYPE_STRING)
    ls = b.colref(P.L_LINESTATUS, GX_TYPE_STRING)
    if func is None:
        val = price
        sr = 2
    else:
        val = b.call(func, GX_TYPE_DECIMAL, sr, price, qty)
    proj = b.projection(src, [rf, ls, val])
    agg = b.hashagg(proj, [b.colref(0, GX_TYPE_STRING), b.colref(1, GX_TYPE_STRING)],
                    [(GX_AGG_SUM, b.colref(2, GX_TYPE_DECIMAL, sr), sr),
                     (GX_AGG_COUNT, -1, 0)])
    return b, src, agg, [GX_TYPE_STRING, GX_TYPE_STRING, GX_TYPE_DECIMAL, GX_TYPE_I64], [0,0,sr,0]

def run(lib, func, sr):
    d = lambda s: str_to_decimal_bytes(lib, s)
    t = lib.gx_time_from_date(1995, 1, 1)
    rows = [(1, d("2.00"), d("10.00"), d("0.30"), d("0.70"), "A", "F", t)]
    chunk = PyChunk(P.LINEITEM_TYPES, len(rows), P.LINEITEM_FRACS,
                    data_caps=[None] * 5 + [16, 16] + [None])
    for r in rows:
        chunk.append_row(list(r))
    b, src, agg, ot, of = make_plan(lib, func, sr)
    ex = b.build(agg)
    ex.bind_chunks(src, [chunk])
    ex.open()
    got = ex.pull_all(ot, of, data_caps=[2048, 2048, None, None])
    ex.close(); ex.free(); b.free()
    return got

for name, lib in (("oracle ", load_oracle()), ("product", load_product())):
    print(name, "sum(price)    ", run(lib, None, 0))
    print(name, "sum(price*qty)", run(lib, GX_F_MUL, 4))
    print(name, "sum(price-qty)", run(lib, GX_F_MINUS, 2))
    print(name, "sum(price/qty)", run(lib, GX_F_DIV, 18))
