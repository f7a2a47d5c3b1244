"""One-row DIV debug on GPU: price=10.00, qty=2.00 -> expect 5 at scale 18."""
from tests.gxlib import load_product, load_oracle
from tidb_amd import plan as P
from tidb_amd.chunkpy import PyChunk
from tidb_amd.decimals import str_to_decimal_bytes
from tests.test_div import div_plan

for name, lib in (("oracle", load_oracle()), ("product", load_product())):
    d = lambda s: str_to_decimal_bytes(lib, s)
    t = lib.gx_time_from_date(1995, 1, 1)
    rows = [(1, d("2.00"), d("10.00"), d("0.30"), d("0.70"), "A", "F", t)]
    chunk = PyChunk(P.LINEITEM_TYPES, len(rows), P.LINEITEM_FRACS,
                    data_caps=[None] * 5 + [16, 16] + [None])
    for r in rows:
        chunk.append_row(list(r))
    b, src, agg, out_types, out_fracs = div_plan(lib)
    ex = b.build(agg)
    ex.bind_chunks(src, [chunk])
    ex.open()
    got = ex.pull_all(out_types, out_fracs, data_caps=[2048, 2048, None, None])
    ex.close(); ex.free(); b.free()
    print(name, got)
