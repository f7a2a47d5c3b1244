import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from tests.gxlib import GX_TYPE_I64, load_product, load_oracle
from tidb_amd import plan as P
from tidb_amd.chunkpy import PyChunk

def run(lib, budget):
    brows = [(1, 10), (2, 20), (None, 30), (5, 50), (6, 60), (7, 70)]
    prows = [(1, 100), (2, 200), (3, 300), (None, 400), (5, 500), (6, 600), (7, 700), (2, 201)]
    b = P.Builder(lib)
    bsrc = b.source([GX_TYPE_I64]*2)
    psrc = b.source([GX_TYPE_I64]*2)
    j = b.hashjoin(bsrc, psrc, [b.colref(0, GX_TYPE_I64)], [b.colref(0, GX_TYPE_I64)])
    ex = b.build(j)
    bch = PyChunk([GX_TYPE_I64]*2, 8)
    for r in brows: bch.append_row(list(r))
    pch = PyChunk([GX_TYPE_I64]*2, 8)
    for r in prows: pch.append_row(list(r))
    ex.bind_chunks(bsrc, [bch]); ex.bind_chunks(psrc, [pch])
    if budget: os.environ["GX_HBM_BUDGET"] = str(budget)
    try:
        ex.open()
        rows = ex.pull_all([GX_TYPE_I64]*4)
    finally:
        os.environ.pop("GX_HBM_BUDGET", None)
        ex.close(); ex.free(); b.free()
    return sorted(rows, key=lambda r: tuple((x is None, x) for x in r))

os.environ["GX_DEBUG"] = "1"
mem = run(load_product(), None)
sp = run(load_product(), 100)
print("in-mem:", mem)
print("spill :", sp)
print("EQUAL" if mem == sp else "DIFF")
