#!/usr/bin/env python3
"""On-box debug driver: exercises the product engine stepwise with verbose
output. Run under gpurun with GX_DEBUG=1."""
import faulthandler
import os
import sys

faulthandler.enable()
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from tests.gxlib import GX_TPCH_LINEITEM, load_oracle, load_product  # noqa: E402
from tidb_amd import plan as P  # noqa: E402

CAPS = [None] * 5 + [2048, 2048] + [None]


def pull(lib, n):
    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    ex = b.build(src)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n)
    ex.open()
    rows = ex.pull_all(P.LINEITEM_TYPES, P.LINEITEM_FRACS, data_caps=CAPS)
    ex.close()
    ex.free()
    b.free()
    return rows


def q1(lib, n):
    b, src, agg, out_types, out_fracs = P.q1_plan(lib)
    ex = b.build(agg)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n)
    ex.open()
    caps = [2048 if t == 4 else None for t in out_types]
    rows = ex.pull_all(out_types, out_fracs, data_caps=caps)
    ex.close()
    ex.free()
    b.free()
    return rows


print("== step 1: product source pull (2000 rows) ==", flush=True)
prows = pull(load_product(), 2000)
print("rows:", len(prows))
print("row0:", prows[0])
print("row1999:", prows[1999])

print("== step 2: oracle source pull + parity ==", flush=True)
orows = pull(load_oracle(), 2000)
print("parity:", prows == orows)
if prows != orows:
    for i, (a, b_) in enumerate(zip(prows, orows)):
        if a != b_:
            print("first mismatch at", i)
            print(" gpu:", a)
            print(" cpu:", b_)
            break

print("== step 3: product Q1 (100000 rows) ==", flush=True)
g = q1(load_product(), 100000)
print("groups:", len(g))
for r in sorted(g):
    print(" ", r)

print("== step 4: oracle Q1 parity ==", flush=True)
o = q1(load_oracle(), 100000)
pm = {(r[0], r[1]): tuple(r[2:]) for r in g}
om = {(r[0], r[1]): tuple(r[2:]) for r in o}
print("parity:", pm == om)
if pm != om:
    for k in om:
        if pm.get(k) != om[k]:
            print("mismatch", k)
            print(" gpu:", pm.get(k))
            print(" cpu:", om[k])
print("DONE")
