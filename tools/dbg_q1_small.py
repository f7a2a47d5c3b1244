import os, sys
sys.path.insert(0, '/root/repo')
from tests.gxlib import load_oracle, load_product, GX_TPCH_LINEITEM
from tests.test_gpu_parity import _run_q1, _as_map
o = _as_map(_run_q1(load_oracle(), 1000))
p = _as_map(_run_q1(load_product(), 1000))
for k in sorted(o):
    if o[k] != p[k]:
        print("MISMATCH", k)
        print("  cpu:", o[k])
        print("  gpu:", p.get(k))
for k in p:
    if k not in o: print("extra gpu group", k, p[k])
