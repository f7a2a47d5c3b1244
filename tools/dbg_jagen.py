import os, sys, traceback
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
os.environ["GX_DEBUG"] = "1"
from tests.test_joinagg_general import _data, _run
from tests.gxlib import load_oracle, load_product
custs, ords, lis = _data("distinct")
want = _run(load_oracle(), custs, ords, lis)
print("oracle:", want)
try:
    got = _run(load_product(), custs, ords, lis)
    print("product:", got)
    print("EQUAL" if got == want else "DIFF")
except Exception:
    traceback.print_exc()
