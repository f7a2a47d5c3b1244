"""Golden known-answer tests for the decimal implementations.

Vectors extracted (tools/extract_golden.py) from the reference's own tests:
pkg/types/mydecimal_test.go — TestAddMyDecimal:598, TestSubMyDecimal:634,
TestMulMyDecimal:671, TestDivModMyDecimal:705, rounding :297-403,
TestToBinFromBin:423, TestToHashKey:187, TestShiftMyDecimal:848,
TestFromStringMyDecimal:966, TestCompareMyDecimal:520.

Runs against BOTH libraries: the CPU oracle and the product engine's host-side
decimal (once built). Either failing is a parity break.
"""
import ctypes
import json
import os

import pytest

from tests.gxlib import dec, dec_str, dec_display, load_oracle

GOLDEN = json.load(open(os.path.join(os.path.dirname(__file__), "golden", "mydecimal.json")))


def _libs():
    libs = [pytest.param("oracle", id="oracle")]
    import tests.gxlib as g
    if os.path.exists(os.path.join(g.REPO, "tidb_amd", "csrc", "libgxexec.so")):
        libs.append(pytest.param("product", id="product"))
    return libs


@pytest.fixture(params=_libs(), scope="module")
def lib(request):
    if request.param == "oracle":
        return load_oracle()
    from tests.gxlib import load_product
    return load_product()


def test_fromstring_tostring(lib):
    for t in GOLDEN["fromstring"]:
        d, err = dec(lib, t["input"])
        assert err == t["err"], t
        assert dec_str(lib, d) == t["output"], t
    for t in GOLDEN["tostring"]:
        d, err = dec(lib, t["input"])
        assert err == 0
        assert dec_str(lib, d) == t["output"], t


@pytest.mark.parametrize("op", ["add", "sub", "mul"])
def test_arith(lib, op):
    fn = getattr(lib, "gx_dec_" + op)
    for t in GOLDEN[op]:
        a, _ = dec(lib, t["a"])
        b, _ = dec(lib, t["b"])
        out = (ctypes.c_uint8 * 40)()
        err = fn(a, b, out)
        assert err == t["err"], t
        if op == "mul":
            # TestMulMyDecimal compares String() (rounded to resultFrac)
            assert dec_display(lib, out) == t["result"], t
        else:
            assert dec_str(lib, out) == t["result"], t


def test_div(lib):
    for t in GOLDEN["div"]:
        a, _ = dec(lib, t["a"])
        b, _ = dec(lib, t["b"])
        out = (ctypes.c_uint8 * 40)()
        err = lib.gx_dec_div(a, b, out, t["frac_incr"])
        assert err == t["err"], t
        if t["err"] == 3:
            continue
        got = dec_display(lib, out) if t["display"] else dec_str(lib, out)
        assert got == t["result"], t


def test_round(lib):
    for t in GOLDEN["round"]:
        d, err = dec(lib, t["input"])
        assert err == 0
        out = (ctypes.c_uint8 * 40)()
        err = lib.gx_dec_round(d, t["scale"], t["mode"], out)
        assert err == t["err"], t
        assert dec_str(lib, out) == t["output"], t


def test_compare(lib):
    for t in GOLDEN["compare"]:
        a, _ = dec(lib, t["a"])
        b, _ = dec(lib, t["b"])
        assert lib.gx_dec_compare(a, b) == t["cmp"], t


def test_shift(lib):
    for t in GOLDEN["shift"]:
        d, err = dec(lib, t["input"])
        assert err == 0, t
        out = (ctypes.c_uint8 * 40)()
        err = lib.gx_dec_shift(d, t["shift"], out)
        assert err == t["err"], t
        assert dec_str(lib, out) == t["output"], t


def test_neg(lib):
    for t in GOLDEN["neg"]:
        d, _ = dec(lib, t["input"])
        out = (ctypes.c_uint8 * 40)()
        zero, _ = dec(lib, "0")
        # neg(x) == 0 - x only for nonzero; use sub from zero and compare value
        err = lib.gx_dec_sub(zero, d, out)
        assert err == 0
        want, _ = dec(lib, t["result"])
        assert lib.gx_dec_compare(out, want) == 0, t


def test_tobin_frombin_roundtrip(lib):
    for t in GOLDEN["tobin"]:
        d, err = dec(lib, t["input"])
        assert err == 0
        binbuf = (ctypes.c_uint8 * 64)()
        blen = ctypes.c_int32(0)
        err = lib.gx_dec_to_bin(d, t["precision"], t["frac"], binbuf, ctypes.byref(blen))
        assert err == t["err"], t
        out = (ctypes.c_uint8 * 40)()
        err2 = lib.gx_dec_from_bin(binbuf, blen.value, t["precision"], t["frac"], out)
        assert err2 == 0, t
        assert dec_str(lib, out) == t["output"], t


def test_hash_key_groups(lib):
    for group in GOLDEN["hashkey_groups"]:
        keys = []
        for num in group:
            d, err = dec(lib, num)
            assert err == 0, num
            buf = (ctypes.c_uint8 * 64)()
            n = ctypes.c_int32(0)
            assert lib.gx_dec_to_hash_key(d, buf, ctypes.byref(n)) == 0
            keys.append(bytes(buf[:n.value]))
        assert all(k == keys[0] for k in keys), group
    for group in GOLDEN["hashkey_bin_groups"]:
        keys = []
        for num in group["hash_numbers"]:
            d, err = dec(lib, num)
            assert err == 0, num
            buf = (ctypes.c_uint8 * 64)()
            n = ctypes.c_int32(0)
            assert lib.gx_dec_to_hash_key(d, buf, ctypes.byref(n)) == 0
            keys.append(bytes(buf[:n.value - 1]))  # strip digit-len byte
        for num in group["bin_numbers"]:
            d, err = dec(lib, num)
            assert err == 0, num
            # PrecisionAndFrac then ToBin — reproduce via to_hash... need
            # precision/frac: derive from string form
            s = dec_str(lib, d)
            neg = s.startswith("-")
            body = s[1:] if neg else s
            if "." in body:
                ip, fp = body.split(".")
            else:
                ip, fp = body, ""
            ip = ip.lstrip("0")
            prec = len(ip) + len(fp)
            if prec == 0:
                prec = 1
            buf = (ctypes.c_uint8 * 64)()
            n = ctypes.c_int32(0)
            err = lib.gx_dec_to_bin(d, prec, len(fp), buf, ctypes.byref(n))
            assert err in (0, 1), num
            keys.append(bytes(buf[:n.value]))
        assert all(k == keys[0] for k in keys), group
