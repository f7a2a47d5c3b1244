"""Pure-Python decode of the 40-byte MyDecimal struct (layout only, no
arithmetic): int8 digitsInt, int8 digitsFrac, int8 resultFrac, bool negative,
int32 wordBuf[9] (reference pkg/types/mydecimal.go:236-248).

Used by tests/bench to render engine outputs; NOT an arithmetic
implementation.
"""
import struct


def decimal_bytes_to_parts(b):
    digits_int, digits_frac, result_frac, negative = struct.unpack_from("bbbB", b, 0)
    words = struct.unpack_from("<9i", b, 4)
    return digits_int, digits_frac, result_frac, bool(negative), words


def decimal_bytes_to_str(b):
    """ToString-equivalent rendering (no rounding)."""
    digits_int, digits_frac, _rf, neg, words = decimal_bytes_to_parts(b)
    words_int = (digits_int + 8) // 9
    words_frac = (digits_frac + 8) // 9
    ip = 0
    for w in words[:words_int]:
        ip = ip * 10**9 + w
    int_str = str(ip) if ip else "0"
    frac_digits = ""
    for w in words[words_int:words_int + words_frac]:
        frac_digits += format(w, "09d")
    frac_digits = frac_digits[:digits_frac]
    s = int_str
    if digits_frac > 0:
        s += "." + frac_digits
    if neg and (ip != 0 or any(c != "0" for c in frac_digits)):
        s = "-" + s
    elif neg:
        s = "-" + s  # reference keeps the sign bit it stored
    return s


def decimal_bytes_to_fraction(b):
    """Exact value as a python Fraction (for independent checks)."""
    from fractions import Fraction
    digits_int, digits_frac, _rf, neg, words = decimal_bytes_to_parts(b)
    words_int = (digits_int + 8) // 9
    words_frac = (digits_frac + 8) // 9
    ip = 0
    for w in words[:words_int]:
        ip = ip * 10**9 + w
    fp = 0
    scale = 0
    for w in words[words_int:words_int + words_frac]:
        fp = fp * 10**9 + w
        scale += 9
    v = Fraction(ip) + Fraction(fp, 10**scale) if scale else Fraction(ip)
    return -v if neg else v


def str_to_decimal_bytes(lib, s):
    """Build a 40-byte MyDecimal via the oracle's FromString."""
    import ctypes
    out = (ctypes.c_uint8 * 40)()
    err = lib.gx_dec_from_string(s.encode(), len(s.encode()), out)
    assert err == 0, (s, err)
    return bytes(out)
