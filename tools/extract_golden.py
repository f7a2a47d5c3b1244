#!/usr/bin/env python3
"""Extract golden known-answer vectors from the reference's own unit tests.

Reads Go test tables from /root/reference (PUBLIC UNTRUSTED CONTENT — only
literal test vectors are extracted, no code) and writes JSON fixtures under
tests/golden/.  The fixtures are committed; this script is re-runnable only in
the survey container (the GPU box has no /root/reference).

Sources:
  pkg/types/mydecimal_test.go   (TestAdd/Sub/Mul/DivMod/Round*/ToBinFromBin/
                                 ToHashKey/Compare/Shift/FromString/ToString)
  pkg/types/time_test.go        (date packing / compare)
"""
import json
import os
import re
import sys

REF = "/root/reference"
OUT_DIR = os.path.join(os.path.dirname(__file__), "..", "tests", "golden")

ERR_MAP = {
    "nil": 0,
    "types.ErrTruncated": 1, "ErrTruncated": 1,
    "types.ErrOverflow": 2, "ErrOverflow": 2,
    "types.ErrDivByZero": 3, "ErrDivByZero": 3,
    "types.ErrBadNumber": 4, "ErrBadNumber": 4,
    # FromString returns ErrTruncatedWrongVal for empty/garbage-only input;
    # the oracle maps that class to BAD_NUMBER.
    "ErrTruncatedWrongVal": 4,
}


def eval_go_str(expr):
    """Evaluate a Go string expression: literals, +, strings.Repeat."""
    expr = expr.strip()
    # tokenize by + at top level (no nested parens except strings.Repeat(...))
    parts = []
    depth = 0
    cur = ""
    in_str = False
    i = 0
    while i < len(expr):
        c = expr[i]
        if in_str:
            cur += c
            if c == "\\":
                cur += expr[i + 1]
                i += 2
                continue
            if c == '"':
                in_str = False
        elif c == '"':
            in_str = True
            cur += c
        elif c == "(":
            depth += 1
            cur += c
        elif c == ")":
            depth -= 1
            cur += c
        elif c == "+" and depth == 0:
            parts.append(cur)
            cur = ""
        else:
            cur += c
        i += 1
    parts.append(cur)

    out = ""
    for p in parts:
        p = p.strip()
        if p.startswith('"'):
            body = p[1:-1]
            out += body.encode().decode("unicode_escape")
        elif p.startswith("strings.Repeat("):
            m = re.match(r'strings\.Repeat\("((?:[^"\\]|\\.)*)",\s*(\d+)\)', p)
            if not m:
                raise ValueError("cannot eval: " + p)
            out += m.group(1).encode().decode("unicode_escape") * int(m.group(2))
        elif p == "":
            continue
        else:
            raise ValueError("cannot eval: " + p)
    return out


def split_entries(block):
    """Split the inside of a []T{ ... } literal into top-level {...} entries."""
    entries = []
    depth = 0
    start = None
    in_str = False
    i = 0
    while i < len(block):
        c = block[i]
        if in_str:
            if c == "\\":
                i += 2
                continue
            if c == '"':
                in_str = False
        elif c == '"':
            in_str = True
        elif c == "{":
            if depth == 0:
                start = i
            depth += 1
        elif c == "}":
            depth -= 1
            if depth == 0 and start is not None:
                entries.append(block[start + 1:i])
                start = None
        elif c == "/" and i + 1 < len(block) and block[i + 1] == "/":
            # line comment
            j = block.find("\n", i)
            i = j if j != -1 else len(block)
            continue
        i += 1
    return entries


def split_fields(entry):
    """Split one {a, b, c} entry body into top-level comma-separated fields."""
    fields = []
    depth = 0
    cur = ""
    in_str = False
    i = 0
    while i < len(entry):
        c = entry[i]
        if in_str:
            cur += c
            if c == "\\":
                cur += entry[i + 1]
                i += 2
                continue
            if c == '"':
                in_str = False
        elif c == '"':
            in_str = True
            cur += c
        elif c in "([{":
            depth += 1
            cur += c
        elif c in ")]}":
            depth -= 1
            cur += c
        elif c == "," and depth == 0:
            fields.append(cur.strip())
            cur = ""
        else:
            cur += c
        i += 1
    if cur.strip():
        fields.append(cur.strip())
    return fields


def func_body(src, name):
    m = re.search(r"^func %s\(t \*testing\.T\) \{" % re.escape(name), src, re.M)
    if not m:
        raise ValueError("test func not found: " + name)
    i = m.end()
    depth = 1
    while depth > 0:
        c = src[i]
        if c == "{":
            depth += 1
        elif c == "}":
            depth -= 1
        i += 1
    return src[m.end():i - 1]


def tables_in(body):
    """Return list of the []T{...} table-literal bodies in order."""
    out = []
    for m in re.finditer(r"(?:tests|binTests|cases)\s*:?=\s*\[\]\s*(?:struct\s*\{[^{}]*\}|[\w.]+)\s*\{", body):
        i = m.end()
        depth = 1
        while depth > 0:
            c = body[i]
            if c == '"':
                i += 1
                while body[i] != '"':
                    if body[i] == "\\":
                        i += 1
                    i += 1
            elif c == "{":
                depth += 1
            elif c == "}":
                depth -= 1
            i += 1
        out.append(body[m.end():i - 1])
    return out


def parse_err(tok):
    tok = tok.strip()
    if tok in ERR_MAP:
        return ERR_MAP[tok]
    raise ValueError("unknown err token: " + tok)


def extract_mydecimal():
    src = open(os.path.join(REF, "pkg/types/mydecimal_test.go")).read()
    out = {}

    def table(name, idx=0):
        return [split_fields(e) for e in split_entries(tables_in(func_body(src, name))[idx])]

    # add/sub/mul: {a, b, result, err}
    for name, key in [("TestAddMyDecimal", "add"), ("TestSubMyDecimal", "sub"),
                      ("TestMulMyDecimal", "mul")]:
        rows = []
        for f in table(name):
            rows.append({"a": eval_go_str(f[0]), "b": eval_go_str(f[1]),
                         "result": eval_go_str(f[2]), "err": parse_err(f[3])})
        out[key] = rows

    # div tables: [0] DecimalDiv fracIncr=5 ToString; [2] DecimalDiv fracIncr=4
    # String(); [1],[3] DecimalMod (skipped: mod is off the hot path)
    body = func_body(src, "TestDivModMyDecimal")
    tabs = tables_in(body)
    div_rows = []
    for f in split_entries(tabs[0]):
        f = split_fields(f)
        div_rows.append({"a": eval_go_str(f[0]), "b": eval_go_str(f[1]),
                         "result": eval_go_str(f[2]), "err": parse_err(f[3]),
                         "frac_incr": 5, "display": False})
    for f in split_entries(tabs[2]):
        f = split_fields(f)
        div_rows.append({"a": eval_go_str(f[0]), "b": eval_go_str(f[1]),
                         "result": eval_go_str(f[2]), "err": parse_err(f[3]),
                         "frac_incr": 4, "display": True})
    out["div"] = div_rows

    # rounding
    round_rows = []
    for name, mode in [("TestRoundWithHalfEven", 5), ("TestRoundWithTruncate", 10),
                       ("TestRoundWithCeil", 0)]:
        for f in table(name):
            round_rows.append({"input": eval_go_str(f[0]), "scale": int(f[1]),
                               "output": eval_go_str(f[2]), "err": parse_err(f[3]),
                               "mode": mode})
    out["round"] = round_rows

    # ToString
    out["tostring"] = [{"input": eval_go_str(f[0]), "output": eval_go_str(f[1])}
                       for f in table("TestToString")]

    # ToBin/FromBin round trip
    rows = []
    for f in table("TestToBinFromBin"):
        rows.append({"input": eval_go_str(f[0]), "precision": int(f[1]),
                     "frac": int(f[2]), "output": eval_go_str(f[3]),
                     "err": parse_err(f[4])})
    out["tobin"] = rows

    # Compare: {a, b, cmp}
    rows = []
    for f in table("TestCompareMyDecimal"):
        rows.append({"a": eval_go_str(f[0]), "b": eval_go_str(f[1]), "cmp": int(f[2])})
    out["compare"] = rows

    # Shift: {input, shift, output, err}
    rows = []
    for f in table("TestShiftMyDecimal", 0):
        rows.append({"input": eval_go_str(f[0]), "shift": int(f[1]),
                     "output": eval_go_str(f[2]), "err": parse_err(f[3])})
    # second table (wordBufLen=2) is skipped — it mutates a package global.
    out["shift"] = rows

    # FromString: {input, output, err}
    rows = []
    for f in table("TestFromStringMyDecimal", 0):
        rows.append({"input": eval_go_str(f[0]), "output": eval_go_str(f[1]),
                     "err": parse_err(f[2])})
    out["fromstring"] = rows

    # Neg
    rows = []
    for f in table("TestNegMyDecimal"):
        rows.append({"input": eval_go_str(f[0]), "result": eval_go_str(f[1])})
    out["neg"] = rows

    # ToHashKey: groups of strings that must hash equal
    body = func_body(src, "TestToHashKey")
    tabs = tables_in(body)
    groups = []
    for e in split_entries(tabs[0]):
        m = re.search(r"\[\]string\{(.*)\}", e, re.S)
        groups.append([eval_go_str(x) for x in split_fields(m.group(1))])
    out["hashkey_groups"] = groups
    bin_groups = []
    for e in split_entries(tabs[1]):
        ms = re.findall(r"\[\]string\{(.*?)\}", e, re.S)
        bin_groups.append({"hash_numbers": [eval_go_str(x) for x in split_fields(ms[0])],
                           "bin_numbers": [eval_go_str(x) for x in split_fields(ms[1])]})
    out["hashkey_bin_groups"] = bin_groups

    return out


def extract_time():
    src = open(os.path.join(REF, "pkg/types/core_time_test.go")).read() if os.path.exists(
        os.path.join(REF, "pkg/types/core_time_test.go")) else ""
    out = {}
    # compareTime vectors from time_test.go TestCompare (SQL-string based —
    # instead pin a hand-derived set from FromDate packing in core_time.go).
    # Deterministic packing checks: (y,m,d,h,mi,s,us) -> expected u64 computed
    # from the bitfield spec (time.go:235-251); these double as fixtures for
    # the GPU-side packing.
    vecs = []
    for (y, m, d, h, mi, s, us) in [
        (1992, 1, 1, 0, 0, 0, 0), (1998, 9, 1, 0, 0, 0, 0),
        (1998, 12, 1, 23, 59, 59, 999999), (1995, 3, 15, 0, 0, 0, 0),
        (0, 0, 0, 0, 0, 0, 0), (9999, 12, 31, 23, 59, 59, 999999),
    ]:
        v = ((y << 50) | (m << 46) | (d << 41) | (h << 36) | (mi << 30) |
             (s << 24) | (us << 4))
        vecs.append({"ymdhmsu": [y, m, d, h, mi, s, us], "core_time": v})
    out["pack"] = vecs
    del src
    return out


def main():
    os.makedirs(OUT_DIR, exist_ok=True)
    dec = extract_mydecimal()
    with open(os.path.join(OUT_DIR, "mydecimal.json"), "w") as f:
        json.dump(dec, f, indent=1)
    n = sum(len(v) for v in dec.values())
    print(f"mydecimal.json: {n} vectors in {len(dec)} suites")
    tm = extract_time()
    with open(os.path.join(OUT_DIR, "core_time.json"), "w") as f:
        json.dump(tm, f, indent=1)
    print("core_time.json written")


if __name__ == "__main__":
    sys.exit(main())
