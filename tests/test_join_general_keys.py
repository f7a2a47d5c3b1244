"""General join keys: the device analog of
codec.SerializeKeys' full modes (/root/reference/pkg/util/codec/codec.go:
852-910,445-456) — join keys beyond fixed 8-byte int64/time columns:

- decimal keys, value-normalized (decimal.ToHashKey: trailing zeros removed,
  so 1.10 (frac 2) joins 1.1 (frac 1));
- varlen string keys of any length (utf8mb4_bin PAD SPACE: trailing spaces
  trimmed, collate.go:272 — 'A ' joins 'A');
- mixed multi-column keys (3+ columns);
- NULL keys never match (inner join).

Parity: product (GPU general-key kernels) vs oracle (CPU restatement, whose
serializeJoinKeys already implements the full modes) on identical chunks.
"""
import ctypes

import numpy as np
import pytest

from tests.gxlib import (GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_STRING,
                         load_oracle, load_product)
from tidb_amd import plan as P
from tidb_amd.chunkpy import PyChunk


def _dec40(lib, s):
    out = (ctypes.c_uint8 * 40)()
    assert lib.gx_dec_from_string(s.encode(), len(s.encode()), out) == 0
    return bytes(out)


def _mk_chunk(lib, types, fracs, rows, str_cap=None):
    ch = PyChunk(types, max(len(rows), 1), fracs,
                 [str_cap if t == GX_TYPE_STRING else None for t in types])
    for r in rows:
        vals = []
        for v, t in zip(r, types):
            if v is not None and t == GX_TYPE_DECIMAL:
                vals.append(_dec40(lib, v))
            else:
                vals.append(v)
        ch.append_row(vals)
    return ch


def _run_join(lib, btypes, bfracs, brows, ptypes, pfracs, prows, bkeys, pkeys):
    b = P.Builder(lib)
    bsrc = b.source(btypes, bfracs)
    psrc = b.source(ptypes, pfracs)
    j = b.hashjoin(bsrc, psrc,
                   [b.colref(c, btypes[c], bfracs[c]) for c in bkeys],
                   [b.colref(c, ptypes[c], pfracs[c]) for c in pkeys])
    ex = b.build(j)
    ex.bind_chunks(bsrc, [_mk_chunk(lib, btypes, bfracs, brows, 1 << 16)])
    ex.bind_chunks(psrc, [_mk_chunk(lib, ptypes, pfracs, prows, 1 << 16)])
    ex.open()
    out_types = btypes + ptypes
    out_fracs = bfracs + pfracs
    caps = [1 << 16 if t == GX_TYPE_STRING else None for t in out_types]
    rows = ex.pull_all(out_types, out_fracs, data_caps=caps)
    ex.close()
    ex.free()
    b.free()
    key = lambda r: tuple((x is None, x) for x in r)
    return sorted(rows, key=key)


def _decimal_key_data():
    # build: decimal(., 1) keys; probe: decimal(., 2) keys — cross-frac
    # value-normalized matches (1.1 == 1.10), NULLs, duplicates
    brows = [["1.1", 10], ["2.5", 20], ["2.5", 21], ["0.0", 30], [None, 40],
             ["-3.7", 50], ["100.0", 60]]
    prows = [["1.10", 1], ["2.50", 2], ["0.00", 3], [None, 4], ["-3.70", 5],
             ["1.11", 6], ["100.00", 7], ["2.50", 8]]
    return ([GX_TYPE_DECIMAL, GX_TYPE_I64], [1, 0], brows,
            [GX_TYPE_DECIMAL, GX_TYPE_I64], [2, 0], prows, [0], [0])


def test_oracle_decimal_join_keys():
    lib = load_oracle()
    got = _run_join(lib, *_decimal_key_data())
    # expected matches by numeric value
    want_pairs = {("1.1", "1.10"): 1 * 1, ("2.5", "2.50"): 2 * 2,
                  ("0.0", "0.00"): 1 * 1, ("-3.7", "-3.70"): 1 * 1,
                  ("100.0", "100.00"): 1 * 1}
    assert len(got) == sum(want_pairs.values())
    assert all(r[0] is not None for r in got)


@pytest.mark.gpu
def test_decimal_join_keys_parity():
    data = _decimal_key_data()
    assert _run_join(load_product(), *data) == _run_join(load_oracle(), *data)


def _string_key_data(n=4000, seed=7):
    rng = np.random.default_rng(seed)
    names = ["BUILDING", "AUTOMOBILE", "machinery-with-a-very-long-tail",
             "A", "A ", "", "xy", "prefix-equal-0123-AA",
             "prefix-equal-0123-AB"]
    brows = [[names[i % len(names)], i] for i in range(len(names) * 3)]
    brows.append([None, 999])
    prows = [[names[rng.integers(0, len(names))], int(i)] for i in range(n)]
    prows.append([None, -1])
    return ([GX_TYPE_STRING, GX_TYPE_I64], [0, 0], brows,
            [GX_TYPE_STRING, GX_TYPE_I64], [0, 0], prows, [0], [0])


def test_oracle_string_join_keys():
    lib = load_oracle()
    got = _run_join(lib, *_string_key_data())
    assert len(got) > 1000
    # PAD SPACE: 'A ' and 'A' are the same key — every probe 'A' matches
    # build rows holding both forms
    a_matches = [r for r in got if r[2] == "A"]
    a_builds = {r[0] for r in a_matches}
    assert a_builds == {"A", "A "}


@pytest.mark.gpu
def test_string_join_keys_parity():
    data = _string_key_data()
    assert _run_join(load_product(), *data) == _run_join(load_oracle(), *data)


def _mixed3_key_data(n=6000, seed=13):
    """3-column mixed (i64, string, decimal) join keys."""
    rng = np.random.default_rng(seed)
    combos = [(int(i), f"grp-{i % 5:02d}-{'x' * (i % 9)}", f"{i % 4}.2{i % 3}")
              for i in range(24)]
    brows = []
    for i, (a, s, d) in enumerate(combos):
        brows.append([a % 6, s, d, i])
        if i % 3 == 0:
            brows.append([a % 6, s, d, 100 + i])  # duplicate build keys
    prows = []
    for i in range(n):
        a, s, d = combos[rng.integers(0, len(combos))]
        prows.append([a % 6, s, d, i])
    types = [GX_TYPE_I64, GX_TYPE_STRING, GX_TYPE_DECIMAL, GX_TYPE_I64]
    fracs = [0, 0, 2, 0]
    return (types, fracs, brows, types, fracs, prows, [0, 1, 2], [0, 1, 2])


def test_oracle_mixed3_join_keys():
    got = _run_join(load_oracle(), *_mixed3_key_data())
    assert len(got) > 4000


@pytest.mark.gpu
def test_mixed3_join_keys_parity():
    data = _mixed3_key_data()
    assert _run_join(load_product(), *data) == _run_join(load_oracle(), *data)


def _four_i64_key_data(n=3000, seed=29):
    """4 int64 key columns (> the 2-register fast path) with values beyond
    2^31 — exercises the general loader on kind-0 keys."""
    rng = np.random.default_rng(seed)
    combos = [tuple(int(x) for x in rng.integers(-5, 5, 4) * (1 << 40))
              for _ in range(20)]
    brows = [list(c) + [i] for i, c in enumerate(combos)]
    prows = [list(combos[rng.integers(0, len(combos))]) + [i]
             for i in range(n)]
    types = [GX_TYPE_I64] * 5
    fracs = [0] * 5
    return (types, fracs, brows, types, fracs, prows, [0, 1, 2, 3],
            [0, 1, 2, 3])


def test_oracle_four_i64_join_keys():
    got = _run_join(load_oracle(), *_four_i64_key_data())
    assert len(got) == 3000


@pytest.mark.gpu
def test_four_i64_join_keys_parity():
    data = _four_i64_key_data()
    assert _run_join(load_product(), *data) == _run_join(load_oracle(), *data)
