"""Oracle executor end-to-end tests: TPC-H Q1 (BASELINE config 1) against an
independent Python computation (exact integer arithmetic) on the same
synthetic rows, plus partial/final (MergePartialResult) equivalence — the
semantics the multi-GPU RCCL merge must reproduce.
"""
from fractions import Fraction

import pytest

from tests.gxlib import GX_AGG_MODE_COMPLETE, GX_AGG_MODE_PARTIAL, GX_TPCH_LINEITEM
from tidb_amd import plan as P
from tidb_amd.decimals import decimal_bytes_to_fraction

N_ROWS = 20000
STRING_CAPS = [None, None, None, None, None, 2048, 2048, None]


def pull_lineitem(lib, n_rows, seed=42, row_offset=0, total_rows=None):
    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    ex = b.build(src)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n_rows, seed, row_offset, total_rows)
    ex.open()
    rows = ex.pull_all(P.LINEITEM_TYPES, P.LINEITEM_FRACS, data_caps=STRING_CAPS)
    ex.close()
    ex.free()
    b.free()
    return rows


def dec_to_cents(dec_str, scale):
    """'123.45' -> integer at 10^-scale units."""
    f = Fraction(dec_str) * 10**scale
    assert f.denominator == 1, (dec_str, scale)
    return f.numerator


def round_half_up(num, den, scale):
    """round(num/den) to `scale` frac digits, half away from zero ->
    integer in 10^-scale units (matches Round ModeHalfUp)."""
    target = num * 10**scale
    sign = -1 if (target < 0) != (den < 0) else 1
    a, b = abs(target), abs(den)
    q, r = divmod(a, b)
    if 2 * r >= b:
        q += 1
    return sign * q


CUTOFF = (1998 << 50) | (9 << 46) | (1 << 41)


def expected_q1(rows):
    groups = {}
    for r in rows:
        (_ok, qty, price, disc, tax, rf, ls, shipdate) = r
        if (shipdate & ~0xF) >= CUTOFF:
            continue
        key = (rf, ls)
        g = groups.setdefault(key, {"sq": 0, "sp": 0, "sdp": 0, "sch": 0,
                                    "sd": 0, "n": 0})
        q = dec_to_cents(qty, 2)
        p = dec_to_cents(price, 2)
        d = dec_to_cents(disc, 2)
        t = dec_to_cents(tax, 2)
        dp = p * (100 - d)          # scale 4
        ch = dp * (100 + t)         # scale 6
        g["sq"] += q
        g["sp"] += p
        g["sdp"] += dp
        g["sch"] += ch
        g["sd"] += d
        g["n"] += 1
    out = {}
    for key, g in groups.items():
        out[key] = (
            g["sq"],                      # sum qty (s2)
            g["sp"],                      # sum price (s2)
            g["sdp"],                     # sum disc_price (s4)
            g["sch"],                     # sum charge (s6)
            round_half_up(g["sq"], g["n"] * 100, 6),   # avg qty (s6)
            round_half_up(g["sp"], g["n"] * 100, 6),   # avg price (s6)
            round_half_up(g["sd"], g["n"] * 100, 6),   # avg disc (s6)
            g["n"],
        )
    return out


def q1_results_to_map(res_rows):
    out = {}
    for r in res_rows:
        rf, ls = r[0], r[1]
        sq = dec_to_cents(r[2], 2)
        sp = dec_to_cents(r[3], 2)
        sdp = dec_to_cents(r[4], 4)
        sch = dec_to_cents(r[5], 6)
        aq = dec_to_cents(r[6], 6)
        ap = dec_to_cents(r[7], 6)
        ad = dec_to_cents(r[8], 6)
        cnt = r[9]
        out[(rf, ls)] = (sq, sp, sdp, sch, aq, ap, ad, cnt)
    return out


def run_q1(lib, n_rows, mode=GX_AGG_MODE_COMPLETE, seed=42, row_offset=0,
           total_rows=None):
    b, src, agg, out_types, out_fracs = P.q1_plan(lib, mode)
    ex = b.build(agg)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n_rows, seed, row_offset, total_rows)
    ex.open()
    caps = [2048 if t == 4 else None for t in out_types]
    rows = ex.pull_all(out_types, out_fracs, data_caps=caps)
    ex.close()
    ex.free()
    b.free()
    return rows


def test_q1_oracle_vs_python(oracle_lib):
    rows = pull_lineitem(oracle_lib, N_ROWS)
    assert len(rows) == N_ROWS
    expect = expected_q1(rows)
    got = q1_results_to_map(run_q1(oracle_lib, N_ROWS))
    assert got == expect
    # sanity: Q1 yields at most 6 groups, selectivity ~0.96
    assert 1 <= len(got) <= 6
    total = sum(v[7] for v in got.values())
    assert 0.90 < total / N_ROWS < 1.0


def test_q1_selectivity_matches_survey(oracle_lib):
    rows = pull_lineitem(oracle_lib, 50000)
    passing = sum(1 for r in rows if (r[7] & ~0xF) < CUTOFF)
    # uniform 1992-01-01..1998-12-01 => ~0.96
    assert 0.95 < passing / 50000 < 0.975


def test_q1_partial_final_merge(oracle_lib):
    """Shard rows into 3 ranges, run PARTIAL per shard, merge via FINAL —
    must equal COMPLETE over the whole table (the 8-GPU merge semantics)."""
    import ctypes
    total = 9000
    shards = [(0, 3000), (3000, 3000), (6000, 3000)]
    partial_rows = []
    for off, n in shards:
        partial_rows.extend(
            run_q1(oracle_lib, n, GX_AGG_MODE_PARTIAL, row_offset=off,
                   total_rows=total))
    # feed the partial rows into the FINAL plan via a bound chunk
    b, src, agg, out_types, out_fracs, part_types, part_fracs = \
        P.q1_final_plan(oracle_lib)
    from tidb_amd.chunkpy import PyChunk
    from tidb_amd.decimals import str_to_decimal_bytes
    chunk = PyChunk(part_types, len(partial_rows), part_fracs,
                    data_caps=[4096] * len(part_types))
    for r in partial_rows:
        vals = []
        for v, t in zip(r, part_types):
            if t == 2 and v is not None:  # decimal: re-encode via FromString
                vals.append(str_to_decimal_bytes(oracle_lib, v))
            else:
                vals.append(v)
        chunk.append_row(vals)
    ex = b.build(agg)
    ex.bind_chunks(src, [chunk])
    ex.open()
    caps = [2048 if t == 4 else None for t in out_types]
    merged = q1_results_to_map(ex.pull_all(out_types, out_fracs, data_caps=caps))
    ex.close()
    ex.free()
    b.free()
    complete = q1_results_to_map(run_q1(oracle_lib, total))
    assert merged == complete


def test_q1_with_nulls(oracle_lib):
    """NULL handling through selection + aggregation via bound chunks."""
    from tidb_amd.chunkpy import PyChunk
    from tidb_amd.decimals import str_to_decimal_bytes
    lib = oracle_lib
    d = lambda s: str_to_decimal_bytes(lib, s)
    date = lambda y, m, dd: lib.gx_time_from_date(y, m, dd)
    rows = [
        # orderkey qty price disc tax rf ls shipdate
        (1, d("10.00"), d("100.00"), d("0.05"), d("0.02"), "A", "F", date(1995, 1, 1)),
        (2, None,       d("200.00"), d("0.00"), d("0.01"), "A", "F", date(1995, 1, 2)),
        (3, d("5.00"),  None,        d("0.10"), d("0.00"), "A", "F", date(1995, 1, 3)),
        (4, d("7.00"),  d("50.00"),  d("0.01"), d("0.08"), "N", "O", date(1999, 1, 1)),  # filtered
        (5, d("3.00"),  d("30.00"),  None,      d("0.03"), "R", "F", date(1996, 5, 5)),
        (6, d("2.00"),  d("20.00"),  d("0.02"), d("0.04"), "R", "F", None),  # NULL date -> filtered
    ]
    chunk = PyChunk(P.LINEITEM_TYPES, len(rows), P.LINEITEM_FRACS,
                    data_caps=[None, None, None, None, None, 64, 64, None])
    for r in rows:
        chunk.append_row(list(r))
    b, src, agg, out_types, out_fracs = P.q1_plan(lib)
    ex = b.build(agg)
    ex.bind_chunks(src, [chunk])
    ex.open()
    caps = [2048 if t == 4 else None for t in out_types]
    got = {(r[0], r[1]): r[2:] for r in
           ex.pull_all(out_types, out_fracs, data_caps=caps)}
    ex.close()
    ex.free()
    b.free()
    # group A,F: rows 1-3. sum_qty = 15.00 (null skipped); sum_price = 300.00;
    # disc_price: row1 100*(0.95)=95.0000; row2 200*1=200.0000; row3 NULL
    # charge: row1 95*(1.02)=96.900000; row2 200*1.01=202.000000
    af = got[("A", "F")]
    assert af[0] == "15.00"
    assert af[1] == "300.00"
    assert af[2] == "295.0000"
    assert af[3] == "298.900000"
    # avg qty = 15/2 = 7.500000; avg price = 300/2=150; avg disc = 0.15/3=0.05
    assert af[4] == "7.500000"
    assert af[5] == "150.000000"
    assert af[6] == "0.050000"
    assert af[7] == 3  # count(*)
    # group R,F: row 5 only (row 6 filtered by NULL shipdate)
    rfg = got[("R", "F")]
    assert rfg[0] == "3.00"
    assert rfg[2] is None  # disc NULL -> disc_price NULL -> sum NULL
    assert rfg[7] == 1
    assert ("N", "O") not in got
