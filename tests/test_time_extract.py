"""YEAR/MONTH/DAY extraction (builtinYearSig / builtinMonthSig /
builtinDaySig, /root/reference/pkg/expression/builtin_time_vec.go):
CoreTime bitfield reads (year@50:14 month@46:4 day@41:5, core_time.go).
Covered: standalone projection (with NULL times) and as MIN/MAX aggregate
args through the fused pipeline."""
import numpy as np
import pytest

from tests.gxlib import (GX_AGG_MAX, GX_AGG_MIN, GX_F_DAY, GX_F_MONTH,
                         GX_F_YEAR, GX_TYPE_I64, GX_TYPE_TIME, load_oracle,
                         load_product)
from tidb_amd import plan as P
from tidb_amd.chunkpy import PyChunk


def _data(lib, n=3000, seed=41):
    rng = np.random.default_rng(seed)
    rows = []
    for i in range(n):
        k = int(rng.integers(0, 10))
        if rng.random() < 0.15:
            rows.append((k, None, None))
        else:
            y, m, d = (int(rng.integers(1992, 1999)),
                       int(rng.integers(1, 13)), int(rng.integers(1, 29)))
            rows.append((k, (y, m, d), lib.gx_time_from_date(y, m, d)))
    return rows


def _chunks(rows):
    out = []
    for base in range(0, len(rows), 1000):
        part = rows[base:base + 1000]
        ch = PyChunk([GX_TYPE_I64, GX_TYPE_TIME], len(part))
        for k, _, t in part:
            ch.append_row([k, t])
        out.append(ch)
    return out


def _run_proj(lib):
    rows = _data(lib)
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64, GX_TYPE_TIME])
    t = b.colref(1, GX_TYPE_TIME)
    proj = b.projection(src, [
        b.colref(0, GX_TYPE_I64),
        b.call(GX_F_YEAR, GX_TYPE_I64, 0, t),
        b.call(GX_F_MONTH, GX_TYPE_I64, 0, t),
        b.call(GX_F_DAY, GX_TYPE_I64, 0, t),
    ])
    ex = b.build(proj)
    ex.bind_chunks(src, _chunks(rows))
    ex.open()
    got = ex.pull_all([GX_TYPE_I64] * 4)
    ex.close()
    ex.free()
    b.free()
    return rows, got


def _run_agg(lib):
    rows = _data(lib)
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64, GX_TYPE_TIME])
    t = b.colref(1, GX_TYPE_TIME)
    agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                    [(GX_AGG_MIN, b.call(GX_F_YEAR, GX_TYPE_I64, 0, t), 0),
                     (GX_AGG_MAX, b.call(GX_F_MONTH, GX_TYPE_I64, 0, t), 0)])
    ex = b.build(agg)
    ex.bind_chunks(src, _chunks(rows))
    ex.open()
    got = sorted(ex.pull_all([GX_TYPE_I64] * 3))
    ex.close()
    ex.free()
    b.free()
    return rows, got


def test_oracle_time_extract():
    lib = load_oracle()
    rows, got = _run_proj(lib)
    for (k, y, m, d), (rk, ymd, _) in zip(got, rows):
        assert k == rk
        assert (y, m, d) == (ymd if ymd is not None else (None, None, None))
    rows, got = _run_agg(lib)
    want = {}
    for k, ymd, _ in rows:
        mn, mx = want.get(k, (None, None))
        if ymd is not None:
            mn = ymd[0] if mn is None else min(mn, ymd[0])
            mx = ymd[1] if mx is None else max(mx, ymd[1])
        want.setdefault(k, (mn, mx))
        want[k] = (mn, mx)
    assert got == sorted((k, mn, mx) for k, (mn, mx) in want.items())


@pytest.mark.gpu
def test_time_extract_parity():
    _, w1 = _run_proj(load_oracle())
    _, g1 = _run_proj(load_product())
    assert g1 == w1
    _, w2 = _run_agg(load_oracle())
    _, g2 = _run_agg(load_product())
    assert g2 == w2


def _run_groupby_year(lib):
    """GROUP BY YEAR(t), MONTH(t) — computed i64 group keys ride the wide
    serialized-key path via stable VM registers (kind 4)."""
    from tests.gxlib import GX_AGG_COUNT
    rows = _data(lib, n=4000, seed=47)
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64, GX_TYPE_TIME])
    t = b.colref(1, GX_TYPE_TIME)
    y = b.call(GX_F_YEAR, GX_TYPE_I64, 0, t)
    m = b.call(GX_F_MONTH, GX_TYPE_I64, 0, t)
    proj = b.projection(src, [y, m, b.colref(0, GX_TYPE_I64)])
    agg = b.hashagg(proj, [b.colref(0, GX_TYPE_I64),
                           b.colref(1, GX_TYPE_I64)],
                    [(GX_AGG_COUNT, b.colref(2, GX_TYPE_I64), 0)])
    ex = b.build(agg)
    ex.bind_chunks(src, _chunks(rows))
    ex.open()
    got = ex.pull_all([GX_TYPE_I64] * 3)
    ex.close()
    ex.free()
    b.free()
    want = {}
    for k, ymd, _ in rows:
        key = (None, None) if ymd is None else (ymd[0], ymd[1])
        want[key] = want.get(key, 0) + 1
    wl = sorted(((y, m, c) for (y, m), c in want.items()),
                key=lambda r: tuple((x is None, x) for x in r))
    gl = sorted(got, key=lambda r: tuple((x is None, x) for x in r))
    assert gl == wl, (gl[:4], wl[:4])
    return gl


def test_oracle_groupby_computed_year():
    _run_groupby_year(load_oracle())


@pytest.mark.gpu
def test_groupby_computed_year_parity():
    want = _run_groupby_year(load_oracle())
    got = _run_groupby_year(load_product())
    assert got == want


def _run_hms_greatest(lib):
    """HOUR/MINUTE/SECOND over datetimes + GREATEST/LEAST over decimals
    and ints (NULL if any arg NULL, builtin_compare_vec.go MergeNulls)."""
    import ctypes

    from tests.gxlib import (GX_F_GREATEST, GX_F_HOUR, GX_F_LEAST,
                             GX_F_MINUTE, GX_F_SECOND, GX_TYPE_DECIMAL)
    rng = np.random.default_rng(53)
    lib.gx_time_from_datetime.restype = __import__("ctypes").c_uint64

    def dec(s):
        out = (ctypes.c_uint8 * 40)()
        assert lib.gx_dec_from_string(s.encode(), len(s.encode()), out) == 0
        return bytes(out)

    rows = []
    for i in range(2000):
        if rng.random() < 0.15:
            rows.append((None, None, None, None))
        else:
            hms = (int(rng.integers(0, 24)), int(rng.integers(0, 60)),
                   int(rng.integers(0, 60)))
            t = lib.gx_time_from_datetime(1995, 6, 7, *hms, 0, 0)
            a = f"{int(rng.integers(-50, 50))}.{int(rng.integers(0, 100)):02d}"
            bb = f"{int(rng.integers(-50, 50))}.{int(rng.integers(0, 100)):02d}"
            rows.append((hms, t, a, bb))
    chunks = []
    types = [GX_TYPE_TIME, GX_TYPE_DECIMAL, GX_TYPE_DECIMAL]
    for base in range(0, len(rows), 1000):
        part = rows[base:base + 1000]
        ch = PyChunk(types, len(part), [0, 2, 2])
        for _, t, a, bb in part:
            ch.append_row([t, None if a is None else dec(a),
                           None if bb is None else dec(bb)])
        chunks.append(ch)
    b = P.Builder(lib)
    src = b.source(types, [0, 2, 2])
    t = b.colref(0, GX_TYPE_TIME)
    da = b.colref(1, GX_TYPE_DECIMAL, 2)
    db = b.colref(2, GX_TYPE_DECIMAL, 2)
    proj = b.projection(src, [
        b.call(GX_F_HOUR, GX_TYPE_I64, 0, t),
        b.call(GX_F_MINUTE, GX_TYPE_I64, 0, t),
        b.call(GX_F_SECOND, GX_TYPE_I64, 0, t),
        b.call(GX_F_GREATEST, GX_TYPE_DECIMAL, 2, da, db),
        b.call(GX_F_LEAST, GX_TYPE_DECIMAL, 2, da, db),
    ])
    ex = b.build(proj)
    ex.bind_chunks(src, chunks)
    ex.open()
    got = ex.pull_all([GX_TYPE_I64] * 3 + [GX_TYPE_DECIMAL] * 2,
                      [0, 0, 0, 2, 2])
    ex.close()
    ex.free()
    b.free()
    return rows, got


def test_oracle_hms_greatest():
    from decimal import Decimal
    lib = load_oracle()
    rows, got = _run_hms_greatest(lib)
    for (h, m, s, g, l), (hms, _, a, bb) in zip(got, rows):
        if hms is None:
            assert (h, m, s, g, l) == (None,) * 5
            continue
        assert (h, m, s) == hms
        assert Decimal(g) == max(Decimal(a), Decimal(bb))
        assert Decimal(l) == min(Decimal(a), Decimal(bb))


@pytest.mark.gpu
def test_hms_greatest_parity():
    _, want = _run_hms_greatest(load_oracle())
    _, got = _run_hms_greatest(load_product())
    assert got == want
