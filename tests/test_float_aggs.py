"""Device f64 aggregates (north_star: "stated tolerance for float
aggregates"): sum/avg over double columns (reference float path:
aggfuncs/func_sum.go float4/float8 sigs, func_avg.go float division).

TOLERANCE (stated): the device accumulates with f64 atomics whose order is
nondeterministic, so the sum rounds differently from the oracle's sequential
sum — parity is |gpu - cpu| <= 1e-12 * max(|cpu|, 1) * sqrt(n). Integer,
decimal and index results everywhere else stay bit/digit-exact; the
tolerance applies ONLY to float aggregates (SURVEY §8c bar).

The ORACLE's sequential sum is pinned bit-exactly against an identical
left-to-right Python float sum.
"""
import math

import numpy as np
import pytest

from tests.gxlib import (GX_AGG_AVG, GX_AGG_COUNT, GX_AGG_MODE_COMPLETE,
                         GX_AGG_MODE_FINAL, GX_AGG_MODE_PARTIAL, GX_AGG_SUM,
                         GX_TYPE_F64, GX_TYPE_I64, load_oracle, load_product)
from tidb_amd import plan as P
from tidb_amd.chunkpy import PyChunk

N = 200_000
NGROUP = 5


def _data(n=N, seed=2):
    rng = np.random.default_rng(seed)
    g = rng.integers(0, NGROUP, n)
    v = rng.standard_normal(n) * 100.0
    nulls = rng.random(n) < 0.05
    return g, v, nulls


def _run(lib, mode=GX_AGG_MODE_COMPLETE, n=N):
    g, v, nulls = _data(n)
    ch = PyChunk([GX_TYPE_I64, GX_TYPE_F64], n)
    for i in range(n):
        ch.append_row([int(g[i]), None if nulls[i] else float(v[i])])
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64, GX_TYPE_F64])
    agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                    [(GX_AGG_SUM, b.colref(1, GX_TYPE_F64), 0),
                     (GX_AGG_AVG, b.colref(1, GX_TYPE_F64), 0),
                     (GX_AGG_COUNT, -1, 0)], mode)
    ex = b.build(agg)
    ex.bind_chunks(src, [ch])
    ex.open()
    if mode == GX_AGG_MODE_PARTIAL:
        types = [GX_TYPE_I64, GX_TYPE_F64, GX_TYPE_I64, GX_TYPE_F64,
                 GX_TYPE_I64, GX_TYPE_I64]
    else:
        types = [GX_TYPE_I64, GX_TYPE_F64, GX_TYPE_F64, GX_TYPE_I64]
    rows = ex.pull_all(types, [0] * len(types))
    ex.close()
    ex.free()
    b.free()
    return sorted(rows), types


def _expected_sequential(n=N):
    """Left-to-right Python float sums — the oracle's exact order."""
    g, v, nulls = _data(n)
    sums = {k: 0.0 for k in range(NGROUP)}
    cnts = {k: 0 for k in range(NGROUP)}
    rows = {k: 0 for k in range(NGROUP)}
    for i in range(n):
        rows[int(g[i])] += 1
        if not nulls[i]:
            sums[int(g[i])] += float(v[i])
            cnts[int(g[i])] += 1
    return sorted((k, sums[k], sums[k] / cnts[k], rows[k])
                  for k in range(NGROUP))


def test_oracle_f64_sum_avg_bitexact_sequential():
    got, _ = _run(load_oracle())
    want = _expected_sequential()
    assert got == want  # same left-to-right double additions -> bit equal


TOL = lambda ref, n: 1e-12 * max(abs(ref), 1.0) * math.sqrt(max(n, 1))


@pytest.mark.gpu
def test_f64_sum_avg_parity_tolerance():
    want, _ = _run(load_oracle())
    got, _ = _run(load_product())
    assert len(got) == len(want)
    for (gk, s, a, c), (wk, ws, wa, wc) in zip(got, want):
        assert gk == wk and c == wc  # group + count exact
        assert abs(s - ws) <= TOL(ws, c), (gk, s, ws)
        assert abs(a - wa) <= TOL(wa, 1), (gk, a, wa)


@pytest.mark.gpu
def test_f64_partial_final_parity():
    """PARTIAL states (f64 sum + count) through the FINAL host merge: the
    multi-GPU path for float aggregates."""
    from tests.gxlib import GxChunk
    oracle = load_oracle()
    product = load_product()
    partial_p, ptypes = _run(product, GX_AGG_MODE_PARTIAL)
    partial_o, _ = _run(oracle, GX_AGG_MODE_PARTIAL)

    def final(lib, partials):
        b = P.Builder(lib)
        src = b.source(ptypes, [0] * len(ptypes))
        agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                        [(GX_AGG_SUM, b.colref(1, GX_TYPE_F64), 0),
                         (GX_AGG_AVG, b.colref(3, GX_TYPE_F64), 0),
                         (GX_AGG_COUNT, -1, 0)], GX_AGG_MODE_FINAL)
        ch = PyChunk(ptypes, max(len(partials), 1), [0] * len(ptypes))
        for r in partials:
            ch.append_row(list(r))
        ex = b.build(agg)
        ex.bind_chunks(src, [ch])
        ex.open()
        rows = ex.pull_all([GX_TYPE_I64, GX_TYPE_F64, GX_TYPE_F64,
                            GX_TYPE_I64], [0] * 4)
        ex.close()
        ex.free()
        b.free()
        return sorted(rows)

    got = final(product, partial_p)
    want = final(oracle, partial_o)
    complete, _ = _run(oracle)
    assert len(got) == len(want) == len(complete)
    for (gk, s, a, c), (wk, ws, wa, wc) in zip(got, want):
        assert gk == wk and c == wc
        assert abs(s - ws) <= TOL(ws, c)
        assert abs(a - wa) <= TOL(wa, 1)


def test_oracle_f64_all_null_group():
    """A group whose every arg is NULL: sum/avg NULL, count 0-arg rows still
    counted by count(*)."""
    lib = load_oracle()
    ch = PyChunk([GX_TYPE_I64, GX_TYPE_F64], 4)
    for r in [[1, None], [1, None], [2, 2.5]]:
        ch.append_row(r)
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64, GX_TYPE_F64])
    agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                    [(GX_AGG_SUM, b.colref(1, GX_TYPE_F64), 0),
                     (GX_AGG_COUNT, -1, 0)])
    ex = b.build(agg)
    ex.bind_chunks(src, [ch])
    ex.open()
    rows = sorted(ex.pull_all([GX_TYPE_I64, GX_TYPE_F64, GX_TYPE_I64],
                              [0, 0, 0]))
    ex.close()
    ex.free()
    b.free()
    assert rows == [(1, None, 2), (2, 2.5, 1)]


@pytest.mark.gpu
def test_f64_all_null_group_parity():
    lib = load_product()
    ch = PyChunk([GX_TYPE_I64, GX_TYPE_F64], 4)
    for r in [[1, None], [1, None], [2, 2.5]]:
        ch.append_row(r)
    b = P.Builder(lib)
    src = b.source([GX_TYPE_I64, GX_TYPE_F64])
    agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                    [(GX_AGG_SUM, b.colref(1, GX_TYPE_F64), 0),
                     (GX_AGG_COUNT, -1, 0)])
    ex = b.build(agg)
    ex.bind_chunks(src, [ch])
    ex.open()
    rows = sorted(ex.pull_all([GX_TYPE_I64, GX_TYPE_F64, GX_TYPE_I64],
                              [0, 0, 0]))
    ex.close()
    ex.free()
    b.free()
    assert rows == [(1, None, 2), (2, 2.5, 1)]
