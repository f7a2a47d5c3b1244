"""GPU parity suite (needs a real MI355X): the product engine's results must
match the CPU oracle restatement on the same data — bit/digit-exact for
int/decimal/date (SURVEY §8 bar).
"""
import ctypes

import pytest

from tests.gxlib import (GX_AGG_MODE_COMPLETE, GX_AGG_MODE_PARTIAL,
                         GX_TPCH_LINEITEM, load_oracle, load_product)
from tidb_amd import plan as P

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def libs():
    return load_oracle(), load_product()


def _pull_lineitem(lib, n_rows, seed=42, row_offset=0, total_rows=None):
    b = P.Builder(lib)
    src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
    ex = b.build(src)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, n_rows, seed, row_offset, total_rows)
    ex.open()
    rows = ex.pull_all(P.LINEITEM_TYPES, P.LINEITEM_FRACS,
                       data_caps=[None] * 5 + [2048, 2048] + [None])
    ex.close()
    ex.free()
    b.free()
    return rows


def test_generator_parity(libs):
    """The device lineitem generator must produce bit-identical chunks to the
    CPU spec (oracle/tpch.cpp)."""
    oracle, product = libs
    a = _pull_lineitem(oracle, 5000)
    b = _pull_lineitem(product, 5000)
    assert a == b
    # sharded offset parity too
    a = _pull_lineitem(oracle, 1000, row_offset=4000, total_rows=5000)
    b = _pull_lineitem(product, 1000, row_offset=4000, total_rows=5000)
    assert a == b


def _run_q1(lib, n_rows, mode=GX_AGG_MODE_COMPLETE, seed=42, row_offset=0,
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


def _as_map(rows):
    return {(r[0], r[1]): tuple(r[2:]) for r in rows}


@pytest.mark.parametrize("n", [1000, 65536, 1000000])
def test_q1_complete_parity(libs, n):
    oracle, product = libs
    assert _as_map(_run_q1(oracle, n)) == _as_map(_run_q1(product, n))


def test_q1_partial_parity(libs):
    oracle, product = libs
    got_o = _as_map(_run_q1(oracle, 30000, GX_AGG_MODE_PARTIAL))
    got_p = _as_map(_run_q1(product, 30000, GX_AGG_MODE_PARTIAL))
    assert got_o == got_p


def test_q1_bound_chunks_with_nulls(libs):
    """Upload host chunks with NULLs; product fused kernel must honor the
    null bitmaps exactly like the oracle."""
    oracle, product = libs
    from tidb_amd.chunkpy import PyChunk
    from tidb_amd.decimals import str_to_decimal_bytes

    def make_chunk(lib):
        d = lambda s: str_to_decimal_bytes(lib, s)
        date = lambda y, m, dd: lib.gx_time_from_date(y, m, dd)
        rows = [
            (1, d("10.00"), d("100.00"), d("0.05"), d("0.02"), "A", "F", date(1995, 1, 1)),
            (2, None, d("200.00"), d("0.00"), d("0.01"), "A", "F", date(1995, 1, 2)),
            (3, d("5.00"), None, d("0.10"), d("0.00"), "A", "F", date(1995, 1, 3)),
            (4, d("7.00"), d("50.00"), d("0.01"), d("0.08"), "N", "O", date(1999, 1, 1)),
            (5, d("3.00"), d("30.00"), None, d("0.03"), "R", "F", date(1996, 5, 5)),
            (6, d("2.00"), d("20.00"), d("0.02"), d("0.04"), "R", "F", None),
        ]
        chunk = PyChunk(P.LINEITEM_TYPES, len(rows), P.LINEITEM_FRACS,
                        data_caps=[None] * 5 + [64, 64] + [None])
        for r in rows:
            chunk.append_row(list(r))
        return chunk

    def run(lib):
        b, src, agg, out_types, out_fracs = P.q1_plan(lib)
        ex = b.build(agg)
        ex.bind_chunks(src, [make_chunk(lib)])
        ex.open()
        caps = [2048 if t == 4 else None for t in out_types]
        rows = ex.pull_all(out_types, out_fracs, data_caps=caps)
        ex.close()
        ex.free()
        b.free()
        return _as_map(rows)

    assert run(oracle) == run(product)


def test_q1_shards_merge_parity(libs):
    """PARTIAL per shard on the product engine -> FINAL merge (host) ==
    oracle COMPLETE over the full table. This is exactly the 8-GPU merge."""
    oracle, product = libs
    total = 40000
    shard_rows = []
    for off in range(0, total, 10000):
        shard_rows.extend(
            _run_q1(product, 10000, GX_AGG_MODE_PARTIAL, row_offset=off,
                    total_rows=total))
    from tests.test_dist_merge import merge_partials
    merged = merge_partials(product, shard_rows)
    complete = _as_map(_run_q1(oracle, total))
    assert merged == complete


def test_native_code_loaded(libs):
    _, product = libs
    assert product.gx_engine_is_gpu() == 1
    assert product.gx_engine_name() == b"gxexec-mi355x"


def test_q3_parity(libs):
    """Q3 (3-table join + grouped sum + TopN) on the device join-aggregate
    pipeline vs the oracle executor."""
    oracle, product = libs
    from tests.test_oracle_q3 import run_q3
    got = run_q3(product)
    want = run_q3(oracle)
    assert len(got) == len(want) > 0
    assert got == want


def test_q3_parity_larger(libs):
    oracle, product = libs
    import tests.test_oracle_q3 as q3
    old = (q3.N_LI, q3.N_ORD, q3.N_CUST)
    try:
        q3.N_LI, q3.N_ORD, q3.N_CUST = 400000, 100000, 10000
        assert q3.run_q3(product) == q3.run_q3(oracle)
    finally:
        q3.N_LI, q3.N_ORD, q3.N_CUST = old


def test_q1_interpreted_fallback_parity(libs, monkeypatch):
    """GX_NO_JIT=1 exercises the interpreted fused kernel (the fallback when
    hipRTC is unavailable); results must match the oracle bit-for-bit too."""
    oracle, product = libs
    monkeypatch.setenv("GX_NO_JIT", "1")
    got = _as_map(_run_q1(product, 65536))
    monkeypatch.delenv("GX_NO_JIT")
    assert got == _as_map(_run_q1(oracle, 65536))


def test_q1_glds_staged_parity(libs, monkeypatch):
    """GX_GLDS=1 exercises the glds-staged (LDS-DMA) kernel variant — kept
    opt-in (slower than the specialized kernels) but parity must hold."""
    oracle, product = libs
    monkeypatch.setenv("GX_GLDS", "1")
    monkeypatch.setenv("GX_NO_JIT", "1")  # glds path is the interpreted engine
    got = _as_map(_run_q1(product, 65536))
    monkeypatch.delenv("GX_GLDS")
    monkeypatch.delenv("GX_NO_JIT")
    assert got == _as_map(_run_q1(oracle, 65536))
