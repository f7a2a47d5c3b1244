"""Multi-process merge semantics on CPU (gloo, world_size 2): each rank
produces canonical Q1 partial states (oracle executor — the CPU stand-in for
the per-GPU fused kernel), ranks all_gather the tiny partial chunks, and every
rank runs the FINAL merge — MergePartialResult semantics (aggfuncs.go:250-255).
The same merge path runs on the product library (host side, no GPU needed),
which is what bench.py uses after RCCL all_gather.
"""
import os

import pytest

from tests.gxlib import (GX_AGG_MODE_FINAL, GX_AGG_MODE_PARTIAL,
                         GX_TPCH_LINEITEM, GX_TYPE_DECIMAL, GX_TYPE_I64,
                         load_oracle)
from tidb_amd import plan as P
from tidb_amd.chunkpy import PyChunk


def merge_partials(lib, partial_rows):
    """Feed canonical partial rows through the FINAL plan of `lib`;
    returns {group: aggs} map."""
    from tidb_amd.chunkpy import PyChunk
    from tidb_amd.decimals import str_to_decimal_bytes
    b, src, agg, out_types, out_fracs, part_types, part_fracs = \
        P.q1_final_plan(lib)
    chunk = PyChunk(part_types, max(len(partial_rows), 1), part_fracs,
                    data_caps=[4096] * len(part_types))
    for r in partial_rows:
        vals = []
        for v, t in zip(r, part_types):
            if t == 2 and v is not None:
                vals.append(str_to_decimal_bytes(lib, v))
            else:
                vals.append(v)
        chunk.append_row(vals)
    ex = b.build(agg)
    ex.bind_chunks(src, [chunk])
    ex.open()
    caps = [2048 if t == 4 else None for t in out_types]
    rows = ex.pull_all(out_types, out_fracs, data_caps=caps)
    ex.close()
    ex.free()
    b.free()
    return {(r[0], r[1]): tuple(r[2:]) for r in rows}


def _worker(rank, world, result_q):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = "29711"
    dist.init_process_group("gloo", rank=rank, world_size=world)
    lib = load_oracle()
    total = 20000
    per = total // world
    b, src, agg, out_types, out_fracs = P.q1_plan(lib, GX_AGG_MODE_PARTIAL)
    ex = b.build(agg)
    ex.bind_tpch(src, GX_TPCH_LINEITEM, per, 42, rank * per, total)
    ex.open()
    caps = [2048 if t == 4 else None for t in out_types]
    rows = ex.pull_all(out_types, out_fracs, data_caps=caps)
    ex.close()
    ex.free()
    b.free()
    # all_gather the partial rows (object collective: payload is KBs)
    gathered = [None] * world
    dist.all_gather_object(gathered, rows)
    all_rows = [r for part in gathered for r in part]
    merged = merge_partials(lib, all_rows)
    dist.destroy_process_group()
    result_q.put((rank, merged))


def test_gloo_two_rank_merge():
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, merged = q.get(timeout=180)
        results[rank] = merged
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert results[0] == results[1]
    # equals single-process COMPLETE
    lib = load_oracle()
    from tests.test_oracle_q1 import run_q1, q1_results_to_map
    complete = q1_results_to_map(run_q1(lib, 20000))
    # convert merged decimal strings to the same normal form
    from fractions import Fraction
    def norm(m):
        out = {}
        for k, v in m.items():
            out[k] = tuple(Fraction(x) if isinstance(x, str) else x for x in v)
        return out
    got = norm(results[0])
    want = {}
    for k, v in complete.items():
        # complete map is in integer units; reconstruct fractions
        scales = [2, 2, 4, 6, 6, 6, 6, None]
        want[k] = tuple(
            Fraction(x, 10 ** s) if s is not None else x
            for x, s in zip(v, scales))
    assert got == want


def test_product_final_merge_matches_oracle():
    """The product library's host-side FINAL merge (used after the RCCL
    gather) must equal the oracle's on identical partial chunks. Runs on CPU."""
    from tests.gxlib import load_product
    from tests.test_oracle_q1 import run_q1
    oracle = load_oracle()
    product = load_product()
    partials = []
    for off in (0, 7000, 14000):
        partials.extend(run_q1(oracle, 7000, GX_AGG_MODE_PARTIAL,
                               row_offset=off, total_rows=21000))
    assert merge_partials(product, partials) == merge_partials(oracle, partials)


def test_final_merge_minmax_firstrow_both_libs():
    """MergePartialResult for MIN/MAX/FIRSTROW (func_max_min.go, aggfuncs.go
    merge semantics) on BOTH libraries' host FINAL path: extreme of per-shard
    extremes (NULL shard partials skipped), firstrow = first partial in input
    order (its value may be NULL)."""
    from tests.gxlib import (GX_AGG_FIRSTROW, GX_AGG_MAX, GX_AGG_MIN,
                             GX_AGG_MODE_FINAL, GX_TYPE_DECIMAL, GX_TYPE_I64,
                             load_oracle, load_product)
    from tidb_amd import plan as P
    from tidb_amd.chunkpy import PyChunk
    from tidb_amd.decimals import str_to_decimal_bytes

    part_types = [GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_I64]
    part_fracs = [0, 2, 0, 0]
    # rows: group, min-partial(dec), max-partial(i64), firstrow-partial(i64)
    partials = [
        (1, "2.50", 10, 7),
        (1, "-1.25", 40, None),
        (2, None, None, None),  # all-NULL shard partials
        (2, "9.00", -5, 3),
        (3, "0.00", 0, 0),
    ]
    expected = {
        1: ("-1.25", 40, 7),
        2: ("9.00", -5, None),  # firstrow: group 2's FIRST partial was NULL
        3: ("0.00", 0, 0),
    }

    def run(lib):
        b = P.Builder(lib)
        src = b.source(part_types, part_fracs)
        agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                        [(GX_AGG_MIN, b.colref(1, GX_TYPE_DECIMAL, 2), 2),
                         (GX_AGG_MAX, b.colref(2, GX_TYPE_I64), 0),
                         (GX_AGG_FIRSTROW, b.colref(3, GX_TYPE_I64), 0)],
                        GX_AGG_MODE_FINAL)
        chunk = PyChunk(part_types, len(partials), part_fracs)
        for r in partials:
            vals = [r[0],
                    None if r[1] is None else str_to_decimal_bytes(lib, r[1]),
                    r[2], r[3]]
            chunk.append_row(vals)
        ex = b.build(agg)
        ex.bind_chunks(src, [chunk])
        ex.open()
        rows = ex.pull_all([GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64,
                            GX_TYPE_I64], [0, 2, 0, 0])
        ex.close()
        ex.free()
        b.free()
        return {r[0]: tuple(r[1:]) for r in rows}

    got_o = run(load_oracle())
    got_p = run(load_product())
    assert got_o == got_p == expected


def test_final_empty_scalar_default_row():
    """FINAL with NO group-by over ZERO partial rows must emit the scalar
    default row (count=0, sum/min NULL) — HashAggExec's empty-input
    semantics, same as SELECT count(*), sum(x) FROM empty_t. Both libraries'
    host FINAL paths, CPU."""
    from tests.gxlib import (GX_AGG_COUNT, GX_AGG_MIN, GX_AGG_SUM,
                             load_product)
    part_types = [GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_I64, GX_TYPE_DECIMAL]
    part_fracs = [2, 0, 0, 2]

    def run(lib):
        b = P.Builder(lib)
        src = b.source(part_types, part_fracs)
        agg = b.hashagg(src, [],
                        [(GX_AGG_SUM, b.colref(0, GX_TYPE_DECIMAL, 2), 2),
                         (GX_AGG_COUNT, -1, 0),
                         (GX_AGG_MIN, b.colref(3, GX_TYPE_DECIMAL, 2), 2)],
                        GX_AGG_MODE_FINAL)
        chunk = PyChunk(part_types, 1, part_fracs)  # bound but EMPTY
        ex = b.build(agg)
        ex.bind_chunks(src, [chunk])
        ex.open()
        rows = ex.pull_all([GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_DECIMAL],
                           [2, 0, 2])
        ex.close()
        ex.free()
        b.free()
        return rows

    want = [(None, 0, None)]
    assert run(load_oracle()) == want
    assert run(load_product()) == want


def test_shards_final_merge_full_family():
    """2 row-range shards, PARTIAL per shard, FINAL merge == COMPLETE, for the
    full aggregate family incl. min/max/firstrow (oracle, CPU)."""
    from tests.gxlib import (GX_AGG_COUNT, GX_AGG_FIRSTROW, GX_AGG_MAX,
                             GX_AGG_MIN, GX_AGG_MODE_COMPLETE,
                             GX_AGG_MODE_FINAL, GX_AGG_MODE_PARTIAL,
                             GX_AGG_SUM, GX_TYPE_DECIMAL, GX_TYPE_I64,
                             GX_TYPE_STRING, load_oracle)
    from tidb_amd import plan as P
    from tidb_amd.chunkpy import PyChunk
    from tidb_amd.decimals import str_to_decimal_bytes
    lib = load_oracle()
    total = 16000

    AGGS = lambda b: [
        (GX_AGG_SUM, b.colref(P.L_QUANTITY, GX_TYPE_DECIMAL, 2), 2),
        (GX_AGG_MIN, b.colref(P.L_EXTPRICE, GX_TYPE_DECIMAL, 2), 2),
        (GX_AGG_MAX, b.colref(P.L_ORDERKEY, GX_TYPE_I64), 0),
        (GX_AGG_FIRSTROW, b.colref(P.L_RETFLAG, GX_TYPE_STRING), 0),
        (GX_AGG_COUNT, -1, 0),
    ]
    GROUP = lambda b: [b.colref(P.L_RETFLAG, GX_TYPE_STRING),
                       b.colref(P.L_LINESTATUS, GX_TYPE_STRING)]

    def run(mode, rows, offset):
        from tests.gxlib import GX_TPCH_LINEITEM
        b = P.Builder(lib)
        src = b.source(P.LINEITEM_TYPES, P.LINEITEM_FRACS)
        agg = b.hashagg(src, GROUP(b), AGGS(b), mode)
        ex = b.build(agg)
        ex.bind_tpch(src, GX_TPCH_LINEITEM, rows, 42, offset, total)
        ex.open()
        if mode == GX_AGG_MODE_PARTIAL:
            types = [GX_TYPE_STRING] * 2 + [GX_TYPE_DECIMAL, GX_TYPE_I64,
                                            GX_TYPE_DECIMAL, GX_TYPE_I64,
                                            GX_TYPE_STRING, GX_TYPE_I64]
            fracs = [0, 0, 2, 0, 2, 0, 0, 0]
        else:
            types = [GX_TYPE_STRING] * 2 + [GX_TYPE_DECIMAL, GX_TYPE_DECIMAL,
                                            GX_TYPE_I64, GX_TYPE_STRING,
                                            GX_TYPE_I64]
            fracs = [0, 0, 2, 2, 0, 0, 0]
        out = ex.pull_all(types, fracs, data_caps=[2048] * len(types))
        ex.close()
        ex.free()
        b.free()
        return out, types, fracs

    want, _, _ = run(GX_AGG_MODE_COMPLETE, total, 0)
    p0, pt, pf = run(GX_AGG_MODE_PARTIAL, total // 2, 0)
    p1, _, _ = run(GX_AGG_MODE_PARTIAL, total - total // 2, total // 2)

    # FINAL merge of the gathered partial rows
    b = P.Builder(lib)
    src = b.source(pt, pf)
    agg = b.hashagg(src, [b.colref(0, GX_TYPE_STRING),
                          b.colref(1, GX_TYPE_STRING)],
                    [(GX_AGG_SUM, b.colref(2, GX_TYPE_DECIMAL, 2), 2),
                     (GX_AGG_MIN, b.colref(4, GX_TYPE_DECIMAL, 2), 2),
                     (GX_AGG_MAX, b.colref(5, GX_TYPE_I64), 0),
                     (GX_AGG_FIRSTROW, b.colref(6, GX_TYPE_STRING), 0),
                     (GX_AGG_COUNT, -1, 0)], GX_AGG_MODE_FINAL)
    chunk = PyChunk(pt, len(p0) + len(p1), pf, data_caps=[4096] * len(pt))
    for r in p0 + p1:
        vals = []
        for v, t in zip(r, pt):
            if t == GX_TYPE_DECIMAL and v is not None:
                vals.append(str_to_decimal_bytes(lib, v))
            else:
                vals.append(v)
        chunk.append_row(vals)
    ex = b.build(agg)
    ex.bind_chunks(src, [chunk])
    ex.open()
    types = [GX_TYPE_STRING] * 2 + [GX_TYPE_DECIMAL, GX_TYPE_DECIMAL,
                                    GX_TYPE_I64, GX_TYPE_STRING, GX_TYPE_I64]
    got = ex.pull_all(types, [0, 0, 2, 2, 0, 0, 0], data_caps=[2048] * 7)
    ex.close()
    ex.free()
    b.free()
    as_map = lambda rows: {(r[0], r[1]): tuple(r[2:]) for r in rows}
    assert as_map(got) == as_map(want)
    assert len(got) >= 4


def test_final_merge_string_extremes_both_libs():
    """FINAL merge of MIN/MAX over VARCHAR partials: binary collation with
    PAD SPACE ('abc ' == 'abc' for compare), NULL shard partials skipped.
    Both libraries' host FINAL paths, CPU."""
    from tests.gxlib import (GX_AGG_MAX, GX_AGG_MIN, GX_TYPE_STRING,
                             load_product)
    part_types = [GX_TYPE_I64, GX_TYPE_STRING, GX_TYPE_STRING]
    partials = [
        (1, "apple", "pear"),
        (1, "apple  ", "zebra"),   # PAD SPACE: ties 'apple'
        (1, None, None),           # all-NULL shard partial
        (2, "kiwi", "kiwi"),
    ]

    def run(lib):
        b = P.Builder(lib)
        src = b.source(part_types)
        agg = b.hashagg(src, [b.colref(0, GX_TYPE_I64)],
                        [(GX_AGG_MIN, b.colref(1, GX_TYPE_STRING), 0),
                         (GX_AGG_MAX, b.colref(2, GX_TYPE_STRING), 0)],
                        GX_AGG_MODE_FINAL)
        chunk = PyChunk(part_types, len(partials), None, [None, 256, 256])
        for r in partials:
            chunk.append_row(list(r))
        ex = b.build(agg)
        ex.bind_chunks(src, [chunk])
        ex.open()
        rows = sorted(ex.pull_all([GX_TYPE_I64, GX_TYPE_STRING,
                                   GX_TYPE_STRING]))
        ex.close()
        ex.free()
        b.free()
        return rows

    got_o = run(load_oracle())
    got_p = run(load_product())
    assert got_o == got_p
    assert got_o[0][0] == 1 and got_o[0][2] == "zebra"
    assert got_o[0][1] in ("apple", "apple  ")  # PAD-SPACE tie, either rep
    assert got_o[1] == (2, "kiwi", "kiwi")
