"""Plan construction + pull-loop driver over the gx C-ABI.

Builds the operator trees the reference's executorBuilder would build
(pkg/executor/builder.go:283-311) for the benchmark queries, and drives them
through the Open/Next/Close contract.
"""
import ctypes

from tests.gxlib import (GX_AGG_AVG, GX_AGG_COUNT, GX_AGG_FIRSTROW,
                         GX_AGG_MODE_COMPLETE, GX_AGG_MODE_FINAL,
                         GX_AGG_MODE_PARTIAL, GX_AGG_SUM, GX_F_LT, GX_F_MINUS,
                         GX_F_MUL, GX_F_PLUS, GX_TYPE_DECIMAL, GX_TYPE_I64,
                         GX_TYPE_STRING, GX_TYPE_TIME, GX_TPCH_LINEITEM)
from tidb_amd.chunkpy import PyChunk

LINEITEM_TYPES = [GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_DECIMAL, GX_TYPE_DECIMAL,
                  GX_TYPE_DECIMAL, GX_TYPE_STRING, GX_TYPE_STRING, GX_TYPE_TIME]
LINEITEM_FRACS = [0, 2, 2, 2, 2, 0, 0, 0]
# column indexes
L_ORDERKEY, L_QUANTITY, L_EXTPRICE, L_DISCOUNT, L_TAX, L_RETFLAG, L_LINESTATUS, L_SHIPDATE = range(8)

ORDERS_TYPES = [GX_TYPE_I64, GX_TYPE_I64, GX_TYPE_TIME, GX_TYPE_I64]
CUSTOMER_TYPES = [GX_TYPE_I64, GX_TYPE_STRING]


def _arr(vals):
    return (ctypes.c_int32 * len(vals))(*vals)


def _u8arr(vals):
    return (ctypes.c_uint8 * len(vals))(*vals)


class Builder:
    def __init__(self, lib):
        self.lib = lib
        self.pb = lib.gx_pb_new()

    def colref(self, idx, typ, frac=0):
        return self.lib.gx_pb_colref(self.pb, idx, typ, frac)

    def const_i64(self, v):
        return self.lib.gx_pb_const_i64(self.pb, v)

    def const_time(self, v):
        return self.lib.gx_pb_const_time(self.pb, v)

    def const_dec(self, dec40):
        return self.lib.gx_pb_const_dec(self.pb, (ctypes.c_uint8 * 40)(*dec40))

    def const_str(self, s):
        b = s.encode() if isinstance(s, str) else s
        return self.lib.gx_pb_const_str(self.pb, b, len(b))

    def call(self, func, ret_type, ret_frac, *args):
        return self.lib.gx_pb_call(self.pb, func, ret_type, ret_frac, _arr(args), len(args))

    def source(self, types, fracs=None):
        fr = fracs or [0] * len(types)
        return self.lib.gx_pb_source(self.pb, _arr(types), _arr(fr), len(types))

    def selection(self, child, conds):
        return self.lib.gx_pb_selection(self.pb, child, _arr(conds), len(conds))

    def projection(self, child, exprs):
        return self.lib.gx_pb_projection(self.pb, child, _arr(exprs), len(exprs))

    def hashagg(self, child, group_exprs, aggs, mode=GX_AGG_MODE_COMPLETE):
        funcs = _arr([a[0] for a in aggs])
        args = _arr([a[1] for a in aggs])
        fracs = _arr([a[2] for a in aggs])
        return self.lib.gx_pb_hashagg(self.pb, child, _arr(group_exprs),
                                      len(group_exprs), funcs, args, fracs,
                                      len(aggs), mode)

    def streamagg(self, child, group_exprs, aggs):
        funcs = _arr([a[0] for a in aggs])
        args = _arr([a[1] for a in aggs])
        fracs = _arr([a[2] for a in aggs])
        return self.lib.gx_pb_streamagg(self.pb, child, _arr(group_exprs),
                                        len(group_exprs), funcs, args, fracs,
                                        len(aggs))

    def topn(self, child, keys, desc, limit, offset=0):
        return self.lib.gx_pb_topn(self.pb, child, _arr(keys), _u8arr(desc),
                                   len(keys), limit, offset)

    def sort(self, child, keys, desc):
        return self.lib.gx_pb_topn(self.pb, child, _arr(keys), _u8arr(desc),
                                   len(keys), -1, 0)

    def mergejoin(self, build, probe, build_keys, probe_keys, join_type=0):
        return self.lib.gx_pb_mergejoin(self.pb, build, probe,
                                        _arr(build_keys), _arr(probe_keys),
                                        len(build_keys), join_type)

    def hashjoin(self, build, probe, build_keys, probe_keys, join_type=0):
        return self.lib.gx_pb_hashjoin(self.pb, build, probe, _arr(build_keys),
                                       _arr(probe_keys), len(build_keys), join_type)

    def build(self, root, device=-1):
        ex = self.lib.gx_build(self.pb, root, device)
        assert ex, "gx_build failed"
        return Executor(self.lib, ex)

    def free(self):
        self.lib.gx_pb_free(self.pb)


class Executor:
    def __init__(self, lib, ex):
        self.lib = lib
        self.ex = ex

    def bind_tpch(self, node, table, n_rows, seed=42, row_offset=0, total_rows=None):
        if total_rows is None:
            total_rows = n_rows
        rc = self.lib.gx_bind_tpch_sharded(self.ex, node, table, n_rows, seed,
                                           row_offset, total_rows)
        assert rc == 0, self.error()

    def bind_chunks(self, node, pychunks):
        from tests.gxlib import GxChunk
        arr = (GxChunk * len(pychunks))()
        for i, pc in enumerate(pychunks):
            arr[i] = pc.as_gx()
        rc = self.lib.gx_bind_chunks(self.ex, node, arr, len(pychunks))
        assert rc == 0, self.error()

    def open(self):
        rc = self.lib.gx_open(self.ex)
        assert rc == 0, f"open failed ({rc}): {self.error()}"

    def error(self):
        return self.lib.gx_last_error(self.ex).decode()

    def pull_all(self, out_types, out_fracs=None, max_rows=1024, data_caps=None,
                 reuse_chunk=None):
        """Drive Next until EOF; returns list of row tuples (decoded).
        reuse_chunk: a PyChunk from a previous call (avoids buffer realloc
        in benchmark loops)."""
        rows = []
        while True:
            if reuse_chunk is not None:
                chunk = reuse_chunk
                for col in chunk.columns:
                    col.length = 0
            else:
                chunk = PyChunk(out_types, max_rows, out_fracs, data_caps)
            g = chunk.as_gx()
            n = ctypes.c_int32(0)
            rc = self.lib.gx_next(self.ex, ctypes.byref(g), ctypes.byref(n))
            assert rc == 0, f"next failed ({rc}): {self.error()}"
            if n.value == 0:
                break
            for c, col in enumerate(chunk.columns):
                col.length = g.cols[c].length
            rows.extend(chunk.rows(n.value))
        return rows

    def pull_one(self, out_types, out_fracs=None, max_rows=1024,
                 data_caps=None):
        """One Next call (e.g. to trigger a lazy device sort without reading
        the whole table back); returns the row count."""
        chunk = PyChunk(out_types, max_rows, out_fracs, data_caps)
        g = chunk.as_gx()
        n = ctypes.c_int32(0)
        rc = self.lib.gx_next(self.ex, ctypes.byref(g), ctypes.byref(n))
        assert rc == 0, f"next failed ({rc}): {self.error()}"
        return n.value

    def close(self):
        self.lib.gx_close(self.ex)

    def free(self):
        self.lib.gx_exec_free(self.ex)


def q1_plan(lib, mode=GX_AGG_MODE_COMPLETE, firstrow=False):
    """TPC-H Q1 operator tree per the reference's golden plan
    (pkg/planner/core/casetest/tpch/testdata/tpch_suite_out.json TestQ1):
      Selection(shipdate <= DATE'1998-09-01' - 90d ~ here: < 1998-09-01 per
      BASELINE config) -> Projection(cols + disc_price + charge) ->
      HashAgg(group by returnflag, linestatus; sums + avgs + count).
    Returns (builder, source_node, root_node, out_types, out_fracs).
    """
    b = Builder(lib)
    src = b.source(LINEITEM_TYPES, LINEITEM_FRACS)
    shipdate = b.colref(L_SHIPDATE, GX_TYPE_TIME)
    cutoff = b.const_time(lib.gx_time_from_date(1998, 9, 1))
    cond = b.call(GX_F_LT, GX_TYPE_I64, 0, shipdate, cutoff)
    sel = b.selection(src, [cond])

    qty = b.colref(L_QUANTITY, GX_TYPE_DECIMAL, 2)
    price = b.colref(L_EXTPRICE, GX_TYPE_DECIMAL, 2)
    disc = b.colref(L_DISCOUNT, GX_TYPE_DECIMAL, 2)
    tax = b.colref(L_TAX, GX_TYPE_DECIMAL, 2)
    rf = b.colref(L_RETFLAG, GX_TYPE_STRING)
    ls = b.colref(L_LINESTATUS, GX_TYPE_STRING)
    one = _const_dec_one(lib, b)
    one_minus_disc = b.call(GX_F_MINUS, GX_TYPE_DECIMAL, 2, one, disc)
    disc_price = b.call(GX_F_MUL, GX_TYPE_DECIMAL, 4, price, one_minus_disc)
    one_plus_tax = b.call(GX_F_PLUS, GX_TYPE_DECIMAL, 2, one, tax)
    charge = b.call(GX_F_MUL, GX_TYPE_DECIMAL, 6, disc_price, one_plus_tax)
    # projection emits: rf, ls, qty, price, disc, disc_price, charge
    proj = b.projection(sel, [rf, ls, qty, price, disc, disc_price, charge])

    # post-projection col indexes
    p_rf = b.colref(0, GX_TYPE_STRING)
    p_ls = b.colref(1, GX_TYPE_STRING)
    p_qty = b.colref(2, GX_TYPE_DECIMAL, 2)
    p_price = b.colref(3, GX_TYPE_DECIMAL, 2)
    p_disc = b.colref(4, GX_TYPE_DECIMAL, 2)
    p_dp = b.colref(5, GX_TYPE_DECIMAL, 4)
    p_ch = b.colref(6, GX_TYPE_DECIMAL, 6)
    aggs = [
        (GX_AGG_SUM, p_qty, 2),       # sum_qty
        (GX_AGG_SUM, p_price, 2),     # sum_base_price
        (GX_AGG_SUM, p_dp, 4),        # sum_disc_price
        (GX_AGG_SUM, p_ch, 6),        # sum_charge
        (GX_AGG_AVG, p_qty, 6),       # avg_qty (frac 2+4)
        (GX_AGG_AVG, p_price, 6),     # avg_price
        (GX_AGG_AVG, p_disc, 6),      # avg_disc
        (GX_AGG_COUNT, -1, 0),        # count_order
    ]
    if firstrow:
        # the golden plan also carries firstrow(rf), firstrow(ls)
        # (tpch_suite_out.json TestQ1; aggfuncs/builder.go)
        aggs += [(GX_AGG_FIRSTROW, p_rf, 0), (GX_AGG_FIRSTROW, p_ls, 0)]
    agg = b.hashagg(proj, [p_rf, p_ls], aggs, mode)
    out_types = [GX_TYPE_STRING, GX_TYPE_STRING]
    out_fracs = [0, 0]
    if mode == GX_AGG_MODE_PARTIAL:
        for f, _a, fr in aggs:
            if f == GX_AGG_COUNT:
                out_types += [GX_TYPE_I64]
                out_fracs += [0]
            elif f == GX_AGG_FIRSTROW:
                out_types += [GX_TYPE_STRING]
                out_fracs += [0]
            else:
                out_types += [GX_TYPE_DECIMAL, GX_TYPE_I64]
                out_fracs += [fr, 0]
    else:
        for f, _a, fr in aggs:
            if f == GX_AGG_COUNT:
                out_types += [GX_TYPE_I64]
            elif f == GX_AGG_FIRSTROW:
                out_types += [GX_TYPE_STRING]
            else:
                out_types += [GX_TYPE_DECIMAL]
            out_fracs += [fr]
    return b, src, agg, out_types, out_fracs


def _const_dec_one(lib, b):
    out = (ctypes.c_uint8 * 40)()
    lib.gx_dec_from_string(b"1", 1, out)
    return b.const_dec(bytes(out))


def q1_final_plan(lib):
    """FINAL-mode agg consuming canonical partial-state chunks (the RCCL-merge
    reduce step; MergePartialResult semantics, aggfuncs.go:250-255)."""
    b = Builder(lib)
    part_types = [GX_TYPE_STRING, GX_TYPE_STRING,
                  GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64,
                  GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64,
                  GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64,
                  GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_I64]
    part_fracs = [0, 0, 2, 0, 2, 0, 4, 0, 6, 0, 2, 0, 2, 0, 2, 0, 0]
    src = b.source(part_types, part_fracs)
    rf = b.colref(0, GX_TYPE_STRING)
    ls = b.colref(1, GX_TYPE_STRING)
    aggs = [
        (GX_AGG_SUM, b.colref(2, GX_TYPE_DECIMAL, 2), 2),
        (GX_AGG_SUM, b.colref(4, GX_TYPE_DECIMAL, 2), 2),
        (GX_AGG_SUM, b.colref(6, GX_TYPE_DECIMAL, 4), 4),
        (GX_AGG_SUM, b.colref(8, GX_TYPE_DECIMAL, 6), 6),
        (GX_AGG_AVG, b.colref(10, GX_TYPE_DECIMAL, 2), 6),
        (GX_AGG_AVG, b.colref(12, GX_TYPE_DECIMAL, 2), 6),
        (GX_AGG_AVG, b.colref(14, GX_TYPE_DECIMAL, 2), 6),
        (GX_AGG_COUNT, -1, 0),
    ]
    agg = b.hashagg(src, [rf, ls], aggs, GX_AGG_MODE_FINAL)
    out_types = [GX_TYPE_STRING, GX_TYPE_STRING] + \
        [GX_TYPE_DECIMAL] * 7 + [GX_TYPE_I64]
    out_fracs = [0, 0, 2, 2, 4, 6, 6, 6, 6, 0]
    return b, src, agg, out_types, out_fracs, part_types, part_fracs


# ---- TPC-H Q3 (BASELINE config 3) ----
# Plan shape pinned by the reference's TPC-H golden
# (pkg/planner/core/casetest/tpch/tpch_test.go:454-467):
#   HashJoin(customer filtered on c_mktsegment = 'BUILDING'
#            ⋈ orders filtered on o_orderdate < 1995-03-15)
#   ⋈ lineitem filtered on l_shipdate > 1995-03-15
#   → HashAgg(group by l_orderkey, o_orderdate, o_shippriority;
#             sum(l_extendedprice * (1 - l_discount)) as revenue)
#   → TopN(revenue desc, o_orderdate asc, limit 10)
O_ORDERKEY, O_CUSTKEY, O_ORDERDATE, O_SHIPPRIORITY = range(4)
C_CUSTKEY, C_MKTSEGMENT = range(2)


def q3_merge_plan(lib, limit=10):
    """Q3 with merge joins (join/merge_join.go): each join's inputs sorted on
    its join key by an explicit full Sort, as the planner would arrange.
    Results are identical to q3_plan's hash joins."""
    from tests.gxlib import GX_F_GT, GX_F_EQ
    b = Builder(lib)
    cust = b.source(CUSTOMER_TYPES)
    seg = b.colref(C_MKTSEGMENT, GX_TYPE_STRING)
    cond_c = b.call(GX_F_EQ, GX_TYPE_I64, 0, seg,
                    lib.gx_pb_const_str(b.pb, b"BUILDING", 8))
    sel_c = b.selection(cust, [cond_c])
    sort_c = b.sort(sel_c, [b.colref(C_CUSTKEY, GX_TYPE_I64)], [0])

    orders = b.source(ORDERS_TYPES)
    odate = b.colref(O_ORDERDATE, GX_TYPE_TIME)
    cond_o = b.call(GX_F_LT, GX_TYPE_I64, 0, odate,
                    b.const_time(lib.gx_time_from_date(1995, 3, 15)))
    sel_o = b.selection(orders, [cond_o])
    sort_o = b.sort(sel_o, [b.colref(O_CUSTKEY, GX_TYPE_I64)], [0])

    j1 = b.mergejoin(sort_c, sort_o,
                     [b.colref(C_CUSTKEY, GX_TYPE_I64)],
                     [b.colref(O_CUSTKEY, GX_TYPE_I64)])
    sort_j1 = b.sort(j1, [b.colref(2, GX_TYPE_I64)], [0])  # by o_orderkey

    li = b.source(LINEITEM_TYPES, LINEITEM_FRACS)
    sdate = b.colref(L_SHIPDATE, GX_TYPE_TIME)
    cond_l = b.call(GX_F_GT, GX_TYPE_I64, 0, sdate,
                    b.const_time(lib.gx_time_from_date(1995, 3, 15)))
    sel_l = b.selection(li, [cond_l])
    sort_l = b.sort(sel_l, [b.colref(L_ORDERKEY, GX_TYPE_I64)], [0])

    j2 = b.mergejoin(sort_j1, sort_l,
                     [b.colref(2, GX_TYPE_I64)],
                     [b.colref(L_ORDERKEY, GX_TYPE_I64)])

    jo_orderkey = b.colref(6 + L_ORDERKEY, GX_TYPE_I64)
    jo_odate = b.colref(4, GX_TYPE_TIME)
    jo_prio = b.colref(5, GX_TYPE_I64)
    price = b.colref(6 + L_EXTPRICE, GX_TYPE_DECIMAL, 2)
    disc = b.colref(6 + L_DISCOUNT, GX_TYPE_DECIMAL, 2)
    one = _const_dec_one(lib, b)
    om_d = b.call(GX_F_MINUS, GX_TYPE_DECIMAL, 2, one, disc)
    rev = b.call(GX_F_MUL, GX_TYPE_DECIMAL, 4, price, om_d)
    proj = b.projection(j2, [jo_orderkey, jo_odate, jo_prio, rev])

    from tests.gxlib import GX_AGG_SUM
    agg = b.hashagg(proj,
                    [b.colref(0, GX_TYPE_I64), b.colref(1, GX_TYPE_TIME),
                     b.colref(2, GX_TYPE_I64)],
                    [(GX_AGG_SUM, b.colref(3, GX_TYPE_DECIMAL, 4), 4)])
    topn = b.topn(agg, [b.colref(3, GX_TYPE_DECIMAL, 4),
                        b.colref(1, GX_TYPE_TIME)],
                  [1, 0], limit)
    out_types = [GX_TYPE_I64, GX_TYPE_TIME, GX_TYPE_I64, GX_TYPE_DECIMAL]
    out_fracs = [0, 0, 0, 4]
    return b, (cust, orders, li), topn, out_types, out_fracs


def q3_plan(lib, limit=10):
    from tests.gxlib import GX_F_GT, GX_F_EQ, GX_TPCH_ORDERS, GX_TPCH_CUSTOMER
    b = Builder(lib)
    cust = b.source(CUSTOMER_TYPES)
    seg = b.colref(C_MKTSEGMENT, GX_TYPE_STRING)
    cond_c = b.call(GX_F_EQ, GX_TYPE_I64, 0, seg,
                    lib.gx_pb_const_str(b.pb, b"BUILDING", 8))
    sel_c = b.selection(cust, [cond_c])

    orders = b.source(ORDERS_TYPES)
    odate = b.colref(O_ORDERDATE, GX_TYPE_TIME)
    cutoff = b.const_time(lib.gx_time_from_date(1995, 3, 15))
    cond_o = b.call(GX_F_LT, GX_TYPE_I64, 0, odate, cutoff)
    sel_o = b.selection(orders, [cond_o])

    # join1: build = filtered customer, probe = filtered orders
    # output cols: [c_custkey, c_mktsegment, o_orderkey, o_custkey,
    #               o_orderdate, o_shippriority]
    j1 = b.hashjoin(sel_c, sel_o,
                    [b.colref(C_CUSTKEY, GX_TYPE_I64)],
                    [b.colref(O_CUSTKEY, GX_TYPE_I64)])

    li = b.source(LINEITEM_TYPES, LINEITEM_FRACS)
    sdate = b.colref(L_SHIPDATE, GX_TYPE_TIME)
    cond_l = b.call(GX_F_GT, GX_TYPE_I64, 0, sdate,
                    b.const_time(lib.gx_time_from_date(1995, 3, 15)))
    sel_l = b.selection(li, [cond_l])

    # join2: build = join1 output, probe = filtered lineitem
    # output cols: join1 cols (6) ++ lineitem cols (8)
    j2 = b.hashjoin(j1, sel_l,
                    [b.colref(2, GX_TYPE_I64)],          # o_orderkey in join1
                    [b.colref(L_ORDERKEY, GX_TYPE_I64)])

    # projection: l_orderkey, o_orderdate, o_shippriority, revenue-term
    jo_orderkey = b.colref(6 + L_ORDERKEY, GX_TYPE_I64)
    jo_odate = b.colref(4, GX_TYPE_TIME)
    jo_prio = b.colref(5, GX_TYPE_I64)
    price = b.colref(6 + L_EXTPRICE, GX_TYPE_DECIMAL, 2)
    disc = b.colref(6 + L_DISCOUNT, GX_TYPE_DECIMAL, 2)
    one = _const_dec_one(lib, b)
    om_d = b.call(GX_F_MINUS, GX_TYPE_DECIMAL, 2, one, disc)
    rev = b.call(GX_F_MUL, GX_TYPE_DECIMAL, 4, price, om_d)
    proj = b.projection(j2, [jo_orderkey, jo_odate, jo_prio, rev])

    from tests.gxlib import GX_AGG_SUM
    agg = b.hashagg(proj,
                    [b.colref(0, GX_TYPE_I64), b.colref(1, GX_TYPE_TIME),
                     b.colref(2, GX_TYPE_I64)],
                    [(GX_AGG_SUM, b.colref(3, GX_TYPE_DECIMAL, 4), 4)])
    # agg out: orderkey, orderdate, shippriority, revenue(dec s4)
    topn = b.topn(agg, [b.colref(3, GX_TYPE_DECIMAL, 4),
                        b.colref(1, GX_TYPE_TIME)],
                  [1, 0], limit)
    out_types = [GX_TYPE_I64, GX_TYPE_TIME, GX_TYPE_I64, GX_TYPE_DECIMAL]
    out_fracs = [0, 0, 0, 4]
    return b, (cust, orders, li), topn, out_types, out_fracs
