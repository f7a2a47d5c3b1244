// oracle/capi_exec.cpp — executor C-ABI implementation (oracle flavor).
// ORACLE / TEST INFRASTRUCTURE ONLY.
#include <cstring>
#include <map>
#include <memory>
#include <string>

#include "../include/gx_executor.h"
#include "exec.h"
#include "core_time.h"

using namespace oracle;

struct gx_pb {
  Plan plan;
  std::string err;
};

struct gx_exec {
  Plan plan;
  int root = -1;
  std::map<int, SourceBinding> bindings;
  std::unique_ptr<Exec> exec;
  std::string err;
  bool opened = false;
};

extern "C" {

gx_pb* gx_pb_new(void) { return new gx_pb(); }
void gx_pb_free(gx_pb* pb) { delete pb; }

static int32_t addExpr(gx_pb* pb, Expr e) {
  pb->plan.exprs.push_back(std::move(e));
  return (int32_t)pb->plan.exprs.size() - 1;
}

int32_t gx_pb_colref(gx_pb* pb, int32_t col_idx, int32_t type, int32_t frac) {
  Expr e;
  e.kind = EK_COLREF;
  e.colIdx = col_idx;
  e.retType = type;
  e.retFrac = frac;
  return addExpr(pb, std::move(e));
}
int32_t gx_pb_const_i64(gx_pb* pb, int64_t v) {
  Expr e;
  e.kind = EK_CONST;
  e.retType = GX_TYPE_I64;
  e.constI64 = v;
  return addExpr(pb, std::move(e));
}
int32_t gx_pb_const_f64(gx_pb* pb, double v) {
  Expr e;
  e.kind = EK_CONST;
  e.retType = GX_TYPE_F64;
  e.constF64 = v;
  return addExpr(pb, std::move(e));
}
int32_t gx_pb_const_time(gx_pb* pb, uint64_t v) {
  Expr e;
  e.kind = EK_CONST;
  e.retType = GX_TYPE_TIME;
  e.constTime = v;
  return addExpr(pb, std::move(e));
}
int32_t gx_pb_const_dec(gx_pb* pb, const uint8_t dec40[40]) {
  Expr e;
  e.kind = EK_CONST;
  e.retType = GX_TYPE_DECIMAL;
  std::memcpy(&e.constDec, dec40, 40);
  e.retFrac = e.constDec.resultFrac;
  return addExpr(pb, std::move(e));
}
int32_t gx_pb_const_str(gx_pb* pb, const char* s, int32_t len) {
  Expr e;
  e.kind = EK_CONST;
  e.retType = GX_TYPE_STRING;
  e.constStr.assign(s, len);
  return addExpr(pb, std::move(e));
}
int32_t gx_pb_call(gx_pb* pb, int32_t func, int32_t ret_type, int32_t ret_frac,
                   const int32_t* args, int32_t n_args) {
  Expr e;
  e.kind = EK_CALL;
  e.func = func;
  e.retType = ret_type;
  e.retFrac = ret_frac < 0 ? 0 : ret_frac;
  for (int i = 0; i < n_args; i++) {
    if (args[i] < 0 || args[i] >= (int32_t)pb->plan.exprs.size()) return GX_ERR_INVALID;
    e.args.push_back(args[i]);
  }
  return addExpr(pb, std::move(e));
}

static int32_t addNode(gx_pb* pb, PlanNode n) {
  pb->plan.nodes.push_back(std::move(n));
  return (int32_t)pb->plan.nodes.size() - 1;
}

int32_t gx_pb_source(gx_pb* pb, const int32_t* col_types, const int32_t* col_fracs,
                     int32_t n_cols) {
  PlanNode n;
  n.kind = PK_SOURCE;
  for (int i = 0; i < n_cols; i++) {
    n.colTypes.push_back(col_types[i]);
    n.colFracs.push_back(col_fracs ? col_fracs[i] : 0);
  }
  return addNode(pb, std::move(n));
}
int32_t gx_pb_selection(gx_pb* pb, int32_t child, const int32_t* conds, int32_t n_conds) {
  PlanNode n;
  n.kind = PK_SELECTION;
  n.child = child;
  for (int i = 0; i < n_conds; i++) n.exprs.push_back(conds[i]);
  return addNode(pb, std::move(n));
}
int32_t gx_pb_projection(gx_pb* pb, int32_t child, const int32_t* exprs, int32_t n_exprs) {
  PlanNode n;
  n.kind = PK_PROJECTION;
  n.child = child;
  for (int i = 0; i < n_exprs; i++) n.exprs.push_back(exprs[i]);
  return addNode(pb, std::move(n));
}
int32_t gx_pb_hashagg(gx_pb* pb, int32_t child, const int32_t* group_exprs,
                      int32_t n_group, const int32_t* agg_funcs,
                      const int32_t* agg_args, const int32_t* agg_fracs,
                      int32_t n_aggs, int32_t mode) {
  PlanNode n;
  n.kind = PK_HASHAGG;
  n.child = child;
  n.aggMode = mode;
  for (int i = 0; i < n_group; i++) n.exprs.push_back(group_exprs[i]);
  for (int i = 0; i < n_aggs; i++) {
    n.aggFuncs.push_back(agg_funcs[i]);
    n.aggArgs.push_back(agg_args[i]);
    n.aggFracs.push_back(agg_fracs ? agg_fracs[i] : 0);
  }
  return addNode(pb, std::move(n));
}
int32_t gx_pb_streamagg(gx_pb* pb, int32_t child, const int32_t* group_exprs,
                        int32_t n_group, const int32_t* agg_funcs,
                        const int32_t* agg_args, const int32_t* agg_fracs,
                        int32_t n_aggs) {
  PlanNode n;
  n.kind = PK_STREAMAGG;
  n.child = child;
  n.aggMode = GX_AGG_MODE_COMPLETE;
  for (int i = 0; i < n_group; i++) n.exprs.push_back(group_exprs[i]);
  for (int i = 0; i < n_aggs; i++) {
    n.aggFuncs.push_back(agg_funcs[i]);
    n.aggArgs.push_back(agg_args[i]);
    n.aggFracs.push_back(agg_fracs ? agg_fracs[i] : 0);
  }
  return addNode(pb, std::move(n));
}
int32_t gx_pb_topn(gx_pb* pb, int32_t child, const int32_t* key_exprs,
                   const uint8_t* key_desc, int32_t n_keys, int64_t limit,
                   int64_t offset) {
  PlanNode n;
  n.kind = limit < 0 ? PK_SORT : PK_TOPN;
  n.child = child;
  for (int i = 0; i < n_keys; i++) {
    n.exprs.push_back(key_exprs[i]);
    n.keyDesc.push_back(key_desc ? key_desc[i] : 0);
  }
  n.limit = limit;
  n.offset = offset;
  return addNode(pb, std::move(n));
}
int32_t gx_pb_mergejoin(gx_pb* pb, int32_t build_child, int32_t probe_child,
                        const int32_t* build_keys, const int32_t* probe_keys,
                        int32_t n_keys, int32_t join_type) {
  PlanNode n;
  n.kind = PK_MERGEJOIN;
  n.child = build_child;
  n.child2 = probe_child;
  for (int i = 0; i < n_keys; i++) {
    n.buildKeys.push_back(build_keys[i]);
    n.probeKeys.push_back(probe_keys[i]);
  }
  n.joinType = join_type;
  return addNode(pb, std::move(n));
}
int32_t gx_pb_hashjoin(gx_pb* pb, int32_t build_child, int32_t probe_child,
                       const int32_t* build_keys, const int32_t* probe_keys,
                       int32_t n_keys, int32_t join_type) {
  PlanNode n;
  n.kind = PK_HASHJOIN;
  n.child = build_child;
  n.child2 = probe_child;
  n.joinType = join_type;
  for (int i = 0; i < n_keys; i++) {
    n.buildKeys.push_back(build_keys[i]);
    n.probeKeys.push_back(probe_keys[i]);
  }
  return addNode(pb, std::move(n));
}

gx_exec* gx_build(gx_pb* pb, int32_t root, int32_t device) {
  (void)device;  // oracle is CPU-only
  if (!pb || root < 0 || root >= (int32_t)pb->plan.nodes.size()) return nullptr;
  auto* ex = new gx_exec();
  ex->plan = pb->plan;
  ex->root = root;
  return ex;
}

int32_t gx_bind_chunks(gx_exec* ex, int32_t source_node, const gx_chunk* chunks,
                       int32_t n_chunks) {
  if (!ex || source_node < 0 || source_node >= (int32_t)ex->plan.nodes.size())
    return GX_ERR_INVALID;
  const PlanNode& node = ex->plan.nodes[source_node];
  if (node.kind != PK_SOURCE) return GX_ERR_INVALID;
  SourceBinding b;
  b.haveChunks = true;
  for (int i = 0; i < n_chunks; i++) {
    Chunk c;
    if (chunks[i].n_cols != (int32_t)node.colTypes.size()) return GX_ERR_INVALID;
    for (int j = 0; j < chunks[i].n_cols; j++)
      c.cols.push_back(Column::fromGx(chunks[i].cols[j], node.colTypes[j], node.colFracs[j]));
    b.chunks.push_back(std::move(c));
  }
  ex->bindings[source_node] = std::move(b);
  return GX_OK;
}

int32_t gx_bind_tpch_sharded(gx_exec* ex, int32_t source_node, int32_t table,
                             int64_t n_rows, uint64_t seed, int64_t row_offset,
                             int64_t total_rows) {
  if (!ex || source_node < 0 || source_node >= (int32_t)ex->plan.nodes.size())
    return GX_ERR_INVALID;
  if (ex->plan.nodes[source_node].kind != PK_SOURCE) return GX_ERR_INVALID;
  SourceBinding b;
  b.tpchTable = table;
  b.tpchRows = n_rows;
  b.tpchSeed = seed;
  b.tpchRowOffset = row_offset;
  // stash total rows in tpchRows semantics: generator needs it for key ranges
  b.tpchTotalRows = total_rows;
  ex->bindings[source_node] = std::move(b);
  return GX_OK;
}

int32_t gx_bind_tpch(gx_exec* ex, int32_t source_node, int32_t table,
                     int64_t n_rows, uint64_t seed, int64_t row_offset) {
  return gx_bind_tpch_sharded(ex, source_node, table, n_rows, seed, row_offset,
                              n_rows);
}

int32_t gx_open(gx_exec* ex) {
  if (!ex) return GX_ERR_INVALID;
  ex->exec = BuildExec(ex->plan, ex->root, &ex->bindings, &ex->err);
  if (!ex->exec) return GX_ERR_INVALID;
  int32_t ec = ex->exec->open();
  if (ec) {
    ex->err = ex->exec->err;
    return ec;
  }
  ex->opened = true;
  return GX_OK;
}

// copy an oracle Chunk into the caller's gx_chunk buffers
static int32_t chunkToGx(const Chunk& in, gx_chunk* out, std::string* err) {
  int n = in.numRows();
  if (in.cols.empty() || n == 0) {  // EOF (0 rows)
    out->n_rows = 0;
    return GX_OK;
  }
  if (out->n_cols != (int32_t)in.cols.size()) {
    *err = "output chunk column count mismatch";
    return GX_ERR_INVALID;
  }
  for (size_t c = 0; c < in.cols.size(); c++) {
    const Column& col = in.cols[c];
    gx_col* g = &out->cols[c];
    int nb = (n + 7) / 8;
    if (col.isVarlen()) {
      if (g->offsets_cap < n + 1 || g->data_cap < (int64_t)col.data.size()) {
        *err = "output buffer too small";
        return GX_ERR_INVALID;
      }
      std::memcpy(g->offsets, col.offsets.data(), (n + 1) * 8);
      std::memcpy(g->data, col.data.data(), col.data.size());
    } else {
      int64_t bytes = (int64_t)n * col.elemSize();
      if (g->data_cap < bytes) {
        *err = "output buffer too small";
        return GX_ERR_INVALID;
      }
      std::memcpy(g->data, col.data.data(), bytes);
    }
    if (g->null_bitmap) {
      std::memset(g->null_bitmap, 0, nb);
      std::memcpy(g->null_bitmap, col.nullBitmap.data(),
                  std::min((size_t)nb, col.nullBitmap.size()));
    }
    g->length = n;
    g->elem_size = col.elemSize();
  }
  out->n_rows = n;
  return GX_OK;
}

int32_t gx_next(gx_exec* ex, gx_chunk* out, int32_t* rows_out) {
  if (!ex || !ex->opened) return GX_ERR_INVALID;
  Chunk c;
  int32_t ec = ex->exec->next(c);
  if (ec) {
    ex->err = ex->exec->err;
    *rows_out = 0;
    return ec;
  }
  ec = chunkToGx(c, out, &ex->err);
  if (ec) {
    *rows_out = 0;
    return ec;
  }
  *rows_out = c.numRows();
  return GX_OK;
}

int32_t gx_close(gx_exec* ex) {
  if (!ex || !ex->exec) return GX_ERR_INVALID;
  return ex->exec->close();
}

void gx_exec_free(gx_exec* ex) { delete ex; }

const char* gx_last_error(gx_exec* ex) {
  if (!ex) return "null exec";
  return ex->err.c_str();
}

}  // extern "C"

// ---- CPU-baseline timing entries (bench.py cpu_baseline leg) ----
// Generates `rows` lineitem rows into host-resident chunks, then times ONE
// Q1 pass of the oracle executor over the bound chunks. The MT variant
// shards rows across threads (PARTIAL mode per shard) and merges through a
// FINAL-mode agg -- the same partial/final split the multi-GPU path uses.
#include <chrono>
#include <thread>

// builds the Q1 plan over `bind` (mirrors tidb_amd/plan.py q1_plan);
// returns the agg node id, plan in *planOut
static int buildQ1Plan(Plan& plan, int aggMode, int* srcOut) {

  auto expr = [&](Expr e) { plan.exprs.push_back(e); return (int)plan.exprs.size() - 1; };
  auto node = [&](PlanNode n) { plan.nodes.push_back(n); return (int)plan.nodes.size() - 1; };
  PlanNode src;
  src.kind = PK_SOURCE;
  TpchSchema(GX_TPCH_LINEITEM, &src.colTypes, &src.colFracs);
  int nsrc = node(src);
  Expr shipdate; shipdate.kind = EK_COLREF; shipdate.colIdx = 7; shipdate.retType = GX_TYPE_TIME;
  Expr cutoff; cutoff.kind = EK_CONST; cutoff.retType = GX_TYPE_TIME;
  cutoff.constTime = TimeFromDate(1998, 9, 1);
  Expr lt; lt.kind = EK_CALL; lt.func = GX_F_LT; lt.retType = GX_TYPE_I64;
  lt.args = {expr(shipdate), expr(cutoff)};
  PlanNode sel; sel.kind = PK_SELECTION; sel.child = nsrc; sel.exprs = {expr(lt)};
  int nsel = node(sel);
  auto colref = [&](int idx, int t, int f) {
    Expr e; e.kind = EK_COLREF; e.colIdx = idx; e.retType = t; e.retFrac = f;
    return expr(e);
  };
  MyDecimal one; one.FromString("1", 1);
  Expr onee; onee.kind = EK_CONST; onee.retType = GX_TYPE_DECIMAL; onee.constDec = one;
  int eOne = expr(onee);
  int eQty = colref(1, GX_TYPE_DECIMAL, 2), ePrice = colref(2, GX_TYPE_DECIMAL, 2);
  int eDisc = colref(3, GX_TYPE_DECIMAL, 2), eTax = colref(4, GX_TYPE_DECIMAL, 2);
  int eRf = colref(5, GX_TYPE_STRING, 0), eLs = colref(6, GX_TYPE_STRING, 0);
  Expr sub; sub.kind = EK_CALL; sub.func = GX_F_MINUS; sub.retType = GX_TYPE_DECIMAL;
  sub.retFrac = 2; sub.args = {eOne, eDisc};
  int eSub = expr(sub);
  Expr mul1; mul1.kind = EK_CALL; mul1.func = GX_F_MUL; mul1.retType = GX_TYPE_DECIMAL;
  mul1.retFrac = 4; mul1.args = {ePrice, eSub};
  int eDp = expr(mul1);
  Expr add; add.kind = EK_CALL; add.func = GX_F_PLUS; add.retType = GX_TYPE_DECIMAL;
  add.retFrac = 2; add.args = {eOne, eTax};
  int eAdd = expr(add);
  Expr mul2; mul2.kind = EK_CALL; mul2.func = GX_F_MUL; mul2.retType = GX_TYPE_DECIMAL;
  mul2.retFrac = 6; mul2.args = {eDp, eAdd};
  int eCh = expr(mul2);
  PlanNode proj; proj.kind = PK_PROJECTION; proj.child = nsel;
  proj.exprs = {eRf, eLs, eQty, ePrice, eDisc, eDp, eCh};
  int nproj = node(proj);
  PlanNode agg; agg.kind = PK_HASHAGG; agg.child = nproj;
  agg.aggMode = aggMode;
  agg.exprs = {colref(0, GX_TYPE_STRING, 0), colref(1, GX_TYPE_STRING, 0)};
  int aQty = colref(2, GX_TYPE_DECIMAL, 2), aPrice = colref(3, GX_TYPE_DECIMAL, 2);
  int aDisc = colref(4, GX_TYPE_DECIMAL, 2), aDp = colref(5, GX_TYPE_DECIMAL, 4);
  int aCh = colref(6, GX_TYPE_DECIMAL, 6);
  agg.aggFuncs = {GX_AGG_SUM, GX_AGG_SUM, GX_AGG_SUM, GX_AGG_SUM,
                  GX_AGG_AVG, GX_AGG_AVG, GX_AGG_AVG, GX_AGG_COUNT};
  agg.aggArgs = {aQty, aPrice, aDp, aCh, aQty, aPrice, aDisc, -1};
  agg.aggFracs = {2, 2, 4, 6, 6, 6, 6, 0};
  int nagg = node(agg);
  *srcOut = nsrc;
  return nagg;
}

// FINAL-mode plan over canonical 17-col partial-state chunks (mirrors
// tidb_amd/plan.py q1_final_plan; MergePartialResult, aggfuncs.go:250-255)
static int buildQ1FinalPlan(Plan& plan, int* srcOut) {
  auto expr = [&](Expr e) { plan.exprs.push_back(e); return (int)plan.exprs.size() - 1; };
  auto node = [&](PlanNode n) { plan.nodes.push_back(n); return (int)plan.nodes.size() - 1; };
  PlanNode src;
  src.kind = PK_SOURCE;
  src.colTypes = {GX_TYPE_STRING, GX_TYPE_STRING,
                  GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64,
                  GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64,
                  GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_I64,
                  GX_TYPE_DECIMAL, GX_TYPE_I64, GX_TYPE_I64};
  src.colFracs = {0, 0, 2, 0, 2, 0, 4, 0, 6, 0, 2, 0, 2, 0, 2, 0, 0};
  int nsrc = node(src);
  auto colref = [&](int idx, int t, int f) {
    Expr e; e.kind = EK_COLREF; e.colIdx = idx; e.retType = t; e.retFrac = f;
    return expr(e);
  };
  PlanNode agg;
  agg.kind = PK_HASHAGG;
  agg.child = nsrc;
  agg.aggMode = GX_AGG_MODE_FINAL;
  agg.exprs = {colref(0, GX_TYPE_STRING, 0), colref(1, GX_TYPE_STRING, 0)};
  agg.aggFuncs = {GX_AGG_SUM, GX_AGG_SUM, GX_AGG_SUM, GX_AGG_SUM,
                  GX_AGG_AVG, GX_AGG_AVG, GX_AGG_AVG, GX_AGG_COUNT};
  agg.aggArgs = {colref(2, GX_TYPE_DECIMAL, 2), colref(4, GX_TYPE_DECIMAL, 2),
                 colref(6, GX_TYPE_DECIMAL, 4), colref(8, GX_TYPE_DECIMAL, 6),
                 colref(10, GX_TYPE_DECIMAL, 2), colref(12, GX_TYPE_DECIMAL, 2),
                 colref(14, GX_TYPE_DECIMAL, 2), -1};
  agg.aggFracs = {2, 2, 4, 6, 6, 6, 6, 0};
  int nagg = node(agg);
  *srcOut = nsrc;
  return nagg;
}

static void genShard(int64_t lo, int64_t hi, int64_t total, uint64_t seed,
                     SourceBinding* bind) {
  bind->haveChunks = true;
  int64_t pos = lo;
  while (pos < hi) {
    int n = (int)std::min<int64_t>(hi - pos, kMaxChunkSize);
    Chunk c;
    TpchGenChunk(GX_TPCH_LINEITEM, pos, n, seed, total, c);
    bind->chunks.push_back(std::move(c));
    pos += n;
  }
}

extern "C" int32_t gx_oracle_bench_q1(int64_t rows, uint64_t seed,
                                      double* gen_ms, double* exec_ms,
                                      int64_t* groups_out) {
  using clk = std::chrono::steady_clock;
  auto t0 = clk::now();
  SourceBinding bind;
  genShard(0, rows, rows, seed, &bind);
  auto t1 = clk::now();
  Plan plan;
  int nsrc = 0;
  int nagg = buildQ1Plan(plan, GX_AGG_MODE_COMPLETE, &nsrc);
  std::map<int, SourceBinding> bindings;
  bindings[nsrc] = std::move(bind);
  std::string err;
  auto execp = BuildExec(plan, nagg, &bindings, &err);
  if (!execp) return GX_ERR_INTERNAL;
  auto t2 = clk::now();
  execp->open();
  int64_t groups = 0;
  for (;;) {
    Chunk out;
    if (execp->next(out) != GX_OK) return GX_ERR_INTERNAL;
    if (out.numRows() == 0) break;
    groups += out.numRows();
  }
  execp->close();
  auto t3 = clk::now();
  *gen_ms = std::chrono::duration<double, std::milli>(t1 - t0).count();
  *exec_ms = std::chrono::duration<double, std::milli>(t3 - t2).count();
  *groups_out = groups;
  return GX_OK;
}

// All-cores leg: PARTIAL-mode Q1 per shard on `threads` host threads, then a
// FINAL-mode merge of the partial chunks (the CPU analog of the 8-GPU
// partial/merge split). exec_ms covers shard execs + merge.
extern "C" int32_t gx_oracle_bench_q1_mt(int64_t rows, uint64_t seed,
                                         int32_t threads, double* gen_ms,
                                         double* exec_ms, int64_t* groups_out) {
  using clk = std::chrono::steady_clock;
  if (threads < 1) threads = 1;
  auto t0 = clk::now();
  std::vector<SourceBinding> binds(threads);
  {
    std::vector<std::thread> ts;
    int64_t per = (rows + threads - 1) / threads;
    for (int t = 0; t < threads; t++) {
      int64_t lo = std::min<int64_t>((int64_t)t * per, rows);
      int64_t hi = std::min<int64_t>(lo + per, rows);
      ts.emplace_back(genShard, lo, hi, rows, seed, &binds[t]);
    }
    for (auto& th : ts) th.join();
  }
  auto t1 = clk::now();
  std::vector<SourceBinding> partials(threads);
  std::vector<int32_t> rcs(threads, GX_OK);
  auto worker = [&](int t) {
    Plan plan;
    int nsrc = 0;
    int nagg = buildQ1Plan(plan, GX_AGG_MODE_PARTIAL, &nsrc);
    std::map<int, SourceBinding> bindings;
    bindings[nsrc] = std::move(binds[t]);
    std::string err;
    auto execp = BuildExec(plan, nagg, &bindings, &err);
    if (!execp) { rcs[t] = GX_ERR_INTERNAL; return; }
    execp->open();
    partials[t].haveChunks = true;
    for (;;) {
      Chunk out;
      if (execp->next(out) != GX_OK) { rcs[t] = GX_ERR_INTERNAL; return; }
      if (out.numRows() == 0) break;
      partials[t].chunks.push_back(std::move(out));
    }
    execp->close();
  };
  auto t2 = clk::now();
  {
    std::vector<std::thread> ts;
    for (int t = 0; t < threads; t++) ts.emplace_back(worker, t);
    for (auto& th : ts) th.join();
  }
  for (int32_t rc : rcs)
    if (rc != GX_OK) return rc;
  // FINAL merge of all partial chunks
  Plan fplan;
  int fsrc = 0;
  int fagg = buildQ1FinalPlan(fplan, &fsrc);
  std::map<int, SourceBinding> fbind;
  fbind[fsrc].haveChunks = true;
  for (auto& p : partials)
    for (auto& c : p.chunks) fbind[fsrc].chunks.push_back(std::move(c));
  std::string err;
  auto fex = BuildExec(fplan, fagg, &fbind, &err);
  if (!fex) return GX_ERR_INTERNAL;
  fex->open();
  int64_t groups = 0;
  for (;;) {
    Chunk out;
    if (fex->next(out) != GX_OK) return GX_ERR_INTERNAL;
    if (out.numRows() == 0) break;
    groups += out.numRows();
  }
  fex->close();
  auto t3 = clk::now();
  *gen_ms = std::chrono::duration<double, std::milli>(t1 - t0).count();
  *exec_ms = std::chrono::duration<double, std::milli>(t3 - t2).count();
  *groups_out = groups;
  return GX_OK;
}
