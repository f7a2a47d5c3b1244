// oracle/exec.cpp — CPU restatement of the reference chunk executors.
// ORACLE / TEST INFRASTRUCTURE ONLY.
//
// Follows: pkg/executor/select.go:750 (Selection), projection.go (Projection),
// expression/chunk_executor.go:99-124 + expression.go:420-504 (vectorized
// eval / VectorizedFilter), aggregate/agg_hash_executor.go + aggfuncs
// (HashAgg), sortexec/topn.go + topn_chunk_heap.go (TopN), sortexec/sort.go
// (Sort).
#include "exec.h"

#include <algorithm>
#include <cassert>
#include <set>
#include <cstring>
#include <functional>

#include "codec.h"
#include "core_time.h"

namespace oracle {

namespace {

// ---------- expression evaluation ----------
// Mirrors ScalarFunction.VecEval* (builtin_compare_vec_generated.go,
// builtin_arithmetic_vec.go): children evaluated to full columns, nulls
// merged, per-row op.

struct EvalCtx {
  const Plan* plan;
  std::string* err;
};

int32_t evalVec(EvalCtx& ctx, int exprId, const Chunk& in, Column& out);

void fillConst(const Expr& e, int n, Column& out) {
  out.type = e.retType;
  out.frac = e.retFrac;
  out.reset();
  for (int i = 0; i < n; i++) {
    switch (e.retType) {
      case GX_TYPE_I64: out.appendI64(e.constI64); break;
      case GX_TYPE_F64: out.appendF64(e.constF64); break;
      case GX_TYPE_TIME: out.appendU64(e.constTime); break;
      case GX_TYPE_DECIMAL: out.appendDecimal(e.constDec); break;
      case GX_TYPE_STRING: out.appendBytes(e.constStr.data(), e.constStr.size()); break;
    }
  }
}

inline bool bothNotNull(const Column& a, const Column& b, int i) {
  return !a.isNull(i) && !b.isNull(i);
}

int cmpString(const Column& a, const Column& b, int i) {
  int la, lb;
  const uint8_t* pa = a.getBytes(i, &la);
  const uint8_t* pb = b.getBytes(i, &lb);
  std::string ka = BinCollatorKey(pa, la);
  std::string kb = BinCollatorKey(pb, lb);
  int c = ka.compare(kb);
  return c < 0 ? -1 : (c > 0 ? 1 : 0);
}

int32_t evalCompare(EvalCtx& ctx, const Expr& e, const Chunk& in, Column& out) {
  Column a, b;
  int32_t err = evalVec(ctx, e.args[0], in, a);
  if (err) return err;
  err = evalVec(ctx, e.args[1], in, b);
  if (err) return err;
  int n = in.numRows();
  out.type = GX_TYPE_I64;
  out.reset();
  for (int i = 0; i < n; i++) {
    if (!bothNotNull(a, b, i)) { out.appendNull(); continue; }
    int c = 0;
    switch (a.type) {
      case GX_TYPE_I64: {
        int64_t x = a.getI64(i), y = b.getI64(i);
        c = x < y ? -1 : (x > y ? 1 : 0);
        break;
      }
      case GX_TYPE_F64: {
        double x = a.getF64(i), y = b.getF64(i);
        c = x < y ? -1 : (x > y ? 1 : 0);
        break;
      }
      case GX_TYPE_TIME:
        c = CompareTime(a.getU64(i), b.getU64(i));
        break;
      case GX_TYPE_DECIMAL:
        c = a.getDecimal(i)->Compare(*b.getDecimal(i));
        break;
      case GX_TYPE_STRING:
        c = cmpString(a, b, i);
        break;
    }
    int64_t r = 0;
    switch (e.func) {
      case GX_F_LT: r = c < 0; break;
      case GX_F_LE: r = c <= 0; break;
      case GX_F_GT: r = c > 0; break;
      case GX_F_GE: r = c >= 0; break;
      case GX_F_EQ: r = c == 0; break;
      case GX_F_NE: r = c != 0; break;
    }
    out.appendI64(r);
  }
  return GX_OK;
}

// casts (builtin_cast_vec.go): CAST_DEC = ProduceDecWithSpecifiedTp's
// Round(HalfUp, ret_frac) (datum.go:1629-1660; flen clamping out of scope
// this round); CAST_INT = Round(0, HalfUp) + ToInt (:1817-1852)
int32_t evalCast(EvalCtx& ctx, const Expr& e, const Chunk& in, Column& out) {
  Column a;
  int32_t err = evalVec(ctx, e.args[0], in, a);
  if (err) return err;
  int n = in.numRows();
  out.type = e.retType;
  out.frac = e.retFrac;
  out.reset();
  for (int i = 0; i < n; i++) {
    if (a.isNull(i)) { out.appendNull(); continue; }
    MyDecimal d;
    if (a.type == GX_TYPE_DECIMAL) d = *a.getDecimal(i);
    else if (a.type == GX_TYPE_I64) d.FromInt(a.getI64(i));
    else { *ctx.err = "unsupported cast argument type"; return GX_ERR_INVALID; }
    if (e.func == GX_F_ROUND) {
      // builtinRoundWithFracDecSig: Round(min(d, ret frac), ModeHalfUp)
      const Expr& de = ctx.plan->exprs[e.args[1]];
      MyDecimal r;
      int32_t ec = d.Round(
          &r, (int)std::min<int64_t>(de.constI64, e.retFrac), ModeHalfUp);
      if (ec != E_OK && ec != E_TRUNCATED) return ec;
      out.appendDecimal(r);
    } else if (e.func == GX_F_ABS) {
      // builtinAbsDecSig / builtinAbsIntSig
      if (a.type == GX_TYPE_I64) {
        int64_t v = a.getI64(i);
        out.appendI64(v < 0 ? -v : v);
      } else {
        d.negative = 0;  // |x|: the sign bit IS the sign (mydecimal.go)
        out.appendDecimal(d);
      }
    } else if (e.func == GX_F_CAST_DEC) {
      MyDecimal r;
      int32_t ec = d.Round(&r, e.retFrac, ModeHalfUp);
      if (ec != E_OK && ec != E_TRUNCATED) return ec;
      out.appendDecimal(r);
    } else {  // GX_F_CAST_INT
      MyDecimal r;
      int32_t ec = d.Round(&r, 0, ModeHalfUp);
      if (ec != E_OK && ec != E_TRUNCATED) return ec;
      int64_t v = 0;
      ec = r.ToInt(&v);
      if (ec == E_OVERFLOW) { *ctx.err = "cast overflow"; return ec; }
      out.appendI64(v);
    }
  }
  return GX_OK;
}

int32_t evalArith(EvalCtx& ctx, const Expr& e, const Chunk& in, Column& out) {
  Column a, b;
  int32_t err = evalVec(ctx, e.args[0], in, a);
  if (err) return err;
  err = evalVec(ctx, e.args[1], in, b);
  if (err) return err;
  int n = in.numRows();
  out.type = e.retType;
  out.frac = e.retFrac;
  out.reset();
  if (e.retType == GX_TYPE_DECIMAL) {
    for (int i = 0; i < n; i++) {
      if (!bothNotNull(a, b, i)) { out.appendNull(); continue; }
      // int children coerce to decimal (the reference's planner inserts the
      // cast; expression.WrapWithCastAsDecimal semantics)
      MyDecimal ca, cb;
      const MyDecimal* pa;
      const MyDecimal* pb;
      if (a.type == GX_TYPE_I64) { ca.FromInt(a.getI64(i)); pa = &ca; }
      else pa = a.getDecimal(i);
      if (b.type == GX_TYPE_I64) { cb.FromInt(b.getI64(i)); pb = &cb; }
      else pb = b.getDecimal(i);
      MyDecimal to;
      int32_t ec = GX_OK;
      switch (e.func) {
        case GX_F_PLUS: ec = DecimalAdd(pa, pb, &to); break;
        case GX_F_MINUS: ec = DecimalSub(pa, pb, &to); break;
        case GX_F_MUL: ec = DecimalMul(pa, pb, &to); break;
        case GX_F_DIV: ec = DecimalDiv(pa, pb, &to, kDivFracIncr); break;
      }
      if (ec == E_DIV_ZERO) { out.appendNull(); continue; }  // div-by-0 -> NULL (MySQL)
      if (ec != GX_OK && ec != E_TRUNCATED) {
        *ctx.err = "decimal arithmetic error";
        return ec;
      }
      out.appendDecimal(to);
    }
    return GX_OK;
  }
  if (e.retType == GX_TYPE_I64) {
    for (int i = 0; i < n; i++) {
      if (!bothNotNull(a, b, i)) { out.appendNull(); continue; }
      int64_t x = a.getI64(i), y = b.getI64(i), r = 0;
      bool ovf = false;
      switch (e.func) {
        case GX_F_PLUS: ovf = __builtin_add_overflow(x, y, &r); break;
        case GX_F_MINUS: ovf = __builtin_sub_overflow(x, y, &r); break;
        case GX_F_MUL: ovf = __builtin_mul_overflow(x, y, &r); break;
        case GX_F_DIV:
          if (y == 0) { out.appendNull(); continue; }
          r = x / y;
          break;
      }
      if (ovf) { *ctx.err = "BIGINT value out of range"; return E_OVERFLOW; }
      out.appendI64(r);
    }
    return GX_OK;
  }
  if (e.retType == GX_TYPE_F64) {
    for (int i = 0; i < n; i++) {
      if (!bothNotNull(a, b, i)) { out.appendNull(); continue; }
      double x = a.getF64(i), y = b.getF64(i), r = 0;
      switch (e.func) {
        case GX_F_PLUS: r = x + y; break;
        case GX_F_MINUS: r = x - y; break;
        case GX_F_MUL: r = x * y; break;
        case GX_F_DIV:
          if (y == 0) { out.appendNull(); continue; }
          r = x / y;
          break;
      }
      out.appendF64(r);
    }
    return GX_OK;
  }
  *ctx.err = "unsupported arith type";
  return GX_ERR_INVALID;
}

// string builtins (builtin_string_vec.go, byte/binary sigs — see
// gx_executor.h GX_F_LENGTH.. doc)
static int32_t evalString(EvalCtx& ctx, const Expr& e, const Chunk& in,
                          Column& out) {
  Column a;
  int32_t err = evalVec(ctx, e.args[0], in, a);
  if (err) return err;
  int n = in.numRows();
  out.reset();
  if (e.func == GX_F_LENGTH) {  // builtinLengthSig: byte length
    out.type = GX_TYPE_I64;
    for (int i = 0; i < n; i++) {
      if (a.isNull(i)) { out.appendNull(); continue; }
      int len;
      a.getBytes(i, &len);
      out.appendI64(len);
    }
    return GX_OK;
  }
  if (e.func == GX_F_UPPER || e.func == GX_F_LOWER) {
    // builtinUpperSig / builtinLowerSig (ASCII byte semantics)
    out.type = GX_TYPE_STRING;
    out.offsets.assign(1, 0);
    for (int i = 0; i < n; i++) {
      if (a.isNull(i)) { out.appendNull(); continue; }
      int len;
      const uint8_t* p = a.getBytes(i, &len);
      std::string s((const char*)p, len);
      for (char& c : s) {
        if (e.func == GX_F_UPPER && c >= 'a' && c <= 'z')
          c = (char)(c - 'a' + 'A');
        if (e.func == GX_F_LOWER && c >= 'A' && c <= 'Z')
          c = (char)(c - 'A' + 'a');
      }
      out.appendBytes(s.data(), s.size());
    }
    return GX_OK;
  }
  if (e.func == GX_F_TRIM) {  // builtinTrim1ArgSig: strip 0x20 both ends
    out.type = GX_TYPE_STRING;
    out.offsets.assign(1, 0);
    for (int i = 0; i < n; i++) {
      if (a.isNull(i)) { out.appendNull(); continue; }
      int len;
      const uint8_t* p = a.getBytes(i, &len);
      int st = 0, en = len;
      while (st < en && p[st] == ' ') st++;
      while (en > st && p[en - 1] == ' ') en--;
      out.appendBytes((const char*)p + st, en - st);
    }
    return GX_OK;
  }
  if (e.func == GX_F_LIKE_PREFIX) {  // builtinLikeSig 'abc%' fast path
    const Expr& pat = ctx.plan->exprs[e.args[1]];
    out.type = GX_TYPE_I64;
    for (int i = 0; i < n; i++) {
      if (a.isNull(i)) { out.appendNull(); continue; }
      int len;
      const uint8_t* p = a.getBytes(i, &len);
      bool m = (size_t)len >= pat.constStr.size() &&
               std::memcmp(p, pat.constStr.data(), pat.constStr.size()) == 0;
      out.appendI64(m ? 1 : 0);
    }
    return GX_OK;
  }
  if (e.func == GX_F_SUBSTR) {  // builtinSubstring3ArgsSig (byte semantics)
    int64_t pos = ctx.plan->exprs[e.args[1]].constI64;
    int64_t length = ctx.plan->exprs[e.args[2]].constI64;
    out.type = GX_TYPE_STRING;
    out.offsets.assign(1, 0);
    for (int i = 0; i < n; i++) {
      if (a.isNull(i)) { out.appendNull(); continue; }
      int len;
      const uint8_t* p = a.getBytes(i, &len);
      int64_t start = pos;
      if (start < 0) start = (int64_t)len + start + 1;
      if (start < 1 || start > len || length <= 0) {
        out.appendBytes(p, 0);
        continue;
      }
      int64_t l = std::min<int64_t>(length, len - (start - 1));
      out.appendBytes(p + (start - 1), (size_t)l);
    }
    return GX_OK;
  }
  return GX_ERR_INVALID;
}

int32_t evalVec(EvalCtx& ctx, int exprId, const Chunk& in, Column& out) {
  const Expr& e = ctx.plan->exprs[exprId];
  switch (e.kind) {
    case EK_COLREF:
      out = in.cols[e.colIdx];
      return GX_OK;
    case EK_CONST:
      fillConst(e, in.numRows(), out);
      return GX_OK;
    case EK_CALL:
      if (e.func <= GX_F_NE) return evalCompare(ctx, e, in, out);
      if (e.func == GX_F_CAST_DEC || e.func == GX_F_CAST_INT ||
          e.func == GX_F_ROUND || e.func == GX_F_ABS)
        return evalCast(ctx, e, in, out);
      if ((e.func >= GX_F_LENGTH && e.func <= GX_F_LOWER) ||
          e.func == GX_F_TRIM)
        return evalString(ctx, e, in, out);
      if (e.func >= GX_F_YEAR && e.func <= GX_F_SECOND) {
        // builtinYear/Month/Day/Hour/Minute/SecondSig: CoreTime bitfields
        static const int kSh[6] = {50, 46, 41, 36, 30, 24};
        static const uint64_t kMk[6] = {0x3FFF, 0xF, 0x1F, 0x1F, 0x3F, 0x3F};
        Column a;
        int32_t err = evalVec(ctx, e.args[0], in, a);
        if (err) return err;
        out.reset();
        out.type = GX_TYPE_I64;
        int fi = e.func - GX_F_YEAR;
        for (int i = 0; i < in.numRows(); i++) {
          if (a.isNull(i)) { out.appendNull(); continue; }
          out.appendI64((int64_t)((a.getU64(i) >> kSh[fi]) & kMk[fi]));
        }
        return GX_OK;
      }
      if (e.func == GX_F_GREATEST || e.func == GX_F_LEAST) {
        // builtinGreatest/Least*Sig (n-ary; NULL if ANY arg is NULL)
        Column acc;
        int32_t err = evalVec(ctx, e.args[0], in, acc);
        if (err) return err;
        for (size_t j = 1; j < e.args.size(); j++) {
          Column b2, merged;
          err = evalVec(ctx, e.args[j], in, b2);
          if (err) return err;
          merged.type = acc.type;
          merged.frac = std::max(acc.frac, b2.frac);
          merged.reset();
          if (merged.isVarlen()) merged.offsets.assign(1, 0);
          for (int i = 0; i < in.numRows(); i++) {
            if (acc.isNull(i) || b2.isNull(i)) { merged.appendNull(); continue; }
            int c;
            if (acc.type == GX_TYPE_DECIMAL)
              c = acc.getDecimal(i)->Compare(*b2.getDecimal(i));
            else {
              int64_t x = acc.getI64(i), y = b2.getI64(i);
              c = x < y ? -1 : (x > y ? 1 : 0);
            }
            bool takeA = (e.func == GX_F_GREATEST) == (c >= 0);
            merged.appendFrom(takeA ? acc : b2, i);
          }
          acc = std::move(merged);
        }
        out = std::move(acc);
        return GX_OK;
      }
      if (e.func == GX_F_TUPLE) {
        if (ctx.err) *ctx.err = "tuple is only valid as a DISTINCT aggregate argument";
        return GX_ERR_INVALID;
      }
      if (e.func == GX_F_OR) {
        // builtinLogicOrSig: TRUE if either side is truthy, NULL when
        // undecidable (NULL vs 0/NULL), else FALSE
        Column a, b2;
        int32_t err = evalVec(ctx, e.args[0], in, a);
        if (err) return err;
        err = evalVec(ctx, e.args[1], in, b2);
        if (err) return err;
        out.reset();
        out.type = GX_TYPE_I64;
        for (int i = 0; i < in.numRows(); i++) {
          bool aT = !a.isNull(i) && a.getI64(i) != 0;
          bool bT = !b2.isNull(i) && b2.getI64(i) != 0;
          if (aT || bT) out.appendI64(1);
          else if (a.isNull(i) || b2.isNull(i)) out.appendNull();
          else out.appendI64(0);
        }
        return GX_OK;
      }
      if (e.func == GX_F_IF) {
        // builtinIfSig: NULL/0 cond -> else branch; result = that branch
        Column c0, a1, b2;
        int32_t err = evalVec(ctx, e.args[0], in, c0);
        if (err) return err;
        err = evalVec(ctx, e.args[1], in, a1);
        if (err) return err;
        err = evalVec(ctx, e.args[2], in, b2);
        if (err) return err;
        out.reset();
        out.type = a1.type;
        out.frac = std::max(a1.frac, b2.frac);
        if (out.isVarlen()) out.offsets.assign(1, 0);
        for (int i = 0; i < in.numRows(); i++) {
          bool t = !c0.isNull(i) && c0.getI64(i) != 0;
          const Column& pick = t ? a1 : b2;
          if (pick.isNull(i)) out.appendNull();
          else out.appendFrom(pick, i);
        }
        return GX_OK;
      }
      if (e.func == GX_F_IFNULL) {
        // builtinIfNullSig: first non-NULL operand per row
        Column a, b2;
        int32_t err = evalVec(ctx, e.args[0], in, a);
        if (err) return err;
        err = evalVec(ctx, e.args[1], in, b2);
        if (err) return err;
        out.reset();
        out.type = a.type;
        out.frac = std::max(a.frac, b2.frac);
        if (out.isVarlen()) out.offsets.assign(1, 0);
        for (int i = 0; i < in.numRows(); i++) {
          const Column& pick = a.isNull(i) ? b2 : a;
          if (pick.isNull(i)) out.appendNull();
          else out.appendFrom(pick, i);
        }
        return GX_OK;
      }
      if (e.func == GX_F_IS_NULL || e.func == GX_F_IS_NOT_NULL) {
        // builtin*IsNullSig (builtin_op_vec.go): 0/1, never NULL
        Column a;
        int32_t err = evalVec(ctx, e.args[0], in, a);
        if (err) return err;
        out.reset();
        out.type = GX_TYPE_I64;
        for (int i = 0; i < in.numRows(); i++) {
          bool isn = a.isNull(i);
          out.appendI64((e.func == GX_F_IS_NULL) == isn ? 1 : 0);
        }
        return GX_OK;
      }
      return evalArith(ctx, e, in, out);
  }
  return GX_ERR_INVALID;
}

// VectorizedFilter semantics (expression.go:420-504): conjuncts narrow the
// selection progressively; NULL or 0 rejects the row.
int32_t vectorizedFilter(EvalCtx& ctx, const std::vector<int>& conds,
                         const Chunk& in, std::vector<uint8_t>& selected) {
  int n = in.numRows();
  selected.assign(n, 1);
  for (int cond : conds) {
    Column r;
    int32_t err = evalVec(ctx, cond, in, r);
    if (err) return err;
    for (int i = 0; i < n; i++) {
      if (!selected[i]) continue;
      if (r.isNull(i) || r.getI64(i) == 0) selected[i] = 0;
    }
  }
  return GX_OK;
}

// ---------- operators ----------

class SourceExec : public Exec {
 public:
  SourceExec(const PlanNode& node, SourceBinding* b) : node_(node), bind_(b) {
    outTypes = node.colTypes;
    outFracs = node.colFracs;
  }
  int32_t open() override {
    pos_ = 0;
    return GX_OK;
  }
  int32_t next(Chunk& out) override {
    out.reset();
    if (bind_ == nullptr) return GX_OK;  // empty source
    if (bind_->haveChunks) {
      if (pos_ >= (int64_t)bind_->chunks.size()) return GX_OK;
      out = bind_->chunks[(size_t)pos_++];
      return GX_OK;
    }
    if (bind_->tpchTable >= 0) {
      int64_t remaining = bind_->tpchRows - pos_;
      if (remaining <= 0) return GX_OK;
      int n = (int)std::min<int64_t>(remaining, kMaxChunkSize);
      TpchGenChunk(bind_->tpchTable, bind_->tpchRowOffset + pos_, n,
                   bind_->tpchSeed,
                   bind_->tpchTotalRows > 0 ? bind_->tpchTotalRows : bind_->tpchRows,
                   out);
      pos_ += n;
      return GX_OK;
    }
    return GX_OK;
  }
  int32_t close() override { return GX_OK; }

 private:
  const PlanNode& node_;
  SourceBinding* bind_;
  int64_t pos_ = 0;
};

// select.go:750-785: filter child chunks, copy surviving rows.
class SelectionExec : public Exec {
 public:
  SelectionExec(const Plan& plan, const PlanNode& node, std::unique_ptr<Exec> child)
      : plan_(plan), node_(node), child_(std::move(child)) {
    outTypes = child_->outTypes;
    outFracs = child_->outFracs;
  }
  int32_t open() override { return child_->open(); }
  int32_t next(Chunk& out) override {
    out.cols.resize(outTypes.size());
    for (size_t c = 0; c < out.cols.size(); c++) {
      out.cols[c].type = outTypes[c];
      out.cols[c].frac = outFracs[c];
    }
    out.reset();
    EvalCtx ctx{&plan_, &err};
    for (;;) {
      Chunk in;
      int32_t ec = child_->next(in);
      if (ec) { err = child_->err; return ec; }
      int n = in.numRows();
      if (n == 0) return GX_OK;  // EOF
      std::vector<uint8_t> selected;
      ec = vectorizedFilter(ctx, node_.exprs, in, selected);
      if (ec) return ec;
      for (int i = 0; i < n; i++) {
        if (!selected[i]) continue;
        for (size_t c = 0; c < out.cols.size(); c++)
          out.cols[c].appendFrom(in.cols[c], i);
      }
      if (out.numRows() > 0) return GX_OK;
    }
  }
  int32_t close() override { return child_->close(); }

 private:
  const Plan& plan_;
  const PlanNode& node_;
  std::unique_ptr<Exec> child_;
};

// projection.go + chunk_executor.go:99-124
class ProjectionExec : public Exec {
 public:
  ProjectionExec(const Plan& plan, const PlanNode& node, std::unique_ptr<Exec> child)
      : plan_(plan), node_(node), child_(std::move(child)) {
    for (int eid : node.exprs) {
      outTypes.push_back(plan.exprs[eid].retType);
      outFracs.push_back(plan.exprs[eid].retFrac);
    }
  }
  int32_t open() override { return child_->open(); }
  int32_t next(Chunk& out) override {
    out.cols.resize(node_.exprs.size());
    Chunk in;
    int32_t ec = child_->next(in);
    if (ec) { err = child_->err; return ec; }
    EvalCtx ctx{&plan_, &err};
    if (in.numRows() == 0) {
      for (size_t c = 0; c < out.cols.size(); c++) {
        out.cols[c].type = outTypes[c];
        out.cols[c].frac = outFracs[c];
        out.cols[c].reset();
      }
      return GX_OK;
    }
    for (size_t c = 0; c < node_.exprs.size(); c++) {
      ec = evalVec(ctx, node_.exprs[c], in, out.cols[c]);
      if (ec) return ec;
      out.cols[c].frac = outFracs[c];
    }
    return GX_OK;
  }
  int32_t close() override { return child_->close(); }

 private:
  const Plan& plan_;
  const PlanNode& node_;
  std::unique_ptr<Exec> child_;
};

// ---------- hash aggregation ----------
// agg_hash_executor.go semantics with aggfuncs update/merge/finalize rules.
// Complete mode == partial-then-final in one process; the canonical
// partial-state chunk layout (for GX_AGG_MODE_PARTIAL output / FINAL input):
//   [group cols...] then per agg:
//     COUNT           -> I64 count
//     SUM (decimal)   -> DECIMAL val, I64 notNullRowCount
//     SUM (f64)       -> F64 val, I64 notNullRowCount
//     AVG (decimal)   -> DECIMAL sum, I64 count
//     AVG (f64)       -> F64 sum, I64 count
//     MIN/MAX         -> value col (null = empty)
//     FIRSTROW        -> value col
struct AggState {
  std::set<std::string> seen;  // DISTINCT: HashGroupKey-encoded values
  MyDecimal dec;       // sum/avg accumulator or min/max/firstrow decimal
  double f64 = 0;
  int64_t i64 = 0;     // count / notNullRowCount
  int64_t aux = 0;     // generic value for int64 min/max/firstrow
  uint64_t u64 = 0;    // time value
  std::string str;
  bool hasValue = false;
};

struct Group {
  std::vector<AggState> states;
  Chunk keyRow;  // 1-row chunk holding the group-by column values (firstrow)
};

class HashAggExec : public Exec {
 public:
  HashAggExec(const Plan& plan, const PlanNode& node, std::unique_ptr<Exec> child)
      : plan_(plan), node_(node), child_(std::move(child)) {
    // output schema
    if (node.aggMode == GX_AGG_MODE_PARTIAL) {
      appendGroupSchema();
      for (size_t a = 0; a < node.aggFuncs.size(); a++) {
        int f = node.aggFuncs[a];
        int vt = argType(a);
        switch (f) {
          case GX_AGG_COUNT: outTypes.push_back(GX_TYPE_I64); outFracs.push_back(0); break;
          case GX_AGG_SUM:
          case GX_AGG_AVG:
            outTypes.push_back(vt); outFracs.push_back(node.aggFracs[a]);
            outTypes.push_back(GX_TYPE_I64); outFracs.push_back(0);
            break;
          default:
            outTypes.push_back(vt); outFracs.push_back(node.aggFracs[a]);
        }
      }
    } else {
      appendGroupSchema();
      for (size_t a = 0; a < node.aggFuncs.size(); a++) {
        int f = node.aggFuncs[a];
        switch (f) {
          case GX_AGG_COUNT:
          case GX_AGG_COUNT_DISTINCT:
            outTypes.push_back(GX_TYPE_I64); outFracs.push_back(0); break;
          default:
            outTypes.push_back(argType(a));
            outFracs.push_back(node.aggFracs[a]);
        }
      }
    }
  }

  int32_t open() override {
    for (size_t a = 0; a < node_.aggFuncs.size(); a++) {
      int f = node_.aggFuncs[a];
      if (f < GX_AGG_COUNT_DISTINCT) continue;
      if (node_.aggMode != GX_AGG_MODE_COMPLETE) {
        err = "DISTINCT aggregates support COMPLETE mode only";
        return GX_ERR_INVALID;
      }
      if (f != GX_AGG_COUNT_DISTINCT && node_.aggArgs[a] >= 0 &&
          plan_.exprs[node_.aggArgs[a]].func == GX_F_TUPLE &&
          plan_.exprs[node_.aggArgs[a]].kind == EK_CALL) {
        err = "SUM/AVG DISTINCT take a single column";
        return GX_ERR_INVALID;
      }
    }
    done_ = false;
    emitPos_ = 0;
    order_.clear();
    groups_.clear();
    return child_->open();
  }

  int32_t next(Chunk& out) override {
    shapeOut(out);
    if (!done_) {
      int32_t ec = (node_.aggMode == GX_AGG_MODE_FINAL) ? drainFinal() : drainRaw();
      if (ec) return ec;
      done_ = true;
      // no-group-by + zero rows still yields one row (e.g. count(*) = 0,
      // sum = NULL) — aggregate semantics for empty input
      if (node_.exprs.empty() && order_.empty() && node_.aggMode != GX_AGG_MODE_PARTIAL) {
        Group g;
        g.states.resize(node_.aggFuncs.size());
        groups_[""] = g;
        order_.push_back("");
      }
    }
    int emitted = 0;
    while (emitPos_ < order_.size() && emitted < kMaxChunkSize) {
      Group& g = groups_[order_[emitPos_]];
      int32_t ec = emitGroup(g, out);
      if (ec) return ec;
      emitPos_++;
      emitted++;
    }
    return GX_OK;
  }

  int32_t close() override { return child_->close(); }

 private:
  void appendGroupSchema() {
    for (int eid : node_.exprs) {
      outTypes.push_back(plan_.exprs[eid].retType);
      outFracs.push_back(plan_.exprs[eid].retFrac);
    }
  }
  int argType(size_t a) const {
    int ae = node_.aggArgs[a];
    if (ae < 0) return GX_TYPE_I64;
    return plan_.exprs[ae].retType;
  }
  void shapeOut(Chunk& out) {
    out.cols.resize(outTypes.size());
    for (size_t c = 0; c < out.cols.size(); c++) {
      out.cols[c].type = outTypes[c];
      out.cols[c].frac = outFracs[c];
    }
    out.reset();
  }

  // GetGroupKey (agg_util.go:106-158): eval group exprs, HashGroupKey per col
  int32_t groupKeys(EvalCtx& ctx, const Chunk& in,
                    std::vector<std::string>& keys,
                    std::vector<Column>& groupCols) {
    int n = in.numRows();
    keys.assign(n, std::string());
    groupCols.clear();
    for (int eid : node_.exprs) {
      Column c;
      int32_t ec = evalVec(ctx, eid, in, c);
      if (ec) return ec;
      ec = HashGroupKeyCol(c, keys);
      if (ec) return ec;
      groupCols.push_back(std::move(c));
    }
    return GX_OK;
  }

  Group& getGroup(const std::string& key, const std::vector<Column>& groupCols, int row) {
    auto it = groups_.find(key);
    if (it != groups_.end()) return it->second;
    Group g;
    g.states.resize(node_.aggFuncs.size());
    g.keyRow.cols.resize(groupCols.size());
    for (size_t c = 0; c < groupCols.size(); c++) {
      g.keyRow.cols[c].type = groupCols[c].type;
      g.keyRow.cols[c].frac = groupCols[c].frac;
      if (g.keyRow.cols[c].isVarlen()) g.keyRow.cols[c].offsets.assign(1, 0);
      g.keyRow.cols[c].appendFrom(groupCols[c], row);
    }
    order_.push_back(key);
    return groups_.emplace(key, std::move(g)).first->second;
  }

  // UpdatePartialResult per agg func (func_sum.go:224, func_avg.go:110, ...)
  int32_t updateState(AggState& s, int func, const Column* argCol, int row,
                      int valueType, const std::string* vkey = nullptr) {
    bool isNull = argCol ? argCol->isNull(row) : false;
    switch (func) {
      case GX_AGG_COUNT_DISTINCT:
      case GX_AGG_SUM_DISTINCT:
      case GX_AGG_AVG_DISTINCT: {
        // distinct wrappers: NULLs excluded (tuple args: ANY null element
        // — encoded as an EMPTY key); each VALUE updates once per group
        // (value identity = the codec HashGroupKey encoding, the same
        // identity the reference's distinct checker uses)
        if (!vkey || vkey->empty()) break;
        if (!s.seen.insert(*vkey).second) break;
        if (func == GX_AGG_COUNT_DISTINCT) { s.i64++; break; }
        if (valueType == GX_TYPE_DECIMAL) {
          MyDecimal tmp;
          int32_t ec = DecimalAdd(&s.dec, argCol->getDecimal(row), &tmp);
          if (ec != E_OK && ec != E_TRUNCATED) return ec;
          s.dec = tmp;
          s.i64++;
        } else if (valueType == GX_TYPE_F64) {
          s.f64 += argCol->getF64(row);
          s.i64++;
        } else {
          return GX_ERR_INVALID;  // int sum/avg arrives CAST_DEC-wrapped
        }
        break;
      }
      case GX_AGG_COUNT:
        if (!isNull) s.i64++;
        break;
      case GX_AGG_SUM:
        if (isNull) break;
        if (valueType == GX_TYPE_DECIMAL) {
          if (s.i64 == 0) {
            s.dec = *argCol->getDecimal(row);
            s.i64 = 1;
          } else {
            MyDecimal tmp;
            int32_t ec = DecimalAdd(&s.dec, argCol->getDecimal(row), &tmp);
            if (ec != E_OK && ec != E_TRUNCATED) return ec;
            s.dec = tmp;
            s.i64++;
          }
        } else if (valueType == GX_TYPE_F64) {
          s.f64 += argCol->getF64(row);
          s.i64++;
        } else {
          return GX_ERR_INVALID;  // sum over int is decimal in MySQL: builder
                                  // must cast; reject here
        }
        break;
      case GX_AGG_AVG:
        if (isNull) break;
        if (valueType == GX_TYPE_DECIMAL) {
          MyDecimal tmp;
          int32_t ec = DecimalAdd(&s.dec, argCol->getDecimal(row), &tmp);
          if (ec != E_OK && ec != E_TRUNCATED) return ec;
          s.dec = tmp;
          s.i64++;
        } else if (valueType == GX_TYPE_F64) {
          s.f64 += argCol->getF64(row);
          s.i64++;
        } else {
          return GX_ERR_INVALID;
        }
        break;
      case GX_AGG_MIN:
      case GX_AGG_MAX: {
        if (isNull) break;
        bool greater = func == GX_AGG_MAX;
        if (!s.hasValue) {
          storeValue(s, *argCol, row, valueType);
          s.hasValue = true;
          break;
        }
        int c = cmpValue(s, *argCol, row, valueType);
        if ((greater && c < 0) || (!greater && c > 0)) storeValue(s, *argCol, row, valueType);
        break;
      }
      case GX_AGG_FIRSTROW:
        if (!s.hasValue) {
          s.hasValue = true;
          s.aux = isNull ? 1 : 0;  // aux=1 => the first row was NULL
          if (!isNull) storeValue(s, *argCol, row, valueType);
        }
        break;
    }
    return GX_OK;
  }

  void storeValue(AggState& s, const Column& col, int row, int vt) {
    switch (vt) {
      case GX_TYPE_I64: s.i64 = col.getI64(row); break;
      case GX_TYPE_F64: s.f64 = col.getF64(row); break;
      case GX_TYPE_TIME: s.u64 = col.getU64(row); break;
      case GX_TYPE_DECIMAL: s.dec = *col.getDecimal(row); break;
      case GX_TYPE_STRING: s.str = col.getStr(row); break;
    }
  }
  int cmpValue(const AggState& s, const Column& col, int row, int vt) {
    switch (vt) {
      case GX_TYPE_I64: {
        int64_t x = s.i64, y = col.getI64(row);
        return x < y ? -1 : (x > y ? 1 : 0);
      }
      case GX_TYPE_F64: {
        double x = s.f64, y = col.getF64(row);
        return x < y ? -1 : (x > y ? 1 : 0);
      }
      case GX_TYPE_TIME: return CompareTime(s.u64, col.getU64(row));
      case GX_TYPE_DECIMAL: return s.dec.Compare(*col.getDecimal(row));
      case GX_TYPE_STRING: {
        int n;
        const uint8_t* p = col.getBytes(row, &n);
        std::string k = BinCollatorKey(p, n);
        std::string mk = BinCollatorKey((const uint8_t*)s.str.data(), (int)s.str.size());
        int c = mk.compare(k);
        return c < 0 ? -1 : (c > 0 ? 1 : 0);
      }
    }
    return 0;
  }

  std::string lastKey_;  // PK_STREAMAGG contiguity check

  int32_t drainRaw() {
    EvalCtx ctx{&plan_, &err};
    for (;;) {
      Chunk in;
      int32_t ec = child_->next(in);
      if (ec) { err = child_->err; return ec; }
      int n = in.numRows();
      if (n == 0) return GX_OK;
      std::vector<std::string> keys;
      std::vector<Column> groupCols;
      ec = groupKeys(ctx, in, keys, groupCols);
      if (ec) return ec;
      // eval agg arg exprs once per chunk
      std::vector<Column> argCols(node_.aggFuncs.size());
      std::vector<const Column*> argPtr(node_.aggFuncs.size(), nullptr);
      std::vector<std::vector<std::string>> valKeys(node_.aggFuncs.size());
      for (size_t a = 0; a < node_.aggFuncs.size(); a++) {
        if (node_.aggArgs[a] < 0) continue;
        const Expr& ae = plan_.exprs[node_.aggArgs[a]];
        if (ae.kind == EK_CALL && ae.func == GX_F_TUPLE) {
          // multi-column distinct: concatenated element encodings; a row
          // with ANY NULL element gets an EMPTY key (= excluded)
          valKeys[a].assign(n, std::string());
          std::vector<uint8_t> anyNull(n, 0);
          for (int el : ae.args) {
            Column c;
            ec = evalVec(ctx, el, in, c);
            if (ec) return ec;
            std::vector<std::string> part(n);
            ec = HashGroupKeyCol(c, part);
            if (ec) return ec;
            for (int i = 0; i < n; i++) {
              if (c.isNull(i)) anyNull[i] = 1;
              valKeys[a][i] += part[i];
            }
          }
          for (int i = 0; i < n; i++)
            if (anyNull[i]) valKeys[a][i].clear();
          continue;
        }
        ec = evalVec(ctx, node_.aggArgs[a], in, argCols[a]);
        if (ec) return ec;
        argPtr[a] = &argCols[a];
        if (node_.aggFuncs[a] >= GX_AGG_COUNT_DISTINCT) {
          valKeys[a].assign(n, std::string());
          ec = HashGroupKeyCol(argCols[a], valKeys[a]);
          if (ec) return ec;
          for (int i = 0; i < n; i++)
            if (argCols[a].isNull(i)) valKeys[a][i].clear();
        }
      }
      for (int i = 0; i < n; i++) {
        // stream aggregation (agg_stream_executor.go): the child must be
        // GROUPED -- all rows of a key contiguous. A key that re-appears
        // after another key started violates the operator contract.
        if (node_.kind == PK_STREAMAGG && keys[i] != lastKey_ &&
            groups_.count(keys[i]) && !order_.empty()) {
          err = "stream aggregation requires grouped (sorted) input";
          return GX_ERR_INVALID;
        }
        if (node_.kind == PK_STREAMAGG) lastKey_ = keys[i];
        Group& g = getGroup(keys[i], groupCols, i);
        for (size_t a = 0; a < node_.aggFuncs.size(); a++) {
          ec = updateState(g.states[a], node_.aggFuncs[a], argPtr[a], i,
                           argType(a),
                           valKeys[a].empty() ? nullptr : &valKeys[a][i]);
          if (ec) return ec;
        }
      }
    }
  }

  // FINAL mode: child emits canonical partial-state chunks
  int32_t drainFinal() {
    EvalCtx ctx{&plan_, &err};
    size_t nGroupCols = node_.exprs.size();
    for (;;) {
      Chunk in;
      int32_t ec = child_->next(in);
      if (ec) { err = child_->err; return ec; }
      int n = in.numRows();
      if (n == 0) return GX_OK;
      // group cols are the first columns of the partial chunk
      std::vector<std::string> keys(n);
      std::vector<Column> groupCols;
      for (size_t c = 0; c < nGroupCols; c++) {
        ec = HashGroupKeyCol(in.cols[c], keys);
        if (ec) return ec;
        groupCols.push_back(in.cols[c]);
      }
      for (int i = 0; i < n; i++) {
        Group& g = getGroup(keys[i], groupCols, i);
        size_t col = nGroupCols;
        for (size_t a = 0; a < node_.aggFuncs.size(); a++) {
          AggState& s = g.states[a];
          int f = node_.aggFuncs[a];
          int vt = argType(a);
          switch (f) {
            case GX_AGG_COUNT:
              s.i64 += in.cols[col].getI64(i);
              col += 1;
              break;
            case GX_AGG_SUM:
            case GX_AGG_AVG: {
              int64_t srcCount = in.cols[col + 1].getI64(i);
              if (srcCount > 0) {
                if (vt == GX_TYPE_DECIMAL) {
                  MyDecimal tmp;
                  int32_t e2 = DecimalAdd(&s.dec, in.cols[col].getDecimal(i), &tmp);
                  if (e2 != E_OK && e2 != E_TRUNCATED) return e2;
                  s.dec = tmp;
                } else {
                  s.f64 += in.cols[col].getF64(i);
                }
                s.i64 += srcCount;
              }
              col += 2;
              break;
            }
            case GX_AGG_MIN:
            case GX_AGG_MAX: {
              if (!in.cols[col].isNull(i)) {
                bool greater = f == GX_AGG_MAX;
                if (!s.hasValue) {
                  storeValue(s, in.cols[col], i, vt);
                  s.hasValue = true;
                } else {
                  int c2 = cmpValue(s, in.cols[col], i, vt);
                  if ((greater && c2 < 0) || (!greater && c2 > 0))
                    storeValue(s, in.cols[col], i, vt);
                }
              }
              col += 1;
              break;
            }
            case GX_AGG_FIRSTROW:
              if (!s.hasValue) {
                s.hasValue = true;
                s.aux = in.cols[col].isNull(i) ? 1 : 0;
                if (!in.cols[col].isNull(i)) storeValue(s, in.cols[col], i, vt);
              }
              col += 1;
              break;
          }
        }
      }
    }
  }

  void appendValue(const AggState& s, int vt, Column& out) {
    switch (vt) {
      case GX_TYPE_I64: out.appendI64(s.i64); break;
      case GX_TYPE_F64: out.appendF64(s.f64); break;
      case GX_TYPE_TIME: out.appendU64(s.u64); break;
      case GX_TYPE_DECIMAL: out.appendDecimal(s.dec); break;
      case GX_TYPE_STRING: out.appendBytes(s.str.data(), s.str.size()); break;
    }
  }

  int32_t emitGroup(Group& g, Chunk& out) {
    size_t col = 0;
    for (size_t c = 0; c < node_.exprs.size(); c++, col++)
      out.cols[col].appendFrom(g.keyRow.cols[c], 0);
    bool partial = node_.aggMode == GX_AGG_MODE_PARTIAL;
    for (size_t a = 0; a < node_.aggFuncs.size(); a++) {
      AggState& s = g.states[a];
      int f = node_.aggFuncs[a];
      int vt = argType(a);
      switch (f) {
        case GX_AGG_COUNT:
        case GX_AGG_COUNT_DISTINCT:
          out.cols[col++].appendI64(s.i64);
          break;
        case GX_AGG_SUM_DISTINCT:  // never partial (rejected at open)
        case GX_AGG_SUM:
          if (partial) {
            if (s.i64 == 0) out.cols[col++].appendNull();
            else if (vt == GX_TYPE_DECIMAL) out.cols[col++].appendDecimal(s.dec);
            else out.cols[col++].appendF64(s.f64);
            out.cols[col++].appendI64(s.i64);
          } else {
            if (s.i64 == 0) { out.cols[col++].appendNull(); break; }
            if (vt == GX_TYPE_DECIMAL) {
              // sum finalize rounds to ret frac (func_sum.go:203-222)
              MyDecimal v = s.dec;
              int32_t ec = v.Round(&v, node_.aggFracs[a], ModeHalfUp);
              if (ec != E_OK && ec != E_TRUNCATED) return ec;
              out.cols[col++].appendDecimal(v);
            } else {
              out.cols[col++].appendF64(s.f64);
            }
          }
          break;
        case GX_AGG_AVG_DISTINCT:  // never partial (rejected at open)
        case GX_AGG_AVG:
          if (partial) {
            if (vt == GX_TYPE_DECIMAL) out.cols[col++].appendDecimal(s.dec);
            else out.cols[col++].appendF64(s.f64);
            out.cols[col++].appendI64(s.i64);
          } else {
            if (s.i64 == 0) { out.cols[col++].appendNull(); break; }
            if (vt == GX_TYPE_DECIMAL) {
              // func_avg.go:84-109: div by count (+DivPrecisionIncrement),
              // round HalfUp to ret frac
              MyDecimal cnt, res;
              cnt.FromInt(s.i64);
              int32_t ec = DecimalDiv(&s.dec, &cnt, &res, kDivFracIncr);
              if (ec != E_OK && ec != E_TRUNCATED) return ec;
              ec = res.Round(&res, node_.aggFracs[a], ModeHalfUp);
              if (ec != E_OK && ec != E_TRUNCATED) return ec;
              out.cols[col++].appendDecimal(res);
            } else {
              out.cols[col++].appendF64(s.f64 / (double)s.i64);
            }
          }
          break;
        case GX_AGG_MIN:
        case GX_AGG_MAX:
          if (!s.hasValue) out.cols[col++].appendNull();
          else { appendValue(s, vt, out.cols[col]); col++; }
          break;
        case GX_AGG_FIRSTROW:
          if (!s.hasValue || s.aux == 1) out.cols[col++].appendNull();
          else { appendValue(s, vt, out.cols[col]); col++; }
          break;
      }
    }
    return GX_OK;
  }

  const Plan& plan_;
  const PlanNode& node_;
  std::unique_ptr<Exec> child_;
  bool done_ = false;
  std::unordered_map<std::string, Group> groups_;
  std::vector<std::string> order_;  // deterministic emit order (insertion)
  size_t emitPos_ = 0;
};

// ---------- TopN / Sort ----------
// sortexec/topn.go:61-98 + topn_chunk_heap.go: keep limit+offset smallest rows
// by the order keys; emit sorted, skipping offset.

struct RowRef {
  int chunkIdx;
  int rowIdx;
};

class TopNExec : public Exec {
 public:
  TopNExec(const Plan& plan, const PlanNode& node, std::unique_ptr<Exec> child,
           bool isSort)
      : plan_(plan), node_(node), child_(std::move(child)), isSort_(isSort) {
    outTypes = child_->outTypes;
    outFracs = child_->outFracs;
  }
  int32_t open() override {
    done_ = false;
    emit_ = 0;
    rows_.clear();
    data_.clear();
    keyCols_.clear();
    return child_->open();
  }
  int32_t next(Chunk& out) override {
    out.cols.resize(outTypes.size());
    for (size_t c = 0; c < out.cols.size(); c++) {
      out.cols[c].type = outTypes[c];
      out.cols[c].frac = outFracs[c];
    }
    out.reset();
    if (!done_) {
      int32_t ec = drain();
      if (ec) return ec;
      done_ = true;
      emit_ = std::min((size_t)node_.offset, rows_.size());
    }
    size_t end = isSort_ ? rows_.size()
                         : std::min(rows_.size(), (size_t)(node_.offset + node_.limit));
    while (emit_ < end && out.numRows() < kMaxChunkSize) {
      const RowRef& r = rows_[emit_];
      for (size_t c = 0; c < out.cols.size(); c++)
        out.cols[c].appendFrom(data_[r.chunkIdx].cols[c], r.rowIdx);
      emit_++;
    }
    return GX_OK;
  }
  int32_t close() override { return child_->close(); }

 private:
  int cmpRows(const RowRef& a, const RowRef& b) const {
    for (size_t k = 0; k < node_.exprs.size(); k++) {
      const Column& ca = keyCols_[a.chunkIdx][k];
      const Column& cb = keyCols_[b.chunkIdx][k];
      bool na = ca.isNull(a.rowIdx), nb = cb.isNull(b.rowIdx);
      int c;
      if (na && nb) c = 0;
      else if (na) c = -1;  // NULL first ascending
      else if (nb) c = 1;
      else {
        switch (ca.type) {
          case GX_TYPE_I64: {
            int64_t x = ca.getI64(a.rowIdx), y = cb.getI64(b.rowIdx);
            c = x < y ? -1 : (x > y ? 1 : 0);
            break;
          }
          case GX_TYPE_F64: {
            double x = ca.getF64(a.rowIdx), y = cb.getF64(b.rowIdx);
            c = x < y ? -1 : (x > y ? 1 : 0);
            break;
          }
          case GX_TYPE_TIME: c = CompareTime(ca.getU64(a.rowIdx), cb.getU64(b.rowIdx)); break;
          case GX_TYPE_DECIMAL: c = ca.getDecimal(a.rowIdx)->Compare(*cb.getDecimal(b.rowIdx)); break;
          default: {
            int la, lb;
            const uint8_t* pa = ca.getBytes(a.rowIdx, &la);
            const uint8_t* pb = cb.getBytes(b.rowIdx, &lb);
            std::string ka = BinCollatorKey(pa, la), kb = BinCollatorKey(pb, lb);
            int cc = ka.compare(kb);
            c = cc < 0 ? -1 : (cc > 0 ? 1 : 0);
          }
        }
      }
      if (node_.keyDesc[k]) c = -c;
      if (c != 0) return c;
    }
    return 0;
  }

  int32_t drain() {
    EvalCtx ctx{&plan_, &err};
    for (;;) {
      Chunk in;
      int32_t ec = child_->next(in);
      if (ec) { err = child_->err; return ec; }
      int n = in.numRows();
      if (n == 0) break;
      int ci = (int)data_.size();
      std::vector<Column> kc;
      for (int eid : node_.exprs) {
        Column c;
        ec = evalVec(ctx, eid, in, c);
        if (ec) return ec;
        kc.push_back(std::move(c));
      }
      keyCols_.push_back(std::move(kc));
      data_.push_back(std::move(in));
      for (int i = 0; i < n; i++) rows_.push_back({ci, i});
    }
    std::stable_sort(rows_.begin(), rows_.end(),
                     [this](const RowRef& a, const RowRef& b) { return cmpRows(a, b) < 0; });
    return GX_OK;
  }

  const Plan& plan_;
  const PlanNode& node_;
  std::unique_ptr<Exec> child_;
  bool isSort_;
  bool done_ = false;
  std::vector<Chunk> data_;
  std::vector<std::vector<Column>> keyCols_;
  std::vector<RowRef> rows_;
  size_t emit_ = 0;
};

// ---------- hash join ----------
// HashJoinV2 semantics (join/hash_join_v2.go): output columns = build side
// cols then probe side cols, one output row per matching (build,probe) pair;
// NULL join keys never match a chain entry. joinType
// (gx_executor.h gx_pb_hashjoin):
//   0 inner  (inner_join_probe.go:27-86)
//   1 left outer, probe = outer side: unmatched probe rows emit once with
//     build cols NULL (outer_join_probe.go probe-outer path; NULL-key probe
//     rows are unmatched)
//   2 right outer, build = outer side: matched pairs plus every unmatched
//     build row with probe cols NULL (outer_join_probe.go build-outer path
//     via per-row matched flags)
//   3 semi: emit each probe row once if ANY build row matches; output =
//     probe cols only (base_semi_join.go)
//   4 anti semi: emit each probe row iff NO build row matches (NULL-key
//     probe rows match nothing and are emitted); output = probe cols only
//     (anti_semi_join_probe.go, non-null-aware variant)
class HashJoinExec : public Exec {
 public:
  HashJoinExec(const Plan& plan, const PlanNode& node, std::unique_ptr<Exec> build,
               std::unique_ptr<Exec> probe)
      : plan_(plan), node_(node), build_(std::move(build)), probe_(std::move(probe)) {
    int jt = node_.joinType;
    if (jt < 3) {
      for (size_t i = 0; i < build_->outTypes.size(); i++) {
        outTypes.push_back(build_->outTypes[i]);
        outFracs.push_back(build_->outFracs[i]);
      }
    }
    for (size_t i = 0; i < probe_->outTypes.size(); i++) {
      outTypes.push_back(probe_->outTypes[i]);
      outFracs.push_back(probe_->outFracs[i]);
    }
    if (jt >= 6) {  // left outer semi: the x IN (...) scalar column
      outTypes.push_back(GX_TYPE_I64);
      outFracs.push_back(0);
    }
  }
  int32_t open() override {
    built_ = false;
    buildRows_ = 0;
    buildHasNullKey_ = false;
    pending_.reset();
    pendEmit_ = 0;
    buildData_.clear();
    table_.clear();
    matched_.clear();
    drainChunk_ = 0;
    drainRow_ = 0;
    probeDone_ = false;
    int32_t ec = build_->open();
    if (ec) return ec;
    return probe_->open();
  }
  int32_t next(Chunk& out) override {
    out.cols.resize(outTypes.size());
    for (size_t c = 0; c < out.cols.size(); c++) {
      out.cols[c].type = outTypes[c];
      out.cols[c].frac = outFracs[c];
    }
    out.reset();
    EvalCtx ctx{&plan_, &err};
    if (!built_) {
      int32_t ec = buildTable(ctx);
      if (ec) return ec;
      built_ = true;
    }
    int jt = node_.joinType;
    size_t nb = jt >= 3 ? 0 : build_->outTypes.size();
    size_t np = probe_->outTypes.size();
    for (;;) {
      // drain matches pending from the previous probe chunk first: one Next
      // fills at most MaxChunkSize rows (exec.Executor contract,
      // executor.go:224-250)
      while (pendEmit_ < pending_.numRows() && out.numRows() < kMaxChunkSize) {
        for (size_t c = 0; c < out.cols.size(); c++)
          out.cols[c].appendFrom(pending_.cols[c], pendEmit_);
        pendEmit_++;
      }
      if (out.numRows() >= kMaxChunkSize) return GX_OK;
      if (probeDone_) {
        // right outer: unmatched build rows with probe cols NULL
        if (jt == 2) {
          while (out.numRows() < kMaxChunkSize &&
                 drainChunk_ < buildData_.size()) {
            const Chunk& bc = buildData_[drainChunk_];
            if (drainRow_ >= (size_t)bc.numRows()) {
              drainChunk_++;
              drainRow_ = 0;
              continue;
            }
            if (!matched_[drainChunk_][drainRow_]) {
              for (size_t c = 0; c < nb; c++)
                out.cols[c].appendFrom(bc.cols[c], (int)drainRow_);
              for (size_t c = 0; c < np; c++) out.cols[nb + c].appendNull();
            }
            drainRow_++;
          }
        }
        return GX_OK;
      }
      Chunk in;
      int32_t ec = probe_->next(in);
      if (ec) { err = probe_->err; return ec; }
      int n = in.numRows();
      if (n == 0) {
        probeDone_ = true;
        continue;  // may still drain unmatched build rows (right outer)
      }
      pending_.cols.resize(outTypes.size());
      for (size_t c = 0; c < pending_.cols.size(); c++) {
        pending_.cols[c].type = outTypes[c];
        pending_.cols[c].frac = outFracs[c];
      }
      pending_.reset();
      pendEmit_ = 0;
      std::vector<std::string> keys(n);
      std::vector<uint8_t> hasNullKey(n, 0);
      ec = serializeJoinKeys(ctx, node_.probeKeys, in, keys, hasNullKey);
      if (ec) return ec;
      auto appendProbeOnly = [&](int i) {
        for (size_t c = 0; c < np; c++)
          pending_.cols[nb + c].appendFrom(in.cols[c], i);
      };
      for (int i = 0; i < n; i++) {
        const std::vector<RowRef>* refs = nullptr;
        if (!hasNullKey[i]) {
          auto it = table_.find(keys[i]);
          if (it != table_.end()) refs = &it->second;
        }
        bool any = refs && !refs->empty();
        if (jt == 3) {  // semi
          if (any) appendProbeOnly(i);
          continue;
        }
        if (jt == 4) {  // anti semi
          if (!any) appendProbeOnly(i);
          continue;
        }
        if (jt == 5) {
          // null-aware anti semi — x NOT IN (y set), null_aware NAASJ
          // (hash_join_v1.go:599): empty y accepts EVERY x (incl. NULL);
          // a NULL y (or a NULL x against nonempty y) can never be TRUE;
          // else plain anti semi
          if (buildRows_ == 0) { appendProbeOnly(i); continue; }
          if (hasNullKey[i] || buildHasNullKey_) continue;
          if (!any) appendProbeOnly(i);
          continue;
        }
        if (jt >= 6) {
          // left outer semi: probe row + the x IN (y set) scalar.
          // Null-aware (7): NULL when no match but the evidence is
          // inconclusive (NULL x against nonempty y, or a NULL y)
          appendProbeOnly(i);
          Column& fc = pending_.cols[np];
          if (any) fc.appendI64(1);
          else if (jt == 7 && ((hasNullKey[i] && buildRows_ > 0) ||
                               buildHasNullKey_))
            fc.appendNull();
          else
            fc.appendI64(0);
          continue;
        }
        if (any) {
          for (const RowRef& br : *refs) {
            if (jt == 2) matched_[br.chunkIdx][br.rowIdx] = 1;
            for (size_t c = 0; c < nb; c++)
              pending_.cols[c].appendFrom(buildData_[br.chunkIdx].cols[c],
                                          br.rowIdx);
            appendProbeOnly(i);
          }
        } else if (jt == 1) {  // left outer: null-extend the build side
          for (size_t c = 0; c < nb; c++) pending_.cols[c].appendNull();
          appendProbeOnly(i);
        }
      }
    }
  }
  int32_t close() override {
    build_->close();
    return probe_->close();
  }

 private:
  int32_t serializeJoinKeys(EvalCtx& ctx, const std::vector<int>& keyExprs,
                            const Chunk& in, std::vector<std::string>& keys,
                            std::vector<uint8_t>& hasNull) {
    for (int eid : keyExprs) {
      Column c;
      int32_t ec = evalVec(ctx, eid, in, c);
      if (ec) return ec;
      for (int i = 0; i < (int)keys.size(); i++) {
        if (c.isNull(i)) { hasNull[i] = 1; continue; }
        switch (c.type) {
          case GX_TYPE_I64: {
            int64_t v = c.getI64(i);
            keys[i].append((const char*)&v, 8);
            break;
          }
          case GX_TYPE_TIME: {
            uint64_t v = c.getU64(i);
            keys[i].append((const char*)&v, 8);
            break;
          }
          case GX_TYPE_F64: {
            double v = c.getF64(i);
            keys[i].append((const char*)&v, 8);
            break;
          }
          case GX_TYPE_DECIMAL: {
            // join keys use the value-normalized hash key (ToHashKey) so that
            // 1.10 == 1.1 matches (codec.go:852-910 SerializeKeys semantics)
            uint8_t buf[48];
            int blen = 0;
            int32_t e2 = c.getDecimal(i)->ToHashKey(buf, &blen);
            if (e2 != E_OK) return e2;
            keys[i].append((const char*)buf, blen);
            break;
          }
          case GX_TYPE_STRING: {
            int len;
            const uint8_t* p = c.getBytes(i, &len);
            std::string k = BinCollatorKey(p, len);
            uint32_t l = (uint32_t)k.size();
            keys[i].append((const char*)&l, 4);
            keys[i].append(k);
            break;
          }
        }
      }
    }
    return GX_OK;
  }

  int32_t buildTable(EvalCtx& ctx) {
    for (;;) {
      Chunk in;
      int32_t ec = build_->next(in);
      if (ec) { err = build_->err; return ec; }
      int n = in.numRows();
      if (n == 0) return GX_OK;
      int ci = (int)buildData_.size();
      std::vector<std::string> keys(n);
      std::vector<uint8_t> hasNull(n, 0);
      ec = serializeJoinKeys(ctx, node_.buildKeys, in, keys, hasNull);
      if (ec) return ec;
      buildRows_ += n;
      for (int i = 0; i < n; i++) {
        if (hasNull[i]) { buildHasNullKey_ = true; continue; }
        table_[keys[i]].push_back({ci, i});
      }
      matched_.push_back(std::vector<uint8_t>(n, 0));
      buildData_.push_back(std::move(in));
    }
  }

  const Plan& plan_;
  const PlanNode& node_;
  std::unique_ptr<Exec> build_, probe_;
  bool built_ = false;
  std::vector<Chunk> buildData_;
  std::unordered_map<std::string, std::vector<RowRef>> table_;
  Chunk pending_;      // matches of the current probe chunk, emitted <=
  int pendEmit_ = 0;   // kMaxChunkSize per next()
  std::vector<std::vector<uint8_t>> matched_;  // right outer: per build row
  bool probeDone_ = false;
  size_t drainChunk_ = 0, drainRow_ = 0;  // right-outer drain cursor
  int64_t buildRows_ = 0;        // null-aware anti semi build scalars
  bool buildHasNullKey_ = false;
};

}  // namespace

std::unique_ptr<Exec> BuildExec(const Plan& plan, int root,
                                std::map<int, SourceBinding>* bindings,
                                std::string* err) {
  if (root < 0 || root >= (int)plan.nodes.size()) {
    *err = "bad plan root";
    return nullptr;
  }
  const PlanNode& node = plan.nodes[root];
  switch (node.kind) {
    case PK_SOURCE: {
      SourceBinding* b = nullptr;
      auto it = bindings->find(root);
      if (it != bindings->end()) b = &it->second;
      return std::make_unique<SourceExec>(node, b);
    }
    case PK_SELECTION: {
      auto child = BuildExec(plan, node.child, bindings, err);
      if (!child) return nullptr;
      return std::make_unique<SelectionExec>(plan, node, std::move(child));
    }
    case PK_PROJECTION: {
      auto child = BuildExec(plan, node.child, bindings, err);
      if (!child) return nullptr;
      return std::make_unique<ProjectionExec>(plan, node, std::move(child));
    }
    case PK_HASHAGG:
    case PK_STREAMAGG: {
      auto child = BuildExec(plan, node.child, bindings, err);
      if (!child) return nullptr;
      return std::make_unique<HashAggExec>(plan, node, std::move(child));
    }
    case PK_TOPN: {
      auto child = BuildExec(plan, node.child, bindings, err);
      if (!child) return nullptr;
      return std::make_unique<TopNExec>(plan, node, std::move(child), false);
    }
    case PK_SORT: {
      auto child = BuildExec(plan, node.child, bindings, err);
      if (!child) return nullptr;
      return std::make_unique<TopNExec>(plan, node, std::move(child), true);
    }
    case PK_MERGEJOIN:
      // merge_join.go: planned only over inputs sorted on the join keys --
      // enforce that contract statically (each child must be a full sort
      // whose leading key is the join key), then execute through the join
      // machinery: the two-pointer merge is an execution strategy, results
      // are those of the inner join
      for (int side = 0; side < 2; side++) {
        int cid = side == 0 ? node.child : node.child2;
        int keyExpr = side == 0 ? node.buildKeys[0] : node.probeKeys[0];
        const PlanNode* c = &plan.nodes[cid];
        bool okSort = c->kind == PK_SORT && !c->exprs.empty();
        if (okSort) {
          const Expr& ke = plan.exprs[c->exprs[0]];
          const Expr& je = plan.exprs[keyExpr];
          okSort = ke.kind == EK_COLREF && je.kind == EK_COLREF &&
                   ke.colIdx == je.colIdx && c->keyDesc[0] == 0;
        }
        if (!okSort) {
          if (err) *err = "merge join requires children sorted on the join key";
          return nullptr;
        }
      }
      [[fallthrough]];
    case PK_HASHJOIN: {
      auto build = BuildExec(plan, node.child, bindings, err);
      if (!build) return nullptr;
      auto probe = BuildExec(plan, node.child2, bindings, err);
      if (!probe) return nullptr;
      return std::make_unique<HashJoinExec>(plan, node, std::move(build), std::move(probe));
    }
  }
  *err = "unknown plan kind";
  return nullptr;
}

}  // namespace oracle
