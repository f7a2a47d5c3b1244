/* gx_executor.h — C-ABI drop-in boundary for the TiDB analytical hot path.
 *
 * This header declares the operator surface a cgo shim in TiDB would bind in
 * place of the CPU executors built by `executorBuilder.build`
 * (reference: pkg/executor/builder.go:283-311 dispatch on
 * PhysicalSelection/PhysicalProjection/PhysicalHashAgg/PhysicalTopN/...).
 *
 * The operator contract mirrors `exec.Executor`
 * (reference: pkg/executor/internal/exec/executor.go:224-250):
 *   Open() -> repeated Next(chunk) filling <= 1024 rows, 0 rows = EOF -> Close().
 * gx_next is single-threaded per executor handle; the engine may use internal
 * streams/threads (reference threading contract, executor.go:238).
 *
 * Chunk layout matches pkg/util/chunk exactly
 * (reference: pkg/util/chunk/column.go:74-82, chunk.go:35-54):
 *   - fixed-width columns: data = length*elem_size raw little-endian bytes
 *     (8 B int64/uint64/float64/Time, 40 B MyDecimal);
 *   - null bitmap: 1 bit per row, LSB-first within each byte, 1 = NOT NULL
 *     (column.go:225-267);
 *   - var-len columns: row i spans data[offsets[i] .. offsets[i+1]).
 *
 * Two shared libraries implement this ABI:
 *   - liboracle.so   (oracle/…):  CPU restatement of the reference executor —
 *                                 TEST INFRASTRUCTURE ONLY (parity anchor).
 *   - libgxexec.so   (tidb_amd/csrc/…): the MI355X product engine (HIP/CDNA4).
 */
#ifndef GX_EXECUTOR_H
#define GX_EXECUTOR_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- status codes (mirror reference error classes) ---- */
enum {
  GX_OK = 0,
  GX_ERR_TRUNCATED = 1, /* types.ErrTruncated  */
  GX_ERR_OVERFLOW = 2,  /* types.ErrOverflow   */
  GX_ERR_DIV_ZERO = 3,  /* types.ErrDivByZero  */
  GX_ERR_BAD_NUMBER = 4,/* types.ErrBadNumber  */
  GX_ERR_INVALID = -1,  /* bad plan / usage    */
  GX_ERR_NO_GPU = -2,   /* product engine built without / cannot reach a GPU */
  GX_ERR_INTERNAL = -3
};

/* ---- column element types (subset of mysql types on the hot path) ---- */
enum {
  GX_TYPE_I64 = 0,     /* TypeLonglong, 8 B */
  GX_TYPE_F64 = 1,     /* TypeDouble, 8 B */
  GX_TYPE_DECIMAL = 2, /* TypeNewDecimal, 40 B MyDecimal struct (mydecimal.go:236-248) */
  GX_TYPE_TIME = 3,    /* TypeDate/Datetime, 8 B packed CoreTime (core_time.go:25, time.go:233-265) */
  GX_TYPE_STRING = 4   /* var-len (char/varchar), offsets + bytes */
};

/* ---- chunk ---- */
typedef struct gx_col {
  void*    data;        /* fixed: elem data; varlen: byte payload */
  uint8_t* null_bitmap; /* ceil(length/8) bytes; LSB-first; 1 = NOT NULL; may be NULL => all not-null */
  int64_t* offsets;     /* varlen only: length+1 entries, offsets[0] = 0; NULL for fixed */
  int32_t  length;      /* number of rows */
  int32_t  elem_size;   /* fixed elem bytes; -1 for varlen */
  int64_t  data_cap;    /* capacity of data in bytes (output chunks: caller-provided) */
  int64_t  offsets_cap; /* capacity of offsets in entries (output chunks) */
} gx_col;

typedef struct gx_chunk {
  gx_col* cols;
  int32_t n_cols;
  int32_t n_rows;
} gx_chunk;

/* ---- scalar function codes (vectorized builtins on the hot path) ----
 * compare family: reference pkg/expression/builtin_compare_vec_generated.go:64,138-170
 * decimal arithmetic: builtin_arithmetic_vec.go:529 (Mul), :328 (Minus), :973 (Plus), :67 (Div)
 */
enum {
  GX_F_LT = 0, GX_F_LE = 1, GX_F_GT = 2, GX_F_GE = 3, GX_F_EQ = 4, GX_F_NE = 5,
  GX_F_PLUS = 16, GX_F_MINUS = 17, GX_F_MUL = 18, GX_F_DIV = 19,
  /* casts (builtin_cast_vec.go; types.ProduceDecWithSpecifiedTp
   * datum.go:1629): CAST_DEC rounds HalfUp to the call's ret_frac (flen
   * clamping unimplemented this round); CAST_INT = Round(0, HalfUp) + ToInt
   * (builtin_cast_vec.go:1817-1852). */
  GX_F_CAST_DEC = 20, GX_F_CAST_INT = 21,
  /* string builtins (pkg/expression/builtin_string_vec.go), binary/byte
   * semantics (the non-UTF8 sigs; test data is ASCII where they coincide):
   * LENGTH  (builtinLengthSig): byte length -> i64; NULL propagates.
   * SUBSTR  (builtinSubstring3ArgsSig): args (str, pos i64 const, len i64
   *          const); MySQL 1-based pos, negative pos counts from the end,
   *          pos 0 / |pos|>len / len<=0 -> empty string.
   * LIKE_PREFIX (builtinLikeSig's 'abc%' fast path): args (str, const
   *          prefix WITHOUT the trailing %); 1 iff str starts with prefix
   *          (binary collation, case-sensitive); NULL str -> NULL.
   * UPPER   (builtinUpperSig): ASCII a-z upcased, other bytes unchanged.
   * LOWER   (builtinLowerSig): ASCII A-Z downcased, other bytes unchanged
   *          (binary/ASCII charset scope, like UPPER). */
  GX_F_LENGTH = 32, GX_F_SUBSTR = 33, GX_F_LIKE_PREFIX = 34, GX_F_UPPER = 35,
  GX_F_LOWER = 36,
  /* IS [NOT] NULL (builtinIntIsNullSig / builtinDecimalIsNullSig /
   * builtinStringIsNullSig family, builtin_op_vec.go): unary, any column
   * type; result i64 0/1 and NEVER NULL (the null bit is the value). */
  GX_F_IS_NULL = 37, GX_F_IS_NOT_NULL = 38,
  /* TRIM(str) (builtinTrim1ArgSig, builtin_string.go spaceChars = " "):
   * removes leading AND trailing 0x20 bytes only. */
  GX_F_TRIM = 39,
  /* IFNULL(a, b) (builtinIfNullSig, builtin_control_vec_generated.go):
   * first non-NULL operand; NULL only when both are. COALESCE(a,b,c,...)
   * is the chain IFNULL(a, IFNULL(b, c)). */
  GX_F_IFNULL = 40,
  /* argument TUPLE for multi-column DISTINCT aggregates —
   * count(distinct a, b, ...) (aggregation descriptor with multiple args;
   * rows with ANY NULL element are excluded, aggregate.result's
   * count(distinct b,c,d) golden). NOT a value expression: valid only as
   * a COUNT_DISTINCT arg. */
  GX_F_TUPLE = 41,
  /* ROUND(x, d) (builtinRoundWithFracDecSig, builtin_math_vec.go:977):
   * decimal round half-away-from-zero at scale min(d, ret_frac); const
   * d >= 0 this round.
   * ABS(x) (builtinAbsDecSig / builtinAbsIntSig): absolute value. */
  GX_F_ROUND = 42, GX_F_ABS = 43,
  /* YEAR/MONTH/DAY(t) (builtinYearSig / builtinMonthSig / builtinDaySig,
   * builtin_time_vec.go): CoreTime bitfield extraction (year@50:14,
   * month@46:4, day@41:5 — core_time.go); arg must be a TIME column;
   * NULL propagates. */
  GX_F_YEAR = 44, GX_F_MONTH = 45, GX_F_DAY = 46,
  GX_F_HOUR = 47, GX_F_MINUTE = 48, GX_F_SECOND = 49,
  /* GREATEST/LEAST (builtinGreatest*Sig / builtinLeast*Sig,
   * builtin_compare_vec.go:27: MergeNulls => NULL if ANY arg is NULL).
   * Two args on the device path; n-ary composes as a chain. */
  GX_F_GREATEST = 50, GX_F_LEAST = 51,
  /* IF(cond, a, b) (builtinIfSig, builtin_control_vec_generated.go):
   * a when cond is non-NULL and != 0, else b; the result is the chosen
   * branch's value/NULL. CASE WHEN c1 THEN v1 WHEN c2 THEN v2 ELSE e END
   * is the chain IF(c1, v1, IF(c2, v2, e)). In value context the compare
   * family (GX_F_LT..GX_F_NE) evaluates to i64 0/1 (NULL if either
   * operand is NULL), so conditions compose from comparisons. */
  GX_F_IF = 52,
  /* logical OR (builtinLogicOrSig) INSIDE a Selection conjunct: each
   * top-level condition may be a disjunction of simple predicates
   * (col cmp const / col cmp col / LIKE / IS NULL); nested ORs flatten.
   * Supported on the standalone Selection and inner-join other-condition
   * paths this round (a NULL leaf behaves as FALSE, which matches
   * VecEvalBool's NULL-rejects within both OR and AND). */
  GX_F_OR = 53
};

/* ---- aggregate function codes (pkg/executor/aggfuncs) ---- */
enum {
  GX_AGG_COUNT = 0,    /* arg expr = -1 => count(*) */
  GX_AGG_SUM = 1,      /* sum4Decimal / sum float (func_sum.go:44,224) */
  GX_AGG_AVG = 2,      /* avgOriginal4Decimal: state {sum, count} (func_avg.go:69-135) */
  GX_AGG_MIN = 3,
  GX_AGG_MAX = 4,
  GX_AGG_FIRSTROW = 5,
  /* DISTINCT variants (aggFuncDesc.HasDistinct, aggregation.go;
   * executor/aggfuncs distinct wrappers): aggregate over the DISTINCT
   * non-NULL values of the single arg column. COMPLETE mode only (a
   * distinct partial state is a set and is not exchanged this round).
   * SUM/AVG DISTINCT take decimal (or f64) args — int args must arrive
   * wrapped in CAST_DEC like their non-distinct forms. */
  GX_AGG_COUNT_DISTINCT = 6,
  GX_AGG_SUM_DISTINCT = 7,
  GX_AGG_AVG_DISTINCT = 8
};

/* ---- hash-agg execution mode (AggFuncDesc partial/final split,
 * reference pkg/executor/builder.go:2239-2258, aggfuncs.go:224-263) ---- */
enum {
  GX_AGG_MODE_COMPLETE = 0, /* raw rows in, final values out */
  GX_AGG_MODE_PARTIAL = 1,  /* raw rows in, canonical partial-state chunk out */
  GX_AGG_MODE_FINAL = 2     /* partial-state chunks in, final values out (MergePartialResult) */
};

/* ---- synthetic TPC-H tables (MockDataSource analog,
 * reference pkg/executor/internal/testutil/testutil.go:45-348) ---- */
enum { GX_TPCH_LINEITEM = 0, GX_TPCH_ORDERS = 1, GX_TPCH_CUSTOMER = 2 };

/* ---- plan builder ---- */
typedef struct gx_pb gx_pb;
typedef struct gx_exec gx_exec;

gx_pb* gx_pb_new(void);
void   gx_pb_free(gx_pb* pb);

/* expressions: return expr id (>=0) or negative status */
int32_t gx_pb_colref(gx_pb* pb, int32_t col_idx, int32_t type, int32_t frac);
int32_t gx_pb_const_i64(gx_pb* pb, int64_t v);
int32_t gx_pb_const_f64(gx_pb* pb, double v);
int32_t gx_pb_const_time(gx_pb* pb, uint64_t packed_core_time);
int32_t gx_pb_const_dec(gx_pb* pb, const uint8_t dec40[40]);
int32_t gx_pb_const_str(gx_pb* pb, const char* s, int32_t len);
/* ret_frac: decimal result fraction metadata (field-type frac); -1 if N/A */
int32_t gx_pb_call(gx_pb* pb, int32_t func, int32_t ret_type, int32_t ret_frac,
                   const int32_t* args, int32_t n_args);

/* plan nodes: return node id (>=0) or negative status */
int32_t gx_pb_source(gx_pb* pb, const int32_t* col_types, const int32_t* col_fracs,
                     int32_t n_cols);
int32_t gx_pb_selection(gx_pb* pb, int32_t child, const int32_t* conds, int32_t n_conds);
int32_t gx_pb_projection(gx_pb* pb, int32_t child, const int32_t* exprs, int32_t n_exprs);
int32_t gx_pb_hashagg(gx_pb* pb, int32_t child,
                      const int32_t* group_exprs, int32_t n_group,
                      const int32_t* agg_funcs, const int32_t* agg_args,
                      const int32_t* agg_fracs, int32_t n_aggs, int32_t mode);
/* stream aggregation (aggregate/agg_stream_executor.go): same semantics as
 * hash aggregation but requires the child stream grouped (all rows of a key
 * contiguous -- e.g. sorted on the group cols); emits groups in stream
 * order. COMPLETE mode this round. */
/* merge join (join/merge_join.go): inner join over inputs the plan sorts on
 * the join keys (children must be full sorts whose leading key is the join
 * key); results identical to the hash join. */
int32_t gx_pb_mergejoin(gx_pb* pb, int32_t build_child, int32_t probe_child,
                        const int32_t* build_keys, const int32_t* probe_keys,
                        int32_t n_keys, int32_t join_type);
int32_t gx_pb_streamagg(gx_pb* pb, int32_t child, const int32_t* group_exprs,
                        int32_t n_group, const int32_t* agg_funcs,
                        const int32_t* agg_args, const int32_t* agg_fracs,
                        int32_t n_aggs);
int32_t gx_pb_topn(gx_pb* pb, int32_t child, const int32_t* key_exprs,
                   const uint8_t* key_desc, int32_t n_keys,
                   int64_t limit, int64_t offset);
/* join_type (HashJoinV2 equivalent, pkg/executor/join/hash_join_v2.go):
 *  0 inner (inner_join_probe.go:27-86)
 *  1 left outer, probe = outer side: unmatched probe rows null-extend the
 *    build columns (outer_join_probe.go probe-outer path)
 *  2 right outer, build = outer side: matched pairs plus unmatched build
 *    rows with probe columns NULL (outer_join_probe.go matched flags)
 *  3 semi: each probe row once on any match; output = probe columns only
 *    (base_semi_join.go)
 *  4 anti semi: each probe row once on NO match (NULL-key probe rows match
 *    nothing and emit); output = probe columns only
 *    (anti_semi_join_probe.go, non-null-aware variant)
 *  5 null-aware anti semi — the x NOT IN (y set) shape (null_aware NAASJ,
 *    hash_join_v1.go:599): an EMPTY (post-filter) build side accepts every
 *    probe row incl. NULL keys; any NULL build key accepts none; otherwise
 *    anti semi that also rejects NULL-key probe rows. Probe columns only.
 *  6 left outer semi (LeftOuterSemiJoin): EVERY probe row once, with a
 *    trailing i64 scalar column = 1 if any build match else 0 (the
 *    IN-subquery-in-select-list shape, non-null-aware).
 *  7 null-aware left outer semi: scalar = x IN (y set) with NULL
 *    semantics — 1 on match; NULL when no match but x is NULL (nonempty y)
 *    or y contains NULL; else 0 (joinNAALOSJ, hash_join_v1.go). */
int32_t gx_pb_hashjoin(gx_pb* pb, int32_t build_child, int32_t probe_child,
                       const int32_t* build_keys, const int32_t* probe_keys,
                       int32_t n_keys, int32_t join_type);

/* build: device = -1 for the library's default device (oracle: CPU; product:
 * current HIP device). The PRODUCT library fails loudly (GX_ERR_NO_GPU) if no
 * MI355X is reachable — it has no CPU fallback. */
gx_exec* gx_build(gx_pb* pb, int32_t root, int32_t device);

/* bind a source node to caller-owned host chunks (uploaded once at Open) */
int32_t gx_bind_chunks(gx_exec* ex, int32_t source_node,
                       const gx_chunk* chunks, int32_t n_chunks);
/* bind a source node to the deterministic synthetic TPC-H generator
 * (seeded; row_offset for multi-GPU row-range sharding — the region-shard
 * analog of store/copr/coprocessor.go:525) */
int32_t gx_bind_tpch(gx_exec* ex, int32_t source_node, int32_t table,
                     int64_t n_rows, uint64_t seed, int64_t row_offset);
/* like gx_bind_tpch but with the TOTAL table row count (for cross-table key
 * ranges under row-range sharding); gx_bind_tpch == total_rows = n_rows. */
int32_t gx_bind_tpch_sharded(gx_exec* ex, int32_t source_node, int32_t table,
                             int64_t n_rows, uint64_t seed, int64_t row_offset,
                             int64_t total_rows);

int32_t gx_open(gx_exec* ex);
/* fills out-chunk (caller-allocated buffers, data_cap/offsets_cap honored),
 * rows_out = rows appended; 0 = EOF. */
int32_t gx_next(gx_exec* ex, gx_chunk* out, int32_t* rows_out);

/* ---- chunk wire codec (util/chunk/codec.go:41-141) -------------------
 * Per column: [u32 LE length][u32 LE nullCount][nullBitmap iff nullCount>0,
 * ceil(len/8) bytes][offsets iff varlen, (len+1)*8 bytes][data bytes].
 * Encode returns bytes written (or -needed if cap too small).
 * Decode consumes one chunk of n_cols columns into caller-allocated buffers
 * (out->cols[i].data/null_bitmap/offsets with caps set); returns bytes
 * consumed or a negative GX_ERR_* status. */
int64_t gx_chunk_encode(const gx_chunk* chunk, uint8_t* out, int64_t cap);
int64_t gx_chunk_decode(const uint8_t* buf, int64_t len, gx_chunk* out);
int32_t gx_close(gx_exec* ex);
void    gx_exec_free(gx_exec* ex);
const char* gx_last_error(gx_exec* ex);

/* engine info: returns 1 if this library executes on GPU, 0 if CPU oracle */
int32_t gx_engine_is_gpu(void);
const char* gx_engine_name(void);
/* duration (ms) of the last fused compute kernel on this executor, measured
 * with HIP events on the launch stream (0 if none ran / CPU engine), and the
 * row count that passed the filter in the last run. */
double  gx_last_kernel_ms(gx_exec* ex);
int64_t gx_last_sel_count(gx_exec* ex);

/* ---- decimal helpers (each library's own implementation; used by golden
 * vector tests against pkg/types/mydecimal_test.go answers) ---- */
int32_t gx_dec_from_string(const char* s, int32_t len, uint8_t out40[40]);
/* ToString (no rounding), returns length written (mydecimal.go:328) */
int32_t gx_dec_to_string(const uint8_t dec40[40], char* buf, int32_t buf_len);
/* String() = Round(resultFrac, HalfUp) + ToString (mydecimal.go:277) */
int32_t gx_dec_display_string(const uint8_t dec40[40], char* buf, int32_t buf_len);
int32_t gx_dec_add(const uint8_t a[40], const uint8_t b[40], uint8_t out[40]);
int32_t gx_dec_sub(const uint8_t a[40], const uint8_t b[40], uint8_t out[40]);
int32_t gx_dec_mul(const uint8_t a[40], const uint8_t b[40], uint8_t out[40]);
int32_t gx_dec_div(const uint8_t a[40], const uint8_t b[40], uint8_t out[40], int32_t frac_incr);
/* round_mode: 5 = ModeHalfUp, 10 = ModeTruncate, 0 = ModeCeiling */
int32_t gx_dec_round(const uint8_t in[40], int32_t frac, int32_t round_mode, uint8_t out[40]);
int32_t gx_dec_compare(const uint8_t a[40], const uint8_t b[40]);
int32_t gx_dec_to_bin(const uint8_t in[40], int32_t precision, int32_t frac,
                      uint8_t* out, int32_t* out_len);
int32_t gx_dec_from_bin(const uint8_t* bin, int32_t bin_len, int32_t precision,
                        int32_t frac, uint8_t out[40]);
int32_t gx_dec_to_hash_key(const uint8_t in[40], uint8_t* out, int32_t* out_len);
int32_t gx_dec_from_i64(int64_t v, uint8_t out[40]);
int32_t gx_dec_shift(const uint8_t in[40], int32_t shift, uint8_t out[40]);
int32_t gx_dec_result_frac(const uint8_t dec40[40]);

/* ---- time helpers (packed CoreTime, core_time.go:25 / time.go:233-265) ---- */
uint64_t gx_time_from_date(int32_t year, int32_t month, int32_t day);
uint64_t gx_time_from_datetime(int32_t year, int32_t month, int32_t day,
                               int32_t hour, int32_t minute, int32_t second,
                               int32_t microsecond, int32_t type_and_fsp);
int32_t  gx_time_compare(uint64_t a, uint64_t b); /* compareTime, core_time.go:256 */

#ifdef __cplusplus
}
#endif
#endif /* GX_EXECUTOR_H */
