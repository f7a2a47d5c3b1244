// tidb_amd/csrc/gx_common.h — product engine: host/device shared descriptors.
//
// The engine compiles the C-ABI plan (include/gx_executor.h) into these
// descriptors; gx_kernels.hip executes them in a single fused pass
// (scan -> filter -> project -> hash-aggregate), the MI355X-native form of
// the reference pipeline SelectionExec -> ProjectionExec -> HashAggExec
// (pkg/executor/select.go:750, projection.go:77, aggregate/*).
#ifndef GX_COMMON_H
#define GX_COMMON_H

#include <cstdint>

namespace gxp {

// ---- device column (reference chunk.Column layout resident in HBM) ----
struct DevCol {
  void* data = nullptr;          // fixed: N*elem; varlen: bytes
  uint8_t* nullBitmap = nullptr; // LSB-first, 1 = NOT NULL; null => no NULLs
  int64_t* offsets = nullptr;    // varlen: N+1
  int32_t elemSize = 8;          // -1 varlen
  int32_t type = 0;              // gx type
  int32_t frac = 0;
  uint8_t hasNulls = 0;
  // varlen col whose offsets are the identity (offsets[i] == i, i.e. 1 byte
  // per row — char(1)): the scan needs no offsets reads. Proven at bind time.
  uint8_t denseOffsets = 0;
};

constexpr int kMaxCols = 16;

struct DevTable {
  DevCol cols[kMaxCols];
  int32_t nCols = 0;
  int64_t nRows = 0;
};

// ---- per-row expression VM (fixed-point decimal) ----
// Registers are int128 values at a STATIC scale decided at plan-compile time;
// null-ness is a per-register bit. Engine guarantees operand magnitudes fit:
// column loads are validated (digitsInt <= 18, digitsFrac <= 9) and overflow
// in MUL sets the kernel error flag (no silent wrap, no CPU fallback).
enum VmOp : int32_t {
  VM_LOAD_DEC = 0,   // dst <- decimal column a (units at its frac)
  VM_LOAD_I64 = 1,   // dst <- int64 column a (scale 0)
  VM_LOAD_CONST = 2, // dst <- const[a] (pre-scaled by engine)
  VM_ADD = 3,        // dst <- a + b (same scale, engine-aligned)
  VM_SUB = 4,
  VM_MUL = 5,        // dst <- a * b (scale = sa + sb)
  VM_SCALE_UP = 6,   // dst <- a * 10^b (align scales)
  VM_DIV = 7,        // dst <- trunc(a * 10^c / b) (DecimalDiv semantics:
                     // result scale word-granular, mydecimal.go doDiv;
                     // div-by-zero -> NULL; engine forces the wide VM)
  VM_ROUND_SCALE = 8,  // dst <- round_half_up(a, scale b -> scale... encoded:
                       // ins.b = target scale, ins.c = source scale (the
                       // cast family, ProduceDecWithSpecifiedTp/ToInt)
  VM_STRLEN = 9,       // dst <- byte length of string column a
                       // (builtinLengthSig; offsets fetch slot in c)
  VM_IFNULL = 10,      // dst <- a if a not NULL else b (builtinIfNullSig;
                       // scales engine-aligned; COALESCE = chained IFNULL)
  VM_ABS = 11,         // dst <- |a| (builtinAbs*Sig; narrow INT64_MIN
                       // retries wide)
  VM_TIME_EXTRACT = 12,  // dst <- CoreTime field b of a (0 year@50:14,
                         // 1 month@46:4, 2 day@41:5, 3 hour@36:5,
                         // 4 minute@30:6, 5 second@24:6 — core_time.go)
  VM_MAX2 = 13,          // dst <- max(a, b) (builtinGreatest*Sig; NULL if
                         // either is NULL; scales engine-aligned)
  VM_MIN2 = 14,          // dst <- min(a, b) (builtinLeast*Sig)
  VM_CMP = 15,           // dst <- (a <op> b) as i64 0/1, op in ins.c
                         // (GX_F_LT..GX_F_NE); NULL if either is NULL
  VM_IF = 16,            // dst <- cond(a) truthy ? b : c (builtinIfSig;
                         // NULL cond counts false; branch scales aligned)
};

struct VmIns {
  int32_t op;
  int32_t dst;  // register index
  int32_t a;    // column index / const index / register
  int32_t b;    // register / power
  int32_t c;    // LOAD ops: raw fetch slot (see FetchDesc)
};

constexpr int kMaxVmIns = 48;
constexpr int kMaxVmRegs = 14;
constexpr int kMaxVmConsts = 16;

// ---- filter (CNF of simple predicates; Q1-class) ----
enum PredKind : int32_t {
  PRED_TIME_CMP_CONST = 0,  // masked-u64 compare (core_time.go:256 semantics)
  PRED_I64_CMP_CONST = 1,
  PRED_DEC_CMP_CONST = 2,   // units compare at engine-aligned scale
};

struct PredDesc {
  int32_t orWith = 0;  // this pred ORs with the NEXT orWith entries
  int32_t kind;
  int32_t col;
  int32_t cmp;        // GX_F_LT..GX_F_NE
  int32_t slot;       // raw fetch slot (TIME/I64 preds; -1 = load at use)
  uint64_t constU64;  // time value or i64/units bits
  // string predicates (PRED_STR_EQ_CONST / PRED_STR_LIKE_PREFIX): the
  // constant lives inline so every conjunct can carry its own
  uint8_t strC[16] = {0};
  int32_t strCLen = 0;
};

constexpr int kMaxPreds = 8;

// ---- batched row fetch plan ----
// The engine lists every 8/16-byte fetch the row pipeline needs; the kernel
// issues them for R rows back-to-back before consuming (load-latency
// pipelining — the kernel is otherwise wait-bound on the per-row load chain).
enum FetchKind : int32_t {
  FETCH_8B = 0,       // 8-byte element at data + row*8
  FETCH_DEC16 = 1,    // 16 bytes at data + row*40 (decimal header+3 words)
  FETCH_OFFSETS = 2,  // offsets[row], offsets[row+1]
  FETCH_B1 = 3,       // 1 byte per row (dense char(1); glds variant only)
  FETCH_8B_CHAR2 = 4, // x = 8-byte element; y = 1-2 dense char(1) cols packed
                      // (plain kernel: prefetches group chars with the row)
  FETCH_CHAR2 = 5,    // y = 1-2 dense char(1) cols packed; x unused
};

struct FetchDesc {
  int32_t kind;
  int32_t col;
  int32_t ldsOff;  // glds variant: byte offset of this stream in a tile buffer
                   // FETCH_*CHAR2: packed char col ids (c0 | c1<<8 | n<<16)
};

constexpr int kMaxFetch = 8;

// ---- aggregation ----
// Per-group state layout (all aggs): int128 acc + int64 count per agg slot.
// SUM/AVG: acc = exact fixed-point sum, count = notNullRowCount;
// COUNT: count only; MIN/MAX/FIRSTROW: acc = value units, count = hasValue.
struct AggDesc {
  int32_t func;   // GX_AGG_*
  int32_t srcReg; // VM register holding the arg value (-1 for count(*))
  int32_t scale;  // scale of the accumulated units
  int32_t fr = -1;  // FIRSTROW over a group-by column: index into gkey cols
                    // (value decoded from the group key; no per-row state)
  int32_t fcol = -1;  // f64 sum/avg: source column (bypasses the integer VM;
                      // the value loads raw through a fetch slot and
                      // accumulates with f64 atomics — oracle/exec.cpp f64
                      // path, stated-tolerance parity)
};

constexpr int kMaxAggs = 12;

// Physical accumulator plan: aggregates sharing a source value (e.g. Q1's
// sum(qty) and avg(qty)) share ONE acc slot -- every supported device agg
// (count/sum/avg) is additive, so the per-row atomic runs once per unique
// srcReg instead of once per aggregate. accMap is host-decode-side: agg a
// reads acc slot accMap[a] (-1 for COUNT). When sharedCnt is set (no
// consumed column has NULLs) every agg's count equals the group row count
// and the kernel bumps only cnt[0].

// ---- group keys ----
// Two device key paths:
//  PACKED (round 1, kept for the Q1 shape): <= 2 dense char(1) / short
//  string / small-i64 columns pack into ONE u64 key.
//  WIDE (serialized keys, codec.HashGroupKey semantics codec.go:1791-1879):
//  arbitrary column sets (strings any length, decimals, time, i64, up to
//  kMaxGroupKeyCols). Per row a 64-bit hash over the canonical column
//  values probes the global table; the slot key packs (hash32 << 32 |
//  recordIdx) and equality is VERIFIED against a fixed-stride key record
//  (exact — hash collisions probe on). Records: [0] u64 null bits,
//  [8] u64 owner row (the row that claimed the group; string bytes beyond
//  the 16 B inline prefix compare/decode through it), [16 + 24*k] per-col
//  {u64 a, u64 b, u64 c}: i64/time value | decimal units lo/hi | string
//  trimmed len + 16 B prefix. utf8mb4_bin PAD SPACE: trailing spaces are
//  trimmed before hashing/comparing (collate.go:272), and the emitted group
//  value is the trimmed form.
constexpr int kMaxGroupKeyCols = 6;
struct GroupKeyDesc {
  int32_t nCols;
  int32_t col[kMaxGroupKeyCols];
  int32_t kind[kMaxGroupKeyCols];  // packed: 0 = short string, 1 = small i64
                                   // (<2^31), 2 = dense char(1);
                                   // wide: 3 = raw 8B (i64/time, fetch slot),
                                   // 4 = decimal (VM register), 5 = varlen
                                   // string (offsets fetch slot)
  int32_t slot[kMaxGroupKeyCols];  // fetch slot; kind 4: VM register
  int32_t kscale[kMaxGroupKeyCols] = {0};  // kind 4: units scale (decode)
  int32_t rawSlot[2] = {-1, -1};  // plain-kernel prefetch slot for dense chars:
                                  // char k lives at byte k of that slot's v.y
  int32_t wideMode = 0;       // 1 = serialized wide-key path
  int32_t recBytes = 0;       // record stride = 16 + 24 * nCols
  int64_t recCap = 0;         // record capacity (grown with the table)
  uint8_t* keyStore = nullptr;    // recCap * recBytes
  uint64_t* recCursor = nullptr;  // device allocation cursor (1 u64)
};

constexpr uint64_t kEmptyKey = ~0ULL;
constexpr int kLdsGroups = 64;     // per-workgroup LDS table capacity (19 KiB state => 8 blocks/CU, full 32-wave occupancy)
constexpr int kGlobalGroups = 8192; // DEFAULT global table capacity (2^13);
                                    // FusedQueryDesc.globalGroupsLog2 scales
                                    // it — on kErrGlobalFull the engine
                                    // reruns with an 8x larger table (the
                                    // spill-partition growth analog of
                                    // agg_spill.go's 256 partitions)

// group state sized for kMaxAggs
struct GroupSlot {
  uint64_t key;
  uint64_t accLo[kMaxAggs];
  int64_t accHi[kMaxAggs];
  int64_t cnt[kMaxAggs];
};

struct FusedQueryDesc {
  DevTable table;
  // filter
  PredDesc preds[kMaxPreds];
  int32_t nPreds = 0;
  // projection/arg VM (engine orders LOAD ops first: [0, nLoadIns))
  VmIns ins[kMaxVmIns];
  int32_t nIns = 0;
  int32_t nLoadIns = 0;
  // per-instruction precomputed constants (scalar-loaded; avoids per-row
  // power-of-ten select trees / table loads):
  //   LOAD_DEC i: insP10[i] = 10^expectedFrac, insMagic[i] = ceil(2^62/10^(9-f))
  //   SCALE_UP i: insP10[i] = 10^shift
  int64_t insP10[kMaxVmIns];
  uint64_t insMagic[kMaxVmIns];
  // batched fetch plan
  FetchDesc fetch[kMaxFetch];
  int32_t nFetch = 0;
  int64_t constLo[kMaxVmConsts];
  int64_t constHi[kMaxVmConsts];
  int32_t nConsts = 0;
  // aggs
  AggDesc aggs[kMaxAggs];
  int32_t nAggs = 0;
  int32_t nAccSlots = 0;
  int32_t accReg[kMaxAggs];   // phys acc slot -> VM register
  int32_t accMap[kMaxAggs];   // agg index -> phys acc slot (-1 for COUNT)
  // phys slot accumulation kind: 0 = int128 sum, 1 = max over the
  // order-preserving biased-u64 encoding (min stores the complement, so
  // BOTH min and max accumulate with unsigned max from a zero-initialized
  // table; func_max_min.go semantics, narrow int64 values only),
  // 3 = f64 sum (accLo holds the double bits; accReg = raw fetch slot,
  // accFcol = source column for the NULL check)
  int32_t accKind[kMaxAggs];
  int32_t accFcol[kMaxAggs];
  int32_t sharedCnt = 0;
  int32_t hasDiv = 0;
  int32_t nVmRegs = 0;  // registers the compiled VM uses (>12 selects the
                        // wide VmState14 kernel variant)  // launch the DIVOK kernel variant (division code is
                       // compiled out of the common kernels: its register
                       // demand alone costs a wave/SIMD of occupancy)
  GroupKeyDesc gkey;
  // outputs
  GroupSlot* globalTable = nullptr;  // 1<<globalGroupsLog2 slots
  int32_t globalGroupsLog2 = 13;     // current global table capacity (log2)
  uint32_t* errorFlag = nullptr;     // != 0 => abort with error
  uint64_t* selCount = nullptr;      // rows passing the filter (stats)
  // perf ablation (GX_ABLATE env; results are WRONG when nonzero — timing
  // experiments only): 1 = skip LDS agg update, 2 = skip VM+agg (filter only)
  int32_t ablate = 0;
  // 0 = narrow int64 VM (fast path); 1 = wide int128 VM. The narrow kernel
  // reports kErrRetryWide on overflow and the engine relaunches wide.
  int32_t wide = 0;
  // rows per thread batch in the fetch pipeline (engine-chosen: large fetch
  // plans use a smaller R to stay inside the VGPR budget)
  int32_t rbatch = 2;
  // 1 = skip the per-workgroup LDS table and aggregate straight into the
  // global table (engine retries with this set when a workgroup's LDS table
  // overflows — NDV above kLdsGroups)
  int32_t noLds = 0;
  // noLds accumulator BANKS: low-NDV tables replicate the accumulator
  // arrays accBanks x so concurrent blocks hammer different cache lines
  // (bank = blockIdx & (B-1)); keys/inserts live in bank 0 only and the
  // host decode merges banks (all supported states are mergeable: int128 /
  // f64 sums add, biased min/max takes the extreme, counts add)
  int32_t accBanks = 1;
  // glds (LDS-DMA) staged variant: streams are DMA'd tile-by-tile into LDS
  // (double-buffered, counted vmcnt waits) so compute overlaps the memory
  // stream. Engine enables it when every fetch kind is stageable and no
  // consumed column carries NULLs.
  int32_t useGlds = 0;
  int32_t tileBytes = 0;  // per-tile LDS bytes (sum of stream slots, 16-aligned)
};

// ---- join-aggregate pipeline (TPC-H Q3 class) ----
// customer(filter) -> key set; orders(filter) x set -> slot table keyed by
// orderkey carrying <=2 payload cols + one int128 accumulator; lineitem
// (filter) probes the table and accumulates; top-N selected by radix
// threshold + host final sort. This is HashJoinV2 build/probe
// (join/hash_join_v2.go) fused with the grouped sum, MI355X-shaped: the
// group-by keys are the probe key + build-side payload, so the aggregate
// lands in the build row (one random-HBM touch per matching probe row).
struct JoinAggSlot {
  uint64_t key;      // build join key (kEmptyKey = vacant)
  uint64_t payload0; // e.g. o_orderdate
  int64_t payload1;  // e.g. o_shippriority
  uint64_t accLo;
  int64_t accHi;
  uint64_t cnt;
};

struct JoinAggDesc {
  // phase tables
  DevTable build0;   // e.g. customer
  DevTable build1;   // e.g. orders
  DevTable probe;    // e.g. lineitem
  // filters (one pred per table this round; PRED_STR_EQ_CONST via strConst)
  PredDesc pred0, pred1, predP;
  int32_t nPred0 = 0, nPred1 = 0, nPredP = 0;
  uint8_t strConst[16];
  int32_t strConstLen = 0;
  // join columns
  int32_t b0KeyCol;   // customer.c_custkey
  int32_t b1ProbeCol; // orders.o_custkey (probes the b0 set)
  int32_t b1KeyCol;   // orders.o_orderkey (keys the slot table)
  int32_t pKeyCol;    // lineitem.l_orderkey
  int32_t payloadCol0, payloadCol1;  // orders cols carried into slots (-1 none)
  // probe-side value expression (VM over `probe` columns)
  VmIns ins[kMaxVmIns];
  int32_t nIns = 0;
  int64_t constLo[kMaxVmConsts];
  int64_t constHi[kMaxVmConsts];
  int32_t nConsts = 0;
  int32_t valueReg = -1;  // register holding the summed value
  int64_t insP10[kMaxVmIns];     // same meaning as FusedQueryDesc::insP10
  uint64_t insMagic[kMaxVmIns];
  FetchDesc fetch[kMaxFetch];
  int32_t nFetch = 0;
  // device state
  uint64_t* keySet = nullptr;  // open-addressed key set (build0)
  int32_t keySetLog2 = 0;
  JoinAggSlot* slots = nullptr;
  int32_t slotsLog2 = 0;
  // Bloom filter over the build0 key set (customer keys), probed by every
  // build1 row before the 64 MB key-set random access -- ~20% hit rate on
  // Q3 makes this the build phases' main traffic cut
  uint32_t* bloom0 = nullptr;
  int32_t bloom0Log2 = 0;
  // Bloom filter over the build keys (2 hashes; sized ~8 bits/key so it stays
  // L2-resident): rejects the ~90% non-matching probes without touching the
  // HBM-random slot table
  uint32_t* bloom = nullptr;
  int32_t bloomLog2 = 0;  // log2(bits)
  uint64_t* counters = nullptr;  // [0] build0 pass, [1] build1 pass, [2] probe match
  uint32_t* errorFlag = nullptr;
  int32_t wide = 0;
  // extra filter conjuncts per table beyond the specialized first one
  // (evaluated through evalSimplePred; the JIT emits a runtime loop)
  PredDesc pred0x[3], pred1x[3], predPx[3];
  int32_t nPred0x = 0, nPred1x = 0, nPredPx = 0;
  // duplicate build1 keys: row-indexed slots (one per build1 row) chained
  // from a pow2 heads table — the fused-pipeline analog of the standalone
  // join's heads+next (hash_table_v2.go chains). Engine retries with
  // chained=1 when the unique-key insert sees a duplicate (kErrBadKey).
  int32_t chained = 0;
  uint32_t* b1Heads = nullptr;  // build1 row + 1, 0 = empty
  int32_t b1HeadsLog2 = 0;
  uint32_t* b1Next = nullptr;   // per build1 row
  int64_t nSlots = 0;           // slot entries to scan (chained: build1 rows;
                                // else 1 << slotsLog2)
  // 128-bit top-N: the radix-threshold select runs on the order-preserving
  // projection key64 = (acc128 >> topnShift); candidates compact with the
  // exact 128-bit acc and the host sorts them exactly
  int32_t topnShift = 0;
};

enum { PRED_STR_EQ_CONST = 3 };  // extra PredKind for the join path
enum { PRED_STR_LIKE_PREFIX = 6 };  // LIKE 'abc%' fast path (builtinLikeSig)
enum { PRED_IS_NULL = 7 };  // IS [NOT] NULL (builtin*IsNullSig: result is
                            // the null bit itself, never NULL; cmp EQ = IS
                            // NULL, NE = IS NOT NULL; any column type)

// one conjunct of a post-join filter (NULL operand rejects the row, the
// VecEvalBool NULL semantics, expression.go:420-504)
struct JoinPostPred {
  int32_t orWith = 0;  // this pred ORs with the NEXT orWith entries (a
                       // disjunctive conjunct: LogicOr inside the CNF)
  int32_t kind = 0;    // 0 = side-local <col cmp const>; 1 = <col cmp col>
  int32_t side = 0;    // kind 0: 0 = build, 1 = probe (pd.col is side-local)
  PredDesc pd{};       // kind 0
  uint8_t strC[16] = {0};
  int32_t strCLen = 0;
  int32_t lcol = 0, rcol = 0;  // kind 1: join OUTPUT column indexes
  int32_t cmp = 0;             // kind 1: GX_F_LT..GX_F_NE
  int32_t ctype = 0;           // kind 1: GX_TYPE_I64 or GX_TYPE_TIME
};

constexpr int kMaxJoinKeys = 4;

// ---- standalone hash join (inner, duplicate build keys) ----
// HashJoinV2 equivalent (join/hash_join_v2.go): chained hash table over the
// build side — heads + intrusive per-row next links (hash_table_v2.go:22-53
// tagged-ptr chains / join_row_table.go next_row_ptr), lock-free CAS head
// insert (hash_table_v2.go:94-105), chain-walk probe comparing keys
// (base_join_probe.go:289-331), output = build cols ++ probe cols, one row
// per matching pair, NULL keys never match (inner_join_probe.go:27-86).
// MI355X shape: the "row table" is the resident columnar build table itself —
// no serialized row copy; output columns gather straight from HBM through the
// match-pair index, so each build/probe byte is read once per emitted value.
// Output order is unspecified (the reference's concurrent probe workers make
// its join output order nondeterministic too).
struct HashJoinDesc {
  DevTable build, probe;
  // one filter conjunct per side this round (SelectionExec pushed into the
  // build/probe scans)
  PredDesc predB, predP;
  int32_t nPredB = 0, nPredP = 0;
  uint8_t strConstB[16];
  int32_t strConstBLen = 0;
  uint8_t strConstP[16];
  int32_t strConstPLen = 0;
  // up to kMaxJoinKeys key columns per side (SerializeKeys semantics,
  // codec.go:852-910,445-456). keyKind per column:
  //   0 = raw 8 B (int64 / packed CoreTime; equality is bitwise — the round-1
  //       fast path, kernels unchanged)
  //   1 = decimal, value-normalized (ToHashKey semantics: units with trailing
  //       zeros removed + residual scale, so 1.10 == 1.1 across fracs)
  //   2 = varlen string (utf8mb4_bin PAD SPACE: trailing spaces trimmed,
  //       BinCollatorKey semantics)
  // generalKeys = 1 selects the general build/probe kernel variants.
  int32_t nKeys = 1;
  int32_t bKeyCol[kMaxJoinKeys] = {0, 0, 0, 0}, pKeyCol[kMaxJoinKeys] = {0, 0, 0, 0};
  int32_t keyKind[kMaxJoinKeys] = {0, 0, 0, 0};
  int32_t generalKeys = 0;
  uint32_t* heads = nullptr;  // 1<<headsLog2 entries: build row+1, 0 = empty
  int32_t headsLog2 = 0;
  uint32_t* next = nullptr;   // per build row: next chain row+1, 0 = end
  uint32_t* outBuild = nullptr;  // match pairs (fill phase)
  uint32_t* outProbe = nullptr;
  uint64_t* counters = nullptr;  // [0] count-phase total, [1] fill cursor,
                                 // [2] post-filter cursor
  uint32_t* errorFlag = nullptr;
  // left outer semi scalar (joinType 6/7): the appended i64 flag is NULL
  // when the no-match case cannot be decided (x IN S with NULL evidence):
  // flag2 iff (probe key NULL && naNullIfKeyNull) || naNullAlways
  // (naNullIfKeyNull = build nonempty, naNullAlways = build has a NULL key)
  int32_t naNullIfKeyNull = 0;
  int32_t naNullAlways = 0;
  // post-join filter (the join's "other conditions": VectorizedFilter over
  // the joined chunk, inner_join_probe.go:75 — expressed as a Selection
  // above the join). CNF; evaluated on the match pairs BEFORE the gather so
  // rejected rows never touch HBM output.
  JoinPostPred post[8];  // CNF conjuncts; OR groups flatten into entries
  int32_t nPost = 0;
  int64_t nPairs = 0;            // fill-phase total (filter input size)
  uint32_t* outBuild2 = nullptr; // filter-surviving pairs
  uint32_t* outProbe2 = nullptr;
  // deterministic fill (no shared-cursor atomics): the count pass caches
  // per-probe-row match info in `hits` (0 = eligible-no-match, brow+1 =
  // single match, kHjMulti = several matches -> fill re-walks the chain,
  // kHjIneligible = pred-failed row) and writes each 64-row wave tile's
  // emit total into tileCounts; the engine exclusive-scans those into
  // tileBases, so the fill pass streams hits + bases and walks no chain for
  // unique matches (the r01 fill was 12.5 ms of wave-serialized cursor
  // atomics + a full chain re-walk)
  uint32_t* hits = nullptr;      // per probe row
  int64_t* tileCounts = nullptr; // per 64-row tile: emit total
  int64_t* tileBases = nullptr;  // exclusive scan of tileCounts (+ total)
  // join type (gx_executor.h gx_pb_hashjoin): 0 inner, 1 left outer
  // (probe = outer: unmatched probe rows null-extend the build side),
  // 2 right outer (build = outer: matched flags + unmatched-build drain),
  // 3 semi / 4 anti semi (probe cols only). Null-extended pairs carry
  // kHjNullRow on the missing side; the gathers decode it to NULL.
  int32_t joinType = 0;
  uint32_t* matched = nullptr;   // right outer: 1 bit per build row
};

constexpr uint32_t kHjNullRow = 0xFFFFFFFFu;
constexpr uint32_t kHjIneligible = 0xFFFFFFFFu;  // hits[]: pred-failed row
constexpr uint32_t kHjMulti = 0xFFFFFFFEu;       // hits[]: >1 match

// row-pack gather: the build side's fixed-width output columns pack into a
// row-major staging buffer (one pass of sequential reads), so the random
// per-output-row fetch touches ONE cache line per row instead of one line
// PER COLUMN — the reference's row-table layout (join_row_table.go) applied
// where it actually pays on MI355X: only for the random-access gather.
struct RowPackDesc {
  const void* src[kMaxCols];
  void* dst[kMaxCols];
  int32_t width[kMaxCols];  // 1 (dense char), 8, or 40
  int32_t off[kMaxCols];    // byte offset inside the packed row
  int32_t nCols = 0;
  int32_t stride = 0;       // packed row bytes (8-aligned)
  int64_t nRows = 0;        // pack: source rows
  const uint32_t* idx = nullptr;  // unpack: match index (kHjNullRow -> zeros)
  int64_t total = 0;        // unpack: output rows
  uint8_t* staging = nullptr;
};
int gxPackRows(const RowPackDesc& d, void* stream);
int gxUnpackRows(const RowPackDesc& d, void* stream);

// out-of-core join: per-row partition ids for side 0 (build) / 1 (probe)
int gxHjBuildStats(const HashJoinDesc* devDesc, const HashJoinDesc& h,
                   uint64_t* out2, void* stream);
int gxHjFlagCol(const uint32_t* enc, int64_t n, int64_t* data,
                uint8_t* nullBitmap, void* stream);
int gxHjPartIds(const HashJoinDesc* devDesc, const HashJoinDesc& h, int side,
                int nParts, uint32_t* out, void* stream);

// phases: 0 = build (chain insert), 1 = count matches (+ matched flags),
// 2 = fill match pairs, 3 = post filter, 4 = count unmatched build rows
// (right outer), 5 = fill unmatched build rows
int gxHashJoinPhase(int phase, const HashJoinDesc* devDesc,
                    const HashJoinDesc& h, void* stream);
// standalone Selection compaction over h.probe (post[] = the CNF):
// 0 = count survivors, 1 = fill survivor indices into outProbe
int gxSelectPhase(int phase, const HashJoinDesc* devDesc,
                  const HashJoinDesc& h, void* stream);
// ascending u32 radix sort (survivor indices -> input row order)
int gxSortU32Keys(const uint32_t* in, uint32_t* out, int64_t n, void* tmp,
                  size_t* tmpBytes, void* stream);

// ---- standalone Projection (ProjectionExec, projection.go:77) ----
// Computed expressions evaluate per row in the direct-load VM and MATERIALIZE
// as device columns — decimals encode to the 40-byte MyDecimal struct ON
// DEVICE (canonical word layout, mydecimal.go:236-248); passthrough column
// references alias the source buffers (zero copy). Null flags write as one
// byte per row, packed to the LSB-first bitmap in a second pass.
// string projection program (builtin_string_vec.go SUBSTR/UPPER over a
// string column): windows compose into ONE (start,len) view per row —
// no intermediate materialization; bytes copy once at emit (with optional
// ASCII upcase). LENGTH rides the integer VM (VM_STRLEN) instead.
constexpr int kMaxStrWin = 4;
struct StrProg {
  int32_t col = -1;
  int32_t nWin = 0;
  int64_t winPos[kMaxStrWin];  // MySQL 1-based; negative counts from the end
  int64_t winLen[kMaxStrWin];  // -1 = a TRIM step (scan 0x20 off both ends)
  int32_t upper = 0;
  int32_t lower = 0;  // outermost case op wins (compileStrProg clears the other)
  // per-run device temps (engine-allocated)
  int64_t* starts = nullptr;
  int64_t* lens = nullptr;
  uint8_t* notNull = nullptr;
};

struct ProjDesc {
  DevTable table;
  VmIns ins[kMaxVmIns];
  int32_t nIns = 0;
  int64_t constLo[kMaxVmConsts];
  int64_t constHi[kMaxVmConsts];
  int32_t nConsts = 0;
  int64_t insP10[kMaxVmIns];
  uint64_t insMagic[kMaxVmIns];
  int32_t nOut = 0;           // computed outputs only
  int32_t outReg[kMaxCols];   // VM register holding the value
  int32_t outScale[kMaxCols]; // units scale of that register
  int32_t outType[kMaxCols];  // GX_TYPE_DECIMAL or GX_TYPE_I64
  void* outData[kMaxCols];    // 40 B/row (decimal) or 8 B/row (i64)
  uint8_t* outNotNull[kMaxCols];  // byte per row, 1 = NOT NULL
  uint32_t* errorFlag = nullptr;
  int32_t wide = 0;
  StrProg sprog[kMaxCols];
  int32_t nSprog = 0;
};

int gxProject(const ProjDesc* devDesc, const ProjDesc& h, void* stream);
// string program: pass 1 computes per-row (start,len,notNull) views;
// pass 2 copies the bytes through the scanned offsets (ASCII upcase opt.)
int gxStrWindow(const ProjDesc* devDesc, const ProjDesc& h, int progIdx,
                void* stream);
int gxStrEmit(const ProjDesc* devDesc, const ProjDesc& h, int progIdx,
              const int64_t* outOffsets, uint8_t* outData, void* stream);
// pack byte-per-row not-null flags into the LSB-first bitmap
int gxPackNulls(const uint8_t* notNullBytes, uint8_t* bitmap, int64_t n,
                void* stream);
// gather a null bitmap through the match index (one thread per output byte)
int gxGatherNulls(const uint8_t* inBitmap, const uint32_t* idx, uint8_t* out,
                  int64_t n, void* stream);
// identity offsets ramp for gathered dense-char columns
int gxIotaOffsets(int64_t* p, int64_t n, void* stream);
// varlen gather, pass 1: per-output-row byte lengths from the source offsets
int gxGatherVarlenLens(const int64_t* inOffsets, const uint32_t* idx,
                       int64_t* lens, int64_t n, void* stream);
// exclusive prefix sum of lens[0..n) -> outOffsets[0..n] (hipcub; outOffsets
// gets n+1 entries, outOffsets[n] = total bytes)
int gxExclusiveSumI64(const int64_t* lens, int64_t* outOffsets, int64_t n,
                      void* tmp, size_t* tmpBytes, void* stream);
// varlen gather, pass 2: copy each row's bytes through the match index
int gxGatherVarlenBytes(const uint8_t* inData, const int64_t* inOffsets,
                        const uint32_t* idx, const int64_t* outOffsets,
                        uint8_t* outData, int64_t n, void* stream);

// top-N selection scratch
struct TopNOut {
  uint64_t key, payload0;
  int64_t payload1;
  uint64_t accLo;
  int64_t accHi;
};

// host-side launch wrappers (defined in gx_kernels.hip)
int gxLaunchTpchGen(int table, DevTable* devTab, int64_t rowBegin, int64_t nRows,
                    uint64_t seed, int64_t totalRows, void* stream);
int gxLaunchFusedAgg(const FusedQueryDesc& desc, const FusedQueryDesc* devDesc,
                     void* stream, int skipInit = 0);
int gxLaunchMemset(void* p, int v, size_t n, void* stream);
int gxFusedGrid(int64_t rows);
int gxLaunchInitTable(GroupSlot* table, int64_t nSlots, void* stream);
int gxDumpDesc(const FusedQueryDesc* devDesc, void* stream);

// ---- device full sort (sortexec/sort.go analog): LSD stable radix passes
// over composed order-preserving u64 keys, then a device gather of every
// column into sorted order ----
struct SortKeyCompose {
  int32_t col;
  int32_t kind;   // 0 = i64 (sign-biased), 1 = time (masked), 2 = dense char,
                  // 3 = decimal (parsed to int64 units at column frac)
  int32_t desc;   // 1 = descending (key complemented)
};
// one stable pass: keys[i] = orderKey(table[idx[i]].col); err: sets
// kErrRetryWide-style flag bit 1 on any unrepresentable decimal key
int gxSortComposeKeys(const DevTable* tab, const DevTable& htab,
                      SortKeyCompose k, const uint32_t* idx, uint64_t* keys,
                      int64_t n, uint32_t* errFlag, void* stream);
int gxSortPairs(uint64_t* keysIn, uint64_t* keysOut, uint32_t* idxIn,
                uint32_t* idxOut, int64_t n, void* tmp, size_t* tmpBytes,
                int beginBit, int endBit, void* stream);
// writes {OR, AND} of all keys into devOrAnd[0..1]
int gxSortKeyBits(const uint64_t* keys, int64_t n, uint64_t* devOrAnd,
                  void* stream);
int gxSortIota(uint32_t* idx, int64_t n, void* stream);
// null-bit key pass for nullable sort keys (NULL < any value)
int gxSortComposeNullKeys(const uint8_t* bitmap, const uint32_t* idx,
                          uint64_t* keys, int64_t n, int desc, void* stream);
// gather one column into out (same layout; elemSize 1, 8 or 40)
int gxSortGatherCol(const void* in, void* out, const uint32_t* idx, int64_t n,
                    int elemSize, void* stream);

// join-agg pipeline steps (gx_kernels.hip)
int gxJoinAggPhase(int phase, const JoinAggDesc* devDesc, const JoinAggDesc& h,
                   void* stream);  // 0 count0 1 build0 2 count1 3 build1 4 probe
int gxJoinAggMax(const JoinAggDesc* devDesc, const JoinAggDesc& h, uint64_t* devMax,
                 void* stream);
// pass A of the 128-bit top-N: max accHi over occupied slots -> devMaxHi
int gxJoinAggMaxHi(const JoinAggDesc* devDesc, const JoinAggDesc& h,
                   uint64_t* devMaxHi, void* stream);
// chained mode: merge duplicate-(key,payload) slots into one group
int gxJoinAggMergeDups(const JoinAggDesc* devDesc, const JoinAggDesc& h,
                       void* stream);
int gxJoinAggHist(const JoinAggDesc* devDesc, const JoinAggDesc& h,
                  uint32_t* devHist, int shift, void* stream);
int gxJoinAggCompact(const JoinAggDesc* devDesc, const JoinAggDesc& h,
                     TopNOut* out, uint64_t* outCount, uint64_t thresholdBucket,
                     int shift, uint64_t cap, void* stream);
int gxGenOrders(DevTable* tab, int64_t rowBegin, int64_t nRows, uint64_t seed,
                int64_t totalRows, void* stream);
// customer: pass 1 fills offsets (device scan) and returns total data bytes
// (synchronizes); pass 2 fills key + segment bytes
int gxGenCustomerOffsets(DevTable* tab, int64_t rowBegin, int64_t nRows,
                         uint64_t seed, void* stream, long long* totalBytes);
int gxGenCustomerFill(DevTable* tab, int64_t rowBegin, int64_t nRows,
                      uint64_t seed, void* stream);

}  // namespace gxp

#endif
