// oracle/exec.h — CPU restatement of the reference's chunk executors.
// ORACLE / TEST INFRASTRUCTURE ONLY.
//
// Operator contract mirrors exec.Executor (internal/exec/executor.go:224-250):
// Open -> Next(chunk<=1024 rows; 0 rows = EOF) -> Close.
#ifndef ORACLE_EXEC_H
#define ORACLE_EXEC_H

#include <map>
#include <memory>
#include <string>
#include <unordered_map>
#include <vector>

#include "chunk.h"
#include "mydecimal.h"

namespace oracle {

constexpr int kMaxChunkSize = 1024;  // vardef DefMaxChunkSize (tidb_vars.go:1559)

enum ExprKind { EK_COLREF = 0, EK_CONST = 1, EK_CALL = 2 };
enum PlanKind {
  PK_SOURCE = 0, PK_SELECTION, PK_PROJECTION, PK_HASHAGG, PK_TOPN, PK_HASHJOIN,
  PK_SORT,
  PK_STREAMAGG,  // sorted/grouped-input aggregation (agg_stream_executor.go)
  PK_MERGEJOIN   // sorted-input inner join (join/merge_join.go)
};

struct Expr {
  int kind = EK_COLREF;
  int retType = GX_TYPE_I64;
  int retFrac = 0;
  int colIdx = -1;
  int func = -1;
  std::vector<int> args;
  int64_t constI64 = 0;
  double constF64 = 0;
  uint64_t constTime = 0;
  MyDecimal constDec;
  std::string constStr;
};

struct PlanNode {
  int kind = PK_SOURCE;
  int child = -1, child2 = -1;
  std::vector<int> colTypes, colFracs;       // source schema
  std::vector<int> exprs;                    // conds / projections / group-bys / sort keys
  std::vector<int> aggFuncs, aggArgs, aggFracs;
  int aggMode = 0;
  std::vector<uint8_t> keyDesc;              // topn/sort: 1 = desc
  int64_t limit = 0, offset = 0;
  std::vector<int> buildKeys, probeKeys;
  int joinType = 0;
};

struct Plan {
  std::vector<Expr> exprs;
  std::vector<PlanNode> nodes;
};

struct SourceBinding {
  // either bound host chunks...
  std::vector<Chunk> chunks;
  bool haveChunks = false;
  // ...or the synthetic TPC-H table
  int tpchTable = -1;
  int64_t tpchRows = 0;       // rows this source emits (the shard)
  uint64_t tpchSeed = 0;
  int64_t tpchRowOffset = 0;  // global row index of the shard's first row
  int64_t tpchTotalRows = 0;  // whole-table rows (cross-table key ranges)
};

class Exec {
 public:
  virtual ~Exec() = default;
  virtual int32_t open() = 0;
  virtual int32_t next(Chunk& out) = 0;  // out pre-shaped by schema; reset by callee
  virtual int32_t close() = 0;
  std::vector<int> outTypes, outFracs;
  std::string err;
};

// Builds the executor tree for `root`. bindings: per-node-id source binding.
std::unique_ptr<Exec> BuildExec(const Plan& plan, int root,
                                std::map<int, SourceBinding>* bindings,
                                std::string* err);

// tpch.cpp — deterministic synthetic generators (§8d distributions; seed 42
// default). Fills `out` with rows [rowBegin, rowBegin+n) of the table.
void TpchGenChunk(int table, int64_t rowBegin, int n, uint64_t seed,
                  int64_t totalRowsHint, Chunk& out);
void TpchSchema(int table, std::vector<int>* types, std::vector<int>* fracs);

}  // namespace oracle
#endif
