// runtime kernel specialization via hipRTC (see gx_jit.cpp)
#pragma once
#include <string>
#include "gx_common.h"

namespace gxjit {
struct JitProg;
// compile (process-cached) a specialized fused-agg kernel for this plan;
// nullptr (with *whyNot set) on any failure -- caller falls back to the
// interpreted kernel
const JitProg* compile(const gxp::FusedQueryDesc& d, std::string* whyNot);
int launch(const JitProg* prog, bool wide, const gxp::FusedQueryDesc* devDesc,
           int grid, void* stream);
std::string generateSource(const gxp::FusedQueryDesc& d);
// join-aggregate probe kernel specialization
const JitProg* compileJa(const gxp::JoinAggDesc& d, std::string* whyNot);
int launchJa(const JitProg* prog, bool wide, const gxp::JoinAggDesc* devDesc,
             int grid, void* stream);
// join-aggregate build-phase kernels (count0/build0/count1/build1)
struct JaBuildProg;
const JaBuildProg* compileJaBuild(const gxp::JoinAggDesc& d,
                                  std::string* whyNot);
int launchJaBuild(const JaBuildProg* prog, int phase,
                  const gxp::JoinAggDesc* devDesc, int grid, void* stream);
}  // namespace gxjit
