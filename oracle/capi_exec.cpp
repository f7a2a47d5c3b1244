// oracle/capi_exec.cpp — executor C-ABI (oracle flavor). Implementation lands
// with exec.cpp; stubs below are replaced incrementally.
#include "../include/gx_executor.h"
extern "C" {
gx_pb* gx_pb_new(void) { return nullptr; }
void gx_pb_free(gx_pb*) {}
int32_t gx_pb_colref(gx_pb*, int32_t, int32_t, int32_t) { return GX_ERR_INVALID; }
int32_t gx_pb_const_i64(gx_pb*, int64_t) { return GX_ERR_INVALID; }
int32_t gx_pb_const_f64(gx_pb*, double) { return GX_ERR_INVALID; }
int32_t gx_pb_const_time(gx_pb*, uint64_t) { return GX_ERR_INVALID; }
int32_t gx_pb_const_dec(gx_pb*, const uint8_t*) { return GX_ERR_INVALID; }
int32_t gx_pb_const_str(gx_pb*, const char*, int32_t) { return GX_ERR_INVALID; }
int32_t gx_pb_call(gx_pb*, int32_t, int32_t, int32_t, const int32_t*, int32_t) { return GX_ERR_INVALID; }
int32_t gx_pb_source(gx_pb*, const int32_t*, const int32_t*, int32_t) { return GX_ERR_INVALID; }
int32_t gx_pb_selection(gx_pb*, int32_t, const int32_t*, int32_t) { return GX_ERR_INVALID; }
int32_t gx_pb_projection(gx_pb*, int32_t, const int32_t*, int32_t) { return GX_ERR_INVALID; }
int32_t gx_pb_hashagg(gx_pb*, int32_t, const int32_t*, int32_t, const int32_t*, const int32_t*, const int32_t*, int32_t, int32_t) { return GX_ERR_INVALID; }
int32_t gx_pb_topn(gx_pb*, int32_t, const int32_t*, const uint8_t*, int32_t, int64_t, int64_t) { return GX_ERR_INVALID; }
int32_t gx_pb_hashjoin(gx_pb*, int32_t, int32_t, const int32_t*, const int32_t*, int32_t, int32_t) { return GX_ERR_INVALID; }
gx_exec* gx_build(gx_pb*, int32_t, int32_t) { return nullptr; }
int32_t gx_bind_chunks(gx_exec*, int32_t, const gx_chunk*, int32_t) { return GX_ERR_INVALID; }
int32_t gx_bind_tpch(gx_exec*, int32_t, int32_t, int64_t, uint64_t, int64_t) { return GX_ERR_INVALID; }
int32_t gx_open(gx_exec*) { return GX_ERR_INVALID; }
int32_t gx_next(gx_exec*, gx_chunk*, int32_t*) { return GX_ERR_INVALID; }
int32_t gx_close(gx_exec*) { return GX_ERR_INVALID; }
void gx_exec_free(gx_exec*) {}
const char* gx_last_error(gx_exec*) { return "not implemented"; }
}
