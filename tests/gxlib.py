"""ctypes bindings for the gx C-ABI (include/gx_executor.h).

Loads either liboracle.so (CPU restatement, parity anchor) or libgxexec.so
(MI355X product engine). Both export the same symbols.
"""
import ctypes
import os
import subprocess

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

GX_OK = 0
GX_ERR_TRUNCATED = 1
GX_ERR_OVERFLOW = 2
GX_ERR_DIV_ZERO = 3
GX_ERR_BAD_NUMBER = 4

GX_TYPE_I64 = 0
GX_TYPE_F64 = 1
GX_TYPE_DECIMAL = 2
GX_TYPE_TIME = 3
GX_TYPE_STRING = 4

GX_F_LT, GX_F_LE, GX_F_GT, GX_F_GE, GX_F_EQ, GX_F_NE = 0, 1, 2, 3, 4, 5
GX_F_PLUS, GX_F_MINUS, GX_F_MUL, GX_F_DIV = 16, 17, 18, 19
GX_F_CAST_DEC, GX_F_CAST_INT = 20, 21
GX_F_LENGTH, GX_F_SUBSTR, GX_F_LIKE_PREFIX, GX_F_UPPER = 32, 33, 34, 35
GX_F_LOWER = 36
GX_F_IS_NULL, GX_F_IS_NOT_NULL = 37, 38
GX_F_TRIM = 39
GX_F_IFNULL = 40
GX_F_TUPLE = 41
GX_F_ROUND, GX_F_ABS = 42, 43
GX_F_YEAR, GX_F_MONTH, GX_F_DAY = 44, 45, 46
GX_F_HOUR, GX_F_MINUTE, GX_F_SECOND = 47, 48, 49
GX_F_GREATEST, GX_F_LEAST = 50, 51
GX_F_IF = 52
GX_F_OR = 53

GX_AGG_COUNT, GX_AGG_SUM, GX_AGG_AVG, GX_AGG_MIN, GX_AGG_MAX, GX_AGG_FIRSTROW = range(6)
GX_AGG_COUNT_DISTINCT, GX_AGG_SUM_DISTINCT, GX_AGG_AVG_DISTINCT = 6, 7, 8
GX_AGG_MODE_COMPLETE, GX_AGG_MODE_PARTIAL, GX_AGG_MODE_FINAL = 0, 1, 2

GX_TPCH_LINEITEM, GX_TPCH_ORDERS, GX_TPCH_CUSTOMER = 0, 1, 2


class GxCol(ctypes.Structure):
    _fields_ = [
        ("data", ctypes.c_void_p),
        ("null_bitmap", ctypes.POINTER(ctypes.c_uint8)),
        ("offsets", ctypes.POINTER(ctypes.c_int64)),
        ("length", ctypes.c_int32),
        ("elem_size", ctypes.c_int32),
        ("data_cap", ctypes.c_int64),
        ("offsets_cap", ctypes.c_int64),
    ]


class GxChunk(ctypes.Structure):
    _fields_ = [
        ("cols", ctypes.POINTER(GxCol)),
        ("n_cols", ctypes.c_int32),
        ("n_rows", ctypes.c_int32),
    ]


def _decl(lib):
    i8p = ctypes.POINTER(ctypes.c_uint8)
    lib.gx_dec_from_string.argtypes = [ctypes.c_char_p, ctypes.c_int32, i8p]
    lib.gx_dec_to_string.argtypes = [i8p, ctypes.c_char_p, ctypes.c_int32]
    lib.gx_dec_display_string.argtypes = [i8p, ctypes.c_char_p, ctypes.c_int32]
    for f in ("gx_dec_add", "gx_dec_sub", "gx_dec_mul"):
        getattr(lib, f).argtypes = [i8p, i8p, i8p]
    lib.gx_dec_div.argtypes = [i8p, i8p, i8p, ctypes.c_int32]
    lib.gx_dec_round.argtypes = [i8p, ctypes.c_int32, ctypes.c_int32, i8p]
    lib.gx_dec_compare.argtypes = [i8p, i8p]
    lib.gx_dec_to_bin.argtypes = [i8p, ctypes.c_int32, ctypes.c_int32, i8p,
                                  ctypes.POINTER(ctypes.c_int32)]
    lib.gx_dec_from_bin.argtypes = [i8p, ctypes.c_int32, ctypes.c_int32,
                                    ctypes.c_int32, i8p]
    lib.gx_dec_to_hash_key.argtypes = [i8p, i8p, ctypes.POINTER(ctypes.c_int32)]
    lib.gx_dec_from_i64.argtypes = [ctypes.c_int64, i8p]
    lib.gx_dec_result_frac.argtypes = [i8p]
    lib.gx_time_from_date.restype = ctypes.c_uint64
    lib.gx_time_from_date.argtypes = [ctypes.c_int32] * 3
    lib.gx_time_from_datetime.restype = ctypes.c_uint64
    lib.gx_time_from_datetime.argtypes = [ctypes.c_int32] * 8
    lib.gx_time_compare.argtypes = [ctypes.c_uint64, ctypes.c_uint64]
    lib.gx_engine_name.restype = ctypes.c_char_p
    # executor API
    lib.gx_pb_new.restype = ctypes.c_void_p
    lib.gx_pb_free.argtypes = [ctypes.c_void_p]
    i32p = ctypes.POINTER(ctypes.c_int32)
    lib.gx_pb_colref.argtypes = [ctypes.c_void_p, ctypes.c_int32, ctypes.c_int32, ctypes.c_int32]
    lib.gx_pb_const_i64.argtypes = [ctypes.c_void_p, ctypes.c_int64]
    lib.gx_pb_const_f64.argtypes = [ctypes.c_void_p, ctypes.c_double]
    lib.gx_pb_const_time.argtypes = [ctypes.c_void_p, ctypes.c_uint64]
    lib.gx_pb_const_dec.argtypes = [ctypes.c_void_p, i8p]
    lib.gx_pb_const_str.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_int32]
    lib.gx_pb_call.argtypes = [ctypes.c_void_p, ctypes.c_int32, ctypes.c_int32,
                               ctypes.c_int32, i32p, ctypes.c_int32]
    lib.gx_pb_source.argtypes = [ctypes.c_void_p, i32p, i32p, ctypes.c_int32]
    lib.gx_pb_selection.argtypes = [ctypes.c_void_p, ctypes.c_int32, i32p, ctypes.c_int32]
    lib.gx_pb_projection.argtypes = [ctypes.c_void_p, ctypes.c_int32, i32p, ctypes.c_int32]
    lib.gx_pb_hashagg.argtypes = [ctypes.c_void_p, ctypes.c_int32, i32p, ctypes.c_int32,
                                  i32p, i32p, i32p, ctypes.c_int32, ctypes.c_int32]
    lib.gx_pb_topn.argtypes = [ctypes.c_void_p, ctypes.c_int32, i32p, i8p,
                               ctypes.c_int32, ctypes.c_int64, ctypes.c_int64]
    lib.gx_pb_streamagg.argtypes = [ctypes.c_void_p, ctypes.c_int32, i32p,
                                    ctypes.c_int32, i32p, i32p, i32p,
                                    ctypes.c_int32]
    lib.gx_pb_mergejoin.argtypes = [ctypes.c_void_p, ctypes.c_int32,
                                    ctypes.c_int32, i32p, i32p,
                                    ctypes.c_int32, ctypes.c_int32]
    lib.gx_pb_hashjoin.argtypes = [ctypes.c_void_p, ctypes.c_int32, ctypes.c_int32,
                                   i32p, i32p, ctypes.c_int32, ctypes.c_int32]
    lib.gx_build.restype = ctypes.c_void_p
    lib.gx_build.argtypes = [ctypes.c_void_p, ctypes.c_int32, ctypes.c_int32]
    lib.gx_bind_chunks.argtypes = [ctypes.c_void_p, ctypes.c_int32,
                                   ctypes.POINTER(GxChunk), ctypes.c_int32]
    lib.gx_bind_tpch.argtypes = [ctypes.c_void_p, ctypes.c_int32, ctypes.c_int32,
                                 ctypes.c_int64, ctypes.c_uint64, ctypes.c_int64]
    lib.gx_bind_tpch_sharded.argtypes = [ctypes.c_void_p, ctypes.c_int32,
                                         ctypes.c_int32, ctypes.c_int64,
                                         ctypes.c_uint64, ctypes.c_int64,
                                         ctypes.c_int64]
    lib.gx_pb_const_dec.argtypes = [ctypes.c_void_p, ctypes.POINTER(ctypes.c_uint8)]
    lib.gx_dec_shift.argtypes = [ctypes.POINTER(ctypes.c_uint8), ctypes.c_int32,
                                 ctypes.POINTER(ctypes.c_uint8)]
    lib.gx_open.argtypes = [ctypes.c_void_p]
    lib.gx_next.argtypes = [ctypes.c_void_p, ctypes.POINTER(GxChunk), i32p]
    lib.gx_close.argtypes = [ctypes.c_void_p]
    lib.gx_exec_free.argtypes = [ctypes.c_void_p]
    lib.gx_last_error.restype = ctypes.c_char_p
    lib.gx_last_error.argtypes = [ctypes.c_void_p]
    return lib


def _build_oracle():
    subprocess.run(["make", "-s", "-j4"], cwd=os.path.join(REPO, "oracle"), check=True)


def load_oracle():
    path = os.path.join(REPO, "oracle", "liboracle.so")
    srcs = [os.path.join(REPO, "oracle", f) for f in os.listdir(os.path.join(REPO, "oracle"))
            if f.endswith((".cpp", ".h"))]
    if not os.path.exists(path) or any(os.path.getmtime(s) > os.path.getmtime(path) for s in srcs):
        _build_oracle()
    return _decl(ctypes.CDLL(path))


def load_product():
    path = os.path.join(REPO, "tidb_amd", "csrc", "libgxexec.so")
    if not os.path.exists(path):
        raise RuntimeError(
            "libgxexec.so not built — run python -c 'import __graft_entry__; __graft_entry__.build()'")
    return _decl(ctypes.CDLL(path))


# ---- helpers ----

def dec(lib, s):
    out = (ctypes.c_uint8 * 40)()
    err = lib.gx_dec_from_string(s.encode(), len(s.encode()), out)
    return out, err


def dec_str(lib, d):
    buf = ctypes.create_string_buffer(128)
    n = lib.gx_dec_to_string(d, buf, 128)
    assert n >= 0
    return buf.value.decode()


def dec_display(lib, d):
    buf = ctypes.create_string_buffer(128)
    n = lib.gx_dec_display_string(d, buf, 128)
    assert n >= 0
    return buf.value.decode()
