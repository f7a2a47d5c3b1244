// oracle/capi_helpers.cpp — C-ABI decimal/time helper exports (oracle flavor).
// ORACLE / TEST INFRASTRUCTURE ONLY.
#include <cstring>

#include "../include/gx_executor.h"
#include "core_time.h"
#include "mydecimal.h"

using oracle::MyDecimal;

static inline const MyDecimal* D(const uint8_t* p) {
  return reinterpret_cast<const MyDecimal*>(p);
}
static inline MyDecimal* D(uint8_t* p) { return reinterpret_cast<MyDecimal*>(p); }

extern "C" {

int32_t gx_dec_from_string(const char* s, int32_t len, uint8_t out40[40]) {
  MyDecimal d;
  int32_t err = d.FromString(s, len);
  std::memcpy(out40, &d, 40);
  return err;
}

int32_t gx_dec_to_string(const uint8_t dec40[40], char* buf, int32_t buf_len) {
  std::string s = D(dec40)->ToString();
  if ((int32_t)s.size() + 1 > buf_len) return GX_ERR_INVALID;
  std::memcpy(buf, s.data(), s.size());
  buf[s.size()] = 0;
  return (int32_t)s.size();
}

int32_t gx_dec_display_string(const uint8_t dec40[40], char* buf, int32_t buf_len) {
  std::string s = D(dec40)->DisplayString();
  if ((int32_t)s.size() + 1 > buf_len) return GX_ERR_INVALID;
  std::memcpy(buf, s.data(), s.size());
  buf[s.size()] = 0;
  return (int32_t)s.size();
}

int32_t gx_dec_add(const uint8_t a[40], const uint8_t b[40], uint8_t out[40]) {
  return oracle::DecimalAdd(D(a), D(b), D(out));
}
int32_t gx_dec_sub(const uint8_t a[40], const uint8_t b[40], uint8_t out[40]) {
  return oracle::DecimalSub(D(a), D(b), D(out));
}
int32_t gx_dec_mul(const uint8_t a[40], const uint8_t b[40], uint8_t out[40]) {
  return oracle::DecimalMul(D(a), D(b), D(out));
}
int32_t gx_dec_div(const uint8_t a[40], const uint8_t b[40], uint8_t out[40],
                   int32_t frac_incr) {
  return oracle::DecimalDiv(D(a), D(b), D(out), frac_incr);
}
int32_t gx_dec_round(const uint8_t in[40], int32_t frac, int32_t round_mode,
                     uint8_t out[40]) {
  MyDecimal tmp = *D(in);
  int32_t err = tmp.Round(&tmp, frac, (oracle::RoundMode)round_mode);
  std::memcpy(out, &tmp, 40);
  return err;
}
int32_t gx_dec_compare(const uint8_t a[40], const uint8_t b[40]) {
  return D(a)->Compare(*D(b));
}
int32_t gx_dec_to_bin(const uint8_t in[40], int32_t precision, int32_t frac,
                      uint8_t* out, int32_t* out_len) {
  int written = 0;
  int32_t err = D(in)->WriteBin(precision, frac, out, &written);
  *out_len = written;
  return err;
}
int32_t gx_dec_from_bin(const uint8_t* bin, int32_t bin_len, int32_t precision,
                        int32_t frac, uint8_t out[40]) {
  MyDecimal d;
  int binSize = 0;
  int32_t err = d.FromBin(bin, bin_len, precision, frac, &binSize);
  std::memcpy(out, &d, 40);
  return err;
}
int32_t gx_dec_to_hash_key(const uint8_t in[40], uint8_t* out, int32_t* out_len) {
  int written = 0;
  int32_t err = D(in)->ToHashKey(out, &written);
  *out_len = written;
  return err;
}
int32_t gx_dec_from_i64(int64_t v, uint8_t out[40]) {
  MyDecimal d;
  d.FromInt(v);
  std::memcpy(out, &d, 40);
  return GX_OK;
}
int32_t gx_dec_shift(const uint8_t in[40], int32_t shift, uint8_t out[40]) {
  MyDecimal tmp = *D(in);
  int32_t err = tmp.Shift(shift);
  std::memcpy(out, &tmp, 40);
  return err;
}
int32_t gx_dec_result_frac(const uint8_t dec40[40]) {
  return (int32_t)D(dec40)->resultFrac;
}

uint64_t gx_time_from_date(int32_t year, int32_t month, int32_t day) {
  return oracle::TimeFromDate(year, month, day);
}
uint64_t gx_time_from_datetime(int32_t year, int32_t month, int32_t day,
                               int32_t hour, int32_t minute, int32_t second,
                               int32_t microsecond, int32_t type_and_fsp) {
  return oracle::TimeFromDatetime(year, month, day, hour, minute, second,
                                  microsecond, type_and_fsp);
}
int32_t gx_time_compare(uint64_t a, uint64_t b) { return oracle::CompareTime(a, b); }

int32_t gx_engine_is_gpu(void) { return 0; }
const char* gx_engine_name(void) { return "oracle-cpu"; }

}  // extern "C"

extern "C" {
// oracle engine runs no GPU kernels
double gx_last_kernel_ms(gx_exec* ex) { (void)ex; return 0; }
int64_t gx_last_sel_count(gx_exec* ex) { (void)ex; return 0; }
}

// ---- chunk wire codec (restates util/chunk/codec.go:41-141) ----
// Per column: [u32 LE length][u32 LE nullCount][nullBitmap iff nullCount>0]
// [offsets iff varlen][data]. nullCount counts zero bits among the first
// `length` bitmap bits (codec.go:56 nullCount).

static int64_t colNullCount(const gx_col& c) {
  if (!c.null_bitmap) return 0;
  int64_t nulls = 0;
  for (int32_t i = 0; i < c.length; i++)
    if (((c.null_bitmap[i >> 3] >> (i & 7)) & 1) == 0) nulls++;
  return nulls;
}

static int64_t colDataBytes(const gx_col& c) {
  if (c.elem_size < 0) return c.offsets ? c.offsets[c.length] : 0;
  return (int64_t)c.length * c.elem_size;
}

extern "C" int64_t gx_chunk_encode(const gx_chunk* chunk, uint8_t* out,
                                   int64_t cap) {
  if (!chunk) return GX_ERR_INVALID;
  int64_t need = 0;
  for (int32_t ci = 0; ci < chunk->n_cols; ci++) {
    const gx_col& c = chunk->cols[ci];
    need += 8;
    if (colNullCount(c) > 0) need += (c.length + 7) / 8;
    if (c.elem_size < 0) need += (int64_t)(c.length + 1) * 8;
    need += colDataBytes(c);
  }
  if (!out || cap < need) return -need;
  uint8_t* p = out;
  for (int32_t ci = 0; ci < chunk->n_cols; ci++) {
    const gx_col& c = chunk->cols[ci];
    uint32_t len = (uint32_t)c.length;
    uint32_t nulls = (uint32_t)colNullCount(c);
    memcpy(p, &len, 4); p += 4;
    memcpy(p, &nulls, 4); p += 4;
    if (nulls > 0) {
      int64_t nb = (c.length + 7) / 8;
      memcpy(p, c.null_bitmap, nb);
      p += nb;
    }
    if (c.elem_size < 0) {
      memcpy(p, c.offsets, (int64_t)(c.length + 1) * 8);
      p += (int64_t)(c.length + 1) * 8;
    }
    int64_t db = colDataBytes(c);
    memcpy(p, c.data, db);
    p += db;
  }
  return p - out;
}

extern "C" int64_t gx_chunk_decode(const uint8_t* buf, int64_t len,
                                   gx_chunk* out) {
  if (!buf || !out) return GX_ERR_INVALID;
  const uint8_t* p = buf;
  const uint8_t* end = buf + len;
  for (int32_t ci = 0; ci < out->n_cols; ci++) {
    gx_col& c = out->cols[ci];
    if (end - p < 8) return GX_ERR_INVALID;
    uint32_t length, nulls;
    memcpy(&length, p, 4); p += 4;
    memcpy(&nulls, p, 4); p += 4;
    c.length = (int32_t)length;
    int64_t nb = (length + 7) / 8;
    if (nulls > 0) {
      if (end - p < nb || !c.null_bitmap) return GX_ERR_INVALID;
      memcpy(c.null_bitmap, p, nb);
      p += nb;
    } else if (c.null_bitmap) {
      memset(c.null_bitmap, 0xFF, nb);  // codec.go:146 setAllNotNull
    }
    int64_t db;
    if (c.elem_size < 0) {
      int64_t ob = (int64_t)(length + 1) * 8;
      if (end - p < ob || !c.offsets || c.offsets_cap < (int64_t)length + 1)
        return GX_ERR_INVALID;
      memcpy(c.offsets, p, ob);
      p += ob;
      db = c.offsets[length];
    } else {
      db = (int64_t)length * c.elem_size;
    }
    if (end - p < db || c.data_cap < db) return GX_ERR_INVALID;
    memcpy(c.data, p, db);
    p += db;
  }
  out->n_rows = out->n_cols > 0 ? out->cols[0].length : 0;
  return p - buf;
}
