// oracle/codec.cpp — CPU restatement of pkg/util/codec key encodings.
// ORACLE / TEST INFRASTRUCTURE ONLY.
#include "codec.h"

#include "core_time.h"

namespace oracle {

void EncodeUint64BE(std::string& b, uint64_t v) {
  for (int i = 7; i >= 0; i--) b.push_back((char)(uint8_t)(v >> (i * 8)));
}

// Go binary.PutUvarint: little-endian base-128
void EncodeUvarint(std::string& b, uint64_t v) {
  while (v >= 0x80) {
    b.push_back((char)(uint8_t)(v | 0x80));
    v >>= 7;
  }
  b.push_back((char)(uint8_t)v);
}

// Go binary.PutVarint: zigzag then uvarint
void EncodeVarint(std::string& b, int64_t v) {
  uint64_t ux = (uint64_t)v << 1;
  if (v < 0) ux = ~ux;
  EncodeUvarint(b, ux);
}

// float.go:23 encodeFloatToCmpUint64 + EncodeUint
void EncodeFloatCmp(std::string& b, double v) {
  uint64_t u;
  static_assert(sizeof(u) == sizeof(v), "");
  std::memcpy(&u, &v, 8);
  constexpr uint64_t signMask = 0x8000000000000000ULL;
  if (v >= 0)
    u |= signMask;
  else
    u = ~u;
  EncodeUint64BE(b, u);
}

// time.go:646 ToPackedUint
uint64_t TimeToPackedUint(uint64_t t) {
  if ((t & kCoreTimeMask) == 0) return 0;
  uint64_t year = (t >> kYearOff) & ((1ULL << 14) - 1);
  uint64_t month = (t >> kMonthOff) & 15;
  uint64_t day = (t >> kDayOff) & 31;
  uint64_t hour = (t >> kHourOff) & 31;
  uint64_t minute = (t >> kMinuteOff) & 63;
  uint64_t sec = (t >> kSecondOff) & 63;
  uint64_t micro = (t >> kMicroOff) & ((1ULL << 20) - 1);
  uint64_t ymd = ((year * 13 + month) << 5) | day;
  uint64_t hms = (hour << 12) | (minute << 6) | sec;
  return ((ymd << 17 | hms) << 24) | micro;
}

// decimal.go:25 EncodeDecimal: [prec byte][frac byte][WriteBin]
int32_t EncodeDecimalKey(std::string& b, const MyDecimal& d, int precision, int frac) {
  if (precision == 0) d.PrecisionAndFrac(&precision, &frac);
  if (frac > kMaxDecimalScale) frac = kMaxDecimalScale;
  b.push_back((char)(uint8_t)precision);
  b.push_back((char)(uint8_t)frac);
  uint8_t bin[40];
  int written = 0;
  int32_t err = d.WriteBin(precision, frac, bin, &written);
  if (err != E_OK && err != E_TRUNCATED) return err;
  b.append((const char*)bin, written);
  return E_OK;
}

// collate.go:272 truncateTailingSpace (utf8mb4_bin is PAD SPACE)
std::string BinCollatorKey(const uint8_t* s, int len) {
  int i = len - 1;
  while (i >= 0 && s[i] == ' ') i--;
  return std::string((const char*)s, i + 1);
}

int32_t HashGroupKeyCol(const Column& col, std::vector<std::string>& keys) {
  int n = col.length;
  switch (col.type) {
    case GX_TYPE_I64:
      for (int i = 0; i < n; i++) {
        if (col.isNull(i)) keys[i].push_back((char)kNilFlag);
        else {
          keys[i].push_back((char)kVarintFlag);
          EncodeVarint(keys[i], col.getI64(i));
        }
      }
      break;
    case GX_TYPE_F64:
      for (int i = 0; i < n; i++) {
        if (col.isNull(i)) keys[i].push_back((char)kNilFlag);
        else {
          keys[i].push_back((char)kFloatFlag);
          EncodeFloatCmp(keys[i], col.getF64(i));
        }
      }
      break;
    case GX_TYPE_DECIMAL:
      for (int i = 0; i < n; i++) {
        if (col.isNull(i)) keys[i].push_back((char)kNilFlag);
        else {
          keys[i].push_back((char)kDecimalFlag);
          // ft.GetFlen()/GetDecimal() == 0 for computed columns => use
          // PrecisionAndFrac (EncodeDecimal's precision==0 path)
          int32_t err = EncodeDecimalKey(keys[i], *col.getDecimal(i), 0, 0);
          if (err != E_OK) return err;
        }
      }
      break;
    case GX_TYPE_TIME:
      for (int i = 0; i < n; i++) {
        if (col.isNull(i)) keys[i].push_back((char)kNilFlag);
        else {
          keys[i].push_back((char)kUintFlag);
          EncodeUint64BE(keys[i], TimeToPackedUint(col.getU64(i)));
        }
      }
      break;
    case GX_TYPE_STRING:
      for (int i = 0; i < n; i++) {
        if (col.isNull(i)) keys[i].push_back((char)kNilFlag);
        else {
          int len;
          const uint8_t* p = col.getBytes(i, &len);
          std::string key = BinCollatorKey(p, len);
          keys[i].push_back((char)kCompactBytesFlag);
          EncodeVarint(keys[i], (int64_t)key.size());
          keys[i].append(key);
        }
      }
      break;
    default:
      return E_BAD_NUMBER;
  }
  return E_OK;
}

}  // namespace oracle
