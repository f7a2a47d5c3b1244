// oracle/core_time.h — CPU restatement of TiDB's packed time representation.
// ORACLE / TEST INFRASTRUCTURE ONLY.
//
// Follows /root/reference/pkg/types/core_time.go and time.go:
//   CoreTime bitfield: year@50:14 month@46:4 day@41:5 hour@36:5 minute@30:6
//                      second@24:6 microsecond@4:20 fspTt@0:4
//                      (time.go:235-251)
//   Time = CoreTime u64 with fspTt tag in the low 4 bits (time.go:224,266-283);
//   fspTt == 0b1110 marks TypeDate.
//   compareTime (core_time.go:256): datetimeToUint64 then microseconds —
//   because the fields are ordered high-to-low, this is equivalent to
//   comparing (u64 & ~0xF).
#ifndef ORACLE_CORE_TIME_H
#define ORACLE_CORE_TIME_H

#include <cstdint>

namespace oracle {

constexpr uint64_t kYearOff = 50, kMonthOff = 46, kDayOff = 41, kHourOff = 36,
                   kMinuteOff = 30, kSecondOff = 24, kMicroOff = 4;
constexpr uint64_t kFspTtMask = 0xF;
constexpr uint64_t kCoreTimeMask = ~kFspTtMask;
constexpr uint64_t kFspTtForDate = 0b1110;

// core_time.go:190 FromDate (packs without validation)
inline uint64_t CoreTimeFromDate(int year, int month, int day, int hour,
                                 int minute, int second, int microsecond) {
  uint64_t v = 0;
  v |= ((uint64_t)microsecond << kMicroOff) & (((1ULL << 20) - 1) << kMicroOff);
  v |= ((uint64_t)second << kSecondOff) & (((1ULL << 6) - 1) << kSecondOff);
  v |= ((uint64_t)minute << kMinuteOff) & (((1ULL << 6) - 1) << kMinuteOff);
  v |= ((uint64_t)hour << kHourOff) & (((1ULL << 5) - 1) << kHourOff);
  v |= ((uint64_t)day << kDayOff) & (((1ULL << 5) - 1) << kDayOff);
  v |= ((uint64_t)month << kMonthOff) & (((1ULL << 4) - 1) << kMonthOff);
  v |= ((uint64_t)year << kYearOff) & (((1ULL << 14) - 1) << kYearOff);
  return v;
}

// time.go:266 NewTime for TypeDate: fspTt = 0b1110
inline uint64_t TimeFromDate(int year, int month, int day) {
  return (CoreTimeFromDate(year, month, day, 0, 0, 0, 0) & kCoreTimeMask) | kFspTtForDate;
}

// time.go:266 NewTime general: type_and_fsp = (tp == Timestamp) | fsp<<1; for
// datetime pass fsp<<1.
inline uint64_t TimeFromDatetime(int year, int month, int day, int hour,
                                 int minute, int second, int microsecond,
                                 int fspTt) {
  return (CoreTimeFromDate(year, month, day, hour, minute, second, microsecond) &
          kCoreTimeMask) | ((uint64_t)fspTt & kFspTtMask);
}

// core_time.go:256 compareTime. Field order makes this a masked u64 compare.
inline int CompareTime(uint64_t a, uint64_t b) {
  uint64_t ma = a & kCoreTimeMask;
  uint64_t mb = b & kCoreTimeMask;
  if (ma < mb) return -1;
  if (ma > mb) return 1;
  return 0;
}

inline int TimeYear(uint64_t t) { return (int)((t >> kYearOff) & ((1ULL << 14) - 1)); }
inline int TimeMonth(uint64_t t) { return (int)((t >> kMonthOff) & ((1ULL << 4) - 1)); }
inline int TimeDay(uint64_t t) { return (int)((t >> kDayOff) & ((1ULL << 5) - 1)); }

// datetimeToUint64, core_time.go:354 (used by key codec EncodeMySQLTime path)
inline uint64_t DatetimeToUint64(uint64_t t) {
  return (uint64_t)TimeYear(t) * 10000000000ULL +
         (uint64_t)TimeMonth(t) * 100000000ULL +
         (uint64_t)TimeDay(t) * 1000000ULL +
         ((t >> kHourOff) & 31) * 10000ULL +
         ((t >> kMinuteOff) & 63) * 100ULL + ((t >> kSecondOff) & 63);
}

}  // namespace oracle
#endif
