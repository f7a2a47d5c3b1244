// tidb_amd/csrc/gx_decimal.h — PRODUCT host-side MyDecimal.
//
// The product engine's own restatement of MySQL/TiDB decimal semantics
// (reference pkg/types/mydecimal.go; same citations as the oracle's copy).
// It deliberately shares NO build artifacts with oracle/ — the oracle is test
// infrastructure and the product must not link it. Host-side use only:
// plan constants, result finalization (avg div/round), chunk emission; the
// GPU kernels compute in fixed-point int64/int128 (see gx_kernels.hip) and
// their equivalence to this arithmetic is covered by the golden-vector and
// parity suites in tests/.

//
// Follows, function-for-function: /root/reference/pkg/types/mydecimal.go
//   struct layout        mydecimal.go:233-248 (MyDecimalStructSize = 40)
//   DecimalAdd/doAdd     mydecimal.go:1694,1898
//   DecimalSub/doSub     mydecimal.go:1705,1737
//   DecimalMul           mydecimal.go:2052
//   DecimalDiv/doDivMod  mydecimal.go:2178,2214
//   Round                mydecimal.go:822
//   ToString/FromString  mydecimal.go:328,406
//   WriteBin/FromBin     mydecimal.go:1295,1476
//   ToHashKey            mydecimal.go:1416
//   Compare              mydecimal.go:1634
#ifndef GXP_DECIMAL_H
#define GXP_DECIMAL_H

#include <cstdint>
#include <string>

namespace gxp {

constexpr int kDigitsPerWord = 9;
constexpr int kWordSize = 4;
constexpr int kMaxWordBufLen = 9;
constexpr int32_t kWordBase = 1000000000;
constexpr int32_t kWordMax = kWordBase - 1;
constexpr int kNotFixedDec = 31;
constexpr int kMaxDecimalScale = 30;   // mysql.MaxDecimalScale
constexpr int kDivFracIncr = 4;        // vardef.DefDivPrecisionIncrement

enum RoundMode : int32_t {
  ModeCeiling = 0,
  ModeHalfUp = 5,
  ModeTruncate = 10,
};

// status codes match gx_executor.h
enum DecErr : int32_t {
  E_OK = 0,
  E_TRUNCATED = 1,
  E_OVERFLOW = 2,
  E_DIV_ZERO = 3,
  E_BAD_NUMBER = 4,
};

// Bit-compatible with the Go struct (40 bytes): int8 digitsInt, int8
// digitsFrac, int8 resultFrac, bool negative, int32 wordBuf[9].
struct MyDecimal {
  int8_t digitsInt = 0;
  int8_t digitsFrac = 0;
  int8_t resultFrac = 0;
  uint8_t negative = 0;
  int32_t wordBuf[kMaxWordBufLen] = {0};

  bool IsZero() const;
  bool IsNegative() const { return negative != 0; }
  void RemoveLeadingZeros(int* wordIdx, int* digits) const;  // mydecimal.go:289
  void RemoveTrailingZeros(int* lastWordIdx, int* digits) const;  // :305
  std::string ToString() const;                // :328 (no rounding)
  std::string DisplayString() const;           // String(), :277 (round to resultFrac)
  int32_t FromString(const char* s, int len);  // :406
  MyDecimal& FromInt(int64_t v);               // :1068
  MyDecimal& FromUint(uint64_t v);             // :1080
  int32_t ToInt(int64_t* out) const;           // :1100
  int32_t Round(MyDecimal* to, int frac, RoundMode mode) const;  // :822
  int32_t Shift(int shift);                    // :555
  int32_t WriteBin(int precision, int frac, uint8_t* buf, int* written) const; // :1295
  int32_t FromBin(const uint8_t* bin, int binLenAvail, int precision, int frac,
                  int* binSize);               // :1476
  int32_t ToHashKey(uint8_t* buf, int* written) const;  // :1416
  int Compare(const MyDecimal& other) const;   // :1634
  void PrecisionAndFrac(int* precision, int* frac) const;  // :1453
};

static_assert(sizeof(MyDecimal) == 40, "MyDecimal must be 40 bytes");

int32_t DecimalAdd(const MyDecimal* from1, const MyDecimal* from2, MyDecimal* to);
int32_t DecimalSub(const MyDecimal* from1, const MyDecimal* from2, MyDecimal* to);
int32_t DecimalMul(const MyDecimal* from1, const MyDecimal* from2, MyDecimal* to);
int32_t DecimalDiv(const MyDecimal* from1, const MyDecimal* from2, MyDecimal* to,
                   int fracIncr);
MyDecimal DecimalNeg(const MyDecimal& from);
int32_t DecimalBinSize(int precision, int frac, int* size);  // :1582
int DigitsToWords(int digits);

}  // namespace gxp
#endif
