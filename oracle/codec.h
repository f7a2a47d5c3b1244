// oracle/codec.h — CPU restatement of pkg/util/codec key encodings.
// ORACLE / TEST INFRASTRUCTURE ONLY.
//
// HashGroupKey (codec.go:1791-1879): per-EvalType, compact (comparable=false)
// flags — int -> varintFlag+EncodeVarint, real -> floatFlag+EncodeFloat,
// decimal -> decimalFlag+EncodeDecimal (decimal.go:25-35), datetime ->
// uintFlag+EncodeUint(ToPackedUint) (codec.go:212-231), string ->
// compactBytesFlag+EncodeCompactBytes of the collator key, NULL -> NilFlag.
#ifndef ORACLE_CODEC_H
#define ORACLE_CODEC_H

#include <cstdint>
#include <string>
#include <vector>

#include "chunk.h"
#include "mydecimal.h"

namespace oracle {

// codec.go:42-54
constexpr uint8_t kNilFlag = 0, kBytesFlag = 1, kCompactBytesFlag = 2,
                  kIntFlag = 3, kUintFlag = 4, kFloatFlag = 5, kDecimalFlag = 6,
                  kDurationFlag = 7, kVarintFlag = 8, kUvarintFlag = 9;

void EncodeUint64BE(std::string& b, uint64_t v);          // number.go:82
void EncodeVarint(std::string& b, int64_t v);             // number.go:123 (zigzag LEB128)
void EncodeUvarint(std::string& b, uint64_t v);           // number.go:144
void EncodeFloatCmp(std::string& b, double v);            // float.go:23,44
uint64_t TimeToPackedUint(uint64_t timeVal);              // time.go:646
int32_t EncodeDecimalKey(std::string& b, const MyDecimal& d, int precision, int frac);  // decimal.go:25
std::string BinCollatorKey(const uint8_t* s, int len);    // utf8mb4_bin PAD SPACE: trim trailing ' '

// Appends the per-row group-key bytes for column `col` to keys[i].
// loc/timezone: N/A (no timestamp columns on this path).
int32_t HashGroupKeyCol(const Column& col, std::vector<std::string>& keys);

}  // namespace oracle
#endif
