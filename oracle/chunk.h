// oracle/chunk.h — CPU restatement of pkg/util/chunk's Column/Chunk.
// ORACLE / TEST INFRASTRUCTURE ONLY.
//
// Layout (column.go:74-82): fixed-width columns hold raw little-endian
// elements in `data` (Int64s()/Decimals()/Times() are reinterpret casts,
// column.go:633-682); the null bitmap is 1 bit per row, LSB-first within each
// byte, 1 = NOT NULL (column.go:225-267); var-len columns use
// offsets[i]..offsets[i+1] into data.
#ifndef ORACLE_CHUNK_H
#define ORACLE_CHUNK_H

#include <cstdint>
#include <cstring>
#include <memory>
#include <string>
#include <vector>

#include "../include/gx_executor.h"
#include "mydecimal.h"

namespace oracle {

struct Column {
  int type = GX_TYPE_I64;   // gx type
  int frac = 0;             // decimal field frac (metadata)
  int length = 0;
  std::vector<uint8_t> nullBitmap;  // 1 = NOT NULL
  std::vector<int64_t> offsets;     // varlen only (length+1, offsets[0]=0)
  std::vector<uint8_t> data;

  bool isVarlen() const { return type == GX_TYPE_STRING; }
  int elemSize() const {
    switch (type) {
      case GX_TYPE_DECIMAL: return 40;
      case GX_TYPE_STRING: return -1;
      default: return 8;
    }
  }
  void reset() {
    length = 0;
    nullBitmap.clear();
    data.clear();
    if (isVarlen()) offsets.assign(1, 0);
    else offsets.clear();
  }
  bool isNull(int i) const {  // column.go:234
    return (nullBitmap[i / 8] & (1u << (i & 7))) == 0;
  }
  void appendNullBitmap(bool notNull) {  // column.go:255
    int idx = length >> 3;
    if (idx >= (int)nullBitmap.size()) nullBitmap.push_back(0);
    if (notNull) nullBitmap[idx] |= (uint8_t)(1u << (length & 7));
  }
  void appendNull() {
    appendNullBitmap(false);
    if (isVarlen()) offsets.push_back(offsets.back());
    else data.insert(data.end(), elemSize(), 0);
    length++;
  }
  void appendI64(int64_t v) {
    appendNullBitmap(true);
    const uint8_t* p = (const uint8_t*)&v;
    data.insert(data.end(), p, p + 8);
    length++;
  }
  void appendU64(uint64_t v) { appendI64((int64_t)v); }
  void appendF64(double v) {
    appendNullBitmap(true);
    const uint8_t* p = (const uint8_t*)&v;
    data.insert(data.end(), p, p + 8);
    length++;
  }
  void appendDecimal(const MyDecimal& d) {
    appendNullBitmap(true);
    const uint8_t* p = (const uint8_t*)&d;
    data.insert(data.end(), p, p + 40);
    length++;
  }
  void appendBytes(const void* p, size_t n) {
    appendNullBitmap(true);
    data.insert(data.end(), (const uint8_t*)p, (const uint8_t*)p + n);
    offsets.push_back((int64_t)data.size());
    length++;
  }
  int64_t getI64(int i) const { int64_t v; std::memcpy(&v, &data[i * 8], 8); return v; }
  uint64_t getU64(int i) const { return (uint64_t)getI64(i); }
  double getF64(int i) const { double v; std::memcpy(&v, &data[i * 8], 8); return v; }
  const MyDecimal* getDecimal(int i) const {
    return reinterpret_cast<const MyDecimal*>(&data[i * 40]);
  }
  const uint8_t* getBytes(int i, int* len) const {
    *len = (int)(offsets[i + 1] - offsets[i]);
    return data.data() + offsets[i];
  }
  std::string getStr(int i) const {
    int n; const uint8_t* p = getBytes(i, &n);
    return std::string((const char*)p, n);
  }
  // generic cell copy from another column of the same type
  void appendFrom(const Column& src, int row) {
    if (src.isNull(row)) { appendNull(); return; }
    if (isVarlen()) {
      int n; const uint8_t* p = src.getBytes(row, &n);
      appendBytes(p, n);
    } else {
      appendNullBitmap(true);
      int es = elemSize();
      data.insert(data.end(), src.data.begin() + (size_t)row * es,
                  src.data.begin() + (size_t)(row + 1) * es);
      length++;
    }
  }
  // zero-copy view over a caller gx_col (copies into owned storage)
  static Column fromGx(const gx_col& c, int type, int frac) {
    Column col;
    col.type = type;
    col.frac = frac;
    col.length = c.length;
    int nb = (c.length + 7) / 8;
    if (c.null_bitmap) col.nullBitmap.assign(c.null_bitmap, c.null_bitmap + nb);
    else col.nullBitmap.assign(nb, 0xFF);
    if (type == GX_TYPE_STRING) {
      col.offsets.assign(c.offsets, c.offsets + c.length + 1);
      col.data.assign((uint8_t*)c.data, (uint8_t*)c.data + col.offsets.back());
    } else {
      size_t n = (size_t)c.length * col.elemSize();
      col.data.assign((uint8_t*)c.data, (uint8_t*)c.data + n);
    }
    return col;
  }
};

struct Chunk {
  std::vector<Column> cols;
  int numRows() const { return cols.empty() ? 0 : cols[0].length; }
  void reset() {
    for (auto& c : cols) c.reset();
  }
};

}  // namespace oracle
#endif
