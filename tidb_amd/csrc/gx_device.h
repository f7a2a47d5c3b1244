// gx_device.h — shared device-side helpers for the MI355X executor
// kernels. Included by gx_kernels.hip AND by the hipRTC-generated
// specialized kernels (gx_jit.cpp), so both paths compute identically:
// Int128 arithmetic, the MyDecimal raw parser, the raw-fetch machinery,
// group keys, and the LDS/global accumulate helpers.
#pragma once
#ifndef __HIPCC_RTC__
#include <hip/hip_runtime.h>
#endif
#include "gx_common.h"

namespace gxp {

constexpr uint32_t kErrBadDecimal = 1u;
constexpr uint32_t kErrBadKey = 2u;
constexpr uint32_t kErrScale = 4u;
constexpr uint32_t kErrOverflow = 8u;
constexpr uint32_t kErrLdsFull = 16u;
constexpr uint32_t kErrGlobalFull = 32u;
constexpr uint32_t kErrRetryWide = 256u;  // narrow VM overflowed; not an error

__host__ __device__ inline uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ULL;
  uint64_t z = x;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}

struct Int128 {
  uint64_t lo;
  int64_t hi;
};

__device__ inline Int128 i128FromI64(int64_t v) {
  return {(uint64_t)v, v < 0 ? -1 : 0};
}
__device__ inline Int128 i128Add(Int128 a, Int128 b) {
  uint64_t lo = a.lo + b.lo;
  int64_t carry = lo < a.lo;
  return {lo, a.hi + b.hi + carry};
}
__device__ inline Int128 i128Neg(Int128 a) {
  uint64_t lo = ~a.lo + 1;
  int64_t hi = ~a.hi + (lo == 0);
  return {lo, hi};
}
__device__ inline Int128 i128Sub(Int128 a, Int128 b) { return i128Add(a, i128Neg(b)); }
// a(int128) * b(int64) with multiply-only overflow detection (a 128-bit
// divide is a slow software loop on gfx950).
__device__ inline Int128 i128MulI64(Int128 a, int64_t b, bool* ovf) {
  bool neg = false;
  __int128 x = ((__int128)a.hi << 64) | a.lo;
  unsigned __int128 ua;
  if (x < 0) { ua = (unsigned __int128)(-x); neg = true; }
  else ua = (unsigned __int128)x;
  uint64_t ub;
  if (b < 0) { ub = (uint64_t)(-b); neg = !neg; }
  else ub = (uint64_t)b;
  uint64_t alo = (uint64_t)ua;
  uint64_t ahi = (uint64_t)(ua >> 64);
  unsigned __int128 plo = (unsigned __int128)alo * ub;
  unsigned __int128 mid = (plo >> 64) + (unsigned __int128)ahi * ub;
  if (mid >> 63) { *ovf = true; return {0, 0}; }  // |result| >= 2^127
  uint64_t rlo = (uint64_t)plo;
  uint64_t rhi = (uint64_t)mid;
  __int128 sr = ((__int128)(int64_t)rhi << 64) | rlo;
  if (neg) sr = -sr;
  return {(uint64_t)sr, (int64_t)(sr >> 64)};
}

// powers of ten / magic reciprocals. These live in __constant__ memory: a
// per-lane indexed lookup is one (cached) global load. The HOT paths never
// reach them -- engine-precomputed per-instruction constants (insP10/insMagic)
// cover every row whose stored frac matches the declared column frac -- so
// these tables serve only mismatched-frac rows and rescale fallbacks. The
// glds kernel's counted-vmcnt bookkeeping is safe because the staged loop's
// executed path never issues these loads on conforming data.
__constant__ int64_t kP10Tab[19] = {1, 10, 100, 1000, 10000, 100000, 1000000,
                                    10000000, 100000000, 1000000000,
                                    10000000000LL, 100000000000LL,
                                    1000000000000LL, 10000000000000LL,
                                    100000000000000LL, 1000000000000000LL,
                                    10000000000000000LL, 100000000000000000LL,
                                    1000000000000000000LL};
__constant__ uint64_t kDivMagicTab[10] = {
    4611686018427387904ULL, 461168601842738791ULL, 46116860184273880ULL,
    4611686018427388ULL,    461168601842739ULL,    46116860184274ULL,
    4611686018428ULL,       461168601843ULL,       46116860185ULL,
    4611686019ULL};
__device__ inline int64_t kP10(int k) { return kP10Tab[k]; }
__device__ inline uint64_t kDivMagic(int k) { return kDivMagicTab[k]; }
__device__ inline int64_t divP10(uint32_t n, int k) {
  return (int64_t)(((unsigned __int128)n * kDivMagic(k)) >> 62);
}

// ---- value type abstraction (NARROW = int64, WIDE = Int128) ----
template <bool WIDE>
struct VT;

template <>
struct VT<false> {
  using T = int64_t;
  static __device__ T fromI64(int64_t v, bool*) { return v; }
  static __device__ T add(T a, T b, bool* ovf) {
    T r;
    *ovf |= __builtin_add_overflow(a, b, &r);
    return r;
  }
  static __device__ T sub(T a, T b, bool* ovf) {
    T r;
    *ovf |= __builtin_sub_overflow(a, b, &r);
    return r;
  }
  static __device__ T mul(T a, T b, bool* ovf) {
    T r;
    *ovf |= __builtin_mul_overflow(a, b, &r);
    return r;
  }
  static __device__ T scale10(T a, int k, bool* ovf) { return mul(a, kP10(k), ovf); }
  static __device__ T zero() { return 0; }
  static __device__ Int128 toAcc(T v) { return i128FromI64(v); }
  static __device__ int cmp(T a, T b) { return a < b ? -1 : (a > b ? 1 : 0); }
};

template <>
struct VT<true> {
  using T = Int128;
  static __device__ T fromI64(int64_t v, bool*) { return i128FromI64(v); }
  static __device__ T add(T a, T b, bool*) { return i128Add(a, b); }  // 127-bit headroom
  static __device__ T sub(T a, T b, bool*) { return i128Sub(a, b); }
  static __device__ T mul(T a, T b, bool* ovf) {
    bool fits = (b.hi == 0 && (int64_t)b.lo >= 0) || (b.hi == -1 && (int64_t)b.lo < 0);
    if (!fits) { *ovf = true; return {0, 0}; }
    return i128MulI64(a, (int64_t)b.lo, ovf);
  }
  static __device__ T scale10(T a, int k, bool* ovf) { return i128MulI64(a, kP10(k), ovf); }
  static __device__ T zero() { return {0, 0}; }
  static __device__ Int128 toAcc(T v) { return v; }
  static __device__ int cmp(T a, T b) {
    __int128 x = ((__int128)a.hi << 64) | a.lo;
    __int128 y = ((__int128)b.hi << 64) | b.lo;
    return x < y ? -1 : (x > y ? 1 : 0);
  }
};

// parse 16 raw bytes of a 40-byte MyDecimal (digitsInt <= 18, digitsFrac <= 9)
// into units at scale = digitsFrac. Returns false on malformed input
// (kErrBadDecimal) or narrow overflow (kErrRetryWide).
// expFrac/p10exp/magicExp: engine-precomputed for the column's declared
// frac — the common case takes no power-of-ten select tree. expFrac < 0
// forces the generic path.
template <bool WIDE>
__device__ __attribute__((always_inline)) inline bool parseDecimalRaw(ulonglong2 raw, typename VT<WIDE>::T* out,
                                       int* scale, uint32_t* err,
                                       int expFrac = -1, int64_t p10exp = 0,
                                       uint64_t magicExp = 0) {
  uint2 lo2 = {(uint32_t)raw.x, (uint32_t)(raw.x >> 32)};
  uint2 hi2 = {(uint32_t)raw.y, (uint32_t)(raw.y >> 32)};
  uint32_t hdr = lo2.x;
  int digitsInt = (int)(int8_t)(hdr & 0xFF);
  int digitsFrac = (int)(int8_t)((hdr >> 8) & 0xFF);
  bool neg = ((hdr >> 24) & 0xFF) != 0;
  if (digitsInt > 18 || digitsFrac > 9 || digitsInt < 0 || digitsFrac < 0) {
    atomicOr(err, kErrBadDecimal);
    return false;
  }
  int wordsInt = (digitsInt + 8) / 9;
  if (digitsInt == 0) wordsInt = 0;
  int64_t ip = 0;
  if (wordsInt == 1) ip = (int32_t)lo2.y;
  else if (wordsInt == 2) ip = (int64_t)(int32_t)lo2.y * 1000000000 + (int32_t)hi2.x;
  uint32_t fw = wordsInt == 0 ? lo2.y : (wordsInt == 1 ? hi2.x : hi2.y);
  int64_t p10v;
  int64_t fr;
  if (digitsFrac == expFrac) {  // fast path: constants provided
    p10v = p10exp;
    fr = digitsFrac > 0
             ? (int64_t)(((unsigned __int128)fw * magicExp) >> 62)
             : 0;
  } else {
    p10v = kP10(digitsFrac);
    fr = digitsFrac > 0 ? divP10(fw, 9 - digitsFrac) : 0;
  }
  if (WIDE) {
    __int128 units = (__int128)ip * p10v + fr;
    if (neg) units = -units;
    Int128 u = {(uint64_t)units, (int64_t)(units >> 64)};
    *out = *(typename VT<WIDE>::T*)&u;
  } else {
    int64_t units;
    bool ovf = __builtin_mul_overflow(ip, p10v, &units);
    ovf |= __builtin_add_overflow(units, fr, &units);
    if (ovf) {
      atomicOr(err, kErrRetryWide);
      return false;
    }
    if (neg) units = -units;
    *out = *(typename VT<WIDE>::T*)&units;
  }
  *scale = digitsFrac;
  return true;
}

// explicit global (AS1) pointer: loads through the generic desc pointers
// otherwise compile to FLAT loads, which decrement BOTH vmcnt and lgkmcnt --
// every LDS hash-table wait then drains the prefetched row loads and the
// software pipeline serializes.
template <typename T>
__device__ inline const __attribute__((address_space(1))) T* gptr(const void* p) {
  return (const __attribute__((address_space(1))) T*)(uintptr_t)p;
}

template <bool WIDE>
__device__ inline bool loadDecimalUnits(const uint8_t* p, typename VT<WIDE>::T* out,
                                        int* scale, uint32_t* err,
                                        int expFrac = -1, int64_t p10exp = 0,
                                        uint64_t magicExp = 0) {
  // 40-byte stride keeps rows 8-byte aligned: two dwordx2 loads
  ulonglong2 raw;
  raw.x = *gptr<uint64_t>(p);
  raw.y = *gptr<uint64_t>(p + 8);
  return parseDecimalRaw<WIDE>(raw, out, scale, err, expFrac, p10exp, magicExp);
}

// raw per-row fetch buffer: setters use compile-time slot indices (phase A),
// the getter is a wave-uniform switch (runtime-indexed arrays would spill)
struct RawState {
  ulonglong2 s0, s1, s2, s3, s4, s5, s6, s7;
  __device__ ulonglong2 get(int i) const {
    switch (i) {
      case 0: return s0; case 1: return s1; case 2: return s2; case 3: return s3;
      case 4: return s4; case 5: return s5; case 6: return s6; default: return s7;
    }
  }
  __device__ void set(int i, ulonglong2 v) {
    switch (i) {
      case 0: s0 = v; break; case 1: s1 = v; break; case 2: s2 = v; break;
      case 3: s3 = v; break; case 4: s4 = v; break; case 5: s5 = v; break;
      case 6: s6 = v; break; default: s7 = v; break;
    }
  }
};

// 5-slot variant: the register allocator keeps every switch member live, so
// a query using <= 5 fetch slots (TPC-H Q1: 4 decimals + 1 date) runs with a
// 2x5x16B pipelined raw footprint instead of 2x8x16B -- the difference
// between occupancy 3 and 4 waves/SIMD.
struct RawState5 {
  ulonglong2 s0, s1, s2, s3, s4;
  __device__ ulonglong2 get(int i) const {
    switch (i) {
      case 0: return s0; case 1: return s1; case 2: return s2;
      case 3: return s3; default: return s4;
    }
  }
  __device__ void set(int i, ulonglong2 v) {
    switch (i) {
      case 0: s0 = v; break; case 1: s1 = v; break; case 2: s2 = v; break;
      case 3: s3 = v; break; default: s4 = v; break;
    }
  }
};

// phase A: issue every fetch for one row, no consumption (loads overlap).
// The loop is unrolled over the compile-time slot bound so every raw.set has
// a literal index — a runtime-indexed store would be re-rolled into scratch.
template <typename RAWT>
__device__ __attribute__((always_inline)) inline void fetchRow(const DevTable& tab, const FetchDesc* fetch, int nFetch, int64_t row, RAWT& raw) {
#pragma unroll
  for (int f = 0; f < kMaxFetch; f++) {
    if (f >= nFetch) break;
    const FetchDesc& fd = fetch[f];
    if (fd.kind == FETCH_B1) continue;  // staged-variant-only stream
    const DevCol& c = tab.cols[fd.col];
    ulonglong2 v;
    if (fd.kind == FETCH_8B) {
      v.x = gptr<uint64_t>(c.data)[row];
      v.y = 0;
    } else if (fd.kind == FETCH_8B_CHAR2 || fd.kind == FETCH_CHAR2) {
      v.x = fd.kind == FETCH_8B_CHAR2 ? gptr<uint64_t>(c.data)[row] : 0;
      const DevTable& t2 = tab;
      uint64_t chars = (uint64_t)gptr<uint8_t>(t2.cols[fd.ldsOff & 0xFF].data)[row];
      if (((fd.ldsOff >> 16) & 0xFF) > 1)
        chars |= (uint64_t)gptr<uint8_t>(t2.cols[(fd.ldsOff >> 8) & 0xFF].data)[row] << 8;
      v.y = chars;
    } else if (fd.kind == FETCH_DEC16) {
      const uint8_t* p = (const uint8_t*)c.data + row * 40;
      v.x = *gptr<uint64_t>(p);
      v.y = *gptr<uint64_t>(p + 8);
    } else {  // FETCH_OFFSETS
      v.x = (uint64_t)gptr<int64_t>(c.offsets)[row];
      v.y = (uint64_t)gptr<int64_t>(c.offsets)[row + 1];
    }
    raw.set(f, v);
  }
}

__device__ inline bool colIsNull(const DevCol& c, int64_t row) {
  if (!c.hasNulls || c.nullBitmap == nullptr) return false;
  return ((gptr<uint8_t>(c.nullBitmap)[row >> 3] >> (row & 7)) & 1) == 0;
}

__device__ inline int cmpResult(int c, int op) {
  switch (op) {
    case 0: return c < 0;   // LT
    case 1: return c <= 0;  // LE
    case 2: return c > 0;   // GT
    case 3: return c >= 0;  // GE
    case 4: return c == 0;  // EQ
    default: return c != 0; // NE
  }
}

// VM state: named registers (runtime-indexed arrays would spill to scratch;
// the instruction stream is wave-uniform so these switches are cheap scalar
// branches)
template <bool WIDE>
struct VmState {
  using T = typename VT<WIDE>::T;
  T r0, r1, r2, r3, r4, r5, r6, r7, r8, r9, r10, r11;
  uint32_t nullBits;
  __device__ T get(int i) const {
    switch (i) {
      case 0: return r0; case 1: return r1; case 2: return r2; case 3: return r3;
      case 4: return r4; case 5: return r5; case 6: return r6; case 7: return r7;
      case 8: return r8; case 9: return r9; case 10: return r10; default: return r11;
    }
  }
  __device__ void set(int i, T v) {
    switch (i) {
      case 0: r0 = v; break; case 1: r1 = v; break; case 2: r2 = v; break;
      case 3: r3 = v; break; case 4: r4 = v; break; case 5: r5 = v; break;
      case 6: r6 = v; break; case 7: r7 = v; break; case 8: r8 = v; break;
      case 9: r9 = v; break; case 10: r10 = v; break; default: r11 = v; break;
    }
  }
  __device__ bool isNull(int i) const { return (nullBits >> i) & 1; }
  __device__ void setNull(int i, bool n) {
    nullBits = (nullBits & ~(1u << i)) | ((uint32_t)n << i);
  }
};

// 14-register VM state for wide projections (kMaxVmRegs > 12): only the
// kernel variant the launcher picks for such plans pays the extra live
// registers -- common plans keep the 12-register footprint above
template <bool WIDE>
struct VmState14 {
  using T = typename VT<WIDE>::T;
  T r0, r1, r2, r3, r4, r5, r6, r7, r8, r9, r10, r11, r12, r13;
  uint32_t nullBits;
  __device__ T get(int i) const {
    switch (i) {
      case 0: return r0; case 1: return r1; case 2: return r2; case 3: return r3;
      case 4: return r4; case 5: return r5; case 6: return r6; case 7: return r7;
      case 8: return r8; case 9: return r9; case 10: return r10;
      case 11: return r11; case 12: return r12; default: return r13;
    }
  }
  __device__ void set(int i, T v) {
    switch (i) {
      case 0: r0 = v; break; case 1: r1 = v; break; case 2: r2 = v; break;
      case 3: r3 = v; break; case 4: r4 = v; break; case 5: r5 = v; break;
      case 6: r6 = v; break; case 7: r7 = v; break; case 8: r8 = v; break;
      case 9: r9 = v; break; case 10: r10 = v; break; case 11: r11 = v; break;
      case 12: r12 = v; break; default: r13 = v; break;
    }
  }
  __device__ bool isNull(int i) const { return (nullBits >> i) & 1; }
  __device__ void setNull(int i, bool n) {
    nullBits = (nullBits & ~(1u << i)) | ((uint32_t)n << i);
  }
};

// pack the group key (see GroupKeyDesc comment); offsets/values come from the
// batched raw fetch
template <typename RAWT>
__device__ __attribute__((always_inline)) inline bool makeGroupKey(const FusedQueryDesc& d, int64_t row,
                                    const RAWT& raw, uint64_t* keyOut,
                                    uint32_t* err) {
  uint64_t key = 0;
  for (int k = 0; k < d.gkey.nCols; k++) {
    // computed keys (col < 0) never touch the column; clamp the index
    const DevCol& c = d.table.cols[d.gkey.col[k] < 0 ? 0 : d.gkey.col[k]];
    uint32_t lane;
    if (colIsNull(c, row)) {
      lane = 0xFF000000u;
    } else if (d.gkey.kind[k] == 2) {
      // dense char(1): data[row] is the value; PAD SPACE trims a lone space.
      // Prefetched with the row's grouped fetch when a rawSlot is assigned
      // (the direct load here is a serial dependency the pipeline can't hide).
      uint8_t b = d.gkey.rawSlot[k] >= 0
                      ? (uint8_t)(raw.get(d.gkey.rawSlot[k]).y >> (8 * k))
                      : gptr<uint8_t>(c.data)[row];
      lane = b == ' ' ? 0u : ((1u << 24) | b);
    } else if (d.gkey.kind[k] == 0) {
      ulonglong2 off = raw.get(d.gkey.slot[k]);
      int64_t s = (int64_t)off.x, e = (int64_t)off.y;
      // utf8mb4_bin PAD SPACE: trim trailing spaces (collate.go:272)
      auto p = gptr<uint8_t>(c.data);
      while (e > s && p[e - 1] == ' ') e--;
      int64_t len = e - s;
      if (len > 3) { atomicOr(err, kErrBadKey); return false; }
      lane = (uint32_t)len << 24;
      for (int64_t j = 0; j < len; j++) lane |= (uint32_t)p[s + j] << (8 * j);
    } else {
      int64_t v = (int64_t)raw.get(d.gkey.slot[k]).x;
      if (v < 0 || v > 0x7FFFFFFF) { atomicOr(err, kErrBadKey); return false; }
      lane = (uint32_t)v;
    }
    key |= (uint64_t)lane << (32 * k);
  }
  if (d.gkey.nCols == 0) key = 0;
  if (key == kEmptyKey) key = kEmptyKey - 1;  // avoid the sentinel
  *keyOut = key;
  return true;
}

// ---- serialized wide group keys (GroupKeyDesc wideMode) ----
// Exact arbitrary-column grouping: a 64-bit value hash probes the global
// table, equality is verified against the group's key record (hash collisions
// linear-probe on). The record is written COMPLETELY and device-fenced
// BEFORE its (hash32|recordIdx) publishes via CAS, so no reader ever spins
// on a half-written record (wave64 lockstep makes intra-wave spinning a
// deadlock; this protocol never waits).

// trimmed varlen string metadata: byte start, PAD-SPACE-trimmed length and
// the 16-byte inline prefix as two LE words (utf8mb4_bin, collate.go:272)
__device__ inline void wideStrMeta(const DevCol& c, ulonglong2 off,
                                   int64_t* sOut, uint64_t* lenOut,
                                   uint64_t* w0, uint64_t* w1) {
  int64_t s = (int64_t)off.x, e = (int64_t)off.y;
  auto p = gptr<uint8_t>(c.data);
  while (e > s && p[e - 1] == ' ') e--;
  uint64_t len = (uint64_t)(e - s);
  uint64_t a = 0, b = 0;
  int64_t m = (int64_t)(len < 16 ? len : 16);
  for (int64_t t = 0; t < m; t++) {
    uint64_t byte = p[s + t];
    if (t < 8) a |= byte << (8 * t);
    else b |= byte << (8 * (t - 8));
  }
  *sOut = s;
  *lenOut = len;
  *w0 = a;
  *w1 = b;
}

template <bool WIDE, typename VMT, typename RAWT>
__device__ inline uint64_t wideKeyHash(const FusedQueryDesc& d, int64_t row,
                                       const RAWT& raw, const VMT& vm,
                                       uint64_t* nullBitsOut) {
  uint64_t h = 0x243F6A8885A308D3ULL;
  uint64_t nb = 0;
  for (int k = 0; k < d.gkey.nCols; k++) {
    // computed keys (col < 0) never touch the column; clamp the index
    const DevCol& c = d.table.cols[d.gkey.col[k] < 0 ? 0 : d.gkey.col[k]];
    int kind = d.gkey.kind[k];
    bool nul = kind == 4 ? vm.isNull(d.gkey.slot[k]) : colIsNull(c, row);
    if (nul) {
      nb |= 1ULL << k;
      h = splitmix64(h ^ 0xA5A5F00DFEEDBEEFULL);
      continue;
    }
    if (kind == 3) {
      h = splitmix64(h ^ raw.get(d.gkey.slot[k]).x);
    } else if (kind == 4) {
      Int128 u = VT<WIDE>::toAcc(vm.get(d.gkey.slot[k]));
      h = splitmix64(h ^ u.lo);
      h = splitmix64(h ^ (uint64_t)u.hi);
    } else {  // kind 5: varlen string
      ulonglong2 off = raw.get(d.gkey.slot[k]);
      int64_t s = (int64_t)off.x, e = (int64_t)off.y;
      auto p = gptr<uint8_t>(c.data);
      while (e > s && p[e - 1] == ' ') e--;
      for (int64_t j = s; j < e; j += 8) {
        uint64_t w = 0;
        int64_t m = e - j < 8 ? e - j : 8;
        for (int64_t t = 0; t < m; t++) w |= (uint64_t)p[j + t] << (8 * t);
        h = splitmix64(h ^ w);
      }
      h = splitmix64(h ^ ((uint64_t)(e - s) * 0x9E3779B97F4A7C15ULL + 1));
    }
  }
  *nullBitsOut = nb;
  return h;
}

template <bool WIDE, typename VMT, typename RAWT>
__device__ inline void wideKeyWrite(const FusedQueryDesc& d, uint8_t* rec,
                                    int64_t row, const RAWT& raw,
                                    const VMT& vm, uint64_t nullBits) {
  uint64_t* r64 = (uint64_t*)rec;
  r64[0] = nullBits;
  r64[1] = (uint64_t)row;
  for (int k = 0; k < d.gkey.nCols; k++) {
    uint64_t* f = (uint64_t*)(rec + 16 + 24 * k);
    f[0] = f[1] = f[2] = 0;  // deterministic record bytes
    if ((nullBits >> k) & 1) continue;
    int kind = d.gkey.kind[k];
    if (kind == 3) {
      f[0] = raw.get(d.gkey.slot[k]).x;
    } else if (kind == 4) {
      Int128 u = VT<WIDE>::toAcc(vm.get(d.gkey.slot[k]));
      f[0] = u.lo;
      f[1] = (uint64_t)u.hi;
    } else {
      const DevCol& c = d.table.cols[d.gkey.col[k] < 0 ? 0 : d.gkey.col[k]];
      int64_t s;
      wideStrMeta(c, raw.get(d.gkey.slot[k]), &s, &f[0], &f[1], &f[2]);
    }
  }
}

template <bool WIDE, typename VMT, typename RAWT>
__device__ inline bool wideKeyMatches(const FusedQueryDesc& d,
                                      const uint8_t* rec, int64_t row,
                                      const RAWT& raw, const VMT& vm,
                                      uint64_t nullBits) {
  const uint64_t* r64 = (const uint64_t*)rec;
  if (r64[0] != nullBits) return false;
  for (int k = 0; k < d.gkey.nCols; k++) {
    if ((nullBits >> k) & 1) continue;
    const uint64_t* f = (const uint64_t*)(rec + 16 + 24 * k);
    int kind = d.gkey.kind[k];
    if (kind == 3) {
      if (f[0] != raw.get(d.gkey.slot[k]).x) return false;
    } else if (kind == 4) {
      Int128 u = VT<WIDE>::toAcc(vm.get(d.gkey.slot[k]));
      if (f[0] != u.lo || f[1] != (uint64_t)u.hi) return false;
    } else {
      const DevCol& c = d.table.cols[d.gkey.col[k] < 0 ? 0 : d.gkey.col[k]];
      int64_t s;
      uint64_t len, w0, w1;
      wideStrMeta(c, raw.get(d.gkey.slot[k]), &s, &len, &w0, &w1);
      if (f[0] != len || f[1] != w0 || f[2] != w1) return false;
      if (len > 16) {  // tail compares against the owner row's bytes
        int64_t owner = (int64_t)r64[1];
        int64_t os = gptr<int64_t>(c.offsets)[owner];
        auto p = gptr<uint8_t>(c.data);
        for (uint64_t j = 16; j < len; j++)
          if (p[s + (int64_t)j] != p[os + (int64_t)j]) return false;
      }
    }
  }
  return true;
}

// probe/insert: returns the global table slot owning this row's group, or
// false with the error flag set (table or record store full -> the engine
// grows 8x and reruns, the agg_spill.go partition-growth analog).
// The slow path's per-row ACQUIRE load is the L1-invalidating cost, so the
// kernel carries a small LDS RESOLUTION CACHE (hash -> verified slot): a
// cache hit re-verifies the record with plain loads (the bytes entered this
// CU's L1 under the acquire that filled the entry) — exact, no fence.
constexpr int kWkCache = 512;
// slot claimed, record still being written. A published pack's low 32 bits
// are a record index < recCap (<= 2^27), so the all-ones low word can never
// be a real pack and probers can always tell the states apart.
constexpr uint64_t kWkLocked = ~1ULL;

template <bool WIDE, typename VMT, typename RAWT>
__device__ inline bool wideKeyResolve(const FusedQueryDesc& d, int64_t row,
                                      const RAWT& raw, const VMT& vm,
                                      uint64_t h, uint64_t nullBits,
                                      uint32_t* slotOut) {
  // CLAIM-FIRST insert: CAS the slot to kWkLocked and only the winner
  // consumes a record index — so recCursor counts exactly the distinct keys
  // and a startup burst of racers on the same new key cannot leak records
  // and spuriously trip kErrGlobalFull (reserve-then-publish leaked one
  // record per CAS loser: ~1e5 on a 60M-row low-NDV table, blowing recCap
  // and costing full-pass grow retries). Losers re-read the slot; the loop
  // never spins inside a divergent branch (each re-read is a fresh
  // iteration, so a same-wave winner's publish completes first) and the
  // probe cap bounds it absolutely.
  uint32_t gmask = (1u << d.globalGroupsLog2) - 1;
  uint32_t slot = (uint32_t)h & gmask;
  uint32_t h32 = (uint32_t)(h >> 32);
  for (uint32_t probe = 0;; probe++) {
    if (probe > gmask) {
      atomicOr(d.errorFlag, kErrGlobalFull);
      return false;
    }
    uint64_t cur = __hip_atomic_load(
        (unsigned long long*)&d.globalTable[slot].key, __ATOMIC_ACQUIRE,
        __HIP_MEMORY_SCOPE_AGENT);
    if (cur == kEmptyKey) {
      unsigned long long expect = (unsigned long long)kEmptyKey;
      __hip_atomic_compare_exchange_strong(
          (unsigned long long*)&d.globalTable[slot].key, &expect,
          (unsigned long long)kWkLocked, __ATOMIC_ACQ_REL, __ATOMIC_ACQUIRE,
          __HIP_MEMORY_SCOPE_AGENT);
      if (expect == (unsigned long long)kEmptyKey) {  // claimed the slot
        uint64_t idx = atomicAdd((unsigned long long*)d.gkey.recCursor, 1ULL);
        if ((int64_t)idx >= d.gkey.recCap) {
          // record store full: flag it so probers parked on kWkLocked bail
          atomicOr(d.errorFlag, kErrGlobalFull);
          return false;
        }
        wideKeyWrite<WIDE>(d, d.gkey.keyStore + idx * (uint64_t)d.gkey.recBytes,
                           row, raw, vm, nullBits);
        __threadfence();  // record visible before the key publishes
        __hip_atomic_store(
            (unsigned long long*)&d.globalTable[slot].key,
            (unsigned long long)(((uint64_t)h32 << 32) | (uint32_t)idx),
            __ATOMIC_RELEASE, __HIP_MEMORY_SCOPE_AGENT);
        *slotOut = slot;
        return true;
      }
      cur = (uint64_t)expect;  // lost: the winner's state decides
    }
    if (cur == kWkLocked) {
      // the claimant is materializing its record: re-read this slot (it may
      // turn out to be OUR key). If the claimant overflowed the record
      // store it set the flag instead of publishing — bail with it.
      if (__hip_atomic_load(d.errorFlag, __ATOMIC_RELAXED,
                            __HIP_MEMORY_SCOPE_AGENT) &
          kErrGlobalFull)
        return false;
      continue;
    }
    if ((uint32_t)(cur >> 32) == h32) {
      const uint8_t* rec =
          d.gkey.keyStore + (cur & 0xFFFFFFFFu) * (uint64_t)d.gkey.recBytes;
      if (wideKeyMatches<WIDE>(d, rec, row, raw, vm, nullBits)) {
        *slotOut = slot;
        return true;
      }
    }
    slot = (slot + 1) & gmask;
  }
}

typedef __attribute__((address_space(3))) uint64_t WkLds64;
typedef __attribute__((address_space(3))) uint32_t WkLds32;

template <bool WIDE, typename VMT, typename RAWT>
__device__ inline bool makeWideGroupKey(const FusedQueryDesc& d, int64_t row,
                                        const RAWT& raw, const VMT& vm,
                                        WkLds64* cacheH, WkLds32* cacheS,
                                        uint32_t* slotOut) {
  uint64_t nullBits;
  uint64_t h = wideKeyHash<WIDE>(d, row, raw, vm, &nullBits);
  int ci = (int)((h >> 32) & (kWkCache - 1));
  if (cacheH[ci] == h) {
    uint32_t slot = cacheS[ci];
    uint64_t cur = d.globalTable[slot].key;  // plain: slot keys write once
    if ((uint32_t)(cur >> 32) == (uint32_t)(h >> 32)) {
      const uint8_t* rec =
          d.gkey.keyStore + (cur & 0xFFFFFFFFu) * (uint64_t)d.gkey.recBytes;
      if (wideKeyMatches<WIDE>(d, rec, row, raw, vm, nullBits)) {
        *slotOut = slot;
        return true;
      }
    }
    // hash collision or torn entry: exact verify failed -> full resolve
  }
  if (!wideKeyResolve<WIDE>(d, row, raw, vm, h, nullBits, slotOut))
    return false;
  cacheH[ci] = h;  // racy plain writes: any torn pair fails the verify above
  cacheS[ci] = *slotOut;
  return true;
}

// simple single-table predicate (direct loads; build phases are cheap scans)
__device__ inline bool evalSimplePred(const DevTable& tab, const PredDesc& pd,
                                      const uint8_t* strConst, int strConstLen,
                                      int64_t row, uint32_t* err = nullptr) {
  const DevCol& c = tab.cols[pd.col];
  if (pd.kind == PRED_IS_NULL)  // the null bit IS the result (never NULL)
    return colIsNull(c, row) == (pd.cmp == 4 /*GX_F_EQ*/);
  if (colIsNull(c, row)) return false;
  if (pd.kind == PRED_DEC_CMP_CONST) {
    // units compare at the column's declared frac (narrow path; >18-digit
    // rows flag RetryWide on the caller's error flag and reject)
    if (!err) return false;
    int64_t u;
    int sc;
    if (!loadDecimalUnits<false>((const uint8_t*)c.data + row * 40, &u, &sc,
                                 err))
      return false;
    int64_t k = (int64_t)pd.constU64;
    return cmpResult(u < k ? -1 : (u > k ? 1 : 0), pd.cmp);
  }
  if (pd.kind == PRED_TIME_CMP_CONST) {
    uint64_t v = gptr<uint64_t>(c.data)[row] & ~0xFULL;
    uint64_t k = pd.constU64 & ~0xFULL;
    return cmpResult(v < k ? -1 : (v > k ? 1 : 0), pd.cmp);
  }
  if (pd.kind == PRED_I64_CMP_CONST) {
    int64_t v = gptr<int64_t>(c.data)[row];
    int64_t k = (int64_t)pd.constU64;
    return cmpResult(v < k ? -1 : (v > k ? 1 : 0), pd.cmp);
  }
  if (pd.kind == PRED_STR_EQ_CONST) {
    int64_t st, en;
    if (c.denseOffsets) { st = row; en = row + 1; }
    else { st = gptr<int64_t>(c.offsets)[row]; en = gptr<int64_t>(c.offsets)[row + 1]; }
    auto p = gptr<uint8_t>(c.data);
    while (en > st && p[en - 1] == ' ') en--;  // PAD SPACE
    int len = (int)(en - st);
    bool eq = len == strConstLen;
    for (int j = 0; j < len && eq; j++) eq = p[st + j] == strConst[j];
    return cmpResult(eq ? 0 : 1, pd.cmp);
  }
  if (pd.kind == PRED_STR_LIKE_PREFIX) {
    // builtinLikeSig 'abc%' fast path: byte prefix, case-sensitive, no pad
    // trimming (LIKE does not use PAD SPACE semantics)
    int64_t st, en;
    if (c.denseOffsets) { st = row; en = row + 1; }
    else { st = gptr<int64_t>(c.offsets)[row]; en = gptr<int64_t>(c.offsets)[row + 1]; }
    auto p = gptr<uint8_t>(c.data);
    if (en - st < strConstLen) return false;
    for (int j = 0; j < strConstLen; j++)
      if (p[st + j] != strConst[j]) return false;
    return true;
  }
  return false;
}

// atomic int128 + count accumulation into a slot (LDS or global)
template <typename SlotT>
__device__ inline void accumInto(SlotT* slot, int a, Int128 v, int64_t dc) {
  // generic-pointer variant (global table); LDS paths use the AS3 helpers
  // below so the atomics compile to ds_* ops (a flat atomic counts against
  // vmcnt AND lgkmcnt and serializes the prefetch pipeline).
  if (v.lo != 0 || v.hi != 0) {
    uint64_t old = atomicAdd((unsigned long long*)&slot->accLo[a],
                             (unsigned long long)v.lo);
    uint64_t carry = (old + v.lo) < old ? 1 : 0;
    int64_t hiAdd = v.hi + (int64_t)carry;
    if (hiAdd != 0)
      atomicAdd((unsigned long long*)&slot->accHi[a], (unsigned long long)hiAdd);
  }
  if (dc != 0)
    atomicAdd((unsigned long long*)&slot->cnt[a], (unsigned long long)dc);
}

typedef __attribute__((address_space(3))) GroupSlot Lds3GroupSlot;
typedef __attribute__((address_space(3))) uint64_t Lds3U64;

// order-preserving biased encoding for int64 min/max accumulation
__device__ inline uint64_t biasI64(int64_t v) {
  return (uint64_t)v ^ 0x8000000000000000ULL;
}
__device__ inline int64_t unbiasU64(uint64_t u) {
  return (int64_t)(u ^ 0x8000000000000000ULL);
}

__device__ inline void lds3AccumMax(Lds3GroupSlot* slot, int s, uint64_t enc) {
  __hip_atomic_fetch_max((Lds3U64*)&slot->accLo[s], enc, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_WORKGROUP);
}

template <typename SlotT>
__device__ inline void accumMax(SlotT* slot, int s, uint64_t enc) {
  atomicMax((unsigned long long*)&slot->accLo[s], (unsigned long long)enc);
}

__device__ inline void lds3AccumAcc(Lds3GroupSlot* slot, int s, Int128 v) {
  if (v.lo != 0 || v.hi != 0) {
    uint64_t old = __hip_atomic_fetch_add((Lds3U64*)&slot->accLo[s], (uint64_t)v.lo,
                                          __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_WORKGROUP);
    uint64_t carry = (old + v.lo) < old ? 1 : 0;
    int64_t hiAdd = v.hi + (int64_t)carry;
    if (hiAdd != 0)
      __hip_atomic_fetch_add((Lds3U64*)&slot->accHi[s], (uint64_t)hiAdd,
                             __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_WORKGROUP);
  }
}

typedef __attribute__((address_space(3))) double Lds3F64;

__device__ inline void lds3AccumF64(Lds3GroupSlot* slot, int s, double v) {
  __hip_atomic_fetch_add((Lds3F64*)&slot->accLo[s], v, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_WORKGROUP);
}

__device__ inline void lds3AccumCnt(Lds3GroupSlot* slot, int a, int64_t dc) {
  if (dc != 0)
    __hip_atomic_fetch_add((Lds3U64*)&slot->cnt[a], (uint64_t)dc,
                           __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_WORKGROUP);
}

__device__ inline uint64_t lds3CasKey(Lds3GroupSlot* slot, uint64_t expect,
                                      uint64_t val) {
  __hip_atomic_compare_exchange_strong((Lds3U64*)&slot->key, &expect, val,
                                       __ATOMIC_RELAXED, __ATOMIC_RELAXED,
                                       __HIP_MEMORY_SCOPE_WORKGROUP);
  return expect;  // holds the previous value on failure, `expect` on success
}

__constant__ uint64_t kDivArgMax[37][2] = {  // I128_MAX / 10^e: {lo, hi}
    {0xffffffffffffffffULL, 0x7fffffffffffffffULL},
    {0xccccccccccccccccULL, 0xcccccccccccccccULL},
    {0x147ae147ae147ae1ULL, 0x147ae147ae147aeULL},
    {0xced916872b020c49ULL, 0x20c49ba5e353f7ULL},
    {0x94af4f0d844d013aULL, 0x346dc5d638865ULL},
    {0xc21187e7c06e19b9ULL, 0x53e2d6238da3ULL},
    {0xc69b5a63f9a49c2cULL, 0x8637bd05af6ULL},
    {0x7a42bc3d32907604ULL, 0xd6bf94d5e5ULL},
    {0x8c39df9fb841a566ULL, 0x15798ee230ULL},
    {0xdad2965cc5a02a23ULL, 0x225c17d04ULL},
    {0xaf7b756fad5cd103ULL, 0x36f9bfb3ULL},
    {0x5e592557f7bc7b4dULL, 0x57f5ff8ULL},
    {0x96f5088cbf93f87ULL, 0x8cbcccULL},
    {0x3424bb40e132865aULL, 0xe12e1ULL},
    {0xb86a12b9b01ea709ULL, 0x16849ULL},
    {0x5f3dceac2b3643e7ULL, 0x2407ULL},
    {0x5652fb1137856d30ULL, 0x39aULL},
    {0x3bd5191b525a2484ULL, 0x5cULL},
    {0x392ee8e921d5d073ULL, 0x9ULL},
    {0xec1e4a7db69561a5ULL, 0x0ULL},
    {0x179ca10c9242235dULL, 0x0ULL},
    {0x25c768141d369efULL, 0x0ULL},
    {0x3c7240202ebdcbULL, 0x0ULL},
    {0x60b6cd004ac94ULL, 0x0ULL},
    {0x9abe14cd4475ULL, 0x0ULL},
    {0xf79687aed3eULL, 0x0ULL},
    {0x18c240c4aecULL, 0x0ULL},
    {0x279d346de4ULL, 0x0ULL},
    {0x3f61ed7caULL, 0x0ULL},
    {0x65697bfaULL, 0x0ULL},
    {0xa2425ffULL, 0x0ULL},
    {0x1039d66ULL, 0x0ULL},
    {0x19f623ULL, 0x0ULL},
    {0x2989dULL, 0x0ULL},
    {0x4276ULL, 0x0ULL},
    {0x6a5ULL, 0x0ULL},
    {0xaaULL, 0x0ULL},
};

// 128/64 unsigned division with no compiler-rt libcalls (__divti3 is a
// device-library call with its own stack traffic; this inlines to plain
// VALU). Hacker's Delight fig. 9-3 shape: normalized 2-by-1 division with
// 32-bit digits.
__device__ inline uint64_t udiv128by64(uint64_t hi, uint64_t lo, uint64_t d,
                                       uint64_t* rem) {
  // requires hi < d (caller peels the top word)
  int s = __clzll(d);
  d <<= s;
  uint64_t un64 = s ? (hi << s) | (lo >> (64 - s)) : hi;
  uint64_t un10 = lo << s;
  uint32_t vn1 = (uint32_t)(d >> 32), vn0 = (uint32_t)d;
  uint64_t un1 = un10 >> 32, un0 = (uint32_t)un10;
  uint64_t q1 = un64 / vn1;
  uint64_t rhat = un64 - q1 * vn1;
  while (q1 >= 0x100000000ULL || q1 * vn0 > (rhat << 32) + un1) {
    q1--;
    rhat += vn1;
    if (rhat >= 0x100000000ULL) break;
  }
  uint64_t un21 = (un64 << 32) + un1 - q1 * d;
  uint64_t q0 = un21 / vn1;
  rhat = un21 - q0 * vn1;
  while (q0 >= 0x100000000ULL || q0 * vn0 > (rhat << 32) + un0) {
    q0--;
    rhat += vn1;
    if (rhat >= 0x100000000ULL) break;
  }
  if (rem) *rem = ((un21 << 32) + un0 - q0 * d) >> s;
  return (q1 << 32) + q0;
}

// full u128 / u64 -> u128 quotient (64-bit div/mod lower inline on amdgcn)
__device__ inline unsigned __int128 u128DivU64(unsigned __int128 n, uint64_t d) {
  uint64_t nhi = (uint64_t)(n >> 64), nlo = (uint64_t)n;
  uint64_t qhi = nhi / d;
  uint64_t r = nhi % d;
  uint64_t qlo = udiv128by64(r, nlo, d, nullptr);
  return ((unsigned __int128)qhi << 64) | qlo;
}

// u128 / u128 with divisor >= 2^64: quotient < 2^64, shift-subtract
__device__ inline unsigned __int128 u128DivBig(unsigned __int128 n,
                                               unsigned __int128 d) {
  unsigned __int128 q = 0;
  int shift = 0;
  while ((d << 1) <= n && (d >> 126) == 0 && shift < 64) { d <<= 1; shift++; }
  for (; shift >= 0; shift--) {
    if (n >= d) { n -= d; q |= (unsigned __int128)1 << shift; }
    d >>= 1;
  }
  return q;
}

}  // namespace gxp
