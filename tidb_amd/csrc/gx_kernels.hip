// tidb_amd/csrc/gx_kernels.hip — MI355X (gfx950/CDNA4) kernels for the
// TiDB analytical hot path.
//
// Fused scan -> filter -> project -> hash-aggregate in ONE pass over the
// columns (the MI355X-native form of SelectionExec -> ProjectionExec ->
// HashAggExec, pkg/executor/select.go:750 / projection.go:77 /
// aggregate/agg_hash_*.go): every consumed input column is read once from
// HBM — the path is HBM-bandwidth-bound (no MFMA shape here).
//
// Grouping: per-workgroup LDS hash table (LDS-staged accumulator state,
// atomically updated), flushed once per workgroup into a global table with
// device-scope atomics — the device analog of the reference's partial/final
// worker split (agg_hash_executor.go:54-92).
//
// Decimal arithmetic: fixed-point int64/int128 units at static scales; exact,
// with overflow detection -> error flag (never silent). Equivalence to the
// word-based MyDecimal arithmetic is covered by tests/golden + parity suites.
#include <hip/hip_runtime.h>
#include <type_traits>
#include <hipcub/hipcub.hpp>

#include "gx_common.h"

namespace gxp {

// ------------------------------------------------------------------
// shared PRNG/date spec — MUST generate bit-identical data to the CPU
// restatement in oracle/tpch.cpp (parity-tested).
// ------------------------------------------------------------------
__host__ __device__ inline uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ULL;
  uint64_t z = x;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}
__host__ __device__ inline uint64_t fieldRand(uint64_t seed, int64_t row, int field) {
  uint64_t h = splitmix64(seed ^ (0x9E3779B97F4A7C15ULL * (uint64_t)(row + 1)));
  return splitmix64(h ^ (0xBF58476D1CE4E5B9ULL * (uint64_t)(field + 1)));
}
__host__ __device__ inline int64_t daysFromCivil(int y, int m, int d) {
  y -= m <= 2;
  int64_t era = (y >= 0 ? y : y - 399) / 400;
  unsigned yoe = (unsigned)(y - era * 400);
  unsigned doy = (unsigned)((153 * (m + (m > 2 ? -3 : 9)) + 2) / 5 + d - 1);
  unsigned doe = yoe * 365 + yoe / 4 - yoe / 100 + doy;
  return era * 146097 + (int64_t)doe - 719468;
}
__host__ __device__ inline void civilFromDays(int64_t z, int* yy, int* mm, int* dd) {
  z += 719468;
  int64_t era = (z >= 0 ? z : z - 146096) / 146097;
  unsigned doe = (unsigned)(z - era * 146097);
  unsigned yoe = (doe - doe / 1460 + doe / 36524 - doe / 146096) / 365;
  int64_t y = (int64_t)yoe + era * 400;
  unsigned doy = doe - (365 * yoe + yoe / 4 - yoe / 100);
  unsigned mp = (5 * doy + 2) / 153;
  unsigned d = doy - (153 * mp + 2) / 5 + 1;
  unsigned m = mp < 10 ? mp + 3 : mp - 9;
  *yy = (int)(y + (m <= 2));
  *mm = (int)m;
  *dd = (int)d;
}

// packed CoreTime DATE (time.go:235-251,266: fspTt 0b1110 marks TypeDate)
__host__ __device__ inline uint64_t timeFromDate(int y, int m, int d) {
  return ((uint64_t)y << 50) | ((uint64_t)m << 46) | ((uint64_t)d << 41) | 0xEULL;
}

// date-range constants are computed (constant-folded), never hand-written
#define GX_EPOCH_1992 daysFromCivil(1992, 1, 1)
#define GX_SHIPDATE_DAYS (daysFromCivil(1998, 12, 1) - daysFromCivil(1992, 1, 1) + 1)
#define GX_ORDERDATE_DAYS (daysFromCivil(1998, 8, 2) - daysFromCivil(1992, 1, 1) + 1)

// canonical cents -> 40-byte MyDecimal (scale 2), matches oracle/tpch.cpp
__device__ inline void storeDecCents(uint8_t* p, int64_t cents) {
  int64_t ip = cents / 100;
  uint32_t f = (uint32_t)(cents % 100);
  int digits = 1;
  for (int64_t t = ip; t >= 10; t /= 10) digits++;
  uint32_t hdr = (uint32_t)(uint8_t)digits | (2u << 8) | (2u << 16);
  uint32_t* w = (uint32_t*)p;
  w[0] = hdr;
  w[1] = (uint32_t)ip;
  w[2] = f * 10000000u;
  w[3] = 0; w[4] = 0; w[5] = 0; w[6] = 0; w[7] = 0; w[8] = 0; w[9] = 0;
}

// ------------------------------------------------------------------
// lineitem generator: cols 0..7 =
//   orderkey i64, quantity dec, extendedprice dec, discount dec, tax dec,
//   returnflag char(1), linestatus char(1), shipdate date
// ------------------------------------------------------------------
__global__ void genLineitemKernel(DevTable tab, int64_t rowBegin, int64_t nRows,
                                  uint64_t seed, int64_t totalRows) {
  int64_t nOrders = totalRows / 4;
  if (nOrders < 1) nOrders = 1;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nRows;
       i += stride) {
    int64_t row = rowBegin + i;
    ((int64_t*)tab.cols[0].data)[i] =
        1 + (int64_t)(fieldRand(seed, row, 0) % (uint64_t)nOrders);
    storeDecCents((uint8_t*)tab.cols[1].data + i * 40,
                  (1 + (int64_t)(fieldRand(seed, row, 1) % 50)) * 100);
    storeDecCents((uint8_t*)tab.cols[2].data + i * 40,
                  90100 + (int64_t)(fieldRand(seed, row, 2) % (10495000 - 90100 + 1)));
    storeDecCents((uint8_t*)tab.cols[3].data + i * 40,
                  (int64_t)(fieldRand(seed, row, 3) % 11));
    storeDecCents((uint8_t*)tab.cols[4].data + i * 40,
                  (int64_t)(fieldRand(seed, row, 4) % 9));
    const char rf[3] = {'A', 'N', 'R'};
    ((uint8_t*)tab.cols[5].data)[i] = rf[fieldRand(seed, row, 5) % 3];
    tab.cols[5].offsets[i + 1] = i + 1;
    const char ls[2] = {'O', 'F'};
    ((uint8_t*)tab.cols[6].data)[i] = ls[fieldRand(seed, row, 6) % 2];
    tab.cols[6].offsets[i + 1] = i + 1;
    int y, m, d;
    civilFromDays(GX_EPOCH_1992 +
                      (int64_t)(fieldRand(seed, row, 7) % (uint64_t)GX_SHIPDATE_DAYS),
                  &y, &m, &d);
    ((uint64_t*)tab.cols[7].data)[i] = timeFromDate(y, m, d);
    if (i == 0) {
      tab.cols[5].offsets[0] = 0;
      tab.cols[6].offsets[0] = 0;
    }
  }
}

// ------------------------------------------------------------------
// fused filter+project+aggregate
// ------------------------------------------------------------------
//
// Two instantiations: NARROW (int64 VM registers — covers every value with
// <= 18 significant digits, i.e. all real decimal(15,2)-class workloads) and
// WIDE (int128). NARROW detects overflow per operation and sets the
// RETRY_WIDE flag; the engine then relaunches the WIDE variant — exactness
// is never traded away.

constexpr uint32_t kErrBadDecimal = 1u;
constexpr uint32_t kErrBadKey = 2u;
constexpr uint32_t kErrScale = 4u;
constexpr uint32_t kErrOverflow = 8u;
constexpr uint32_t kErrLdsFull = 16u;
constexpr uint32_t kErrGlobalFull = 32u;
constexpr uint32_t kErrRetryWide = 256u;  // narrow VM overflowed; not an error

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

// pack the group key (see GroupKeyDesc comment); offsets/values come from the
// batched raw fetch
template <typename RAWT>
__device__ __attribute__((always_inline)) inline bool makeGroupKey(const FusedQueryDesc& d, int64_t row,
                                    const RAWT& raw, uint64_t* keyOut,
                                    uint32_t* err) {
  uint64_t key = 0;
  for (int k = 0; k < d.gkey.nCols; k++) {
    const DevCol& c = d.table.cols[d.gkey.col[k]];
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

// per-row pipeline after the raw fetch: filter -> VM -> LDS aggregate.
// Returns false on a hard failure (error flag already set).
template <bool WIDE, bool DIVOK, typename RAWT>
__device__ __attribute__((always_inline)) inline bool processRow(const FusedQueryDesc& d, int64_t row,
                                  const RAWT& raw, Lds3GroupSlot* lds,
                                  uint64_t* mySel) {
  // ---- filter (CNF; NULL rejects — expression.go:507 toBool) ----
  bool pass = true;
  for (int p = 0; p < d.nPreds && pass; p++) {
    const PredDesc& pd = d.preds[p];
    const DevCol& c = d.table.cols[pd.col];
    if (colIsNull(c, row)) { pass = false; break; }
    if (pd.kind == PRED_TIME_CMP_CONST) {
      uint64_t v = raw.get(pd.slot).x & ~0xFULL;
      uint64_t k = pd.constU64 & ~0xFULL;
      int cmp = v < k ? -1 : (v > k ? 1 : 0);
      pass = cmpResult(cmp, pd.cmp);
    } else if (pd.kind == PRED_I64_CMP_CONST) {
      int64_t v = (int64_t)raw.get(pd.slot).x;
      int64_t k = (int64_t)pd.constU64;
      int cmp = v < k ? -1 : (v > k ? 1 : 0);
      pass = cmpResult(cmp, pd.cmp);
    } else {  // PRED_DEC_CMP_CONST (engine aligned const to column scale)
      typename VT<WIDE>::T u;
      int sc;
      if (!loadDecimalUnits<WIDE>((const uint8_t*)c.data + row * 40, &u, &sc,
                                  d.errorFlag))
        return false;
      int cmp = VT<WIDE>::cmp(u, VT<WIDE>::fromI64((int64_t)pd.constU64, nullptr));
      pass = cmpResult(cmp, pd.cmp);
    }
  }
  if (!pass) return true;
  (*mySel)++;
  if (d.ablate == 2) return true;  // timing ablation: filter only

  // ---- projection / agg-arg VM ----
  VmState<WIDE> vm;
  vm.nullBits = 0;
  bool bad = false;
  bool ovf = false;
  for (int i = 0; i < d.nIns && !bad; i++) {
    const VmIns& ins = d.ins[i];
    switch (ins.op) {
      case VM_LOAD_DEC: {
        const DevCol& c = d.table.cols[ins.a];
        bool nul = colIsNull(c, row);
        typename VT<WIDE>::T v = VT<WIDE>::zero();
        if (!nul) {
          int sc;
          if (!parseDecimalRaw<WIDE>(raw.get(ins.c), &v, &sc, d.errorFlag,
                                     ins.b, d.insP10[i], d.insMagic[i])) {
            bad = true;
            break;
          }
          if (sc != ins.b) {  // engine encoded expected scale in b
            if (sc < ins.b) v = VT<WIDE>::scale10(v, ins.b - sc, &ovf);
            else { atomicOr(d.errorFlag, kErrScale); bad = true; break; }
          }
        }
        vm.set(ins.dst, v);
        vm.setNull(ins.dst, nul);
        break;
      }
      case VM_LOAD_I64: {
        const DevCol& c = d.table.cols[ins.a];
        bool nul = colIsNull(c, row);
        vm.set(ins.dst, nul ? VT<WIDE>::zero()
                            : VT<WIDE>::fromI64((int64_t)raw.get(ins.c).x, &ovf));
        vm.setNull(ins.dst, nul);
        break;
      }
      case VM_LOAD_CONST: {
        if (WIDE) {
          Int128 cv = {(uint64_t)d.constLo[ins.a], d.constHi[ins.a]};
          vm.set(ins.dst, *(typename VT<WIDE>::T*)&cv);
        } else {
          // engine guarantees narrow-mode consts fit i64 (else it forces WIDE)
          int64_t cv = d.constLo[ins.a];
          vm.set(ins.dst, *(typename VT<WIDE>::T*)&cv);
        }
        vm.setNull(ins.dst, false);
        break;
      }
      case VM_ADD:
        vm.set(ins.dst, VT<WIDE>::add(vm.get(ins.a), vm.get(ins.b), &ovf));
        vm.setNull(ins.dst, vm.isNull(ins.a) || vm.isNull(ins.b));
        break;
      case VM_SUB:
        vm.set(ins.dst, VT<WIDE>::sub(vm.get(ins.a), vm.get(ins.b), &ovf));
        vm.setNull(ins.dst, vm.isNull(ins.a) || vm.isNull(ins.b));
        break;
      case VM_MUL: {
        bool nul = vm.isNull(ins.a) || vm.isNull(ins.b);
        typename VT<WIDE>::T v = VT<WIDE>::zero();
        if (!nul) v = VT<WIDE>::mul(vm.get(ins.a), vm.get(ins.b), &ovf);
        vm.set(ins.dst, v);
        vm.setNull(ins.dst, nul);
        break;
      }
      case VM_SCALE_UP:
        vm.set(ins.dst,
               VT<WIDE>::mul(vm.get(ins.a),
                             VT<WIDE>::fromI64(d.insP10[i], nullptr), &ovf));
        vm.setNull(ins.dst, vm.isNull(ins.a));
        break;
      case VM_DIV: {
        if constexpr (!DIVOK) {
          // engine launches the DIVOK variant for plans containing DIV;
          // reaching here means a dispatch bug -- fail loudly
          atomicOr(d.errorFlag, kErrBadDecimal);
          bad = true;
          break;
        } else {
        // DecimalDiv (mydecimal.go:1311, doDiv:1168): quotient truncated
        // toward zero at the word-granular result scale; ins.c holds the
        // exponent e with result = trunc(a * 10^e / b). Division by zero
        // yields NULL (builtin_arithmetic_vec.go:92 handleDivisionByZero).
        bool nul = vm.isNull(ins.a) || vm.isNull(ins.b);
        typename VT<WIDE>::T v = VT<WIDE>::zero();
        if (!nul) {
          Int128 bi = VT<WIDE>::toAcc(vm.get(ins.b));
          __int128 bv = ((__int128)bi.hi << 64) | (__int128)bi.lo;
          if (bv == 0) {
            nul = true;
          } else {
            Int128 ai = VT<WIDE>::toAcc(vm.get(ins.a));
            __int128 av = ((__int128)ai.hi << 64) | (__int128)ai.lo;
            int e = ins.c;
            unsigned __int128 p10 =
                (unsigned __int128)(uint64_t)kP10(e > 18 ? 18 : e);
            if (e > 18) p10 *= (uint64_t)kP10(e - 18);
            unsigned __int128 aAbs =
                (unsigned __int128)(av < 0 ? -av : av);
            unsigned __int128 lim =
                ((unsigned __int128)kDivArgMax[e][1] << 64) | kDivArgMax[e][0];
            if (aAbs > lim) {
              atomicOr(d.errorFlag, kErrOverflow);
              bad = true;
              break;
            }
            unsigned __int128 bAbs =
                (unsigned __int128)(bv < 0 ? -bv : bv);
            unsigned __int128 num = aAbs * p10;
            unsigned __int128 uq = (bAbs >> 64) != 0
                                       ? u128DivBig(num, bAbs)
                                       : u128DivU64(num, (uint64_t)bAbs);
            bool negq = (av < 0) != (bv < 0);
            __int128 q = negq ? -(__int128)uq : (__int128)uq;
            if (!WIDE &&
                (q > (__int128)INT64_MAX || q < (__int128)INT64_MIN)) {
              atomicOr(d.errorFlag, kErrRetryWide);
              bad = true;
              break;
            }
            if (WIDE) {
              Int128 r = {(uint64_t)q, (int64_t)(q >> 64)};
              v = *(typename VT<WIDE>::T*)&r;
            } else {
              int64_t qq = (int64_t)q;
              v = *(typename VT<WIDE>::T*)&qq;
            }
          }
        }
        vm.set(ins.dst, v);
        vm.setNull(ins.dst, nul);
        break;
        }
      }
    }
  }
  if (ovf) {
    atomicOr(d.errorFlag, WIDE ? kErrOverflow : kErrRetryWide);
    return false;
  }
  if (bad) return false;

  if (d.ablate == 1) {  // timing ablation: keep VM results live, skip agg
    uint64_t sink = 0;
    for (int a = 0; a < d.nAggs; a++)
      if (d.aggs[a].srcReg >= 0)
        sink ^= (uint64_t)VT<WIDE>::toAcc(vm.get(d.aggs[a].srcReg)).lo;
    asm volatile("" ::"v"(sink));
    return true;
  }

  // ---- group lookup / insert (LDS table, or global when NDV > kLdsGroups) ----
  uint64_t key;
  if (!makeGroupKey(d, row, raw, &key, d.errorFlag)) return false;
  if (!d.noLds) {
    uint32_t slot = (uint32_t)(splitmix64(key) & (kLdsGroups - 1));
    for (int probe = 0;; probe++) {
      if (probe >= kLdsGroups) {
        atomicOr(d.errorFlag, kErrLdsFull);
        return false;
      }
      uint64_t cur = lds[slot].key;
      if (cur == key) break;
      if (cur == kEmptyKey) {
        uint64_t prev = lds3CasKey(&lds[slot], kEmptyKey, key);
        if (prev == kEmptyKey || prev == key) break;
      }
      slot = (slot + 1) & (kLdsGroups - 1);
    }
    Lds3GroupSlot* target = &lds[slot];
    // ---- update states: one atomic per unique acc, one or per-agg counts ----
    if (d.sharedCnt) lds3AccumCnt(target, 0, 1);
    for (int s = 0; s < d.nAccSlots; s++) {
      int reg = d.accReg[s];
      if (!vm.isNull(reg)) lds3AccumAcc(target, s, VT<WIDE>::toAcc(vm.get(reg)));
    }
    if (!d.sharedCnt) {
      for (int a = 0; a < d.nAggs; a++) {
        const AggDesc& ad = d.aggs[a];
        if (ad.fr >= 0) continue;  // firstrow(group col): no per-row state
        bool isNull = ad.srcReg >= 0 && vm.isNull(ad.srcReg);
        if (!isNull) lds3AccumCnt(target, a, 1);
      }
    }
    return true;
  }
  // global-direct path (mid/high NDV up to kGlobalGroups)
  uint32_t slot = (uint32_t)(splitmix64(key) & (kGlobalGroups - 1));
  for (int probe = 0;; probe++) {
    if (probe >= kGlobalGroups) {
      atomicOr(d.errorFlag, kErrGlobalFull);
      return false;
    }
    uint64_t cur = d.globalTable[slot].key;
    if (cur == key) break;
    if (cur == kEmptyKey) {
      uint64_t prev = atomicCAS((unsigned long long*)&d.globalTable[slot].key,
                                (unsigned long long)kEmptyKey,
                                (unsigned long long)key);
      if (prev == kEmptyKey || prev == key) break;
    }
    slot = (slot + 1) & (kGlobalGroups - 1);
  }
  GroupSlot* target = &d.globalTable[slot];
  if (d.sharedCnt) accumInto(target, 0, Int128{0, 0}, 1);  // bumps cnt[0] only
  for (int s = 0; s < d.nAccSlots; s++) {
    int reg = d.accReg[s];
    if (!vm.isNull(reg)) accumInto(target, s, VT<WIDE>::toAcc(vm.get(reg)), 0);
  }
  if (!d.sharedCnt) {
    for (int a = 0; a < d.nAggs; a++) {
      const AggDesc& ad = d.aggs[a];
      if (ad.fr >= 0) continue;  // firstrow(group col): no per-row state
      bool isNull = ad.srcReg >= 0 && vm.isNull(ad.srcReg);
      if (!isNull) accumInto(target, a, Int128{0, 0}, 1);
    }
  }
  return true;
}

template <bool WIDE, int R, bool DIVOK = false>
__launch_bounds__(256)
__global__ void fusedAggKernel(const FusedQueryDesc* __restrict__ dp) {
  const FusedQueryDesc& d = *dp;
  bool failed = false;
  __shared__ GroupSlot lds[kLdsGroups];
  Lds3GroupSlot* lds3 = (Lds3GroupSlot*)lds;
  for (int i = threadIdx.x; i < kLdsGroups; i += blockDim.x) {
    lds[i].key = kEmptyKey;
    for (int a = 0; a < kMaxAggs; a++) {
      lds[i].accLo[a] = 0;
      lds[i].accHi[a] = 0;
      lds[i].cnt[a] = 0;
    }
  }
  __syncthreads();

  int64_t n = d.table.nRows;
  int64_t per = (n + gridDim.x - 1) / gridDim.x;
  int64_t begin = (int64_t)blockIdx.x * per;
  int64_t end = begin + per;
  if (end > n) end = n;
  uint64_t mySel = 0;

  // 2-deep software-pipelined grouped-fetch loop: row k+1's loads issue
  // before row k's pipeline runs, so every wave keeps ~2 rows of bytes in
  // flight (the kernel is HBM-latency bound: VALU util ~7%, concurrency =
  // waves x rows-in-flight). Two NAMED RawState structs, never indexed or
  // swapped through a pointer -- indexed access re-rolls to scratch spills
  // (measured 6.1 ms), this form stays in registers.
  {
    // R is the compile-time fetch-slot bound: 5 covers Q1-shaped queries at
    // one extra wave/SIMD of occupancy; 8 is the general variant.
    using RAWT = typename std::conditional<R <= 5, RawState5, RawState>::type;
    const int64_t stride = blockDim.x;
    int64_t row = begin + threadIdx.x;
    RAWT rawA, rawB;
    if (row < end) fetchRow(d.table, d.fetch, d.nFetch, row, rawA);
    for (; row < end && !failed; row += 2 * stride) {
      const int64_t rB = row + stride;
      if (rB < end) fetchRow(d.table, d.fetch, d.nFetch, rB, rawB);
      if (!processRow<WIDE, DIVOK>(d, row, rawA, lds3, &mySel)) { failed = true; break; }
      const int64_t rA2 = row + 2 * stride;
      if (rA2 < end) fetchRow(d.table, d.fetch, d.nFetch, rA2, rawA);
      if (rB < end && !processRow<WIDE, DIVOK>(d, rB, rawB, lds3, &mySel)) failed = true;
    }
  }

  if (d.selCount) {
    // wave-level reduce then one atomic per wave (G12)
    uint64_t total = mySel;
    for (int off = 32; off > 0; off >>= 1)
      total += __shfl_down(total, off, 64);
    if ((threadIdx.x & 63) == 0 && total)
      atomicAdd((unsigned long long*)d.selCount, (unsigned long long)total);
  }
  __syncthreads();

  // ---- flush LDS table into the global table ----
  for (int i = threadIdx.x; i < kLdsGroups; i += blockDim.x) {
    if (lds[i].key == kEmptyKey) continue;
    uint64_t key = lds[i].key;
    uint32_t slot = (uint32_t)(splitmix64(key) & (kGlobalGroups - 1));
    bool ok = true;
    for (int probe = 0;; probe++) {
      if (probe >= kGlobalGroups) { atomicOr(d.errorFlag, kErrGlobalFull); ok = false; break; }
      uint64_t cur = d.globalTable[slot].key;
      if (cur == key) break;
      if (cur == kEmptyKey) {
        uint64_t prev = atomicCAS((unsigned long long*)&d.globalTable[slot].key,
                                  (unsigned long long)kEmptyKey,
                                  (unsigned long long)key);
        if (prev == kEmptyKey || prev == key) break;
      }
      slot = (slot + 1) & (kGlobalGroups - 1);
    }
    if (!ok) continue;
    for (int s = 0; s < d.nAccSlots; s++) {
      Int128 v = {lds[i].accLo[s], lds[i].accHi[s]};
      accumInto(&d.globalTable[slot], s, v, 0);
    }
    int nCnt = d.sharedCnt ? 1 : d.nAggs;
    for (int a = 0; a < nCnt; a++)
      accumInto(&d.globalTable[slot], a, Int128{0, 0}, lds[i].cnt[a]);
  }
}

__global__ void initGlobalTableKernel(GroupSlot* table, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  table[i].key = kEmptyKey;
  for (int a = 0; a < kMaxAggs; a++) {
    table[i].accLo[a] = 0;
    table[i].accHi[a] = 0;
    table[i].cnt[a] = 0;
  }
}


// ==================================================================
// join-aggregate pipeline (Q3 class) — see gx_common.h JoinAggDesc
// ==================================================================

// simple single-table predicate (direct loads; build phases are cheap scans)
__device__ inline bool evalSimplePred(const DevTable& tab, const PredDesc& pd,
                                      const uint8_t* strConst, int strConstLen,
                                      int64_t row) {
  const DevCol& c = tab.cols[pd.col];
  if (colIsNull(c, row)) return false;
  if (pd.kind == PRED_TIME_CMP_CONST) {
    uint64_t v = ((const uint64_t*)c.data)[row] & ~0xFULL;
    uint64_t k = pd.constU64 & ~0xFULL;
    return cmpResult(v < k ? -1 : (v > k ? 1 : 0), pd.cmp);
  }
  if (pd.kind == PRED_I64_CMP_CONST) {
    int64_t v = ((const int64_t*)c.data)[row];
    int64_t k = (int64_t)pd.constU64;
    return cmpResult(v < k ? -1 : (v > k ? 1 : 0), pd.cmp);
  }
  if (pd.kind == PRED_STR_EQ_CONST) {
    int64_t st, en;
    if (c.denseOffsets) { st = row; en = row + 1; }
    else { st = c.offsets[row]; en = c.offsets[row + 1]; }
    const uint8_t* p = (const uint8_t*)c.data;
    while (en > st && p[en - 1] == ' ') en--;  // PAD SPACE
    int len = (int)(en - st);
    bool eq = len == strConstLen;
    for (int j = 0; j < len && eq; j++) eq = p[st + j] == strConst[j];
    return cmpResult(eq ? 0 : 1, pd.cmp);
  }
  return false;
}

__device__ inline uint64_t hashKey(uint64_t k) { return splitmix64(k); }

__device__ inline void bloomSet(const JoinAggDesc& d, uint64_t key) {
  uint64_t h = splitmix64(key);
  uint32_t mask = (1u << d.bloomLog2) - 1;
  uint32_t b1 = (uint32_t)h & mask;
  uint32_t b2 = (uint32_t)(h >> 32) & mask;
  atomicOr(&d.bloom[b1 >> 5], 1u << (b1 & 31));
  atomicOr(&d.bloom[b2 >> 5], 1u << (b2 & 31));
}

__device__ inline bool bloomMayHave(const JoinAggDesc& d, uint64_t key) {
  if (d.bloomLog2 == 0) return true;
  uint64_t h = splitmix64(key);
  uint32_t mask = (1u << d.bloomLog2) - 1;
  uint32_t b1 = (uint32_t)h & mask;
  uint32_t b2 = (uint32_t)(h >> 32) & mask;
  auto bm = gptr<uint32_t>(d.bloom);  // AS1: keep probes off the flat path
  if (!((bm[b1 >> 5] >> (b1 & 31)) & 1)) return false;
  return ((bm[b2 >> 5] >> (b2 & 31)) & 1) != 0;
}

// count rows of build0 passing its predicate
__global__ void jaCountBuild0Kernel(const JoinAggDesc* __restrict__ dp) {
  const JoinAggDesc& d = *dp;
  int64_t n = d.build0.nRows;
  uint64_t my = 0;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += (int64_t)gridDim.x * blockDim.x) {
    bool pass = d.nPred0 == 0 ||
                evalSimplePred(d.build0, d.pred0, d.strConst, d.strConstLen, row);
    if (pass) my++;
  }
  for (int off = 32; off > 0; off >>= 1) my += __shfl_down(my, off, 64);
  if ((threadIdx.x & 63) == 0 && my)
    atomicAdd((unsigned long long*)&d.counters[0], (unsigned long long)my);
}

// insert passing build0 keys into the open-addressed key set
__global__ void jaBuild0Kernel(const JoinAggDesc* __restrict__ dp) {
  const JoinAggDesc& d = *dp;
  int64_t n = d.build0.nRows;
  uint32_t mask = (1u << d.keySetLog2) - 1;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += (int64_t)gridDim.x * blockDim.x) {
    bool pass = d.nPred0 == 0 ||
                evalSimplePred(d.build0, d.pred0, d.strConst, d.strConstLen, row);
    if (!pass) continue;
    if (colIsNull(d.build0.cols[d.b0KeyCol], row)) continue;  // NULL never joins
    uint64_t key = ((const uint64_t*)d.build0.cols[d.b0KeyCol].data)[row];
    if (key == kEmptyKey) key = kEmptyKey - 1;
    uint32_t slot = (uint32_t)(hashKey(key) & mask);
    for (uint32_t probe = 0; probe <= mask; probe++) {
      uint64_t cur = d.keySet[slot];
      if (cur == key) break;
      if (cur == kEmptyKey) {
        uint64_t prev = atomicCAS((unsigned long long*)&d.keySet[slot],
                                  (unsigned long long)kEmptyKey,
                                  (unsigned long long)key);
        if (prev == kEmptyKey || prev == key) break;
      }
      slot = (slot + 1) & mask;
      if (probe == mask) atomicOr(d.errorFlag, kErrGlobalFull);
    }
  }
}

__device__ inline bool keySetHas(const JoinAggDesc& d, uint64_t key) {
  uint32_t mask = (1u << d.keySetLog2) - 1;
  if (key == kEmptyKey) key = kEmptyKey - 1;
  uint32_t slot = (uint32_t)(hashKey(key) & mask);
  for (uint32_t probe = 0; probe <= mask; probe++) {
    uint64_t cur = d.keySet[slot];
    if (cur == key) return true;
    if (cur == kEmptyKey) return false;
    slot = (slot + 1) & mask;
  }
  return false;
}

// count build1 rows passing pred AND matching the key set
__global__ void jaCountBuild1Kernel(const JoinAggDesc* __restrict__ dp) {
  const JoinAggDesc& d = *dp;
  int64_t n = d.build1.nRows;
  uint64_t my = 0;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += (int64_t)gridDim.x * blockDim.x) {
    bool pass = d.nPred1 == 0 ||
                evalSimplePred(d.build1, d.pred1, d.strConst, d.strConstLen, row);
    if (!pass) continue;
    const DevCol& kc = d.build1.cols[d.b1ProbeCol];
    if (colIsNull(kc, row)) continue;
    if (!keySetHas(d, ((const uint64_t*)kc.data)[row])) continue;
    my++;
  }
  for (int off = 32; off > 0; off >>= 1) my += __shfl_down(my, off, 64);
  if ((threadIdx.x & 63) == 0 && my)
    atomicAdd((unsigned long long*)&d.counters[1], (unsigned long long)my);
}

// build the slot table from qualifying build1 rows
__global__ void jaBuild1Kernel(const JoinAggDesc* __restrict__ dp) {
  const JoinAggDesc& d = *dp;
  int64_t n = d.build1.nRows;
  uint32_t mask = (1u << d.slotsLog2) - 1;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += (int64_t)gridDim.x * blockDim.x) {
    bool pass = d.nPred1 == 0 ||
                evalSimplePred(d.build1, d.pred1, d.strConst, d.strConstLen, row);
    if (!pass) continue;
    const DevCol& pc = d.build1.cols[d.b1ProbeCol];
    if (colIsNull(pc, row)) continue;
    if (!keySetHas(d, ((const uint64_t*)pc.data)[row])) continue;
    const DevCol& kc = d.build1.cols[d.b1KeyCol];
    if (colIsNull(kc, row)) continue;
    uint64_t key = ((const uint64_t*)kc.data)[row];
    if (key == kEmptyKey) key = kEmptyKey - 1;
    uint64_t pay0 = d.payloadCol0 >= 0
        ? ((const uint64_t*)d.build1.cols[d.payloadCol0].data)[row] : 0;
    int64_t pay1 = d.payloadCol1 >= 0
        ? ((const int64_t*)d.build1.cols[d.payloadCol1].data)[row] : 0;
    bloomSet(d, key);
    uint32_t slot = (uint32_t)(hashKey(key) & mask);
    for (uint32_t probe = 0; probe <= mask; probe++) {
      uint64_t cur = d.slots[slot].key;
      if (cur == key) {
        // duplicate build key (one row per orderkey in Q3; general inner join
        // with duplicate build keys needs chaining — next round)
        atomicOr(d.errorFlag, kErrBadKey);
        break;
      }
      if (cur == kEmptyKey) {
        uint64_t prev = atomicCAS((unsigned long long*)&d.slots[slot].key,
                                  (unsigned long long)kEmptyKey,
                                  (unsigned long long)key);
        if (prev == kEmptyKey) {
          d.slots[slot].payload0 = pay0;
          d.slots[slot].payload1 = pay1;
          break;
        }
        if (prev == key) { atomicOr(d.errorFlag, kErrBadKey); break; }
      }
      slot = (slot + 1) & mask;
      if (probe == mask) atomicOr(d.errorFlag, kErrGlobalFull);
    }
  }
}

// probe: scan the probe table; pred -> VM value -> probe slots -> accumulate.
// Payload writes may race with probe reads only via the key CAS (published
// before probing starts: build and probe are separate kernel launches, so
// visibility comes from the dispatch boundary, not intra-kernel hand-off).
template <bool WIDE>
__launch_bounds__(256)
__global__ void jaProbeKernel(const JoinAggDesc* __restrict__ dp) {
  const JoinAggDesc& d = *dp;
  int64_t n = d.probe.nRows;
  uint32_t mask = (1u << d.slotsLog2) - 1;
  int64_t per = (n + gridDim.x - 1) / gridDim.x;
  int64_t begin = (int64_t)blockIdx.x * per;
  int64_t end = begin + per;
  if (end > n) end = n;
  bool failed = false;
  uint64_t myMatch = 0;
  for (int64_t row = begin + threadIdx.x; row < end && !failed;
       row += blockDim.x) {
    RawState raw;
    fetchRow(d.probe, d.fetch, d.nFetch, row, raw);
    // predicate (slot-fetched when planned)
    bool pass = d.nPredP == 0;
    if (!pass) {
      const PredDesc& pd = d.predP;
      const DevCol& c = d.probe.cols[pd.col];
      if (colIsNull(c, row)) pass = false;
      else if (pd.kind == PRED_TIME_CMP_CONST) {
        uint64_t v = (pd.slot >= 0 ? raw.get(pd.slot).x
                                   : gptr<uint64_t>(c.data)[row]) & ~0xFULL;
        uint64_t k = pd.constU64 & ~0xFULL;
        pass = cmpResult(v < k ? -1 : (v > k ? 1 : 0), pd.cmp);
      } else if (pd.kind == PRED_I64_CMP_CONST) {
        int64_t v = pd.slot >= 0 ? (int64_t)raw.get(pd.slot).x
                                 : gptr<int64_t>(c.data)[row];
        int64_t k = (int64_t)pd.constU64;
        pass = cmpResult(v < k ? -1 : (v > k ? 1 : 0), pd.cmp);
      } else {
        pass = evalSimplePred(d.probe, pd, d.strConst, d.strConstLen, row);
      }
    }
    if (!pass) continue;
    const DevCol& kc = d.probe.cols[d.pKeyCol];
    if (colIsNull(kc, row)) continue;
    uint64_t key = gptr<uint64_t>(kc.data)[row];
    if (key == kEmptyKey) key = kEmptyKey - 1;
    if (!bloomMayHave(d, key)) continue;  // L2-resident reject
    // probe the slot table (prebuilt: no inserts, miss -> drop row)
    uint32_t slot = (uint32_t)(hashKey(key) & mask);
    bool found = false;
    for (uint32_t probe = 0; probe <= mask; probe++) {
      uint64_t cur = gptr<uint64_t>(&d.slots[slot].key)[0];
      if (cur == key) { found = true; break; }
      if (cur == kEmptyKey) break;
      slot = (slot + 1) & mask;
    }
    if (!found) continue;
    // VM: compute the summed value
    VmState<WIDE> vm;
    vm.nullBits = 0;
    bool bad = false;
    bool ovf = false;
    for (int i = 0; i < d.nIns && !bad; i++) {
      const VmIns& ins = d.ins[i];
      switch (ins.op) {
        case VM_LOAD_DEC: {
          // lazy: only filter-surviving, join-matching rows reach the VM, so
          // value columns are loaded here (~5-10% of rows), not prefetched
          const DevCol& c = d.probe.cols[ins.a];
          bool nul = colIsNull(c, row);
          typename VT<WIDE>::T v = VT<WIDE>::zero();
          if (!nul) {
            int sc;
            bool okp = ins.c >= 0
                ? parseDecimalRaw<WIDE>(raw.get(ins.c), &v, &sc, d.errorFlag)
                : loadDecimalUnits<WIDE>((const uint8_t*)c.data + row * 40, &v,
                                         &sc, d.errorFlag);
            if (!okp) {
              bad = true;
              break;
            }
            if (sc != ins.b) {
              if (sc < ins.b) v = VT<WIDE>::scale10(v, ins.b - sc, &ovf);
              else { atomicOr(d.errorFlag, kErrScale); bad = true; break; }
            }
          }
          vm.set(ins.dst, v);
          vm.setNull(ins.dst, nul);
          break;
        }
        case VM_LOAD_I64: {
          const DevCol& c = d.probe.cols[ins.a];
          bool nul = colIsNull(c, row);
          int64_t lv = 0;
          if (!nul)
            lv = ins.c >= 0 ? (int64_t)raw.get(ins.c).x
                            : ((const int64_t*)c.data)[row];
          vm.set(ins.dst, nul ? VT<WIDE>::zero() : VT<WIDE>::fromI64(lv, &ovf));
          vm.setNull(ins.dst, nul);
          break;
        }
        case VM_LOAD_CONST: {
          if (WIDE) {
            Int128 cv = {(uint64_t)d.constLo[ins.a], d.constHi[ins.a]};
            vm.set(ins.dst, *(typename VT<WIDE>::T*)&cv);
          } else {
            int64_t cv = d.constLo[ins.a];
            vm.set(ins.dst, *(typename VT<WIDE>::T*)&cv);
          }
          vm.setNull(ins.dst, false);
          break;
        }
        case VM_ADD:
          vm.set(ins.dst, VT<WIDE>::add(vm.get(ins.a), vm.get(ins.b), &ovf));
          vm.setNull(ins.dst, vm.isNull(ins.a) || vm.isNull(ins.b));
          break;
        case VM_SUB:
          vm.set(ins.dst, VT<WIDE>::sub(vm.get(ins.a), vm.get(ins.b), &ovf));
          vm.setNull(ins.dst, vm.isNull(ins.a) || vm.isNull(ins.b));
          break;
        case VM_MUL: {
          bool nul = vm.isNull(ins.a) || vm.isNull(ins.b);
          typename VT<WIDE>::T v = VT<WIDE>::zero();
          if (!nul) v = VT<WIDE>::mul(vm.get(ins.a), vm.get(ins.b), &ovf);
          vm.set(ins.dst, v);
          vm.setNull(ins.dst, nul);
          break;
        }
        case VM_SCALE_UP:
          vm.set(ins.dst,
                 VT<WIDE>::mul(vm.get(ins.a),
                               VT<WIDE>::fromI64(d.insP10[i], nullptr), &ovf));
          vm.setNull(ins.dst, vm.isNull(ins.a));
          break;
      }
    }
    if (ovf) {
      atomicOr(d.errorFlag, WIDE ? kErrOverflow : kErrRetryWide);
      failed = true;
      break;
    }
    if (bad) { failed = true; break; }
    if (vm.isNull(d.valueReg)) continue;  // NULL never enters the sum
    Int128 v = VT<WIDE>::toAcc(vm.get(d.valueReg));
    JoinAggSlot* sp = &d.slots[slot];
    uint64_t old = atomicAdd((unsigned long long*)&sp->accLo, (unsigned long long)v.lo);
    uint64_t carry = (old + v.lo) < old ? 1 : 0;
    int64_t hiAdd = v.hi + (int64_t)carry;
    if (hiAdd != 0)
      atomicAdd((unsigned long long*)&sp->accHi, (unsigned long long)hiAdd);
    // the count atomic exists to mark the group matched; a contribution > 0
    // already proves that through acc (sums can only return to zero if some
    // contribution was <= 0, and those always count) -> one atomic per match
    // on all-positive data
    if (v.hi < 0 || (v.hi == 0 && v.lo == 0))
      atomicAdd((unsigned long long*)&sp->cnt, 1ULL);
    myMatch++;
  }
  for (int off = 32; off > 0; off >>= 1) myMatch += __shfl_down(myMatch, off, 64);
  if ((threadIdx.x & 63) == 0 && myMatch)
    atomicAdd((unsigned long long*)&d.counters[2], (unsigned long long)myMatch);
}

// top-N selection support: max accLo (revenues fit u64 in practice; accHi != 0
// falls back to host full sort via the error-free "big" path)
__global__ void jaMaxKernel(const JoinAggDesc* __restrict__ dp, uint64_t* outMax) {
  const JoinAggDesc& d = *dp;
  int64_t n = 1LL << d.slotsLog2;
  uint64_t my = 0;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const JoinAggSlot& sm = d.slots[i];
    if (sm.key == kEmptyKey || (sm.cnt == 0 && sm.accLo == 0 && sm.accHi == 0))
      continue;
    if (sm.accHi != 0) atomicOr(d.errorFlag, kErrOverflow);
    uint64_t v = sm.accLo;
    if (v > my) my = v;
  }
  for (int off = 32; off > 0; off >>= 1) {
    uint64_t o = __shfl_down(my, off, 64);
    if (o > my) my = o;
  }
  if ((threadIdx.x & 63) == 0) atomicMax((unsigned long long*)outMax, my);
}

__global__ void jaHistKernel(const JoinAggDesc* __restrict__ dp, uint32_t* hist,
                             int shift) {
  const JoinAggDesc& d = *dp;
  int64_t n = 1LL << d.slotsLog2;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const JoinAggSlot& sh = d.slots[i];
    if (sh.key == kEmptyKey || (sh.cnt == 0 && sh.accLo == 0 && sh.accHi == 0))
      continue;
    atomicAdd(&hist[(sh.accLo >> shift) & 4095], 1u);
  }
}

__global__ void jaCompactKernel(const JoinAggDesc* __restrict__ dp, TopNOut* out,
                                uint64_t* outCount, uint64_t thresholdBucket,
                                int shift, uint64_t cap) {
  const JoinAggDesc& d = *dp;
  int64_t n = 1LL << d.slotsLog2;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const JoinAggSlot& s = d.slots[i];
    if (s.key == kEmptyKey || (s.cnt == 0 && s.accLo == 0 && s.accHi == 0))
      continue;
    if ((s.accLo >> shift) < thresholdBucket) continue;
    uint64_t idx = atomicAdd((unsigned long long*)outCount, 1ULL);
    if (idx >= cap) { atomicOr(d.errorFlag, kErrGlobalFull); continue; }
    out[idx] = {s.key, s.payload0, s.payload1, s.accLo, s.accHi};
  }
}

// ==================================================================
// glds-staged fused aggregation
// ==================================================================
// Per-wave double-buffered tile pipeline: each wave owns tiles of 64 rows;
// all streams of tile t+1 are DMA'd HBM->LDS (`global_load_lds`, no VGPR
// round-trip) while tile t is consumed from LDS (parse + VM + aggregate).
// Counted `s_waitcnt vmcnt(N)` keeps the next tile's DMAs in flight across
// the compute (guide §5 'Pipelining across barriers'); all LDS lives in ONE
// dynamic __shared__ carve (a second __shared__ object would make hipcc
// drain vmcnt before every ds_read — guide §5 'Three .s-level traps').

typedef __attribute__((address_space(3))) char LdsChar;
typedef __attribute__((address_space(3))) GroupSlot LdsGroupSlot;

// LDS-table accumulate via explicit AS3 atomics (generic-pointer HIP atomics
// lower to flat ops, which count on vmcnt and would break the counted-wait
// DMA pipeline)
__device__ inline void accumIntoLds(LdsGroupSlot* slot, int a, Int128 v,
                                    int64_t dc) {
  if (v.lo != 0 || v.hi != 0) {
    uint64_t old = __hip_atomic_fetch_add(&slot->accLo[a], (uint64_t)v.lo,
                                          __ATOMIC_RELAXED,
                                          __HIP_MEMORY_SCOPE_WORKGROUP);
    uint64_t carry = (old + v.lo) < old ? 1 : 0;
    int64_t hiAdd = v.hi + (int64_t)carry;
    if (hiAdd != 0)
      __hip_atomic_fetch_add((__attribute__((address_space(3)))uint64_t*)&slot->accHi[a], (uint64_t)hiAdd,
                             __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_WORKGROUP);
  }
  if (dc != 0)
    __hip_atomic_fetch_add((__attribute__((address_space(3)))uint64_t*)&slot->cnt[a], (uint64_t)dc,
                           __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_WORKGROUP);
}

__device__ inline void gldsWaitVmcnt(int n) {
  switch (n) {
    case 0: asm volatile("s_waitcnt vmcnt(0)" ::: "memory"); break;
    case 1: asm volatile("s_waitcnt vmcnt(1)" ::: "memory"); break;
    case 2: asm volatile("s_waitcnt vmcnt(2)" ::: "memory"); break;
    case 3: asm volatile("s_waitcnt vmcnt(3)" ::: "memory"); break;
    case 4: asm volatile("s_waitcnt vmcnt(4)" ::: "memory"); break;
    case 5: asm volatile("s_waitcnt vmcnt(5)" ::: "memory"); break;
    case 6: asm volatile("s_waitcnt vmcnt(6)" ::: "memory"); break;
    case 7: asm volatile("s_waitcnt vmcnt(7)" ::: "memory"); break;
    default: asm volatile("s_waitcnt vmcnt(8)" ::: "memory"); break;
  }
}

// issue all stream DMAs for one 64-row tile into tileBuf (wave-uniform base)
__device__ __attribute__((always_inline)) inline void gldsIssueTile(
    const FusedQueryDesc& d, int64_t row0, int64_t clampMax, LdsChar* tileBuf,
    int lane) {
  for (int f0 = 0; f0 < d.nFetch; f0++) {
    int f = __builtin_amdgcn_readfirstlane(f0);  // uniform -> scalar desc loads
    const FetchDesc& fd = d.fetch[f];
    const DevCol& c = d.table.cols[__builtin_amdgcn_readfirstlane(fd.col)];
    auto lptr = (__attribute__((address_space(3))) void*)(tileBuf + fd.ldsOff);
    if (fd.kind == FETCH_DEC16) {
      int64_t r = row0 + lane;
      if (r > clampMax) r = clampMax;  // tail rows re-read the last row
      auto g = (const __attribute__((address_space(1))) void*)
          ((const uint8_t*)c.data + r * 40);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) uint32_t*)g,
          (__attribute__((address_space(3))) uint32_t*)lptr, 16, 0, 0);
    } else if (fd.kind == FETCH_8B) {
      if (lane < 32) {
        int64_t r = row0 + 2 * lane;
        if (r > clampMax) r = clampMax;
        auto g = (const __attribute__((address_space(1))) void*)
            ((const uint8_t*)c.data + r * 8);
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) uint32_t*)g,
            (__attribute__((address_space(3))) uint32_t*)lptr, 16, 0, 0);
      }
    } else {  // FETCH_B1
      if (lane < 16) {
        int64_t r = row0 + 4 * lane;
        if (r > clampMax - 3) r = clampMax - 3 < 0 ? 0 : clampMax - 3;
        auto g = (const __attribute__((address_space(1))) void*)
            ((const uint8_t*)c.data + r);
        __builtin_amdgcn_global_load_lds(
            (const __attribute__((address_space(1))) uint32_t*)g,
            (__attribute__((address_space(3))) uint32_t*)lptr, 4, 0, 0);
      }
    }
  }
}

// LDS-backed raw accessor for one row (lane) of the staged tile. The buffer
// pointer is LDS-qualified so reads lower to ds_read (a generic pointer
// would emit flat loads, which count on vmcnt and break the pipeline).
struct LdsRaw {
  const LdsChar* buf;
  const FusedQueryDesc* d;  // LDS-resident copy (see kernel prologue)
  int lane;
  __device__ ulonglong2 get(int slot) const {
    const FetchDesc& fd = d->fetch[__builtin_amdgcn_readfirstlane(slot)];
    if (fd.kind == FETCH_DEC16) {
      auto p = (const __attribute__((address_space(3))) uint64_t*)(buf + fd.ldsOff + lane * 16);
      ulonglong2 v;
      v.x = p[0];
      v.y = p[1];
      return v;
    }
    if (fd.kind == FETCH_8B) {
      ulonglong2 v;
      v.x = *(const __attribute__((address_space(3))) uint64_t*)(buf + fd.ldsOff + lane * 8);
      v.y = 0;
      return v;
    }
    ulonglong2 v;
    v.x = *(const __attribute__((address_space(3))) uint8_t*)(buf + fd.ldsOff + lane);
    v.y = 0;
    return v;
  }
};

// group key from staged streams (dense-char via B1 slot, i64 via 8B slot)
template <typename RAW>
__device__ inline bool makeGroupKeyStaged(const FusedQueryDesc& d,
                                          const RAW& raw, uint64_t* keyOut,
                                          uint32_t* err) {
  uint64_t key = 0;
  for (int k0 = 0; k0 < d.gkey.nCols; k0++) {
    int k = __builtin_amdgcn_readfirstlane(k0);
    uint32_t lane32;
    if (d.gkey.kind[k] == 2) {
      uint8_t b = (uint8_t)raw.get(d.gkey.slot[k]).x;
      lane32 = b == ' ' ? 0u : ((1u << 24) | b);
    } else {  // kind 1: small i64
      int64_t v = (int64_t)raw.get(d.gkey.slot[k]).x;
      if (v < 0 || v > 0x7FFFFFFF) { atomicOr(err, kErrBadKey); return false; }
      lane32 = (uint32_t)v;
    }
    key |= (uint64_t)lane32 << (32 * k);
  }
  if (d.gkey.nCols == 0) key = 0;
  if (key == kEmptyKey) key = kEmptyKey - 1;
  *keyOut = key;
  return true;
}

// row pipeline over a staged tile row (mirrors processRow; RAW = LdsRaw)
template <bool WIDE>
__device__ __attribute__((always_inline)) inline bool processRowStaged(
    const FusedQueryDesc& d, const LdsRaw& raw, LdsGroupSlot* lds,
    uint64_t* mySel) {
  bool pass = true;
  for (int p0 = 0; p0 < d.nPreds && pass; p0++) {
    const PredDesc& pd = d.preds[__builtin_amdgcn_readfirstlane(p0)];
    if (pd.kind == PRED_TIME_CMP_CONST) {
      uint64_t v = raw.get(pd.slot).x & ~0xFULL;
      uint64_t k = pd.constU64 & ~0xFULL;
      pass = cmpResult(v < k ? -1 : (v > k ? 1 : 0), pd.cmp);
    } else if (pd.kind == PRED_I64_CMP_CONST) {
      int64_t v = (int64_t)raw.get(pd.slot).x;
      int64_t k = (int64_t)pd.constU64;
      pass = cmpResult(v < k ? -1 : (v > k ? 1 : 0), pd.cmp);
    } else {  // decimal pred via its DEC16 slot
      typename VT<WIDE>::T u;
      int sc;
      if (!parseDecimalRaw<WIDE>(raw.get(pd.slot), &u, &sc, d.errorFlag))
        return false;
      int cmp = VT<WIDE>::cmp(u, VT<WIDE>::fromI64((int64_t)pd.constU64, nullptr));
      pass = cmpResult(cmp, pd.cmp);
    }
  }
  if (!pass) return true;
  (*mySel)++;

  VmState<WIDE> vm;
  vm.nullBits = 0;
  bool bad = false;
  bool ovf = false;
  for (int i0 = 0; i0 < d.nIns && !bad; i0++) {
    const VmIns& ins = d.ins[__builtin_amdgcn_readfirstlane(i0)];
    switch (ins.op) {
      case VM_LOAD_DEC: {
        typename VT<WIDE>::T v = VT<WIDE>::zero();
        int sc;
        if (!parseDecimalRaw<WIDE>(raw.get(ins.c), &v, &sc, d.errorFlag,
                                   ins.b, d.insP10[i0], d.insMagic[i0])) {
          bad = true;
          break;
        }
        if (sc != ins.b) {
          if (sc < ins.b) v = VT<WIDE>::scale10(v, ins.b - sc, &ovf);
          else { atomicOr(d.errorFlag, kErrScale); bad = true; break; }
        }
        vm.set(ins.dst, v);
        vm.setNull(ins.dst, false);
        break;
      }
      case VM_LOAD_I64:
        vm.set(ins.dst, VT<WIDE>::fromI64((int64_t)raw.get(ins.c).x, &ovf));
        vm.setNull(ins.dst, false);
        break;
      case VM_LOAD_CONST: {
        if (WIDE) {
          Int128 cv = {(uint64_t)d.constLo[ins.a], d.constHi[ins.a]};
          vm.set(ins.dst, *(typename VT<WIDE>::T*)&cv);
        } else {
          int64_t cv = d.constLo[ins.a];
          vm.set(ins.dst, *(typename VT<WIDE>::T*)&cv);
        }
        vm.setNull(ins.dst, false);
        break;
      }
      case VM_ADD:
        vm.set(ins.dst, VT<WIDE>::add(vm.get(ins.a), vm.get(ins.b), &ovf));
        break;
      case VM_SUB:
        vm.set(ins.dst, VT<WIDE>::sub(vm.get(ins.a), vm.get(ins.b), &ovf));
        break;
      case VM_MUL:
        vm.set(ins.dst, VT<WIDE>::mul(vm.get(ins.a), vm.get(ins.b), &ovf));
        break;
      case VM_SCALE_UP:
        vm.set(ins.dst,
               VT<WIDE>::mul(vm.get(ins.a),
                             VT<WIDE>::fromI64(d.insP10[i0], nullptr), &ovf));
        break;
    }
  }
  if (ovf) {
    atomicOr(d.errorFlag, WIDE ? kErrOverflow : kErrRetryWide);
    return false;
  }
  if (bad) return false;
  if (d.ablate == 1) {
    uint64_t sink = 0;
    for (int a = 0; a < d.nAggs; a++)
      if (d.aggs[a].srcReg >= 0)
        sink ^= (uint64_t)VT<WIDE>::toAcc(vm.get(d.aggs[a].srcReg)).lo;
    asm volatile("" ::"v"(sink));
    return true;
  }

  uint64_t key;
  if (!makeGroupKeyStaged(d, raw, &key, d.errorFlag)) return false;
  // (the glds variant always uses the LDS table; the engine runs the plain
  // kernel for the high-NDV global-direct retry)
  uint32_t slot = (uint32_t)(splitmix64(key) & (kLdsGroups - 1));
  for (int probe = 0;; probe++) {
    if (probe >= kLdsGroups) { atomicOr(d.errorFlag, kErrLdsFull); return false; }
    uint64_t cur = lds[slot].key;
    if (cur == key) break;
    if (cur == kEmptyKey) {
      uint64_t expected = kEmptyKey;
      bool won = __hip_atomic_compare_exchange_strong(
          &lds[slot].key, &expected, key, __ATOMIC_RELAXED, __ATOMIC_RELAXED,
          __HIP_MEMORY_SCOPE_WORKGROUP);
      if (won || expected == key) break;
    }
    slot = (slot + 1) & (kLdsGroups - 1);
  }
  LdsGroupSlot* target = &lds[slot];
  if (d.sharedCnt) accumIntoLds(target, 0, Int128{0, 0}, 1);
  for (int s0 = 0; s0 < d.nAccSlots; s0++) {
    int s = __builtin_amdgcn_readfirstlane(s0);
    int reg = d.accReg[s];
    accumIntoLds(target, s, VT<WIDE>::toAcc(vm.get(reg)), 0);
  }
  if (!d.sharedCnt) {
    for (int a0 = 0; a0 < d.nAggs; a0++) {
      int a = __builtin_amdgcn_readfirstlane(a0);
      if (d.aggs[a].fr >= 0) continue;
      accumIntoLds(target, a, Int128{0, 0}, 1);
    }
  }
  return true;
}

template <bool WIDE>
__launch_bounds__(256)
__global__ void fusedAggGldsKernel(const FusedQueryDesc* __restrict__ dp) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr size_t kTableBytes = (sizeof(GroupSlot) * kLdsGroups + 15) & ~15ULL;
  auto lds = (LdsGroupSlot*)(LdsChar*)smem;
  auto tilesBase = (LdsChar*)smem + kTableBytes;
  for (int i = threadIdx.x; i < kLdsGroups; i += blockDim.x) {
    lds[i].key = kEmptyKey;
    for (int a = 0; a < kMaxAggs; a++) {
      lds[i].accLo[a] = 0;
      lds[i].accHi[a] = 0;
      lds[i].cnt[a] = 0;
    }
  }
  __syncthreads();
  const FusedQueryDesc& d = *dp;

  int64_t n = d.table.nRows;
  int64_t per = ((n + gridDim.x - 1) / gridDim.x + 63) & ~63LL;
  int64_t begin = (int64_t)blockIdx.x * per;
  int64_t end = begin + per;
  if (end > n) end = n;
  int wave = threadIdx.x >> 6;
  int lane = threadIdx.x & 63;
  LdsChar* myTiles = tilesBase + (size_t)wave * 2 * d.tileBytes;
  uint64_t mySel = 0;
  bool failed = false;

  // tiles of 64 rows, strided across the block's 4 waves
  int64_t range = end - begin;
  int64_t nTiles = (range + 63) / 64;
  int64_t t = wave;
  if (t < nTiles)
    gldsIssueTile(d, begin + t * 64, end - 1, myTiles, lane);
  int buf = 0;
  for (; t < nTiles && !failed; t += 4) {
    int64_t nextT = t + 4;
    bool haveNext = nextT < nTiles;
    if (haveNext)
      gldsIssueTile(d, begin + nextT * 64, end - 1, myTiles + (buf ^ 1) * d.tileBytes,
                    lane);
    // wait for tile t's DMAs (the next tile's stay in flight)
    gldsWaitVmcnt(haveNext ? d.nFetch : 0);
    int64_t row = begin + t * 64 + lane;
    if (row < end) {
      LdsRaw raw{myTiles + buf * d.tileBytes, &d, lane};
      if (!processRowStaged<WIDE>(d, raw, lds, &mySel)) failed = true;
    }
    buf ^= 1;
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");

  if (d.selCount) {
    uint64_t total = mySel;
    for (int off = 32; off > 0; off >>= 1)
      total += __shfl_down(total, off, 64);
    if ((threadIdx.x & 63) == 0 && total)
      atomicAdd((unsigned long long*)d.selCount, (unsigned long long)total);
  }
  __syncthreads();

  for (int i = threadIdx.x; i < kLdsGroups; i += blockDim.x) {
    if (lds[i].key == kEmptyKey) continue;
    uint64_t key = lds[i].key;
    uint32_t slot = (uint32_t)(splitmix64(key) & (kGlobalGroups - 1));
    bool ok = true;
    for (int probe = 0;; probe++) {
      if (probe >= kGlobalGroups) { atomicOr(d.errorFlag, kErrGlobalFull); ok = false; break; }
      uint64_t cur = d.globalTable[slot].key;
      if (cur == key) break;
      if (cur == kEmptyKey) {
        uint64_t prev = atomicCAS((unsigned long long*)&d.globalTable[slot].key,
                                  (unsigned long long)kEmptyKey,
                                  (unsigned long long)key);
        if (prev == kEmptyKey || prev == key) break;
      }
      slot = (slot + 1) & (kGlobalGroups - 1);
    }
    if (!ok) continue;
    for (int s = 0; s < d.nAccSlots; s++) {
      Int128 v = {lds[i].accLo[s], lds[i].accHi[s]};
      accumInto(&d.globalTable[slot], s, v, 0);
    }
    int nCnt = d.sharedCnt ? 1 : d.nAggs;
    for (int a = 0; a < nCnt; a++)
      accumInto(&d.globalTable[slot], a, Int128{0, 0}, lds[i].cnt[a]);
  }
}

__global__ void jaInitSlotsKernel(JoinAggSlot* slots, int64_t n) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  slots[i].key = kEmptyKey;
  slots[i].payload0 = 0;
  slots[i].payload1 = 0;
  slots[i].accLo = 0;
  slots[i].accHi = 0;
  slots[i].cnt = 0;
}

// ---- orders / customer generators ----
__global__ void genOrdersKernel(DevTable tab, int64_t rowBegin, int64_t nRows,
                                uint64_t seed, int64_t totalRows) {
  int64_t nCust = totalRows / 10;
  if (nCust < 1) nCust = 1;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nRows;
       i += stride) {
    int64_t row = rowBegin + i;
    ((int64_t*)tab.cols[0].data)[i] = row + 1;
    ((int64_t*)tab.cols[1].data)[i] =
        1 + (int64_t)(fieldRand(seed, row, 1) % (uint64_t)nCust);
    int y, m, dd;
    civilFromDays(GX_EPOCH_1992 +
                      (int64_t)(fieldRand(seed, row, 2) % (uint64_t)GX_ORDERDATE_DAYS),
                  &y, &m, &dd);
    ((uint64_t*)tab.cols[2].data)[i] = timeFromDate(y, m, dd);
    ((int64_t*)tab.cols[3].data)[i] = 0;
  }
}

__device__ __constant__ const char kSegChars[5][11] = {
    "AUTOMOBILE", "BUILDING", "FURNITURE", "MACHINERY", "HOUSEHOLD"};
__device__ __constant__ const int kSegLens[5] = {10, 8, 9, 9, 9};

__global__ void genCustomerLenKernel(int64_t* lens, int64_t rowBegin,
                                     int64_t nRows, uint64_t seed) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nRows;
       i += stride)
    lens[i] = kSegLens[fieldRand(seed, rowBegin + i, 1) % 5];
}

__global__ void genCustomerFillKernel(DevTable tab, int64_t rowBegin,
                                      int64_t nRows, uint64_t seed) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < nRows;
       i += stride) {
    int64_t row = rowBegin + i;
    ((int64_t*)tab.cols[0].data)[i] = row + 1;
    int seg = (int)(fieldRand(seed, row, 1) % 5);
    int64_t off = tab.cols[1].offsets[i];
    for (int j = 0; j < kSegLens[seg]; j++)
      ((uint8_t*)tab.cols[1].data)[off + j] = kSegChars[seg][j];
  }
}

// ---------------- host launch wrappers ----------------

int gxLaunchTpchGen(int table, DevTable* devTab, int64_t rowBegin, int64_t nRows,
                    uint64_t seed, int64_t totalRows, void* stream) {
  hipStream_t s = (hipStream_t)stream;
  int block = 256;
  int grid = (int)((nRows + block - 1) / block);
  if (grid > 4096) grid = 4096;
  if (grid < 1) grid = 1;
  if (table == 0) {
    hipLaunchKernelGGL(genLineitemKernel, dim3(grid), dim3(block), 0, s, *devTab,
                       rowBegin, nRows, seed, totalRows);
  } else {
    return -1;  // orders/customer device generation lands with the Q3 path
  }
  return (int)hipGetLastError();
}

__global__ void dumpDescKernel(const FusedQueryDesc* dp) {
  if (threadIdx.x || blockIdx.x) return;
  const FusedQueryDesc& d = *dp;
  printf("[dev] nIns=%d nAggs=%d nAccSlots=%d sharedCnt=%d wide=%d\n",
         d.nIns, d.nAggs, d.nAccSlots, d.sharedCnt, d.wide);
  for (int i = 0; i < d.nIns; i++)
    printf("[dev] ins[%d] op=%d dst=%d a=%d b=%d c=%d p10=%lld\n", i,
           d.ins[i].op, d.ins[i].dst, d.ins[i].a, d.ins[i].b, d.ins[i].c,
           (long long)d.insP10[i]);
  for (int a = 0; a < d.nAggs; a++)
    printf("[dev] agg[%d] func=%d srcReg=%d scale=%d fr=%d accMap=%d\n", a,
           d.aggs[a].func, d.aggs[a].srcReg, d.aggs[a].scale, d.aggs[a].fr,
           d.accMap[a]);
  for (int s = 0; s < d.nAccSlots; s++)
    printf("[dev] accReg[%d]=%d\n", s, d.accReg[s]);
  printf("[dev] kP10(18)=%lld sizeof(desc)=%d aggs_off=%d ins_off=%d\n",
         (long long)kP10(18), (int)sizeof(FusedQueryDesc),
         (int)((char*)&d.aggs[0] - (char*)&d), (int)((char*)&d.ins[0] - (char*)&d));
}

int gxDumpDesc(const FusedQueryDesc* devDesc, void* stream) {
  hipLaunchKernelGGL(dumpDescKernel, dim3(1), dim3(1), 0, (hipStream_t)stream, devDesc);
  hipStreamSynchronize((hipStream_t)stream);
  return (int)hipGetLastError();
}

int gxLaunchFusedAgg(const FusedQueryDesc& desc, const FusedQueryDesc* devDesc,
                     void* stream) {
  hipStream_t s = (hipStream_t)stream;
  int grid = (int)((desc.table.nRows + 255) / 256);
  // >> 256 workgroups to fill 256 CUs / 8 XCDs; cap and grid-stride per block
  if (grid > 2048) grid = 2048;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(initGlobalTableKernel, dim3((kGlobalGroups + 255) / 256),
                     dim3(256), 0, s, desc.globalTable, kGlobalGroups);
  if (desc.useGlds) {
    size_t shmem = ((sizeof(GroupSlot) * kLdsGroups + 15) & ~15ULL) +
                   (size_t)4 * 2 * desc.tileBytes;
    if (desc.wide)
      hipLaunchKernelGGL((fusedAggGldsKernel<true>), dim3(grid), dim3(256),
                         shmem, s, devDesc);
    else
      hipLaunchKernelGGL((fusedAggGldsKernel<false>), dim3(grid), dim3(256),
                         shmem, s, devDesc);
  } else {
    // pick the 5-slot raw-state variant when every non-staged fetch slot
    // fits (higher occupancy for Q1-shaped queries)
    int maxSlot = -1;
    for (int f = 0; f < desc.nFetch; f++)
      if (desc.fetch[f].kind != FETCH_B1 && f > maxSlot) maxSlot = f;
    bool small = maxSlot < 5;
    if (desc.hasDiv) {
      if (desc.wide) {
        if (small)
          hipLaunchKernelGGL((fusedAggKernel<true, 5, true>), dim3(grid), dim3(256), 0, s, devDesc);
        else
          hipLaunchKernelGGL((fusedAggKernel<true, 8, true>), dim3(grid), dim3(256), 0, s, devDesc);
      } else {
        if (small)
          hipLaunchKernelGGL((fusedAggKernel<false, 5, true>), dim3(grid), dim3(256), 0, s, devDesc);
        else
          hipLaunchKernelGGL((fusedAggKernel<false, 8, true>), dim3(grid), dim3(256), 0, s, devDesc);
      }
    } else if (desc.wide) {
      if (small)
        hipLaunchKernelGGL((fusedAggKernel<true, 5>), dim3(grid), dim3(256), 0, s, devDesc);
      else
        hipLaunchKernelGGL((fusedAggKernel<true, 8>), dim3(grid), dim3(256), 0, s, devDesc);
    } else {
      if (small)
        hipLaunchKernelGGL((fusedAggKernel<false, 5>), dim3(grid), dim3(256), 0, s, devDesc);
      else
        hipLaunchKernelGGL((fusedAggKernel<false, 8>), dim3(grid), dim3(256), 0, s, devDesc);
    }
  }
  return (int)hipGetLastError();
}

int gxLaunchMemset(void* p, int v, size_t n, void* stream) {
  return (int)hipMemsetAsync(p, v, n, (hipStream_t)stream);
}

static int gridFor(int64_t n) {
  int g = (int)((n + 255) / 256);
  if (g > 2048) g = 2048;
  if (g < 1) g = 1;
  return g;
}

int gxJoinAggPhase(int phase, const JoinAggDesc* devDesc, const JoinAggDesc& h,
                   void* stream) {
  hipStream_t s = (hipStream_t)stream;
  switch (phase) {
    case 0:
      hipLaunchKernelGGL(jaCountBuild0Kernel, dim3(gridFor(h.build0.nRows)),
                         dim3(256), 0, s, devDesc);
      break;
    case 1:
      hipLaunchKernelGGL(jaBuild0Kernel, dim3(gridFor(h.build0.nRows)),
                         dim3(256), 0, s, devDesc);
      break;
    case 2:
      hipLaunchKernelGGL(jaCountBuild1Kernel, dim3(gridFor(h.build1.nRows)),
                         dim3(256), 0, s, devDesc);
      break;
    case 3:
      hipLaunchKernelGGL(jaBuild1Kernel, dim3(gridFor(h.build1.nRows)),
                         dim3(256), 0, s, devDesc);
      break;
    case 4:
      if (h.wide)
        hipLaunchKernelGGL((jaProbeKernel<true>), dim3(gridFor(h.probe.nRows)),
                           dim3(256), 0, s, devDesc);
      else
        hipLaunchKernelGGL((jaProbeKernel<false>), dim3(gridFor(h.probe.nRows)),
                           dim3(256), 0, s, devDesc);
      break;
    case 5: {
      int64_t n = 1LL << h.slotsLog2;
      hipLaunchKernelGGL(jaInitSlotsKernel, dim3((n + 255) / 256), dim3(256), 0, s,
                         h.slots, n);
      break;
    }
    default:
      return -1;
  }
  return (int)hipGetLastError();
}

int gxJoinAggMax(const JoinAggDesc* devDesc, const JoinAggDesc& h, uint64_t* devMax,
                 void* stream) {
  hipLaunchKernelGGL(jaMaxKernel, dim3(gridFor(1LL << h.slotsLog2)), dim3(256), 0,
                     (hipStream_t)stream, devDesc, devMax);
  return (int)hipGetLastError();
}

int gxJoinAggHist(const JoinAggDesc* devDesc, const JoinAggDesc& h,
                  uint32_t* devHist, int shift, void* stream) {
  hipLaunchKernelGGL(jaHistKernel, dim3(gridFor(1LL << h.slotsLog2)), dim3(256), 0,
                     (hipStream_t)stream, devDesc, devHist, shift);
  return (int)hipGetLastError();
}

int gxJoinAggCompact(const JoinAggDesc* devDesc, const JoinAggDesc& h,
                     TopNOut* out, uint64_t* outCount, uint64_t thresholdBucket,
                     int shift, uint64_t cap, void* stream) {
  hipLaunchKernelGGL(jaCompactKernel, dim3(gridFor(1LL << h.slotsLog2)), dim3(256),
                     0, (hipStream_t)stream, devDesc, out, outCount,
                     thresholdBucket, shift, cap);
  return (int)hipGetLastError();
}

int gxGenOrders(DevTable* tab, int64_t rowBegin, int64_t nRows, uint64_t seed,
                int64_t totalRows, void* stream) {
  hipLaunchKernelGGL(genOrdersKernel, dim3(gridFor(nRows)), dim3(256), 0,
                     (hipStream_t)stream, *tab, rowBegin, nRows, seed, totalRows);
  return (int)hipGetLastError();
}

int gxGenCustomerOffsets(DevTable* tab, int64_t rowBegin, int64_t nRows,
                         uint64_t seed, void* stream, long long* totalBytes) {
  hipStream_t s = (hipStream_t)stream;
  // lens in the offsets buffer's tail is unsafe; use a scratch buffer
  int64_t* lens = nullptr;
  if (hipMalloc(&lens, (nRows + 1) * 8) != hipSuccess) return -2;
  hipLaunchKernelGGL(genCustomerLenKernel, dim3(gridFor(nRows)), dim3(256), 0, s,
                     lens, rowBegin, nRows, seed);
  hipMemsetAsync(lens + nRows, 0, 8, s);
  void* tmp = nullptr;
  size_t tmpBytes = 0;
  hipcub::DeviceScan::ExclusiveSum(nullptr, tmpBytes, lens, tab->cols[1].offsets,
                                   nRows + 1, s);
  if (hipMalloc(&tmp, tmpBytes) != hipSuccess) { hipFree(lens); return -2; }
  hipcub::DeviceScan::ExclusiveSum(tmp, tmpBytes, lens, tab->cols[1].offsets,
                                   nRows + 1, s);
  long long total = 0;
  hipError_t e = hipMemcpyAsync(&total, tab->cols[1].offsets + nRows, 8,
                                hipMemcpyDeviceToHost, s);
  if (e == hipSuccess) e = hipStreamSynchronize(s);
  hipFree(tmp);
  hipFree(lens);
  if (e != hipSuccess) return (int)e;
  *totalBytes = total;
  return (int)hipGetLastError();
}

int gxGenCustomerFill(DevTable* tab, int64_t rowBegin, int64_t nRows,
                      uint64_t seed, void* stream) {
  hipLaunchKernelGGL(genCustomerFillKernel, dim3(gridFor(nRows)), dim3(256), 0,
                     (hipStream_t)stream, *tab, rowBegin, nRows, seed);
  return (int)hipGetLastError();
}

}  // namespace gxp

namespace gxp {

// ---- device full sort support (sortexec/sort.go analog) ----

__global__ void sortIotaKernel(uint32_t* idx, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    idx[i] = (uint32_t)i;
}

// order-preserving u64 key for one sort column, gathered through idx so each
// stable LSD pass sorts the CURRENT permutation by the next-outer key
__global__ void sortComposeKeysKernel(DevTable tab, SortKeyCompose k,
                                      const uint32_t* __restrict__ idx,
                                      uint64_t* __restrict__ keys, int64_t n,
                                      uint32_t* errFlag) {
  const DevCol& c = tab.cols[k.col];
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t row = idx[i];
    uint64_t key;
    if (k.kind == 0) {  // int64: flip sign bit
      key = gptr<uint64_t>(c.data)[row] ^ 0x8000000000000000ULL;
    } else if (k.kind == 1) {  // packed CoreTime: masked compare order
      key = gptr<uint64_t>(c.data)[row] & ~0xFULL;
    } else if (k.kind == 2) {  // dense char(1); PAD SPACE: ' ' == ''
      uint8_t b = gptr<uint8_t>(c.data)[row];
      key = b == ' ' ? 0 : b;
    } else {  // decimal -> int64 units at the column's declared frac
      typename VT<false>::T v = 0;
      int sc = 0;
      if (!parseDecimalRaw<false>(
              ulonglong2{gptr<uint64_t>(c.data)[row * 40 / 8],
                         gptr<uint64_t>((const uint8_t*)c.data + row * 40 + 8)[0]},
              &v, &sc, errFlag))
        v = 0;  // flag set; result unusable anyway
      if (sc != c.frac) {
        bool ovf = false;
        if (sc < c.frac) v = VT<false>::scale10(v, c.frac - sc, &ovf);
        if (ovf || sc > c.frac) atomicOr(errFlag, kErrScale);
      }
      key = (uint64_t)v ^ 0x8000000000000000ULL;
    }
    if (k.desc) key = ~key;
    keys[i] = key;
  }
}

__global__ void sortGatherKernel(const uint8_t* __restrict__ in,
                                 uint8_t* __restrict__ out,
                                 const uint32_t* __restrict__ idx, int64_t n,
                                 int elemSize) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t src = idx[i];
    if (elemSize == 8) {
      ((uint64_t*)out)[i] = ((const uint64_t*)in)[src];
    } else if (elemSize == 1) {
      out[i] = in[src];
    } else {  // 40B decimal: five 8B words
      const uint64_t* s = (const uint64_t*)(in + src * 40);
      uint64_t* d = (uint64_t*)(out + i * 40);
      d[0] = s[0]; d[1] = s[1]; d[2] = s[2]; d[3] = s[3]; d[4] = s[4];
    }
  }
}

int gxSortIota(uint32_t* idx, int64_t n, void* stream) {
  hipLaunchKernelGGL(sortIotaKernel, dim3(gridFor(n)), dim3(256), 0,
                     (hipStream_t)stream, idx, n);
  return (int)hipGetLastError();
}

int gxSortComposeKeys(const DevTable* tab, const DevTable& htab,
                      SortKeyCompose k, const uint32_t* idx, uint64_t* keys,
                      int64_t n, uint32_t* errFlag, void* stream) {
  (void)tab;
  hipLaunchKernelGGL(sortComposeKeysKernel, dim3(gridFor(n)), dim3(256), 0,
                     (hipStream_t)stream, htab, k, idx, keys, n, errFlag);
  return (int)hipGetLastError();
}

// OR / AND reductions over the composed keys: radix passes can skip bit
// positions that are identical across every key (begin_bit = first
// differing bit, end_bit = last + 1) -- dates pack into bits 41..54 and
// dense int keys into the low bits, cutting 8 passes to 2-3.
__global__ void sortKeyBitsKernel(const uint64_t* __restrict__ keys, int64_t n,
                                  uint64_t* orOut, uint64_t* andOut) {
  uint64_t o = 0, a = ~0ULL;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    o |= keys[i];
    a &= keys[i];
  }
  for (int off = 32; off > 0; off >>= 1) {
    o |= __shfl_down(o, off, 64);
    a &= __shfl_down(a, off, 64);
  }
  if ((threadIdx.x & 63) == 0) {
    atomicOr((unsigned long long*)orOut, (unsigned long long)o);
    atomicAnd((unsigned long long*)andOut, (unsigned long long)a);
  }
}

int gxSortKeyBits(const uint64_t* keys, int64_t n, uint64_t* devOrAnd,
                  void* stream) {
  hipMemsetAsync(devOrAnd, 0, 8, (hipStream_t)stream);
  hipMemsetAsync(devOrAnd + 1, 0xFF, 8, (hipStream_t)stream);
  hipLaunchKernelGGL(sortKeyBitsKernel, dim3(gridFor(n)), dim3(256), 0,
                     (hipStream_t)stream, keys, n, devOrAnd, devOrAnd + 1);
  return (int)hipGetLastError();
}

int gxSortPairs(uint64_t* keysIn, uint64_t* keysOut, uint32_t* idxIn,
                uint32_t* idxOut, int64_t n, void* tmp, size_t* tmpBytes,
                int beginBit, int endBit, void* stream) {
  return (int)hipcub::DeviceRadixSort::SortPairs(tmp, *tmpBytes, keysIn,
                                                 keysOut, idxIn, idxOut,
                                                 (int)n, beginBit, endBit,
                                                 (hipStream_t)stream);
}

int gxSortGatherCol(const void* in, void* out, const uint32_t* idx, int64_t n,
                    int elemSize, void* stream) {
  hipLaunchKernelGGL(sortGatherKernel, dim3(gridFor(n)), dim3(256), 0,
                     (hipStream_t)stream, (const uint8_t*)in, (uint8_t*)out,
                     idx, n, elemSize);
  return (int)hipGetLastError();
}

}  // namespace gxp
