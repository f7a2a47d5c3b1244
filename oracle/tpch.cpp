// oracle/tpch.cpp — deterministic synthetic TPC-H-shaped generators.
// ORACLE / TEST INFRASTRUCTURE + data spec of record.
//
// This file DEFINES the synthetic data (seed, PRNG, distributions); the
// product engine's device generator (tidb_amd/csrc/gx_tpch.hip) restates the
// identical algorithm — parity tests assert the two produce identical chunks.
//
// Distributions follow SURVEY.md §8d (MockDataSource analog,
// pkg/executor/internal/testutil/testutil.go:45-348):
//   lineitem: l_orderkey = 1 + u % n_orders (n_orders = total/4);
//     l_quantity dec(15,2) 1..50; l_extendedprice dec(15,2) 901.00..104950.00;
//     l_discount 0.00..0.10; l_tax 0.00..0.08; l_returnflag in {A,N,R};
//     l_linestatus in {O,F}; l_shipdate uniform 1992-01-01..1998-12-01.
//   orders: o_orderkey dense row+1; o_custkey = 1 + u % n_cust (n_cust =
//     total/10); o_orderdate uniform 1992-01-01..1998-08-02; o_shippriority 0.
//   customer: c_custkey dense row+1; c_mktsegment 1-of-5 segments.
#include <cstdint>

#include "core_time.h"
#include "exec.h"

namespace oracle {

namespace {

// splitmix64 — public-domain PRNG mixer
inline uint64_t splitmix64(uint64_t x) {
  x += 0x9E3779B97F4A7C15ULL;
  uint64_t z = x;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}

// per-(row,field) random stream: deterministic, order-independent,
// trivially parallel — restated identically on the device.
inline uint64_t fieldRand(uint64_t seed, int64_t row, int field) {
  uint64_t h = splitmix64(seed ^ (0x9E3779B97F4A7C15ULL * (uint64_t)(row + 1)));
  return splitmix64(h ^ (0xBF58476D1CE4E5B9ULL * (uint64_t)(field + 1)));
}

// Howard Hinnant's civil-date algorithms (public domain)
inline int64_t daysFromCivil(int y, int m, int d) {
  y -= m <= 2;
  int64_t era = (y >= 0 ? y : y - 399) / 400;
  unsigned yoe = (unsigned)(y - era * 400);
  unsigned doy = (unsigned)((153 * (m + (m > 2 ? -3 : 9)) + 2) / 5 + d - 1);
  unsigned doe = yoe * 365 + yoe / 4 - yoe / 100 + doy;
  return era * 146097 + (int64_t)doe - 719468;
}
inline void civilFromDays(int64_t z, int* yy, int* mm, int* dd) {
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

const int64_t kEpoch19920101 = daysFromCivil(1992, 1, 1);
const int64_t kShipdateDays = daysFromCivil(1998, 12, 1) - kEpoch19920101 + 1;
const int64_t kOrderdateDays = daysFromCivil(1998, 8, 2) - kEpoch19920101 + 1;

// canonical MyDecimal for cents at scale 2, 0 <= cents < 1e11
inline MyDecimal decFromCents(int64_t cents) {
  MyDecimal d;
  int64_t ip = cents / 100;
  int32_t f = (int32_t)(cents % 100);
  int digits = 1;
  for (int64_t t = ip; t >= 10; t /= 10) digits++;
  d.digitsInt = (int8_t)digits;
  d.digitsFrac = 2;
  d.resultFrac = 2;
  d.negative = 0;
  d.wordBuf[0] = (int32_t)ip;
  d.wordBuf[1] = f * 10000000;
  return d;
}

inline uint64_t dateFromDayOffset(int64_t off) {
  int y, m, d;
  civilFromDays(kEpoch19920101 + off, &y, &m, &d);
  return TimeFromDate(y, m, d);
}

const char* kSegments[5] = {"AUTOMOBILE", "BUILDING", "FURNITURE", "MACHINERY",
                            "HOUSEHOLD"};

}  // namespace

void TpchSchema(int table, std::vector<int>* types, std::vector<int>* fracs) {
  switch (table) {
    case GX_TPCH_LINEITEM:
      *types = {GX_TYPE_I64, GX_TYPE_DECIMAL, GX_TYPE_DECIMAL, GX_TYPE_DECIMAL,
                GX_TYPE_DECIMAL, GX_TYPE_STRING, GX_TYPE_STRING, GX_TYPE_TIME};
      *fracs = {0, 2, 2, 2, 2, 0, 0, 0};
      break;
    case GX_TPCH_ORDERS:
      *types = {GX_TYPE_I64, GX_TYPE_I64, GX_TYPE_TIME, GX_TYPE_I64};
      *fracs = {0, 0, 0, 0};
      break;
    case GX_TPCH_CUSTOMER:
      *types = {GX_TYPE_I64, GX_TYPE_STRING};
      *fracs = {0, 0};
      break;
  }
}

void TpchGenChunk(int table, int64_t rowBegin, int n, uint64_t seed,
                  int64_t totalRows, Chunk& out) {
  std::vector<int> types, fracs;
  TpchSchema(table, &types, &fracs);
  out.cols.resize(types.size());
  for (size_t c = 0; c < types.size(); c++) {
    out.cols[c].type = types[c];
    out.cols[c].frac = fracs[c];
  }
  out.reset();
  switch (table) {
    case GX_TPCH_LINEITEM: {
      int64_t nOrders = totalRows / 4;
      if (nOrders < 1) nOrders = 1;
      for (int i = 0; i < n; i++) {
        int64_t row = rowBegin + i;
        out.cols[0].appendI64(1 + (int64_t)(fieldRand(seed, row, 0) % (uint64_t)nOrders));
        out.cols[1].appendDecimal(decFromCents((1 + (int64_t)(fieldRand(seed, row, 1) % 50)) * 100));
        out.cols[2].appendDecimal(decFromCents(90100 + (int64_t)(fieldRand(seed, row, 2) % (10495000 - 90100 + 1))));
        out.cols[3].appendDecimal(decFromCents((int64_t)(fieldRand(seed, row, 3) % 11)));
        out.cols[4].appendDecimal(decFromCents((int64_t)(fieldRand(seed, row, 4) % 9)));
        const char rf[3] = {'A', 'N', 'R'};
        char c5 = rf[fieldRand(seed, row, 5) % 3];
        out.cols[5].appendBytes(&c5, 1);
        const char ls[2] = {'O', 'F'};
        char c6 = ls[fieldRand(seed, row, 6) % 2];
        out.cols[6].appendBytes(&c6, 1);
        out.cols[7].appendU64(dateFromDayOffset((int64_t)(fieldRand(seed, row, 7) % (uint64_t)kShipdateDays)));
      }
      break;
    }
    case GX_TPCH_ORDERS: {
      int64_t nCust = totalRows / 10;
      if (nCust < 1) nCust = 1;
      for (int i = 0; i < n; i++) {
        int64_t row = rowBegin + i;
        out.cols[0].appendI64(row + 1);
        out.cols[1].appendI64(1 + (int64_t)(fieldRand(seed, row, 1) % (uint64_t)nCust));
        out.cols[2].appendU64(dateFromDayOffset((int64_t)(fieldRand(seed, row, 2) % (uint64_t)kOrderdateDays)));
        out.cols[3].appendI64(0);
      }
      break;
    }
    case GX_TPCH_CUSTOMER: {
      for (int i = 0; i < n; i++) {
        int64_t row = rowBegin + i;
        out.cols[0].appendI64(row + 1);
        const char* seg = kSegments[fieldRand(seed, row, 1) % 5];
        out.cols[1].appendBytes(seg, strlen(seg));
      }
      break;
    }
  }
}

}  // namespace oracle
