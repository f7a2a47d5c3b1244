// tidb_amd/csrc/gx_decimal.cpp — PRODUCT host-side MyDecimal.
//
// The product engine's own restatement of MySQL/TiDB decimal semantics
// (reference pkg/types/mydecimal.go; same citations as the oracle's copy).
// It deliberately shares NO build artifacts with oracle/ — the oracle is test
// infrastructure and the product must not link it. Host-side use only:
// plan constants, result finalization (avg div/round), chunk emission; the
// GPU kernels compute in fixed-point int64/int128 (see gx_kernels.hip) and
// their equivalence to this arithmetic is covered by the golden-vector and
// parity suites in tests/.
#include "gx_decimal.h"

#include <algorithm>
#include <cstring>

namespace gxp {

namespace {

const int32_t powers10[10] = {1, 10, 100, 1000, 10000, 100000,
                              1000000, 10000000, 100000000, 1000000000};
const int dig2bytes[10] = {0, 1, 1, 2, 2, 3, 3, 4, 4, 4};
const int32_t fracMax[8] = {900000000, 990000000, 999000000, 999900000,
                            999990000, 999999000, 999999900, 999999990};
constexpr int32_t kDigMask = 100000000;

// mydecimal.go:124 add
inline int32_t addw(int32_t a, int32_t b, int32_t carry, int32_t* newCarry) {
  int32_t sum = a + b + carry;
  if (sum >= kWordBase) { *newCarry = 1; sum -= kWordBase; } else { *newCarry = 0; }
  return sum;
}
// mydecimal.go:137 add2 (carry may become 2)
inline int32_t add2w(int32_t a, int32_t b, int32_t carry, int32_t* newCarry) {
  int64_t sum = (int64_t)a + b + carry;
  int32_t c = 0;
  if (sum >= kWordBase) { c = 1; sum -= kWordBase; }
  if (sum >= kWordBase) { sum -= kWordBase; c++; }
  *newCarry = c;
  return (int32_t)sum;
}
// mydecimal.go:155 sub
inline int32_t subw(int32_t a, int32_t b, int32_t carry, int32_t* newCarry) {
  int32_t diff = a - b - carry;
  if (diff < 0) { *newCarry = 1; diff += kWordBase; } else { *newCarry = 0; }
  return diff;
}
// mydecimal.go:167 sub2 (carry may become 2)
inline int32_t sub2w(int32_t a, int32_t b, int32_t carry, int32_t* newCarry) {
  int32_t diff = a - b - carry;
  int32_t c = 0;
  if (diff < 0) { c = 1; diff += kWordBase; }
  if (diff < 0) { diff += kWordBase; c++; }
  *newCarry = c;
  return diff;
}

// mydecimal.go:184 fixWordCntError
inline int32_t fixWordCntError(int* wordsInt, int* wordsFrac) {
  if (*wordsInt + *wordsFrac > kMaxWordBufLen) {
    if (*wordsInt > kMaxWordBufLen) { *wordsInt = kMaxWordBufLen; *wordsFrac = 0; return E_OVERFLOW; }
    *wordsFrac = kMaxWordBufLen - *wordsInt;
    return E_TRUNCATED;
  }
  return E_OK;
}

// mydecimal.go:196 countLeadingZeroes
inline int countLeadingZeroes(int i, int32_t word) {
  int leading = 0;
  while (word < powers10[i]) { i--; leading++; }
  return leading;
}
// mydecimal.go:208 countTrailingZeroes
inline int countTrailingZeroes(int i, int32_t word) {
  int trailing = 0;
  while (word % powers10[i] == 0) { i++; trailing++; }
  return trailing;
}

// mydecimal.go:2004 maxDecimal
void maxDecimal(int precision, int frac, MyDecimal* to) {
  int digitsInt = precision - frac;
  to->negative = 0;
  to->digitsInt = (int8_t)digitsInt;
  int idx = 0;
  if (digitsInt > 0) {
    int firstWordDigits = digitsInt % kDigitsPerWord;
    if (firstWordDigits > 0) to->wordBuf[idx++] = powers10[firstWordDigits] - 1;
    for (digitsInt /= kDigitsPerWord; digitsInt > 0; digitsInt--) to->wordBuf[idx++] = kWordMax;
  }
  to->digitsFrac = (int8_t)frac;
  if (frac > 0) {
    int lastDigits = frac % kDigitsPerWord;
    for (frac /= kDigitsPerWord; frac > 0; frac--) to->wordBuf[idx++] = kWordMax;
    if (lastDigits > 0) to->wordBuf[idx] = fracMax[lastDigits - 1];
  }
}

MyDecimal zeroWithFrac(int8_t frac) {
  MyDecimal z;
  z.digitsFrac = frac;
  z.resultFrac = frac;
  return z;
}

// mydecimal.go:1715 validateArgs — clear `to` (callers here never alias).
void clearTo(MyDecimal* to) {
  to->digitsFrac = 0;
  to->digitsInt = 0;
  to->resultFrac = 0;
  to->negative = 0;
  std::memset(to->wordBuf, 0, sizeof(to->wordBuf));
}

int32_t readWord(const uint8_t* b, int size) {  // mydecimal.go:1594
  int32_t x = 0;
  switch (size) {
    case 1: x = (int32_t)(int8_t)b[0]; break;
    case 2: x = ((int32_t)(int8_t)b[0] << 8) + (int32_t)b[1]; break;
    case 3:
      if (b[0] & 128)
        x = (int32_t)(((uint32_t)255 << 24) | ((uint32_t)b[0] << 16) | ((uint32_t)b[1] << 8) | (uint32_t)b[2]);
      else
        x = (int32_t)(((uint32_t)b[0] << 16) | ((uint32_t)b[1] << 8) | (uint32_t)b[2]);
      break;
    case 4:
      x = (int32_t)b[3] + ((int32_t)b[2] << 8) + ((int32_t)b[1] << 16) + ((int32_t)(int8_t)b[0] << 24);
      break;
  }
  return x;
}

void writeWord(uint8_t* b, int32_t word, int size) {  // mydecimal.go:1613
  uint32_t v = (uint32_t)word;
  switch (size) {
    case 1: b[0] = (uint8_t)word; break;
    case 2: b[0] = (uint8_t)(v >> 8); b[1] = (uint8_t)v; break;
    case 3: b[0] = (uint8_t)(v >> 16); b[1] = (uint8_t)(v >> 8); b[2] = (uint8_t)v; break;
    case 4: b[0] = (uint8_t)(v >> 24); b[1] = (uint8_t)(v >> 16); b[2] = (uint8_t)(v >> 8); b[3] = (uint8_t)v; break;
  }
}

int32_t doSub(const MyDecimal* from1, const MyDecimal* from2, MyDecimal* to, int* cmpOut);
int32_t doAdd(const MyDecimal* from1, const MyDecimal* from2, MyDecimal* to);

}  // namespace

int DigitsToWords(int digits) {  // mydecimal.go:225
  return (digits + kDigitsPerWord - 1) / kDigitsPerWord;
}

bool MyDecimal::IsZero() const {  // mydecimal.go:1464
  for (int i = 0; i < kMaxWordBufLen; i++)
    if (wordBuf[i] != 0) return false;
  return true;
}

// mydecimal.go:289
void MyDecimal::RemoveLeadingZeros(int* wordIdxOut, int* digitsIntOut) const {
  int digits = (int)digitsInt;
  int i = ((digits - 1) % kDigitsPerWord) + 1;
  int wordIdx = 0;
  while (digits > 0 && wordBuf[wordIdx] == 0) {
    digits -= i;
    i = kDigitsPerWord;
    wordIdx++;
  }
  if (digits > 0)
    digits -= countLeadingZeroes((digits - 1) % kDigitsPerWord, wordBuf[wordIdx]);
  else
    digits = 0;
  *wordIdxOut = wordIdx;
  *digitsIntOut = digits;
}

// mydecimal.go:305
void MyDecimal::RemoveTrailingZeros(int* lastWordIdxOut, int* digitsFracOut) const {
  int digits = (int)digitsFrac;
  int i = ((digits - 1) % kDigitsPerWord) + 1;
  int lastWordIdx = DigitsToWords((int)digitsInt) + DigitsToWords((int)digitsFrac);
  while (digits > 0 && wordBuf[lastWordIdx - 1] == 0) {
    digits -= i;
    i = kDigitsPerWord;
    lastWordIdx--;
  }
  if (digits > 0)
    digits -= countTrailingZeroes(9 - ((digits - 1) % kDigitsPerWord), wordBuf[lastWordIdx - 1]);
  else
    digits = 0;
  *lastWordIdxOut = lastWordIdx;
  *digitsFracOut = digits;
}

// mydecimal.go:328 ToString
std::string MyDecimal::ToString() const {
  int digitsFracLocal = (int)digitsFrac;
  int wordStartIdx, digitsIntLocal;
  RemoveLeadingZeros(&wordStartIdx, &digitsIntLocal);
  if (digitsIntLocal + digitsFracLocal == 0) {
    digitsIntLocal = 1;
    wordStartIdx = 0;
  }
  int digitsIntLen = digitsIntLocal;
  if (digitsIntLen == 0) digitsIntLen = 1;
  int digitsFracLen = digitsFracLocal;
  int length = digitsIntLen + digitsFracLen;
  if (negative) length++;
  if (digitsFracLocal > 0) length++;
  std::string str(length, '0');
  int strIdx = 0;
  if (negative) str[strIdx++] = '-';
  int fill;
  int digitsFracIter = digitsFracLocal;
  if (digitsFracIter > 0) {
    int fracIdx = strIdx + digitsIntLen;
    fill = digitsFracLen - digitsFracIter;
    int wordIdx = wordStartIdx + DigitsToWords(digitsIntLocal);
    str[fracIdx++] = '.';
    for (; digitsFracIter > 0; digitsFracIter -= kDigitsPerWord) {
      int32_t x = wordBuf[wordIdx++];
      for (int i = std::min(digitsFracIter, kDigitsPerWord); i > 0; i--) {
        int32_t y = x / kDigMask;
        str[fracIdx++] = (char)((uint8_t)y + '0');
        x -= y * kDigMask;
        x *= 10;
      }
    }
    for (; fill > 0; fill--) str[fracIdx++] = '0';
  }
  fill = digitsIntLen - digitsIntLocal;
  if (digitsIntLocal == 0) fill--;  // symbol 0 before decimal point
  for (; fill > 0; fill--) str[strIdx++] = '0';
  if (digitsIntLocal > 0) {
    strIdx += digitsIntLocal;
    int wordIdx = wordStartIdx + DigitsToWords(digitsIntLocal);
    for (int di = digitsIntLocal; di > 0; di -= kDigitsPerWord) {
      wordIdx--;
      int32_t x = wordBuf[wordIdx];
      for (int i = std::min(di, kDigitsPerWord); i > 0; i--) {
        int32_t y = x / 10;
        str[--strIdx] = (char)('0' + (uint8_t)(x - y * 10));
        x = y;
      }
    }
  } else {
    str[strIdx] = '0';
  }
  return str;
}

// mydecimal.go:277 String()
std::string MyDecimal::DisplayString() const {
  MyDecimal tmp = *this;
  tmp.Round(&tmp, (int)tmp.resultFrac, ModeHalfUp);
  return tmp.ToString();
}

namespace {
inline bool isSpaceB(uint8_t c) { return c == ' ' || c == '\t'; }
inline bool isDigitB(uint8_t c) { return c >= '0' && c <= '9'; }
}

// mydecimal.go:406 FromString.
// Note: the scientific-notation tail ('e'/'E' + Shift) is restated too because
// golden vectors exercise it.
int32_t MyDecimal::FromString(const char* sIn, int lenIn) {
  const uint8_t* str = (const uint8_t*)sIn;
  int len = lenIn;
  while (len > 0 && isSpaceB(str[0])) { str++; len--; }
  *this = MyDecimal();
  if (len == 0) return E_BAD_NUMBER;
  if (str[0] == '-') { negative = 1; str++; len--; }
  else if (str[0] == '+') { str++; len--; }
  int strIdx = 0;
  while (strIdx < len && isDigitB(str[strIdx])) strIdx++;
  int digitsIntL = strIdx;
  int digitsFracL, endIdx;
  if (strIdx < len && str[strIdx] == '.') {
    endIdx = strIdx + 1;
    while (endIdx < len && isDigitB(str[endIdx])) endIdx++;
    digitsFracL = endIdx - strIdx - 1;
  } else {
    digitsFracL = 0;
    endIdx = strIdx;
  }
  if (digitsIntL + digitsFracL == 0) {
    bool wasNeg = negative;
    *this = MyDecimal();
    (void)wasNeg;
    return E_BAD_NUMBER;
  }
  int wordsInt = DigitsToWords(digitsIntL);
  int wordsFrac = DigitsToWords(digitsFracL);
  int32_t err = fixWordCntError(&wordsInt, &wordsFrac);
  if (err != E_OK) {
    digitsFracL = wordsFrac * kDigitsPerWord;
    if (err == E_OVERFLOW) digitsIntL = wordsInt * kDigitsPerWord;
  }
  digitsInt = (int8_t)digitsIntL;
  digitsFrac = (int8_t)digitsFracL;
  int wordIdx = wordsInt;
  int strIdxTmp = strIdx;
  int32_t word = 0;
  int innerIdx = 0;
  int di = digitsIntL;
  int si = strIdx;
  while (di > 0) {
    di--;
    si--;
    word += (int32_t)(str[si] - '0') * powers10[innerIdx];
    innerIdx++;
    if (innerIdx == kDigitsPerWord) {
      wordIdx--;
      wordBuf[wordIdx] = word;
      word = 0;
      innerIdx = 0;
    }
  }
  if (innerIdx != 0) {
    wordIdx--;
    wordBuf[wordIdx] = word;
  }
  wordIdx = wordsInt;
  si = strIdxTmp;
  word = 0;
  innerIdx = 0;
  int df = digitsFracL;
  while (df > 0) {
    df--;
    si++;
    word = (int32_t)(str[si] - '0') + word * 10;
    innerIdx++;
    if (innerIdx == kDigitsPerWord) {
      wordBuf[wordIdx] = word;
      wordIdx++;
      word = 0;
      innerIdx = 0;
    }
  }
  if (innerIdx != 0) wordBuf[wordIdx] = word * powers10[kDigitsPerWord - innerIdx];
  if (endIdx + 1 <= len) {
    if (str[endIdx] == 'e' || str[endIdx] == 'E') {
      // strToInt restatement (helper.go:133): digits after optional sign;
      // uint64 overflow => value 0 + ErrBadNumber; empty/garbage => ErrTruncated.
      constexpr uint64_t kUintCutOff = UINT64_MAX / 10 + 1;
      constexpr uint64_t kIntCutOff = (uint64_t)INT64_MAX + 1;
      int64_t expv = 0;
      int p = endIdx + 1;
      bool eneg = false;
      bool any = false;
      int32_t expErr = E_OK;
      // TrimSpace
      while (p < len && isSpaceB(str[p])) p++;
      if (p >= len) {
        expErr = E_TRUNCATED;
      } else {
        if (str[p] == '-') { eneg = true; p++; }
        else if (str[p] == '+') { p++; }
        uint64_t r = 0;
        for (; p < len; p++) {
          if (!isDigitB(str[p])) { expErr = E_TRUNCATED; break; }
          any = true;
          if (r >= kUintCutOff) { r = 0; expErr = E_BAD_NUMBER; break; }
          r = r * 10;
          uint64_t r1 = r + (uint64_t)(str[p] - '0');
          if (r1 < r) { r = 0; expErr = E_BAD_NUMBER; break; }
          r = r1;
        }
        if (!any) expErr = E_TRUNCATED;
        if (expErr != E_BAD_NUMBER || r != 0) {
          if (!eneg && r >= kIntCutOff) { expv = INT64_MAX; expErr = E_BAD_NUMBER; }
          else if (eneg && r > kIntCutOff) { expv = INT64_MIN; expErr = E_BAD_NUMBER; }
          else expv = eneg ? -(int64_t)r : (int64_t)r;
        }
      }
      if (expErr != E_OK) {
        err = expErr;
        if (err != E_TRUNCATED) *this = MyDecimal();
      }
      if (expv > 0x7fffffffLL / 2) {
        bool neg = negative;
        maxDecimal(kMaxWordBufLen * kDigitsPerWord, 0, this);
        negative = neg;
        err = E_OVERFLOW;
      }
      if (expv < -(0x80000000LL / 2) && err != E_OVERFLOW) {
        *this = MyDecimal();
        err = E_TRUNCATED;
      }
      if (err != E_OVERFLOW) {
        int32_t shiftErr = Shift((int)expv);
        if (shiftErr != E_OK) {
          if (shiftErr == E_OVERFLOW) {
            bool neg = negative;
            maxDecimal(kMaxWordBufLen * kDigitsPerWord, 0, this);
            negative = neg;
          }
          err = shiftErr;
        }
      }
    } else {
      // trailing garbage
      int p = endIdx;
      while (p < len && isSpaceB(str[p])) p++;
      if (p < len) err = E_TRUNCATED;
    }
  }
  bool allZero = true;
  for (int i = 0; i < kMaxWordBufLen; i++)
    if (wordBuf[i] != 0) { allZero = false; break; }
  if (allZero) negative = 0;
  resultFrac = digitsFrac;
  return err;
}

// mydecimal.go:713 digitBounds
static void digitBounds(const MyDecimal* d, int* startOut, int* endOut) {
  int bufBeg = 0;
  int bufLen = DigitsToWords((int)d->digitsInt) + DigitsToWords((int)d->digitsFrac);
  int bufEnd = bufLen - 1;
  while (bufBeg < bufLen && d->wordBuf[bufBeg] == 0) bufBeg++;
  if (bufBeg >= bufLen) { *startOut = 0; *endOut = 0; return; }
  int i, start;
  if (bufBeg == 0 && d->digitsInt > 0) {
    i = ((int)d->digitsInt - 1) % kDigitsPerWord;
    start = kDigitsPerWord - i - 1;
  } else {
    i = kDigitsPerWord - 1;
    start = bufBeg * kDigitsPerWord;
  }
  if (bufBeg < bufLen) start += countLeadingZeroes(i, d->wordBuf[bufBeg]);
  *startOut = start;
  while (bufEnd > bufBeg && d->wordBuf[bufEnd] == 0) bufEnd--;
  int end;
  if (bufEnd == bufLen - 1 && d->digitsFrac > 0) {
    i = (((int)d->digitsFrac - 1) % kDigitsPerWord) + 1;
    end = bufEnd * kDigitsPerWord + i;
    i = kDigitsPerWord - i + 1;
  } else {
    end = (bufEnd + 1) * kDigitsPerWord;
    i = 1;
  }
  end -= countTrailingZeroes(i, d->wordBuf[bufEnd]);
  *endOut = end;
}

// mydecimal.go:767 doMiniLeftShift
static void doMiniLeftShift(MyDecimal* d, int shift, int beg, int end) {
  int bufFrom = beg / kDigitsPerWord;
  int bufEnd = (end - 1) / kDigitsPerWord;
  int cShift = kDigitsPerWord - shift;
  if (beg % kDigitsPerWord < shift)
    d->wordBuf[bufFrom - 1] = d->wordBuf[bufFrom] / powers10[cShift];
  for (; bufFrom < bufEnd; bufFrom++) {
    d->wordBuf[bufFrom] = (d->wordBuf[bufFrom] % powers10[cShift]) * powers10[shift] +
                          d->wordBuf[bufFrom + 1] / powers10[cShift];
  }
  d->wordBuf[bufFrom] = (d->wordBuf[bufFrom] % powers10[cShift]) * powers10[shift];
}

// mydecimal.go:792 doMiniRightShift
static void doMiniRightShift(MyDecimal* d, int shift, int beg, int end) {
  int bufFrom = (end - 1) / kDigitsPerWord;
  int bufEnd = beg / kDigitsPerWord;
  int cShift = kDigitsPerWord - shift;
  if (kDigitsPerWord - ((end - 1) % kDigitsPerWord + 1) < shift)
    d->wordBuf[bufFrom + 1] = (d->wordBuf[bufFrom] % powers10[shift]) * powers10[cShift];
  for (; bufFrom > bufEnd; bufFrom--) {
    d->wordBuf[bufFrom] = d->wordBuf[bufFrom] / powers10[shift] +
                          (d->wordBuf[bufFrom - 1] % powers10[shift]) * powers10[cShift];
  }
  d->wordBuf[bufFrom] = d->wordBuf[bufFrom] / powers10[shift];
}

// mydecimal.go:555 Shift
int32_t MyDecimal::Shift(int shift) {
  int32_t err = E_OK;
  if (shift == 0) return E_OK;
  int digitBegin, digitEnd;
  int point = DigitsToWords((int)digitsInt) * kDigitsPerWord;
  int newPoint = point + shift;
  digitBounds(this, &digitBegin, &digitEnd);
  if (digitBegin == digitEnd) {
    *this = MyDecimal();
    return E_OK;
  }
  int digitsIntV = std::max(newPoint - digitBegin, 0);
  int digitsFracV = std::max(digitEnd - newPoint, 0);
  int wordsInt = DigitsToWords(digitsIntV);
  int wordsFrac = DigitsToWords(digitsFracV);
  int newLen = wordsInt + wordsFrac;
  if (newLen > kMaxWordBufLen) {
    int lack = newLen - kMaxWordBufLen;
    if (wordsFrac < lack) return E_OVERFLOW;
    /* cut off fraction part to allow new number to fit in our buffer */
    err = E_TRUNCATED;
    wordsFrac -= lack;
    int diff = digitsFracV - wordsFrac * kDigitsPerWord;
    int32_t err1 = Round(this, digitEnd - point - diff, ModeHalfUp);
    if (err1 != E_OK && err1 != E_TRUNCATED) return err1;
    digitEnd -= diff;
    digitsFracV = wordsFrac * kDigitsPerWord;
    if (digitEnd <= digitBegin) {
      *this = MyDecimal();
      return E_TRUNCATED;
    }
  }
  if (shift % kDigitsPerWord != 0) {
    int lMiniShift, rMiniShift, miniShift;
    bool doLeft;
    if (shift > 0) {
      lMiniShift = shift % kDigitsPerWord;
      rMiniShift = kDigitsPerWord - lMiniShift;
      doLeft = lMiniShift <= digitBegin;
    } else {
      rMiniShift = (-shift) % kDigitsPerWord;
      lMiniShift = kDigitsPerWord - rMiniShift;
      doLeft = (kDigitsPerWord * kMaxWordBufLen - digitEnd) < rMiniShift;
    }
    if (doLeft) {
      doMiniLeftShift(this, lMiniShift, digitBegin, digitEnd);
      miniShift = -lMiniShift;
    } else {
      doMiniRightShift(this, rMiniShift, digitBegin, digitEnd);
      miniShift = rMiniShift;
    }
    newPoint += miniShift;
    /* if number is shifted and correctly aligned in buffer we can finish */
    if (shift + miniShift == 0 && (newPoint - digitsIntV) < kDigitsPerWord) {
      digitsInt = (int8_t)digitsIntV;
      digitsFrac = (int8_t)digitsFracV;
      return err;
    }
    digitBegin += miniShift;
    digitEnd += miniShift;
  }
  /* if new 'decimal front' is in first digit, we do not need move digits */
  int newFront = newPoint - digitsIntV;
  if (newFront >= kDigitsPerWord || newFront < 0) {
    int wordShift;
    if (newFront > 0) {
      /* move left */
      wordShift = newFront / kDigitsPerWord;
      int to = digitBegin / kDigitsPerWord - wordShift;
      int barier = (digitEnd - 1) / kDigitsPerWord - wordShift;
      for (; to <= barier; to++) wordBuf[to] = wordBuf[to + wordShift];
      for (barier += wordShift; to <= barier; to++) wordBuf[to] = 0;
      wordShift = -wordShift;
    } else {
      /* move right */
      wordShift = (1 - newFront) / kDigitsPerWord;
      int to = (digitEnd - 1) / kDigitsPerWord + wordShift;
      int barier = digitBegin / kDigitsPerWord + wordShift;
      for (; to >= barier; to--) wordBuf[to] = wordBuf[to - wordShift];
      for (barier -= wordShift; to >= barier; to--) wordBuf[to] = 0;
    }
    int digitShift = wordShift * kDigitsPerWord;
    digitBegin += digitShift;
    digitEnd += digitShift;
    newPoint += digitShift;
  }
  /* fill gaps with 0 */
  int wordIdxBegin = digitBegin / kDigitsPerWord;
  int wordIdxEnd = (digitEnd - 1) / kDigitsPerWord;
  int wordIdxNewPoint = 0;
  if (newPoint != 0) wordIdxNewPoint = (newPoint - 1) / kDigitsPerWord;
  if (wordIdxNewPoint > wordIdxEnd) {
    while (wordIdxNewPoint > wordIdxEnd) {
      wordBuf[wordIdxNewPoint] = 0;
      wordIdxNewPoint--;
    }
  } else {
    for (; wordIdxNewPoint < wordIdxBegin; wordIdxNewPoint++) wordBuf[wordIdxNewPoint] = 0;
  }
  digitsInt = (int8_t)digitsIntV;
  digitsFrac = (int8_t)digitsFracV;
  return err;
}

// mydecimal.go:1068 / 1080
MyDecimal& MyDecimal::FromInt(int64_t val) {
  *this = MyDecimal();
  uint64_t uVal;
  if (val < 0) {
    negative = 1;
    uVal = (uint64_t)(-(val + 1)) + 1;  // safe for INT64_MIN
  } else {
    uVal = (uint64_t)val;
  }
  uint8_t neg = negative;
  FromUint(uVal);
  negative = neg;
  return *this;
}

MyDecimal& MyDecimal::FromUint(uint64_t val) {
  uint64_t x = val;
  int wordIdx = 1;
  while (x >= (uint64_t)kWordBase) {
    wordIdx++;
    x /= (uint64_t)kWordBase;
  }
  digitsFrac = 0;
  digitsInt = (int8_t)(wordIdx * kDigitsPerWord);
  x = val;
  while (wordIdx > 0) {
    wordIdx--;
    uint64_t y = x / (uint64_t)kWordBase;
    wordBuf[wordIdx] = (int32_t)(x - y * (uint64_t)kWordBase);
    x = y;
  }
  return *this;
}

// mydecimal.go:1100 ToInt
int32_t MyDecimal::ToInt(int64_t* out) const {
  int64_t x = 0;
  int wordIdx = 0;
  for (int i = digitsInt; i > 0; i -= kDigitsPerWord) {
    int64_t y = x;
    x = x * kWordBase - (int64_t)wordBuf[wordIdx];
    wordIdx++;
    if (y < INT64_MIN / kWordBase || x > y) {
      if (negative) { *out = INT64_MIN; return E_OVERFLOW; }
      *out = INT64_MAX;
      return E_OVERFLOW;
    }
  }
  if (!negative && x == INT64_MIN) { *out = INT64_MAX; return E_OVERFLOW; }
  if (!negative) x = -x;
  for (int i = digitsFrac; i > 0; i -= kDigitsPerWord) {
    if (wordBuf[wordIdx] != 0) { *out = x; return E_TRUNCATED; }
    wordIdx++;
  }
  *out = x;
  return E_OK;
}

// mydecimal.go:822 Round
int32_t MyDecimal::Round(MyDecimal* to, int frac, RoundMode roundMode) const {
  int32_t err = E_OK;
  int wordsFracTo = (frac + 1) / kDigitsPerWord;
  if (frac > 0) wordsFracTo = DigitsToWords(frac);
  int wordsFrac = DigitsToWords((int)digitsFrac);
  int wordsInt = DigitsToWords((int)digitsInt);
  int32_t roundDigit = (int32_t)roundMode;

  if (wordsInt + wordsFracTo > kMaxWordBufLen) {
    wordsFracTo = kMaxWordBufLen - wordsInt;
    frac = wordsFracTo * kDigitsPerWord;
    err = E_TRUNCATED;
  }
  if ((int)digitsInt + frac < 0) {
    *to = MyDecimal();
    return E_OK;
  }
  if (to != this) {
    std::memcpy(to->wordBuf, wordBuf, sizeof(wordBuf));
    to->negative = negative;
    to->digitsInt = (int8_t)(std::min(wordsInt, kMaxWordBufLen) * kDigitsPerWord);
  }
  if (wordsFracTo > wordsFrac) {
    int idx = wordsInt + wordsFrac;
    while (wordsFracTo > wordsFrac) {
      wordsFracTo--;
      to->wordBuf[idx] = 0;
      idx++;
    }
    to->digitsFrac = (int8_t)frac;
    to->resultFrac = to->digitsFrac;
    return err;
  }
  if (frac >= (int)digitsFrac) {
    to->digitsFrac = (int8_t)frac;
    to->resultFrac = to->digitsFrac;
    return err;
  }
  // Do increment.
  int toIdx = wordsInt + wordsFracTo - 1;
  if (frac == wordsFracTo * kDigitsPerWord) {
    bool doInc = false;
    switch (roundMode) {
      case ModeCeiling: {
        int idx = toIdx + (wordsFrac - wordsFracTo);
        while (idx > toIdx) {
          if (wordBuf[idx] != 0) { doInc = true; break; }
          idx--;
        }
        break;
      }
      case ModeHalfUp: {
        int32_t digAfterScale = wordBuf[toIdx + 1] / kDigMask;
        doInc = digAfterScale >= 5;
        break;
      }
      case ModeTruncate:
        doInc = false;
        break;
    }
    if (doInc) {
      if (toIdx >= 0) {
        to->wordBuf[toIdx]++;
      } else {
        toIdx++;
        to->wordBuf[toIdx] = kWordBase;
      }
    } else if (wordsInt + wordsFracTo == 0) {
      *to = MyDecimal();
      return E_OK;
    }
  } else {
    int pos = wordsFracTo * kDigitsPerWord - frac - 1;
    int32_t shiftedNumber = to->wordBuf[toIdx] / powers10[pos];
    int32_t digAfterScale = shiftedNumber % 10;
    if (digAfterScale > roundDigit || (roundDigit == 5 && digAfterScale == 5))
      shiftedNumber += 10;
    to->wordBuf[toIdx] = powers10[pos] * (shiftedNumber - digAfterScale);
  }
  if (wordsFracTo < wordsFrac) {
    int idx = wordsInt + wordsFracTo;
    if (frac == 0 && wordsInt == 0) idx = 1;
    while (idx < kMaxWordBufLen) {
      to->wordBuf[idx] = 0;
      idx++;
    }
  }
  // Handle carry.
  int32_t carry;
  if (to->wordBuf[toIdx] >= kWordBase) {
    carry = 1;
    to->wordBuf[toIdx] -= kWordBase;
    while (carry == 1 && toIdx > 0) {
      toIdx--;
      to->wordBuf[toIdx] = addw(to->wordBuf[toIdx], 0, carry, &carry);
    }
    if (carry > 0) {
      if (wordsInt + wordsFracTo >= kMaxWordBufLen) {
        wordsFracTo--;
        frac = wordsFracTo * kDigitsPerWord;
        err = E_TRUNCATED;
      }
      for (toIdx = wordsInt + std::max(wordsFracTo, 0); toIdx > 0; toIdx--) {
        if (toIdx < kMaxWordBufLen)
          to->wordBuf[toIdx] = to->wordBuf[toIdx - 1];
        else
          err = E_OVERFLOW;
      }
      to->wordBuf[toIdx] = 1;
      if ((int)to->digitsInt < kDigitsPerWord * kMaxWordBufLen)
        to->digitsInt++;
      else
        err = E_OVERFLOW;
    }
  } else {
    for (;;) {
      if (to->wordBuf[toIdx] != 0) break;
      if (toIdx == 0) {
        // making 'zero' with the proper scale
        int idx = wordsFracTo + 1;
        to->digitsInt = 1;
        to->digitsFrac = (int8_t)std::max(frac, 0);
        to->negative = 0;
        while (toIdx < idx) {
          to->wordBuf[toIdx] = 0;
          toIdx++;
        }
        to->resultFrac = to->digitsFrac;
        return E_OK;
      }
      toIdx--;
    }
  }
  // 999.9 -> 1000 case
  int firstDig = (int)to->digitsInt % kDigitsPerWord;
  if (firstDig > 0 && to->wordBuf[toIdx] >= powers10[firstDig]) to->digitsInt++;
  if (frac < 0) frac = 0;
  to->digitsFrac = (int8_t)frac;
  to->resultFrac = to->digitsFrac;
  return err;
}

// ---- add/sub/mul/div ----

namespace {

// mydecimal.go:1898 doAdd
int32_t doAdd(const MyDecimal* from1, const MyDecimal* from2, MyDecimal* to) {
  int32_t err = E_OK;
  int wordsInt1 = DigitsToWords((int)from1->digitsInt);
  int wordsFrac1 = DigitsToWords((int)from1->digitsFrac);
  int wordsInt2 = DigitsToWords((int)from2->digitsInt);
  int wordsFrac2 = DigitsToWords((int)from2->digitsFrac);
  int wordsIntTo = std::max(wordsInt1, wordsInt2);
  int wordsFracTo = std::max(wordsFrac1, wordsFrac2);

  int32_t x;
  if (wordsInt1 > wordsInt2)
    x = from1->wordBuf[0];
  else if (wordsInt2 > wordsInt1)
    x = from2->wordBuf[0];
  else
    x = from1->wordBuf[0] + from2->wordBuf[0];
  if (x > kWordMax - 1) {
    wordsIntTo++;
    to->wordBuf[0] = 0;
  }
  err = fixWordCntError(&wordsIntTo, &wordsFracTo);
  if (err == E_OVERFLOW) {
    maxDecimal(kMaxWordBufLen * kDigitsPerWord, 0, to);
    return err;
  }
  int idxTo = wordsIntTo + wordsFracTo;
  to->negative = from1->negative;
  to->digitsInt = (int8_t)(wordsIntTo * kDigitsPerWord);
  to->digitsFrac = std::max(from1->digitsFrac, from2->digitsFrac);
  if (err != E_OK) {
    if (to->digitsFrac > (int8_t)(wordsFracTo * kDigitsPerWord))
      to->digitsFrac = (int8_t)(wordsFracTo * kDigitsPerWord);
    if (wordsFrac1 > wordsFracTo) wordsFrac1 = wordsFracTo;
    if (wordsFrac2 > wordsFracTo) wordsFrac2 = wordsFracTo;
    if (wordsInt1 > wordsIntTo) wordsInt1 = wordsIntTo;
    if (wordsInt2 > wordsIntTo) wordsInt2 = wordsIntTo;
  }
  const MyDecimal* dec1 = from1;
  const MyDecimal* dec2 = from2;
  int idx1, idx2, stop, stop2 = 0;
  if (wordsFrac1 > wordsFrac2) {
    idx1 = wordsInt1 + wordsFrac1;
    stop = wordsInt1 + wordsFrac2;
    idx2 = wordsInt2 + wordsFrac2;
    if (wordsInt1 > wordsInt2) stop2 = wordsInt1 - wordsInt2;
  } else {
    idx1 = wordsInt2 + wordsFrac2;
    stop = wordsInt2 + wordsFrac1;
    idx2 = wordsInt1 + wordsFrac1;
    if (wordsInt2 > wordsInt1) stop2 = wordsInt2 - wordsInt1;
    dec1 = from2;
    dec2 = from1;
  }
  while (idx1 > stop) {
    idxTo--;
    idx1--;
    to->wordBuf[idxTo] = dec1->wordBuf[idx1];
  }
  int32_t carry = 0;
  while (idx1 > stop2) {
    idx1--;
    idx2--;
    idxTo--;
    to->wordBuf[idxTo] = addw(dec1->wordBuf[idx1], dec2->wordBuf[idx2], carry, &carry);
  }
  stop = 0;
  if (wordsInt1 > wordsInt2) {
    idx1 = wordsInt1 - wordsInt2;
    dec1 = from1;
  } else {
    idx1 = wordsInt2 - wordsInt1;
    dec1 = from2;
  }
  while (idx1 > stop) {
    idxTo--;
    idx1--;
    to->wordBuf[idxTo] = addw(dec1->wordBuf[idx1], 0, carry, &carry);
  }
  if (carry > 0) {
    idxTo--;
    to->wordBuf[idxTo] = 1;
  }
  return err;
}

// mydecimal.go:1737 doSub. cmpOut != nullptr <=> to == nullptr in Go (compare-only).
int32_t doSub(const MyDecimal* from1In, const MyDecimal* from2In, MyDecimal* to, int* cmpOut) {
  int32_t err = E_OK;
  const MyDecimal* from1 = from1In;
  const MyDecimal* from2 = from2In;
  int wordsInt1 = DigitsToWords((int)from1->digitsInt);
  int wordsFrac1 = DigitsToWords((int)from1->digitsFrac);
  int wordsInt2 = DigitsToWords((int)from2->digitsInt);
  int wordsFrac2 = DigitsToWords((int)from2->digitsFrac);
  int wordsFracTo = std::max(wordsFrac1, wordsFrac2);

  int start1 = 0;
  int stop1 = wordsInt1;
  int idx1 = 0;
  int start2 = 0;
  int stop2 = wordsInt2;
  int idx2 = 0;
  if (from1->wordBuf[idx1] == 0) {
    while (idx1 < stop1 && from1->wordBuf[idx1] == 0) idx1++;
    start1 = idx1;
    wordsInt1 = stop1 - idx1;
  }
  if (from2->wordBuf[idx2] == 0) {
    while (idx2 < stop2 && from2->wordBuf[idx2] == 0) idx2++;
    start2 = idx2;
    wordsInt2 = stop2 - idx2;
  }
  int32_t carry = 0;
  if (wordsInt2 > wordsInt1) {
    carry = 1;
  } else if (wordsInt2 == wordsInt1) {
    int end1 = stop1 + wordsFrac1 - 1;
    int end2 = stop2 + wordsFrac2 - 1;
    while (idx1 <= end1 && from1->wordBuf[end1] == 0) end1--;
    while (idx2 <= end2 && from2->wordBuf[end2] == 0) end2--;
    wordsFrac1 = end1 - stop1 + 1;
    wordsFrac2 = end2 - stop2 + 1;
    while (idx1 <= end1 && idx2 <= end2 && from1->wordBuf[idx1] == from2->wordBuf[idx2]) {
      idx1++;
      idx2++;
    }
    if (idx1 <= end1) {
      if (idx2 <= end2 && from2->wordBuf[idx2] > from1->wordBuf[idx1])
        carry = 1;
      else
        carry = 0;
    } else {
      if (idx2 > end2) {
        if (to == nullptr) {
          if (cmpOut) *cmpOut = 0;
          return E_OK;
        }
        *to = zeroWithFrac(to->resultFrac);
        return E_OK;
      }
      carry = 1;
    }
  }
  if (to == nullptr) {
    if ((carry > 0) == (from1->negative != 0)) {  // from2 is negative too
      if (cmpOut) *cmpOut = 1;
    } else {
      if (cmpOut) *cmpOut = -1;
    }
    return E_OK;
  }
  to->negative = from1->negative;
  if (carry > 0) {
    std::swap(from1, from2);
    std::swap(start1, start2);
    std::swap(wordsInt1, wordsInt2);
    std::swap(wordsFrac1, wordsFrac2);
    to->negative = to->negative ? 0 : 1;
  }
  err = fixWordCntError(&wordsInt1, &wordsFracTo);
  int idxTo = wordsInt1 + wordsFracTo;
  to->digitsFrac = std::max(from1->digitsFrac, from2->digitsFrac);
  to->digitsInt = (int8_t)(wordsInt1 * kDigitsPerWord);
  if (err != E_OK) {
    if (to->digitsFrac > (int8_t)(wordsFracTo * kDigitsPerWord))
      to->digitsFrac = (int8_t)(wordsFracTo * kDigitsPerWord);
    if (wordsFrac1 > wordsFracTo) wordsFrac1 = wordsFracTo;
    if (wordsFrac2 > wordsFracTo) wordsFrac2 = wordsFracTo;
    if (wordsInt2 > wordsInt1) wordsInt2 = wordsInt1;
  }
  carry = 0;
  // part 1 - max(frac) ... min(frac)
  if (wordsFrac1 > wordsFrac2) {
    idx1 = start1 + wordsInt1 + wordsFrac1;
    stop1 = start1 + wordsInt1 + wordsFrac2;
    idx2 = start2 + wordsInt2 + wordsFrac2;
    while (wordsFracTo > wordsFrac1) {
      wordsFracTo--;
      idxTo--;
      to->wordBuf[idxTo] = 0;
    }
    while (idx1 > stop1) {
      idxTo--;
      idx1--;
      to->wordBuf[idxTo] = from1->wordBuf[idx1];
    }
  } else {
    idx1 = start1 + wordsInt1 + wordsFrac1;
    idx2 = start2 + wordsInt2 + wordsFrac2;
    stop2 = start2 + wordsInt2 + wordsFrac1;
    while (wordsFracTo > wordsFrac2) {
      wordsFracTo--;
      idxTo--;
      to->wordBuf[idxTo] = 0;
    }
    while (idx2 > stop2) {
      idxTo--;
      idx2--;
      to->wordBuf[idxTo] = subw(0, from2->wordBuf[idx2], carry, &carry);
    }
  }
  // part 2 - min(frac) ... wordsInt2
  while (idx2 > start2) {
    idxTo--;
    idx1--;
    idx2--;
    to->wordBuf[idxTo] = subw(from1->wordBuf[idx1], from2->wordBuf[idx2], carry, &carry);
  }
  // part 3 - wordsInt2 ... wordsInt1
  while (carry > 0 && idx1 > start1) {
    idxTo--;
    idx1--;
    to->wordBuf[idxTo] = subw(from1->wordBuf[idx1], 0, carry, &carry);
  }
  while (idx1 > start1) {
    idxTo--;
    idx1--;
    to->wordBuf[idxTo] = from1->wordBuf[idx1];
  }
  while (idxTo > 0) {
    idxTo--;
    to->wordBuf[idxTo] = 0;
  }
  return err;
}

}  // namespace

// mydecimal.go:1694
int32_t DecimalAdd(const MyDecimal* from1, const MyDecimal* from2, MyDecimal* to) {
  MyDecimal a = *from1, b = *from2;  // allow aliasing with `to`
  clearTo(to);
  to->resultFrac = std::max(a.resultFrac, b.resultFrac);
  if (a.negative == b.negative) return doAdd(&a, &b, to);
  int cmp;
  return doSub(&a, &b, to, &cmp);
}

// mydecimal.go:1705
int32_t DecimalSub(const MyDecimal* from1, const MyDecimal* from2, MyDecimal* to) {
  MyDecimal a = *from1, b = *from2;
  clearTo(to);
  to->resultFrac = std::max(a.resultFrac, b.resultFrac);
  if (a.negative == b.negative) {
    int cmp;
    return doSub(&a, &b, to, &cmp);
  }
  return doAdd(&a, &b, to);
}

MyDecimal DecimalNeg(const MyDecimal& from) {  // mydecimal.go:1682
  MyDecimal to = from;
  if (from.IsZero()) return to;
  to.negative = to.negative ? 0 : 1;
  return to;
}

// mydecimal.go:1634
int MyDecimal::Compare(const MyDecimal& other) const {
  if (negative == other.negative) {
    int cmp = 0;
    doSub(this, &other, nullptr, &cmp);
    return cmp;
  }
  if (negative) return -1;
  return 1;
}

void MyDecimal::PrecisionAndFrac(int* precision, int* frac) const {
  *frac = (int)digitsFrac;
  int wordIdx, di;
  RemoveLeadingZeros(&wordIdx, &di);
  *precision = di + *frac;
  if (*precision == 0) *precision = 1;
}

// mydecimal.go:2052
int32_t DecimalMul(const MyDecimal* from1In, const MyDecimal* from2In, MyDecimal* to) {
  MyDecimal a = *from1In, b = *from2In;
  const MyDecimal* from1 = &a;
  const MyDecimal* from2 = &b;
  clearTo(to);
  int32_t err = E_OK;
  int wordsInt1 = DigitsToWords((int)from1->digitsInt);
  int wordsFrac1 = DigitsToWords((int)from1->digitsFrac);
  int wordsInt2 = DigitsToWords((int)from2->digitsInt);
  int wordsFrac2 = DigitsToWords((int)from2->digitsFrac);
  int wordsIntTo = DigitsToWords((int)from1->digitsInt + (int)from2->digitsInt);
  int wordsFracTo = wordsFrac1 + wordsFrac2;
  int idx1 = wordsInt1;
  int idx2 = wordsInt2;
  int idxTo = 0;
  int tmp1 = wordsIntTo;
  int tmp2 = wordsFracTo;
  to->resultFrac = (int8_t)std::min((int)from1->resultFrac + (int)from2->resultFrac, kMaxDecimalScale);
  err = fixWordCntError(&wordsIntTo, &wordsFracTo);
  to->negative = (from1->negative != from2->negative) ? 1 : 0;
  to->digitsFrac = (int8_t)std::min((int)from1->digitsFrac + (int)from2->digitsFrac, kNotFixedDec);
  to->digitsInt = (int8_t)(wordsIntTo * kDigitsPerWord);
  if (err == E_OVERFLOW) return err;
  if (err != E_OK) {
    if (to->digitsFrac > (int8_t)(wordsFracTo * kDigitsPerWord))
      to->digitsFrac = (int8_t)(wordsFracTo * kDigitsPerWord);
    if (to->digitsInt > (int8_t)(wordsIntTo * kDigitsPerWord))
      to->digitsInt = (int8_t)(wordsIntTo * kDigitsPerWord);
    if (tmp1 > wordsIntTo) {
      tmp1 -= wordsIntTo;
      tmp2 = tmp1 >> 1;
      wordsInt2 -= tmp1 - tmp2;
      wordsFrac1 = 0;
      wordsFrac2 = 0;
    } else {
      tmp2 -= wordsFracTo;
      tmp1 = tmp2 >> 1;
      if (wordsFrac1 <= wordsFrac2) {
        wordsFrac1 -= tmp1;
        wordsFrac2 -= tmp2 - tmp1;
      } else {
        wordsFrac2 -= tmp1;
        wordsFrac1 -= tmp2 - tmp1;
      }
    }
  }
  int startTo = wordsIntTo + wordsFracTo - 1;
  int start2 = idx2 + wordsFrac2 - 1;
  int stop1 = idx1 - wordsInt1;
  int stop2 = idx2 - wordsInt2;
  std::memset(to->wordBuf, 0, sizeof(to->wordBuf));

  for (idx1 += wordsFrac1 - 1; idx1 >= stop1; idx1--) {
    int32_t carry = 0;
    idxTo = startTo;
    idx2 = start2;
    while (idx2 >= stop2) {
      int64_t p = (int64_t)from1->wordBuf[idx1] * (int64_t)from2->wordBuf[idx2];
      int32_t hi = (int32_t)(p / kWordBase);
      int32_t lo = (int32_t)(p - (int64_t)hi * kWordBase);
      to->wordBuf[idxTo] = add2w(to->wordBuf[idxTo], lo, carry, &carry);
      carry += hi;
      idx2--;
      idxTo--;
    }
    if (carry > 0) {
      if (idxTo < 0) return E_OVERFLOW;
      to->wordBuf[idxTo] = add2w(to->wordBuf[idxTo], 0, carry, &carry);
    }
    for (idxTo--; carry > 0; idxTo--) {
      if (idxTo < 0) return E_OVERFLOW;
      to->wordBuf[idxTo] = addw(to->wordBuf[idxTo], 0, carry, &carry);
    }
    startTo--;
  }

  // -0.000 case
  if (to->negative) {
    int idx = 0;
    int end = wordsIntTo + wordsFracTo;
    for (;;) {
      if (to->wordBuf[idx] != 0) break;
      idx++;
      if (idx == end) {
        *to = zeroWithFrac(to->resultFrac);
        break;
      }
    }
  }

  idxTo = 0;
  int dToMove = wordsIntTo + DigitsToWords((int)to->digitsFrac);
  while (to->wordBuf[idxTo] == 0 && to->digitsInt > kDigitsPerWord) {
    idxTo++;
    to->digitsInt -= kDigitsPerWord;
    dToMove--;
  }
  if (idxTo > 0) {
    int curIdx = 0;
    while (dToMove > 0) {
      to->wordBuf[curIdx] = to->wordBuf[idxTo];
      curIdx++;
      idxTo++;
      dToMove--;
    }
  }
  return err;
}

// mydecimal.go:2214 doDivMod (division only; mod path omitted — DecimalMod is
// off the hot path and not exported).
static int32_t doDiv(const MyDecimal* from1, const MyDecimal* from2, MyDecimal* to, int fracIncr) {
  int32_t err = E_OK;
  int frac1 = DigitsToWords((int)from1->digitsFrac) * kDigitsPerWord;
  int prec1 = (int)from1->digitsInt + frac1;
  int frac2 = DigitsToWords((int)from2->digitsFrac) * kDigitsPerWord;
  int prec2 = (int)from2->digitsInt + frac2;

  // remove leading zeros of from2
  int i = ((prec2 - 1) % kDigitsPerWord) + 1;
  int idx2 = 0;
  while (prec2 > 0 && from2->wordBuf[idx2] == 0) {
    prec2 -= i;
    i = kDigitsPerWord;
    idx2++;
  }
  if (prec2 <= 0) return E_DIV_ZERO;
  prec2 -= countLeadingZeroes((prec2 - 1) % kDigitsPerWord, from2->wordBuf[idx2]);

  i = ((prec1 - 1) % kDigitsPerWord) + 1;
  int idx1 = 0;
  while (prec1 > 0 && from1->wordBuf[idx1] == 0) {
    prec1 -= i;
    i = kDigitsPerWord;
    idx1++;
  }
  if (prec1 <= 0) {
    *to = zeroWithFrac(to->resultFrac);
    return E_OK;
  }
  prec1 -= countLeadingZeroes((prec1 - 1) % kDigitsPerWord, from1->wordBuf[idx1]);

  fracIncr -= frac1 - (int)from1->digitsFrac + frac2 - (int)from2->digitsFrac;
  if (fracIncr < 0) fracIncr = 0;

  int digitsIntTo = (prec1 - frac1) - (prec2 - frac2);
  if (from1->wordBuf[idx1] >= from2->wordBuf[idx2]) digitsIntTo++;
  int wordsIntTo;
  if (digitsIntTo < 0) {
    digitsIntTo /= kDigitsPerWord;
    wordsIntTo = 0;
  } else {
    wordsIntTo = DigitsToWords(digitsIntTo);
  }
  int wordsFracTo = DigitsToWords(frac1 + frac2 + fracIncr);
  err = fixWordCntError(&wordsIntTo, &wordsFracTo);
  to->negative = (from1->negative != from2->negative) ? 1 : 0;
  to->digitsInt = (int8_t)(wordsIntTo * kDigitsPerWord);
  to->digitsFrac = (int8_t)(wordsFracTo * kDigitsPerWord);

  int idxTo = 0;
  int stopTo = wordsIntTo + wordsFracTo;
  while (digitsIntTo < 0 && idxTo < kMaxWordBufLen) {
    to->wordBuf[idxTo] = 0;
    idxTo++;
    digitsIntTo++;
  }
  i = DigitsToWords(prec1);
  int len1 = std::max(i + DigitsToWords(2 * frac2 + fracIncr + 1) + 1, 3);
  int32_t tmp1buf[64] = {0};
  int32_t* tmp1 = tmp1buf;
  for (int k = 0; k < i; k++) tmp1[k] = from1->wordBuf[idx1 + k];

  int start1 = 0;
  int stop1;
  int start2 = idx2;
  int stop2 = idx2 + DigitsToWords(prec2) - 1;
  (void)len1;
  (void)stop1;

  // remove end zeroes of divisor
  while (from2->wordBuf[stop2] == 0 && stop2 >= start2) stop2--;
  int len2 = stop2 - start2;
  stop2++;

  int64_t normFactor = (int64_t)kWordBase / (int64_t)(from2->wordBuf[start2] + 1);
  int32_t norm2 = (int32_t)(normFactor * (int64_t)from2->wordBuf[start2]);
  if (len2 > 0) norm2 += (int32_t)(normFactor * (int64_t)from2->wordBuf[start2 + 1] / kWordBase);
  int32_t dcarry = 0;
  if (tmp1[start1] < from2->wordBuf[start2]) {
    dcarry = tmp1[start1];
    start1++;
  }
  int64_t guess;
  for (; idxTo < stopTo; idxTo++) {
    if (dcarry == 0 && tmp1[start1] < from2->wordBuf[start2]) {
      guess = 0;
    } else {
      int64_t x = (int64_t)tmp1[start1] + (int64_t)dcarry * kWordBase;
      int64_t y = (int64_t)tmp1[start1 + 1];
      guess = (normFactor * x + normFactor * y / kWordBase) / (int64_t)norm2;
      if (guess >= kWordBase) guess = kWordBase - 1;
      if (len2 > 0) {
        if ((int64_t)from2->wordBuf[start2 + 1] * guess >
            (x - guess * (int64_t)from2->wordBuf[start2]) * kWordBase + y)
          guess--;
        if ((int64_t)from2->wordBuf[start2 + 1] * guess >
            (x - guess * (int64_t)from2->wordBuf[start2]) * kWordBase + y)
          guess--;
      }
      // D4: multiply and subtract
      int idx2i = stop2;
      int idx1i = start1 + len2;
      int32_t carry = 0;
      for (carry = 0; idx2i > start2; idx1i--) {
        idx2i--;
        int64_t xm = guess * (int64_t)from2->wordBuf[idx2i];
        int32_t hi = (int32_t)(xm / kWordBase);
        int32_t lo = (int32_t)(xm - (int64_t)hi * kWordBase);
        tmp1[idx1i] = sub2w(tmp1[idx1i], lo, carry, &carry);
        carry += hi;
      }
      if (dcarry < carry)
        carry = 1;
      else
        carry = 0;
      // D5/D6
      if (carry > 0) {
        guess--;
        idx2i = stop2;
        idx1i = start1 + len2;
        for (carry = 0; idx2i > start2; idx1i--) {
          idx2i--;
          tmp1[idx1i] = addw(tmp1[idx1i], from2->wordBuf[idx2i], carry, &carry);
        }
      }
    }
    to->wordBuf[idxTo] = (int32_t)guess;
    dcarry = tmp1[start1];
    start1++;
  }

  int wordIdx, di;
  to->RemoveLeadingZeros(&wordIdx, &di);
  to->digitsInt = (int8_t)di;
  if (wordIdx != 0) {
    for (int k = 0; k + wordIdx < kMaxWordBufLen; k++) to->wordBuf[k] = to->wordBuf[k + wordIdx];
    for (int k = kMaxWordBufLen - wordIdx; k < kMaxWordBufLen; k++) to->wordBuf[k] = 0;
  }
  if (to->IsZero()) to->negative = 0;
  return err;
}

// mydecimal.go:2178
int32_t DecimalDiv(const MyDecimal* from1In, const MyDecimal* from2In, MyDecimal* to, int fracIncr) {
  MyDecimal a = *from1In, b = *from2In;
  clearTo(to);
  to->resultFrac = (int8_t)std::min((int)a.resultFrac + fracIncr, kMaxDecimalScale);
  return doDiv(&a, &b, to, fracIncr);
}

// mydecimal.go:1582
int32_t DecimalBinSize(int precision, int frac, int* size) {
  int digitsIntL = precision - frac;
  int wordsInt = digitsIntL / kDigitsPerWord;
  int wordsFrac = frac / kDigitsPerWord;
  int xInt = digitsIntL - wordsInt * kDigitsPerWord;
  int xFrac = frac - wordsFrac * kDigitsPerWord;
  if (xInt < 0 || xInt >= 10 || xFrac < 0 || xFrac >= 10) return E_BAD_NUMBER;
  *size = wordsInt * kWordSize + dig2bytes[xInt] + wordsFrac * kWordSize + dig2bytes[xFrac];
  return E_OK;
}

// mydecimal.go:1295 WriteBin
int32_t MyDecimal::WriteBin(int precision, int frac, uint8_t* bin, int* written) const {
  if (precision > kDigitsPerWord * kMaxWordBufLen || precision < 0 ||
      frac > kMaxDecimalScale || frac < 0)
    return E_BAD_NUMBER;
  int32_t err = E_OK;
  int32_t mask = negative ? -1 : 0;
  int digitsIntL = precision - frac;
  int wordsInt = digitsIntL / kDigitsPerWord;
  int leadingDigits = digitsIntL - wordsInt * kDigitsPerWord;
  int wordsFrac = frac / kDigitsPerWord;
  int trailingDigits = frac - wordsFrac * kDigitsPerWord;

  int wordsFracFrom = (int)digitsFrac / kDigitsPerWord;
  int trailingDigitsFrom = (int)digitsFrac - wordsFracFrom * kDigitsPerWord;
  int intSize = wordsInt * kWordSize + dig2bytes[leadingDigits];
  int fracSize = wordsFrac * kWordSize + dig2bytes[trailingDigits];
  int fracSizeFrom = wordsFracFrom * kWordSize + dig2bytes[trailingDigitsFrom];
  int originIntSize = intSize;
  int originFracSize = fracSize;
  std::memset(bin, 0, intSize + fracSize);
  *written = intSize + fracSize;
  int binIdx = 0;
  int wordIdxFrom, digitsIntFrom;
  RemoveLeadingZeros(&wordIdxFrom, &digitsIntFrom);
  if (digitsIntFrom + fracSizeFrom == 0) {
    mask = 0;
    digitsIntL = 1;
  }
  int wordsIntFrom = digitsIntFrom / kDigitsPerWord;
  int leadingDigitsFrom = digitsIntFrom - wordsIntFrom * kDigitsPerWord;
  int iSizeFrom = wordsIntFrom * kWordSize + dig2bytes[leadingDigitsFrom];

  if (digitsIntL < digitsIntFrom) {
    wordIdxFrom += wordsIntFrom - wordsInt;
    if (leadingDigitsFrom > 0) wordIdxFrom++;
    if (leadingDigits > 0) wordIdxFrom--;
    wordsIntFrom = wordsInt;
    leadingDigitsFrom = leadingDigits;
    err = E_OVERFLOW;
  } else if (intSize > iSizeFrom) {
    while (intSize > iSizeFrom) {
      intSize--;
      bin[binIdx++] = (uint8_t)mask;
    }
  }

  if (fracSize < fracSizeFrom ||
      (fracSize == fracSizeFrom && (trailingDigits <= trailingDigitsFrom || wordsFrac <= wordsFracFrom))) {
    if (fracSize < fracSizeFrom ||
        (fracSize == fracSizeFrom && trailingDigits < trailingDigitsFrom) ||
        (fracSize == fracSizeFrom && wordsFrac < wordsFracFrom))
      err = E_TRUNCATED;
    wordsFracFrom = wordsFrac;
    trailingDigitsFrom = trailingDigits;
  } else if (fracSize > fracSizeFrom && trailingDigitsFrom > 0) {
    if (wordsFrac == wordsFracFrom) {
      trailingDigitsFrom = trailingDigits;
      fracSize = fracSizeFrom;
    } else {
      wordsFracFrom++;
      trailingDigitsFrom = 0;
    }
  }
  // xIntFrom part
  if (leadingDigitsFrom > 0) {
    int i = dig2bytes[leadingDigitsFrom];
    int32_t x = (wordBuf[wordIdxFrom] % powers10[leadingDigitsFrom]) ^ mask;
    wordIdxFrom++;
    writeWord(bin + binIdx, x, i);
    binIdx += i;
  }
  // full words
  for (int stop = wordIdxFrom + wordsIntFrom + wordsFracFrom; wordIdxFrom < stop; binIdx += kWordSize) {
    int32_t x = wordBuf[wordIdxFrom] ^ mask;
    wordIdxFrom++;
    writeWord(bin + binIdx, x, 4);
  }
  // xFracFrom part
  if (trailingDigitsFrom > 0) {
    int i = dig2bytes[trailingDigitsFrom];
    int lim = trailingDigits;
    if (wordsFracFrom < wordsFrac) lim = kDigitsPerWord;
    while (trailingDigitsFrom < lim && dig2bytes[trailingDigitsFrom] == i) trailingDigitsFrom++;
    int32_t x = (wordBuf[wordIdxFrom] / powers10[kDigitsPerWord - trailingDigitsFrom]) ^ mask;
    writeWord(bin + binIdx, x, i);
    binIdx += i;
  }
  if (fracSize > fracSizeFrom) {
    int binIdxEnd = originIntSize + originFracSize;
    while (fracSize > fracSizeFrom && binIdx < binIdxEnd) {
      fracSize--;
      bin[binIdx++] = (uint8_t)mask;
    }
  }
  bin[0] ^= 0x80;
  return err;
}

// mydecimal.go:1476 FromBin
int32_t MyDecimal::FromBin(const uint8_t* binIn, int binLenAvail, int precision, int frac, int* binSizeOut) {
  if (binLenAvail == 0) {
    *this = MyDecimal();
    return E_BAD_NUMBER;
  }
  int32_t err = E_OK;
  int digitsIntL = precision - frac;
  int wordsInt = digitsIntL / kDigitsPerWord;
  int leadingDigits = digitsIntL - wordsInt * kDigitsPerWord;
  int wordsFrac = frac / kDigitsPerWord;
  int trailingDigits = frac - wordsFrac * kDigitsPerWord;
  int wordsIntTo = wordsInt;
  if (leadingDigits > 0) wordsIntTo++;
  int wordsFracTo = wordsFrac;
  if (trailingDigits > 0) wordsFracTo++;

  int binIdx = 0;
  int32_t mask = -1;
  if (binIn[0] & 0x80) mask = 0;
  int binSize;
  int32_t e = DecimalBinSize(precision, frac, &binSize);
  if (e != E_OK) return e;
  if (binSize < 0 || binSize > 40) return E_BAD_NUMBER;
  uint8_t dCopy[40];
  std::memset(dCopy, 0, sizeof(dCopy));
  int copyLen = std::min(binSize, binLenAvail);
  std::memcpy(dCopy, binIn, copyLen);
  dCopy[0] ^= 0x80;
  const uint8_t* bin = dCopy;
  int oldWordsIntTo = wordsIntTo;
  err = fixWordCntError(&wordsIntTo, &wordsFracTo);
  if (err != E_OK) {
    if (wordsIntTo < oldWordsIntTo) {
      binIdx += dig2bytes[leadingDigits] + (wordsInt - wordsIntTo) * kWordSize;
    } else {
      trailingDigits = 0;
      wordsFrac = wordsFracTo;
    }
  }
  *this = MyDecimal();
  negative = (mask != 0) ? 1 : 0;
  digitsInt = (int8_t)(wordsInt * kDigitsPerWord + leadingDigits);
  digitsFrac = (int8_t)(wordsFrac * kDigitsPerWord + trailingDigits);

  int wordIdx = 0;
  if (leadingDigits > 0) {
    int i = dig2bytes[leadingDigits];
    int32_t x = readWord(bin + binIdx, i);
    binIdx += i;
    wordBuf[wordIdx] = x ^ mask;
    if ((uint64_t)(uint32_t)wordBuf[wordIdx] >= (uint64_t)powers10[leadingDigits + 1]) {
      *this = MyDecimal();
      *binSizeOut = binSize;
      return E_BAD_NUMBER;
    }
    if (wordIdx > 0 || wordBuf[wordIdx] != 0)
      wordIdx++;
    else
      digitsInt -= (int8_t)leadingDigits;
  }
  for (int stop = binIdx + wordsInt * kWordSize; binIdx < stop; binIdx += kWordSize) {
    wordBuf[wordIdx] = readWord(bin + binIdx, 4) ^ mask;
    if ((uint32_t)wordBuf[wordIdx] > (uint32_t)kWordMax) {
      *this = MyDecimal();
      *binSizeOut = binSize;
      return E_BAD_NUMBER;
    }
    if (wordIdx > 0 || wordBuf[wordIdx] != 0)
      wordIdx++;
    else
      digitsInt -= kDigitsPerWord;
  }
  for (int stop = binIdx + wordsFrac * kWordSize; binIdx < stop; binIdx += kWordSize) {
    wordBuf[wordIdx] = readWord(bin + binIdx, 4) ^ mask;
    if ((uint32_t)wordBuf[wordIdx] > (uint32_t)kWordMax) {
      *this = MyDecimal();
      *binSizeOut = binSize;
      return E_BAD_NUMBER;
    }
    wordIdx++;
  }
  if (trailingDigits > 0) {
    int i = dig2bytes[trailingDigits];
    int32_t x = readWord(bin + binIdx, i);
    wordBuf[wordIdx] = (x ^ mask) * powers10[kDigitsPerWord - trailingDigits];
    if ((uint32_t)wordBuf[wordIdx] > (uint32_t)kWordMax) {
      *this = MyDecimal();
      *binSizeOut = binSize;
      return E_BAD_NUMBER;
    }
  }
  if (digitsInt == 0 && digitsFrac == 0) *this = MyDecimal();
  resultFrac = (int8_t)frac;
  *binSizeOut = binSize;
  return err;
}

// mydecimal.go:1416 ToHashKey
int32_t MyDecimal::ToHashKey(uint8_t* buf, int* written) const {
  int dummy, di, df;
  RemoveLeadingZeros(&dummy, &di);
  RemoveTrailingZeros(&dummy, &df);
  int prec = di + df;
  if (prec == 0) prec = 1;
  int w = 0;
  int32_t err = WriteBin(prec, df, buf, &w);
  if (err == E_TRUNCATED) err = E_OK;
  buf[w] = (uint8_t)df;
  *written = w + 1;
  return err;
}

}  // namespace gxp
