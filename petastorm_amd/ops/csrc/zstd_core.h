// From-scratch ZSTD (RFC 8878) frame decoder — single-pass, no
// allocations, caller-provided workspace — shared by the host batch
// decompressor (zstd_host.cpp) and the HIP kernel (zstd.hip).
//
// Replaces the Arrow C++ zstd decoder the reference reaches through
// piece.read() (reference petastorm/arrow_reader_worker.py:358) for
// Parquet pages with codec ZSTD (one zstd frame per page).
//
// Scope (everything the reference's writers can emit for Parquet pages):
// raw/RLE/compressed blocks; literals raw/RLE/Huffman(1&4-stream)/treeless;
// Huffman weights direct or FSE-compressed; sequences with
// predefined/RLE/FSE/repeat modes; repeated-offset history; optional
// content checksum (skipped).  Dictionaries are not used by Parquet pages.
#ifndef PSA_ZSTD_CORE_H
#define PSA_ZSTD_CORE_H

#include <stdint.h>
#include <stddef.h>
#ifdef PSA_ZDBG
#include <stdio.h>
#define ZDBG(...) printf(__VA_ARGS__)
#else
#define ZDBG(...)
#endif

#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define PSA_HD __host__ __device__
#else
#define PSA_HD
#endif

namespace psa {
namespace zstd {

typedef uint8_t u8;
typedef uint16_t u16;
typedef uint32_t u32;
typedef uint64_t u64;

enum {
  ZERR_OK = 0,
  ZERR_MAGIC = -1,
  ZERR_TRUNC = -2,
  ZERR_RESERVED = -3,
  ZERR_DST_SMALL = -4,
  ZERR_CORRUPT = -5,
  ZERR_TABLE = -6,
  ZERR_UNSUPPORTED = -7,
};

PSA_HD static inline int highbit32(u32 v) {  // floor(log2(v)), v != 0
  int r = 0;
  while (v > 1) { v >>= 1; ++r; }
  return r;
}

// ---------------------------------------------------------------------------
// backward bitstream (zstd reads entropy streams from the END)
// ---------------------------------------------------------------------------
struct BackBits {
  const u8* start;
  u64 container;      // bits; next bits are the HIGH bits after shifting
  int bitsConsumed;   // within container
  const u8* ptr;      // position of the 8-byte window (ptr..ptr+7 loaded)

  PSA_HD u64 load8(const u8* p) {
    u64 v = 0;
    for (int i = 0; i < 8; ++i) v |= (u64)p[i] << (8 * i);
    return v;
  }

  // src..src+size is the stream; last byte holds the 1-bit sentinel
  PSA_HD int init(const u8* src, size_t size) {
    if (size == 0) return ZERR_CORRUPT;
    start = src;
    u8 last = src[size - 1];
    if (last == 0) return ZERR_CORRUPT;
    if (size >= 8) {
      ptr = src + size - 8;
      container = load8(ptr);
      bitsConsumed = 8 - highbit32(last);
    } else {
      // small stream: low-aligned container, missing bytes PRE-COUNTED as
      // consumed so the overflow status fires exactly when the payload
      // runs out (mirrors BIT_initDStream)
      ptr = src;
      container = 0;
      for (size_t i = 0; i < size; ++i) container |= (u64)src[i] << (8 * i);
      bitsConsumed = (8 - highbit32(last)) + (int)(8 - size) * 8;
      smallStream = 1;
      return ZERR_OK;
    }
    smallStream = 0;
    return ZERR_OK;
  }
  int smallStream;

  PSA_HD u32 look(int nb) {  // peek nb bits (nb >= 1)
    u64 shifted = container << bitsConsumed;
    return (u32)(shifted >> (64 - nb));
  }
  PSA_HD void consume(int nb) { bitsConsumed += nb; }
  PSA_HD u32 read(int nb) {
    if (nb == 0) return 0;
    u32 v = look(nb);
    consume(nb);
    return v;
  }
  // mirror of BIT_reloadDStream's status protocol
  enum { UNFINISHED = 0, END_OF_BUFFER = 1, COMPLETED = 2, OVERFLOW_ = 3 };
  PSA_HD int reload() {
    if (bitsConsumed > 64) return OVERFLOW_;
    if (smallStream || ptr == start)
      return bitsConsumed == 64 ? COMPLETED : END_OF_BUFFER;
    int bytes = bitsConsumed >> 3;
    int status = UNFINISHED;
    if ((size_t)(ptr - start) < (size_t)bytes) {
      bytes = (int)(ptr - start);
      status = END_OF_BUFFER;
    }
    ptr -= bytes;
    bitsConsumed -= bytes * 8;
    container = load8(ptr);
    return status;
  }
};

// ---------------------------------------------------------------------------
// FSE
// ---------------------------------------------------------------------------
struct FseEntry {
  u16 newStateBase;
  u8 symbol;
  u8 nbBits;
};

// Read a normalized-count table description from a FORWARD bitstream.
// Returns bytes consumed (>=1) or negative error.  maxSymbol in/out.
PSA_HD static inline int fse_read_ncount(
    const u8* src, size_t srcSize, short* norm, int* maxSymbolPtr,
    int* tableLogPtr, int maxAllowedLog) {
  if (srcSize < 1) return ZERR_TRUNC;
  const u8* ip = src;
  const u8* iend = src + srcSize;
  int bitPos = 0;
  u32 bitStream = 0;
  for (int i = 0; i < 4 && ip + i < iend; ++i)
    bitStream |= (u32)ip[i] << (8 * i);

  int tableLog = (bitStream & 15) + 5;
  if (tableLog > maxAllowedLog) return ZERR_TABLE;
  bitPos = 4;
  int remaining = (1 << tableLog) + 1;
  int threshold = 1 << tableLog;
  int nbBits = tableLog + 1;
  int charnum = 0;
  int maxSymbol = *maxSymbolPtr;
  int previous0 = 0;

  for (int i = 0; i <= maxSymbol; ++i) norm[i] = 0;

  while (remaining > 1 && charnum <= maxSymbol) {
    if (previous0) {
      // run of zero-probability symbols, 2 bits at a time (value 3 = more)
      int n0 = charnum;
      while (((bitStream >> bitPos) & 0xFFFF) == 0xFFFF) {
        n0 += 24;
        bitPos += 16;
        // refill
        if (ip + (bitPos >> 3) + 4 <= iend || ip + (bitPos >> 3) < iend) {
          ip += bitPos >> 3;
          bitPos &= 7;
          bitStream = 0;
          for (int k = 0; k < 4 && ip + k < iend; ++k)
            bitStream |= (u32)ip[k] << (8 * k);
        } else {
          return ZERR_TRUNC;
        }
      }
      while (((bitStream >> bitPos) & 3) == 3) {
        n0 += 3;
        bitPos += 2;
      }
      n0 += (bitStream >> bitPos) & 3;
      bitPos += 2;
      if (n0 > maxSymbol) return ZERR_CORRUPT;
      while (charnum < n0) norm[charnum++] = 0;
      ip += bitPos >> 3;
      bitPos &= 7;
      bitStream = 0;
      for (int k = 0; k < 4 && ip + k < iend; ++k)
        bitStream |= (u32)ip[k] << (8 * k);
    }
    {
      int max = (2 * threshold - 1) - remaining;
      int count;
      if ((int)((bitStream >> bitPos) & (threshold - 1)) < max) {
        count = (bitStream >> bitPos) & (threshold - 1);
        bitPos += nbBits - 1;
      } else {
        count = (bitStream >> bitPos) & (2 * threshold - 1);
        if (count >= threshold) count -= max;
        bitPos += nbBits;
      }
      count--;  // -1 means "less than 1" probability
      remaining -= count < 0 ? -count : count;
      norm[charnum++] = (short)count;
      previous0 = (count == 0);
      while (remaining < threshold) {
        nbBits--;
        threshold >>= 1;
      }
      ip += bitPos >> 3;
      bitPos &= 7;
      bitStream = 0;
      for (int k = 0; k < 4 && ip + k < iend; ++k)
        bitStream |= (u32)ip[k] << (8 * k);
    }
  }
  if (remaining != 1) { ZDBG("    ncount remaining=%d charnum=%d\n",
                              remaining, charnum); return ZERR_CORRUPT; }
  *maxSymbolPtr = charnum - 1;
  *tableLogPtr = tableLog;
  int consumed = (int)(ip - src) + ((bitPos + 7) >> 3);
  if (consumed > (int)srcSize) return ZERR_TRUNC;
  return consumed;
}

// Build an FSE decode table from normalized counts.
PSA_HD static inline int fse_build_dtable(
    const short* norm, int maxSymbol, int tableLog, FseEntry* table,
    u16* symbolNext /* scratch >= maxSymbol+1 */) {
  int tableSize = 1 << tableLog;
  int highThreshold = tableSize - 1;
  for (int s = 0; s <= maxSymbol; ++s) {
    if (norm[s] == -1) {
      table[highThreshold--].symbol = (u8)s;
      symbolNext[s] = 1;
    } else {
      symbolNext[s] = (u16)norm[s];
    }
  }
  int step = (tableSize >> 1) + (tableSize >> 3) + 3;
  int mask = tableSize - 1;
  int pos = 0;
  for (int s = 0; s <= maxSymbol; ++s) {
    for (int i = 0; i < norm[s]; ++i) {
      table[pos].symbol = (u8)s;
      do {
        pos = (pos + step) & mask;
      } while (pos > highThreshold);
    }
  }
  if (pos != 0) { ZDBG("    dtable spread pos=%d\n", pos);
                  return ZERR_TABLE; }
  for (int u = 0; u < tableSize; ++u) {
    u8 s = table[u].symbol;
    u16 nextState = symbolNext[s]++;
    int nb = tableLog - highbit32(nextState);
    table[u].nbBits = (u8)nb;
    table[u].newStateBase = (u16)((nextState << nb) - tableSize);
  }
  return ZERR_OK;
}

struct FseState {
  u32 state;
  const FseEntry* table;
  PSA_HD void init(BackBits& bits, const FseEntry* t, int tableLog) {
    table = t;
    state = bits.read(tableLog);
  }
  PSA_HD u8 peekSymbol() const { return table[state].symbol; }
  PSA_HD void update(BackBits& bits) {
    const FseEntry& e = table[state];
    state = e.newStateBase + bits.read(e.nbBits);
  }
};

// ---------------------------------------------------------------------------
// Huffman (single-level table, tableLog <= 11)
// ---------------------------------------------------------------------------
struct HufEntry {
  u8 symbol;
  u8 nbBits;
};

struct HufTable {
  HufEntry e[1 << 11];
  int tableLog;
  int valid;
};

// Read weights (direct or FSE) and build the decode table.
// Returns bytes consumed from src, or negative error.
PSA_HD static inline int huf_read_dtable(const u8* src, size_t srcSize,
                                         HufTable* ht,
                                         FseEntry* wksp /* >= 64 */,
                                         u16* wkspNext /* >= 256 */) {
  if (srcSize < 1) return ZERR_TRUNC;
  u8 weights[256];
  int nWeights = 0;  // number of explicit weights (symbols 0..n-1)
  int consumed;
  u8 hbyte = src[0];
  if (hbyte >= 128) {
    nWeights = hbyte - 127;
    int nBytes = (nWeights + 1) / 2;
    if ((int)srcSize < 1 + nBytes) return ZERR_TRUNC;
    for (int i = 0; i < nWeights; ++i) {
      u8 b = src[1 + i / 2];
      weights[i] = (i & 1) ? (b & 15) : (b >> 4);
    }
    consumed = 1 + nBytes;
  } else {
    // FSE-compressed weights, hbyte = compressed size
    if ((int)srcSize < 1 + hbyte) return ZERR_TRUNC;
    const u8* wsrc = src + 1;
    short norm[256];
    int maxSym = 255, tlog = 0;
    int hdr = fse_read_ncount(wsrc, hbyte, norm, &maxSym, &tlog, 6);
    ZDBG("    wfse hdr=%d maxSym=%d tlog=%d\n", hdr, maxSym, tlog);
    if (hdr < 0) return hdr;
    int err = fse_build_dtable(norm, maxSym, tlog, wksp, wkspNext);
    if (err) { ZDBG("    wfse build err=%d\n", err); return err; }
    BackBits bits;
    if (bits.init(wsrc + hdr, hbyte - hdr)) return ZERR_CORRUPT;
    FseState s1, s2;
    s1.init(bits, wksp, tlog);
    s2.init(bits, wksp, tlog);
    // canonical 2-state interleaved FSE decode (FSE_decompress tail loop):
    // each GETSYMBOL = peek + state update; terminate when a reload
    // overflows, emitting the other state's final symbol
    const int maxW = 255;  // up to 255 explicit weights (symbols 0..254)
    while (bits.reload() == BackBits::UNFINISHED && nWeights + 4 <= maxW) {
      weights[nWeights++] = s1.peekSymbol(); s1.update(bits);
      weights[nWeights++] = s2.peekSymbol(); s2.update(bits);
      weights[nWeights++] = s1.peekSymbol(); s1.update(bits);
      weights[nWeights++] = s2.peekSymbol(); s2.update(bits);
    }
    while (1) {
      if (nWeights >= maxW) return ZERR_CORRUPT;
      weights[nWeights++] = s1.peekSymbol(); s1.update(bits);
      if (bits.reload() == BackBits::OVERFLOW_) {
        if (nWeights >= maxW) return ZERR_CORRUPT;
        weights[nWeights++] = s2.peekSymbol();
        break;
      }
      if (nWeights >= maxW) return ZERR_CORRUPT;
      weights[nWeights++] = s2.peekSymbol(); s2.update(bits);
      if (bits.reload() == BackBits::OVERFLOW_) {
        if (nWeights >= maxW) return ZERR_CORRUPT;
        weights[nWeights++] = s1.peekSymbol();
        break;
      }
    }
    consumed = 1 + hbyte;
  }
  // derive the implicit last weight
  u32 total = 0;
  for (int i = 0; i < nWeights; ++i) {
    if (weights[i] > 11) { ZDBG("    bad weight[%d]=%d\n", i, weights[i]);
                           return ZERR_CORRUPT; }
    if (weights[i]) total += 1u << (weights[i] - 1);
  }
  ZDBG("    nWeights=%d total=%u\n", nWeights, total);
  if (total == 0) return ZERR_CORRUPT;
  int tableLog = highbit32(total) + 1;
  if (tableLog > 11) return ZERR_CORRUPT;
  u32 rest = (1u << tableLog) - total;
  // rest must be a power of two; the last symbol gets weight log2(rest)+1
  if (rest == 0 || (rest & (rest - 1))) {
    ZDBG("    rest=%u not pow2 (tlog=%d)\n", rest, tableLog);
    return ZERR_CORRUPT;
  }
  weights[nWeights++] = (u8)(highbit32(rest) + 1);

  // canonical table: symbols grouped by weight ascending (longest codes
  // first), natural symbol order within a weight (HUF_readDTableX1)
  u32 rankCount[13] = {0};
  for (int i = 0; i < nWeights; ++i) rankCount[weights[i]]++;
  u32 rankStart[14];
  u32 nextStart = 0;
  for (int w = 1; w <= 12; ++w) {
    rankStart[w] = nextStart;
    nextStart += rankCount[w] << (w - 1);
  }
  if (nextStart != (1u << tableLog)) return ZERR_CORRUPT;
  for (int s = 0; s < nWeights; ++s) {
    int w = weights[s];
    if (!w) continue;
    u32 len = 1u << (w - 1);
    u32 startp = rankStart[w];
    for (u32 k = 0; k < len; ++k) {
      ht->e[startp + k].symbol = (u8)s;
      ht->e[startp + k].nbBits = (u8)(tableLog + 1 - w);
    }
    rankStart[w] += len;
  }
  ht->tableLog = tableLog;
  ht->valid = 1;
  return consumed;
}

PSA_HD static inline int huf_decode_stream(const u8* src, size_t srcSize,
                                           const HufTable* ht, u8* dst,
                                           size_t dstSize) {
  BackBits bits;
  if (bits.init(src, srcSize)) return ZERR_CORRUPT;
  int tlog = ht->tableLog;
  for (size_t i = 0; i < dstSize; ++i) {
    if (bits.reload() == BackBits::OVERFLOW_) return ZERR_CORRUPT;
    u32 v = bits.look(tlog);
    const HufEntry& e = ht->e[v];
    bits.consume(e.nbBits);
    dst[i] = e.symbol;
  }
  return ZERR_OK;
}

// ---------------------------------------------------------------------------
// sequences: predefined distributions (RFC 8878 / extracted from libzstd)
// ---------------------------------------------------------------------------
PSA_HD static inline const short* ll_default_norm() {
  static const short t[36] = {4, 3, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 1, 1, 1,
                              2, 2, 2, 2, 2, 2, 2, 2, 2, 3, 2, 1, 1, 1, 1, 1,
                              -1, -1, -1, -1};
  return t;
}
PSA_HD static inline const short* ml_default_norm() {
  static const short t[53] = {1, 4, 3, 2, 2, 2, 2, 2, 2, 1, 1, 1, 1, 1, 1, 1,
                              1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1,
                              1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, -1,
                              -1, -1, -1, -1, -1, -1};
  return t;
}
PSA_HD static inline const short* of_default_norm() {
  static const short t[29] = {1, 1, 1, 1, 1, 1, 2, 2, 2, 1, 1, 1, 1, 1, 1, 1,
                              1, 1, 1, 1, 1, 1, 1, 1, -1, -1, -1, -1, -1};
  return t;
}

// baselines + extra bits for literal-length / match-length codes
PSA_HD static inline u32 ll_base(int code) {
  static const u32 b[36] = {0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14,
                            15, 16, 18, 20, 22, 24, 28, 32, 40, 48, 64, 128,
                            256, 512, 1024, 2048, 4096, 8192, 16384, 32768,
                            65536};
  return b[code];
}
PSA_HD static inline int ll_bits(int code) {
  static const u8 b[36] = {0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
                           1, 1, 1, 1, 2, 2, 3, 3, 4, 6, 7, 8, 9, 10, 11, 12,
                           13, 14, 15, 16};
  return b[code];
}
PSA_HD static inline u32 ml_base(int code) {
  static const u32 b[53] = {3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15, 16,
                            17, 18, 19, 20, 21, 22, 23, 24, 25, 26, 27, 28,
                            29, 30, 31, 32, 33, 34, 35, 37, 39, 41, 43, 47,
                            51, 59, 67, 83, 99, 131, 259, 515, 1027, 2051,
                            4099, 8195, 16387, 32771, 65539};
  return b[code];
}
PSA_HD static inline int ml_bits(int code) {
  static const u8 b[53] = {0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
                           0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
                           1, 1, 1, 1, 2, 2, 3, 3, 4, 4, 5, 7, 8, 9, 10, 11,
                           12, 13, 14, 15, 16};
  return b[code];
}

// ---------------------------------------------------------------------------
// frame decode workspace
// ---------------------------------------------------------------------------
struct ZstdWork {
  HufTable huf;                 // persists across blocks (treeless mode)
  FseEntry llTable[1 << 9];
  FseEntry ofTable[1 << 8];
  FseEntry mlTable[1 << 9];
  int llLog, ofLog, mlLog;
  int llValid, ofValid, mlValid;   // for repeat mode
  FseEntry wksp[1 << 6];           // huffman-weight FSE table
  u16 wkspNext[256];
  short norm[256];
  u8 litBuf[1 << 17];              // decoded literals of one block (128 KiB)
};

// decode the literals section; returns bytes consumed from src or negative.
// *litSize receives the regenerated size (literals in ws->litBuf).
PSA_HD static inline int decode_literals(const u8* src, size_t srcSize,
                                         ZstdWork* ws, size_t* litSize) {
  if (srcSize < 1) return ZERR_TRUNC;
  int type = src[0] & 3;
  int sf = (src[0] >> 2) & 3;
  ZDBG("  lit type=%d sf=%d b0=%02x\n", type, sf, src[0]);
  if (type == 0 || type == 1) {  // raw / RLE
    size_t rs;
    int hdr;
    if ((sf & 1) == 0) {         // 00 or 10: 5-bit size, 1-byte header
      rs = src[0] >> 3;
      hdr = 1;
    } else if (sf == 1) {        // 01: 12-bit
      if (srcSize < 2) return ZERR_TRUNC;
      rs = (src[0] >> 4) | ((size_t)src[1] << 4);
      hdr = 2;
    } else {                     // 11: 20-bit
      if (srcSize < 3) return ZERR_TRUNC;
      rs = (src[0] >> 4) | ((size_t)src[1] << 4) | ((size_t)src[2] << 12);
      hdr = 3;
    }
    if (rs > sizeof(ws->litBuf)) return ZERR_CORRUPT;
    if (type == 0) {
      if (srcSize < (size_t)hdr + rs) return ZERR_TRUNC;
      for (size_t i = 0; i < rs; ++i) ws->litBuf[i] = src[hdr + i];
      *litSize = rs;
      return hdr + (int)rs;
    }
    if (srcSize < (size_t)hdr + 1) return ZERR_TRUNC;
    for (size_t i = 0; i < rs; ++i) ws->litBuf[i] = src[hdr];
    *litSize = rs;
    return hdr + 1;
  }
  // compressed (2) or treeless (3)
  size_t rs, cs;
  int hdr;
  int fourStreams = 1;
  if (sf == 0) {                 // single stream, 10-bit sizes
    if (srcSize < 3) return ZERR_TRUNC;
    rs = (src[0] >> 4) | (((size_t)src[1] & 0x3F) << 4);
    cs = ((size_t)src[1] >> 6) | ((size_t)src[2] << 2);
    hdr = 3;
    fourStreams = 0;
  } else if (sf == 1) {          // 4 streams, 10-bit sizes
    if (srcSize < 3) return ZERR_TRUNC;
    rs = (src[0] >> 4) | (((size_t)src[1] & 0x3F) << 4);
    cs = ((size_t)src[1] >> 6) | ((size_t)src[2] << 2);
    hdr = 3;
  } else if (sf == 2) {          // 4 streams, 14-bit sizes
    if (srcSize < 4) return ZERR_TRUNC;
    rs = (src[0] >> 4) | ((size_t)src[1] << 4) |
         (((size_t)src[2] & 3) << 12);
    cs = ((size_t)src[2] >> 2) | ((size_t)src[3] << 6);
    hdr = 4;
  } else {                       // 4 streams, 18-bit sizes
    if (srcSize < 5) return ZERR_TRUNC;
    rs = (src[0] >> 4) | ((size_t)src[1] << 4) |
         (((size_t)src[2] & 0x3F) << 12);
    cs = ((size_t)src[2] >> 6) | ((size_t)src[3] << 2) |
         ((size_t)src[4] << 10);
    hdr = 5;
  }
  ZDBG("    huf rs=%zu cs=%zu hdr=%d\n", rs, cs, hdr);
  if (rs > sizeof(ws->litBuf)) return ZERR_CORRUPT;
  if (srcSize < (size_t)hdr + cs) return ZERR_TRUNC;
  const u8* lit = src + hdr;
  size_t litCs = cs;
  if (type == 2) {
    int used = huf_read_dtable(lit, litCs, &ws->huf, ws->wksp, ws->wkspNext);
    ZDBG("    huf_read_dtable used=%d tlog=%d\n", used,
         used >= 0 ? ws->huf.tableLog : -1);
    if (used < 0) return used;
    lit += used;
    litCs -= used;
  } else if (!ws->huf.valid) {
    return ZERR_CORRUPT;  // treeless without a previous table
  }
  if (!fourStreams) {
    int err = huf_decode_stream(lit, litCs, &ws->huf, ws->litBuf, rs);
    if (err) return err;
  } else {
    if (litCs < 6) return ZERR_TRUNC;
    size_t s1 = lit[0] | ((size_t)lit[1] << 8);
    size_t s2 = lit[2] | ((size_t)lit[3] << 8);
    size_t s3 = lit[4] | ((size_t)lit[5] << 8);
    if (6 + s1 + s2 + s3 > litCs) return ZERR_TRUNC;
    size_t s4 = litCs - 6 - s1 - s2 - s3;
    size_t r1 = (rs + 3) / 4, r4 = rs - 3 * r1;
    if (rs < 3 * r1) return ZERR_CORRUPT;
    const u8* p = lit + 6;
    int err = huf_decode_stream(p, s1, &ws->huf, ws->litBuf, r1);
    if (!err) err = huf_decode_stream(p + s1, s2, &ws->huf,
                                      ws->litBuf + r1, r1);
    if (!err) err = huf_decode_stream(p + s1 + s2, s3, &ws->huf,
                                      ws->litBuf + 2 * r1, r1);
    if (!err) err = huf_decode_stream(p + s1 + s2 + s3, s4, &ws->huf,
                                      ws->litBuf + 3 * r1, r4);
    if (err) return err;
  }
  *litSize = rs;
  return hdr + (int)cs;
}

// build one of the three sequence tables according to its mode.
// Returns bytes consumed or negative.
PSA_HD static inline int build_seq_table(
    const u8* src, size_t srcSize, int mode, FseEntry* table, int* tlogPtr,
    int* validPtr, const short* defaultNorm, int defaultMax, int defaultLog,
    int maxLog, ZstdWork* ws) {
  if (mode == 0) {  // predefined
    int err = fse_build_dtable(defaultNorm, defaultMax, defaultLog, table,
                               ws->wkspNext);
    if (err) return err;
    *tlogPtr = defaultLog;
    *validPtr = 1;
    return 0;
  }
  if (mode == 1) {  // RLE: single symbol, "table" with 0-bit transitions
    if (srcSize < 1) return ZERR_TRUNC;
    table[0].symbol = src[0];
    table[0].nbBits = 0;
    table[0].newStateBase = 0;
    *tlogPtr = 0;
    *validPtr = 1;
    return 1;
  }
  if (mode == 2) {  // FSE-compressed distribution
    int maxSym = 255, tlog = 0;
    int used = fse_read_ncount(src, srcSize, ws->norm, &maxSym, &tlog,
                               maxLog);
    if (used < 0) return used;
    int err = fse_build_dtable(ws->norm, maxSym, tlog, table, ws->wkspNext);
    if (err) return err;
    *tlogPtr = tlog;
    *validPtr = 1;
    return used;
  }
  // repeat: reuse previous table
  if (!*validPtr) return ZERR_CORRUPT;
  return 0;
}

// decode one compressed block into dst (appending at histEnd of window).
// Returns regenerated size or negative.
PSA_HD static inline long decode_block(const u8* src, size_t srcSize,
                                       u8* dstBase, size_t dstPos,
                                       size_t dstCap, ZstdWork* ws,
                                       u32 rep[3]) {
  size_t litSize = 0;
  int used = decode_literals(src, srcSize, ws, &litSize);
  ZDBG("  literals used=%d litSize=%zu\n", used, litSize);
  if (used < 0) return used;
  const u8* ip = src + used;
  size_t remaining = srcSize - used;

  // sequences header
  if (remaining < 1) return ZERR_TRUNC;
  int nbSeq;
  if (ip[0] < 128) {
    nbSeq = ip[0];
    ip += 1;
    remaining -= 1;
  } else if (ip[0] < 255) {
    if (remaining < 2) return ZERR_TRUNC;
    nbSeq = ((ip[0] - 128) << 8) + ip[1];
    ip += 2;
    remaining -= 2;
  } else {
    if (remaining < 3) return ZERR_TRUNC;
    nbSeq = ip[1] + (ip[2] << 8) + 0x7F00;
    ip += 3;
    remaining -= 3;
  }
  if (nbSeq == 0) {
    // literals only
    if (dstPos + litSize > dstCap) return ZERR_DST_SMALL;
    for (size_t i = 0; i < litSize; ++i) dstBase[dstPos + i] = ws->litBuf[i];
    return (long)litSize;
  }
  if (remaining < 1) return ZERR_TRUNC;
  int modes = ip[0];
  ZDBG("  nbSeq=%d modes=%02x\n", nbSeq, modes);
  if (modes & 3) return ZERR_RESERVED;  // low 2 bits reserved
  int llMode = (modes >> 6) & 3;
  int ofMode = (modes >> 4) & 3;
  int mlMode = (modes >> 2) & 3;
  ip += 1;
  remaining -= 1;

  int used2;
  used2 = build_seq_table(ip, remaining, llMode, ws->llTable, &ws->llLog,
                          &ws->llValid, ll_default_norm(), 35, 6, 9, ws);
  ZDBG("  llMode=%d used=%d log=%d\n", llMode, used2, ws->llLog);
  if (used2 < 0) return used2;
  ip += used2; remaining -= used2;
  used2 = build_seq_table(ip, remaining, ofMode, ws->ofTable, &ws->ofLog,
                          &ws->ofValid, of_default_norm(), 28, 5, 8, ws);
  ZDBG("  ofMode=%d used=%d log=%d\n", ofMode, used2, ws->ofLog);
  if (used2 < 0) return used2;
  ip += used2; remaining -= used2;
  used2 = build_seq_table(ip, remaining, mlMode, ws->mlTable, &ws->mlLog,
                          &ws->mlValid, ml_default_norm(), 52, 6, 9, ws);
  ZDBG("  mlMode=%d used=%d log=%d\n", mlMode, used2, ws->mlLog);
  if (used2 < 0) return used2;
  ip += used2; remaining -= used2;

  BackBits bits;
  if (bits.init(ip, remaining)) return ZERR_CORRUPT;
  FseState ll, of, ml;
  ll.init(bits, ws->llTable, ws->llLog);
  of.init(bits, ws->ofTable, ws->ofLog);
  ml.init(bits, ws->mlTable, ws->mlLog);

  ZDBG("  states: ll=%u of=%u ml=%u consumed=%d small=%d\n",
       ll.state, of.state, ml.state, bits.bitsConsumed, bits.smallStream);
  size_t litPos = 0;
  size_t out = dstPos;
  for (int seq = 0; seq < nbSeq; ++seq) {
    if (bits.reload() == BackBits::OVERFLOW_) return ZERR_CORRUPT;
    int ofCode = of.peekSymbol();
    int mlCode = ml.peekSymbol();
    int llCode = ll.peekSymbol();
    ZDBG("  codes: of=%d ml=%d ll=%d\n", ofCode, mlCode, llCode);
    if (ofCode > 31 || mlCode > 52 || llCode > 35) return ZERR_CORRUPT;
    // extra bits are read OF, ML, LL (RFC 8878 3.1.1.4)
    u32 ofValue;
    if (ofCode == 0) {
      ofValue = 1;  // (1<<0)+0
    } else {
      u32 extra = ofCode > 25
          ? ((bits.read(ofCode - 25) << 25) | bits.read(25))
          : bits.read(ofCode);
      ofValue = (1u << ofCode) + extra;
    }
    if (bits.reload() == BackBits::OVERFLOW_) return ZERR_CORRUPT;
    u32 matchLen = ml_base(mlCode) + bits.read(ml_bits(mlCode));
    u32 litLen = ll_base(llCode) + bits.read(ll_bits(llCode));

    // repeated-offset resolution
    u32 offset;
    if (ofValue > 3) {
      offset = ofValue - 3;
      rep[2] = rep[1];
      rep[1] = rep[0];
      rep[0] = offset;
    } else {
      u32 idx = ofValue + (litLen == 0 ? 1 : 0);
      if (idx == 1) {
        offset = rep[0];
      } else if (idx == 2) {
        offset = rep[1];
        rep[1] = rep[0];
        rep[0] = offset;
      } else if (idx == 3) {
        offset = rep[2];
        rep[2] = rep[1];
        rep[1] = rep[0];
        rep[0] = offset;
      } else {  // idx == 4: rep1 - 1
        offset = rep[0] - 1;
        if (offset == 0) return ZERR_CORRUPT;
        rep[2] = rep[1];
        rep[1] = rep[0];
        rep[0] = offset;
      }
    }

    // copy literals
    if (litPos + litLen > litSize) return ZERR_CORRUPT;
    if (out + litLen + matchLen > dstCap) return ZERR_DST_SMALL;
    for (u32 i = 0; i < litLen; ++i)
      dstBase[out + i] = ws->litBuf[litPos + i];
    out += litLen;
    litPos += litLen;
    // copy match
    if ((size_t)offset > out) return ZERR_CORRUPT;
    for (u32 i = 0; i < matchLen; ++i)
      dstBase[out + i] = dstBase[out - offset + i];
    out += matchLen;

    ZDBG("  seq %d: ll=%u ml=%u of=%u out=%zu\n", seq, litLen, matchLen,
         offset, out);
    // state updates (order LL, ML, OF), skipped after the last sequence
    if (seq < nbSeq - 1) {
      if (bits.reload() == BackBits::OVERFLOW_) return ZERR_CORRUPT;
      ll.update(bits);
      ml.update(bits);
      of.update(bits);
    }
  }
  // trailing literals
  size_t tail = litSize - litPos;
  if (out + tail > dstCap) return ZERR_DST_SMALL;
  for (size_t i = 0; i < tail; ++i) dstBase[out + i] = ws->litBuf[litPos + i];
  out += tail;
  return (long)(out - dstPos);
}

// Decode one complete zstd frame.  Returns total regenerated bytes or a
// negative ZERR code.
PSA_HD static inline long decode_frame(const u8* src, size_t srcSize,
                                       u8* dst, size_t dstCap,
                                       ZstdWork* ws) {
  if (srcSize < 4) return ZERR_TRUNC;
  u32 magic = (u32)src[0] | ((u32)src[1] << 8) | ((u32)src[2] << 16) |
              ((u32)src[3] << 24);
  // skippable frames: magic 0x184D2A5? -> skip (rare in Parquet)
  if ((magic & 0xFFFFFFF0u) == 0x184D2A50u) {
    if (srcSize < 8) return ZERR_TRUNC;
    return 0;
  }
  if (magic != 0xFD2FB528u) return ZERR_MAGIC;
  size_t pos = 4;
  if (pos >= srcSize) return ZERR_TRUNC;
  u8 fhd = src[pos++];
  int fcsFlag = fhd >> 6;
  int singleSegment = (fhd >> 5) & 1;
  int checksum = (fhd >> 2) & 1;
  int dictFlag = fhd & 3;
  if ((fhd >> 3) & 1) return ZERR_RESERVED;
  if (!singleSegment) pos += 1;  // window descriptor (unused: dst is full)
  static const int dictLen[4] = {0, 1, 2, 4};
  pos += dictLen[dictFlag];
  int fcsLen;
  if (fcsFlag == 0) fcsLen = singleSegment ? 1 : 0;
  else if (fcsFlag == 1) fcsLen = 2;
  else if (fcsFlag == 2) fcsLen = 4;
  else fcsLen = 8;
  pos += fcsLen;
  if (pos > srcSize) return ZERR_TRUNC;

  ws->huf.valid = 0;
  ws->llValid = ws->ofValid = ws->mlValid = 0;
  u32 rep[3] = {1, 4, 8};
  size_t out = 0;
  while (1) {
    if (pos + 3 > srcSize) return ZERR_TRUNC;
    u32 bh = (u32)src[pos] | ((u32)src[pos + 1] << 8) |
             ((u32)src[pos + 2] << 16);
    pos += 3;
    int last = bh & 1;
    int btype = (bh >> 1) & 3;
    u32 bsize = bh >> 3;
    ZDBG("block last=%d type=%d size=%u out=%zu\n", last, btype, bsize, out);
    if (btype == 0) {  // raw
      if (pos + bsize > srcSize || out + bsize > dstCap) return ZERR_TRUNC;
      for (u32 i = 0; i < bsize; ++i) dst[out + i] = src[pos + i];
      pos += bsize;
      out += bsize;
    } else if (btype == 1) {  // RLE
      if (pos + 1 > srcSize || out + bsize > dstCap) return ZERR_TRUNC;
      u8 b = src[pos++];
      for (u32 i = 0; i < bsize; ++i) dst[out + i] = b;
      out += bsize;
    } else if (btype == 2) {
      if (pos + bsize > srcSize) return ZERR_TRUNC;
      long produced = decode_block(src + pos, bsize, dst, out, dstCap, ws,
                                   rep);
      if (produced < 0) return produced;
      pos += bsize;
      out += (size_t)produced;
    } else {
      return ZERR_RESERVED;
    }
    if (last) break;
  }
  if (checksum) pos += 4;  // xxh64 low 32 bits — not verified
  (void)pos;
  return (long)out;
}

}  // namespace zstd
}  // namespace psa

#endif  // PSA_ZSTD_CORE_H
