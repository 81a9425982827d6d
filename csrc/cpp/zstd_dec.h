// From-scratch zstd *decoder* (RFC 8878 subset: everything libzstd's
// compressor emits — raw/RLE/compressed blocks, huffman literals with
// direct or FSE-compressed weights, 1/4-stream literals, predefined /
// RLE / FSE / repeat sequence tables, repeat offsets, multi-block
// frames; no dictionaries, no checksum verification).
//
// Written to be portable into a HIP kernel: no STL containers in the
// decode path, fixed-size scratch tables, plain C-style code. The host
// build is differential-tested against libzstd on real parquet pages
// (tests/test_zstd_dec.py); the device port lives in csrc/hip/zstd.hip.
//
// This replaces host-side libzstd in the GPU scan path so page
// decompression scales with the GPU instead of the (cgroup-capped) CPU
// quota — see SURVEY.md §7.2 (reference decompresses on tokio threads,
// arrow-rs parquet; we aim the same bytes at the MI355X).

#pragma once

#include <cstdint>
#include <cstring>

#ifndef LSZ_HD
#if defined(__HIPCC__) || defined(__HIP_DEVICE_COMPILE__)
#define LSZ_HD __host__ __device__
#define LSZ_COLD __host__ __device__ __attribute__((noinline))
#else
#define LSZ_HD
#define LSZ_COLD
#endif
#endif

namespace lszstd {

static const uint32_t kMagic = 0xFD2FB528u;

LSZ_HD inline int highbit32(uint32_t v) {
#if defined(__HIP_DEVICE_COMPILE__)
  return 31 - __clz(v);
#elif defined(__GNUC__)
  return 31 - __builtin_clz(v);
#else
  int r = 0;
  while (v >>= 1) r++;
  return r;
#endif
}

// ---------------------------------------------------------------------- //
// backward bitstream: bits were appended LSB-first; the reader starts at
// the sentinel (highest set bit of the last byte) and reads downward.
// Reads past the start return zero bits and set `overflow`.
// ---------------------------------------------------------------------- //

struct BitBwd {
  const uint8_t* buf;
  int64_t bitpos;    // bits remaining below the cursor
  uint64_t cont;     // register window: stream bits [contEnd-64, contEnd)
  int64_t contEnd;   // byte-aligned top of the window (bit index)
  bool overflow;

  LSZ_HD void refill() {
    // place an 8-byte window ending at the cursor byte; never reads past
    // the buffer end (bits exist => bytes do) and zero-pads small bufs
    int64_t byteHi = (bitpos + 7) >> 3;
    if (byteHi >= 8) {
      memcpy(&cont, buf + byteHi - 8, 8);
      contEnd = byteHi * 8;
    } else {
      uint64_t w = 0;
      for (int64_t i = 0; i < byteHi; i++) w |= (uint64_t)buf[i] << (8 * i);
      contEnd = 64;          // treat bits [0,64); bits >= byteHi*8 unused
      cont = w;
    }
  }

  LSZ_HD bool init(const uint8_t* p, int64_t n) {
    buf = p;
    overflow = false;
    cont = 0;
    contEnd = 0;
    if (n <= 0) { bitpos = 0; overflow = true; return false; }
    uint8_t last = p[n - 1];
    if (last == 0) { bitpos = 0; overflow = true; return false; }
    bitpos = (n - 1) * 8 + highbit32(last);  // sentinel bit excluded
    refill();
    return true;
  }

  // read n bits [bitpos-n, bitpos); zero-padded when under-running
  LSZ_HD uint32_t read(int n) {
    if (n == 0) return 0;
    if (bitpos - n < contEnd - 64) refill();
    bitpos -= n;
    uint64_t mask = (n >= 32) ? 0xFFFFFFFFull : ((1ull << n) - 1);
    int64_t shift = bitpos - (contEnd - 64);
    if (bitpos < 0) overflow = true;
    if (shift >= 0) return (uint32_t)((cont >> shift) & mask);
    // under-run: stream bits below 0 read as zeros
    if (shift <= -64) return 0;
    return (uint32_t)((cont << (-shift)) & mask);
  }

  LSZ_HD bool done() const { return bitpos <= 0; }
};

// forward bitstream (FSE table descriptions), LSB-first
struct BitFwd {
  const uint8_t* buf;
  int64_t n;
  int64_t bitpos = 0;

  LSZ_HD uint32_t read(int nb) {
    uint32_t out = 0;
    for (int i = 0; i < nb; i++) {
      int64_t b = bitpos + i;
      if ((b >> 3) < n) out |= (uint32_t)((buf[b >> 3] >> (b & 7)) & 1) << i;
    }
    bitpos += nb;
    return out;
  }
  LSZ_HD uint32_t peek(int nb) const {
    uint32_t out = 0;
    for (int i = 0; i < nb; i++) {
      int64_t b = bitpos + i;
      if ((b >> 3) < n) out |= (uint32_t)((buf[b >> 3] >> (b & 7)) & 1) << i;
    }
    return out;
  }
  LSZ_HD int64_t bytes_consumed() const { return (bitpos + 7) >> 3; }
};

// ---------------------------------------------------------------------- //
// FSE
// ---------------------------------------------------------------------- //

static const int kMaxTableLog = 9;          // zstd: OF 8, ML 9, LL 9, weights 6
static const int kMaxSymbols = 256;

struct FseEntry {
  uint8_t symbol;
  uint8_t nbBits;
  uint16_t newState;  // baseline; nextState = newState + read(nbBits)
};

struct FseTable {  // no default initializers: instances live in LDS
  FseEntry e[1 << kMaxTableLog];
  int tableLog;
  bool rle;          // degenerate: single symbol, no bits
  uint8_t rleSym;
};

// read normalized counts (RFC 8878 4.1.1) from a forward bitstream.
// returns max symbol (inclusive) or -1 on error.
LSZ_COLD inline int fse_read_ncount(BitFwd& br, int16_t* counts, int maxSymLimit,
                                  int& tableLog) {
  tableLog = (int)br.read(4) + 5;
  if (tableLog > kMaxTableLog) return -1;
  int32_t remaining = (1 << tableLog) + 1;
  int sym = 0;
  bool prev0 = false;
  for (int i = 0; i <= maxSymLimit; i++) counts[i] = 0;
  while (remaining > 1 && sym <= maxSymLimit) {
    if (prev0) {
      // runs of zero-probability symbols: 2-bit repeat counts
      int rpt = (int)br.read(2);
      sym += rpt;
      while (rpt == 3) {
        rpt = (int)br.read(2);
        sym += rpt;
      }
      prev0 = false;
      if (sym > maxSymLimit) return -1;
      continue;
    }
    int nbBits = highbit32((uint32_t)remaining) + 1;
    uint32_t val = br.peek(nbBits);
    uint32_t lowMask = (1u << (nbBits - 1)) - 1;
    uint32_t threshold = (uint32_t)((1 << nbBits) - 1 - remaining);
    if ((val & lowMask) < threshold) {
      br.read(nbBits - 1);
      val &= lowMask;
    } else {
      br.read(nbBits);
      val &= (1u << nbBits) - 1;
      if (val >= lowMask + 1) val -= threshold;  // fold the high range
    }
    int32_t count = (int32_t)val - 1;  // -1 means "less than one"
    if (count == -1) {
      counts[sym] = -1;
      remaining -= 1;
    } else {
      counts[sym] = (int16_t)count;
      remaining -= count;
      if (count == 0) prev0 = true;
    }
    sym++;
  }
  if (remaining != 1) return -1;
  return sym - 1;
}

// build a decode table from normalized counts (FSE_buildDTable).
// symTab/symbolNext are caller scratch (>= 512 / 256 entries) so the
// device build can keep them in LDS instead of per-lane scratch memory.
LSZ_COLD inline bool fse_build(FseTable& t, const int16_t* counts, int maxSym,
                               int tableLog, uint8_t* symTab,
                               uint16_t* symbolNext) {
  t.tableLog = tableLog;
  t.rle = false;
  int tableSize = 1 << tableLog;
  int highThreshold = tableSize - 1;

  for (int s = 0; s <= maxSym; s++) {
    if (counts[s] == -1) {
      symTab[highThreshold--] = (uint8_t)s;
      symbolNext[s] = 1;
    } else {
      symbolNext[s] = (uint16_t)counts[s];
    }
  }
  // spread symbols
  int step = (tableSize >> 1) + (tableSize >> 3) + 3;
  int mask = tableSize - 1;
  int pos = 0;
  for (int s = 0; s <= maxSym; s++) {
    for (int i = 0; i < counts[s]; i++) {
      symTab[pos] = (uint8_t)s;
      pos = (pos + step) & mask;
      while (pos > highThreshold) pos = (pos + step) & mask;
    }
  }
  if (pos != 0) return false;
  // state transitions
  for (int u = 0; u < tableSize; u++) {
    uint8_t s = symTab[u];
    uint16_t nextState = symbolNext[s]++;
    int nbBits = tableLog - highbit32(nextState);
    t.e[u].symbol = s;
    t.e[u].nbBits = (uint8_t)nbBits;
    t.e[u].newState = (uint16_t)((nextState << nbBits) - tableSize);
  }
  return true;
}

LSZ_HD inline bool fse_build(FseTable& t, const int16_t* counts, int maxSym,
                             int tableLog) {
  uint8_t symTab[1 << kMaxTableLog];
  uint16_t symbolNext[kMaxSymbols];
  return fse_build(t, counts, maxSym, tableLog, symTab, symbolNext);
}

LSZ_HD inline void fse_build_rle(FseTable& t, uint8_t sym) {
  t.rle = true;
  t.rleSym = sym;
  t.tableLog = 0;
  t.e[0].symbol = sym;
  t.e[0].nbBits = 0;
  t.e[0].newState = 0;
}

struct FseState {
  uint32_t state;
  LSZ_HD void init(const FseTable& t, BitBwd& br) {
    state = t.rle ? 0 : br.read(t.tableLog);
  }
  LSZ_HD uint8_t symbol(const FseTable& t) const { return t.e[state].symbol; }
  LSZ_HD void update(const FseTable& t, BitBwd& br) {
    if (t.rle) return;
    const FseEntry& e = t.e[state];
    state = e.newState + br.read(e.nbBits);
  }
};

// generic FSE decompression (used for huffman weights): strict s1/s2
// alternation, stop after overflow emits the final symbol (libzstd
// FSE_decompress_usingDTable tail semantics)
LSZ_COLD inline int fse_decompress_ws(const uint8_t* src, int64_t n, uint8_t* dst,
                                    int dstCap, FseTable& table,
                                    int16_t* counts, uint8_t* symTab,
                                    uint16_t* symbolNext) {
  BitFwd hdr{src, n};
  int tableLog;
  int maxSym = fse_read_ncount(hdr, counts, 255, tableLog);
  if (maxSym < 0) return -1;
  if (!fse_build(table, counts, maxSym, tableLog, symTab, symbolNext))
    return -1;
  int64_t consumed = hdr.bytes_consumed();
  BitBwd br;
  if (!br.init(src + consumed, n - consumed)) return -1;
  FseState s1, s2;
  s1.init(table, br);
  s2.init(table, br);
  int out = 0;
  for (;;) {
    if (out >= dstCap) return -1;
    dst[out++] = s1.symbol(table);
    s1.update(table, br);
    if (br.overflow) {
      if (out >= dstCap) return -1;
      dst[out++] = s2.symbol(table);
      break;
    }
    if (out >= dstCap) return -1;
    dst[out++] = s2.symbol(table);
    s2.update(table, br);
    if (br.overflow) {
      if (out >= dstCap) return -1;
      dst[out++] = s1.symbol(table);
      break;
    }
  }
  return out;
}

LSZ_HD inline int fse_decompress(const uint8_t* src, int64_t n, uint8_t* dst,
                                 int dstCap) {
  FseTable table;
  int16_t counts[kMaxSymbols];
  uint8_t symTab[1 << kMaxTableLog];
  uint16_t symbolNext[kMaxSymbols];
  return fse_decompress_ws(src, n, dst, dstCap, table, counts, symTab,
                           symbolNext);
}

// ---------------------------------------------------------------------- //
// Huffman literals
// ---------------------------------------------------------------------- //

static const int kHufMaxBits = 11;  // zstd huffman log limit

struct HufEntry {
  uint8_t symbol;
  uint8_t nbBits;
};

struct HufTable {  // no default initializers: instances live in LDS
  HufEntry e[1 << kHufMaxBits];
  int maxBits;
};

// build decode table from weights[0..nsym-1] (HUF_readDTableX1 layout)
LSZ_COLD inline bool huf_build(HufTable& t, const uint8_t* weights, int nsym) {
  uint32_t rankCount[kHufMaxBits + 2] = {0};
  uint32_t total = 0;
  for (int s = 0; s < nsym; s++) {
    if (weights[s] > kHufMaxBits) return false;
    rankCount[weights[s]]++;
    if (weights[s]) total += 1u << (weights[s] - 1);
  }
  if (total == 0) return false;
  int maxBits = highbit32(total) + 1;
  if (maxBits > kHufMaxBits) return false;
  // implicit last symbol: fills the gap to the next power of two
  uint32_t rest = (1u << maxBits) - total;
  // rest must be a power of two; its weight:
  if (rest & (rest - 1)) return false;
  int lastWeight = highbit32(rest) + 1;
  t.maxBits = maxBits;

  // table fill: symbols of weight w occupy 2^(w-1) scaled cells; cells
  // are assigned in weight order (low weight = long codes first),
  // symbols of equal weight in symbol order
  uint32_t rankStart[kHufMaxBits + 2] = {0};
  {
    uint32_t cur = 0;
    for (int w = 1; w <= maxBits; w++) {
      rankStart[w] = cur;
      uint32_t cells = (rankCount[w] + (w == lastWeight ? 1 : 0))
                       << (w - 1);
      cur += cells;
    }
    if (cur != (1u << maxBits)) return false;
  }
  for (int s = 0; s <= nsym; s++) {
    int w = (s == nsym) ? lastWeight : weights[s];
    if (w == 0) continue;
    uint32_t len = 1u << (w - 1);
    uint32_t start = rankStart[w];
    for (uint32_t i = 0; i < len; i++) {
      t.e[start + i].symbol = (uint8_t)s;
      t.e[start + i].nbBits = (uint8_t)(maxBits + 1 - w);
    }
    rankStart[w] += len;
  }
  return true;
}

// huffman-decode one backward stream into dst[0..outLen)
LSZ_HD inline bool huf_stream(const HufTable& t, const uint8_t* src, int64_t n,
                              uint8_t* dst, int64_t outLen) {
  BitBwd br;
  if (!br.init(src, n)) return outLen == 0;
  uint64_t state = br.read(t.maxBits);  // top maxBits bits
  // reading semantics: state holds the next maxBits bits (MSB-aligned
  // window); after consuming nb bits we shift in nb more from below.
  for (int64_t i = 0; i < outLen; i++) {
    const HufEntry& e = t.e[state];
    dst[i] = e.symbol;
    uint32_t nb = e.nbBits;
    uint32_t more = br.read((int)nb);  // zero-padded past start
    state = ((state << nb) | more) & ((1u << t.maxBits) - 1);
  }
  return true;
}

// ---------------------------------------------------------------------- //
// sequence code tables (RFC 8878 3.1.1.3.2.1)
// ---------------------------------------------------------------------- //

struct CodeExtra {
  uint32_t base;
  uint8_t bits;
};

LSZ_HD inline CodeExtra ll_extra(int code) {
  static const uint32_t base[36] = {
      0,  1,  2,  3,  4,  5,  6,  7,  8,  9,  10, 11,   12,   13,   14,   15,
      16, 18, 20, 22, 24, 28, 32, 40, 48, 64, 128, 256, 512, 1024, 2048, 4096,
      8192, 16384, 32768, 65536};
  static const uint8_t bits[36] = {0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
                                   0, 0, 0, 0, 1, 1, 1, 1, 2, 2, 3, 3,
                                   4, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15, 16};
  return {base[code], bits[code]};
}

LSZ_HD inline CodeExtra ml_extra(int code) {
  static const uint32_t base[53] = {
      3,  4,  5,  6,  7,  8,  9,  10, 11, 12, 13, 14, 15, 16, 17, 18, 19, 20,
      21, 22, 23, 24, 25, 26, 27, 28, 29, 30, 31, 32, 33, 34, 35, 37, 39, 41,
      43, 47, 51, 59, 67, 83, 99, 131, 259, 515, 1027, 2051, 4099, 8195, 16387,
      32771, 65539};
  static const uint8_t bits[53] = {0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
                                   0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
                                   0, 0, 0, 0, 1, 1, 1, 1, 2, 2, 3, 3, 4, 4,
                                   5, 7, 8, 9, 10, 11, 12, 13, 14, 15, 16};
  return {base[code], bits[code]};
}

// predefined distributions (RFC 8878 3.1.1.3.2.2)
LSZ_HD inline void predef_ll(int16_t* c, int& maxSym, int& tlog) {
  static const int16_t d[36] = {4, 3, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2,
                                2, 1, 1, 1, 2, 2, 2, 2, 2, 2, 2, 2,
                                2, 3, 2, 1, 1, 1, 1, 1, -1, -1, -1, -1};
  for (int i = 0; i < 36; i++) c[i] = d[i];
  maxSym = 35;
  tlog = 6;
}
LSZ_HD inline void predef_ml(int16_t* c, int& maxSym, int& tlog) {
  static const int16_t d[53] = {1, 4, 3, 2, 2, 2, 2, 2, 2, 1, 1, 1, 1, 1,
                                1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1,
                                1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1,
                                1, 1, 1, 1, -1, -1, -1, -1, -1, -1, -1};
  for (int i = 0; i < 53; i++) c[i] = d[i];
  maxSym = 52;
  tlog = 6;
}
LSZ_HD inline void predef_of(int16_t* c, int& maxSym, int& tlog) {
  static const int16_t d[29] = {1, 1, 1, 1, 1, 1, 2, 2, 2, 1, 1, 1, 1, 1, 1,
                                1, 1, 1, 1, 1, 1, 1, 1, 1, -1, -1, -1, -1, -1};
  for (int i = 0; i < 29; i++) c[i] = d[i];
  maxSym = 28;
  tlog = 5;
}

// ---------------------------------------------------------------------- //
// decoder context (persists across blocks within a frame)
// ---------------------------------------------------------------------- //

struct Ctx {
  FseTable ll, of, ml;
  bool haveLl = false, haveOf = false, haveMl = false;
  HufTable huf;
  bool haveHuf = false;
  uint32_t rep[3] = {1, 4, 8};
  uint8_t litBuf[1 << 17];   // ≤128 KB literals per block
};

// decode literals section; returns bytes consumed from src or -1.
// litLen receives the regenerated length (data in ctx.litBuf).
LSZ_HD inline int64_t decode_literals(Ctx& ctx, const uint8_t* src, int64_t n,
                                      int64_t& litLen) {
  if (n < 1) return -1;
  int type = src[0] & 3;
  int sizeFormat = (src[0] >> 2) & 3;
  if (type == 0 || type == 1) {  // Raw / RLE
    int64_t rs;
    int64_t hdr;
    if ((sizeFormat & 1) == 0) {           // 00 or 10: 5-bit size
      rs = src[0] >> 3;
      hdr = 1;
    } else if (sizeFormat == 1) {          // 01: 12-bit
      if (n < 2) return -1;
      rs = (src[0] >> 4) | ((int64_t)src[1] << 4);
      hdr = 2;
    } else {                               // 11: 20-bit
      if (n < 3) return -1;
      rs = (src[0] >> 4) | ((int64_t)src[1] << 4) | ((int64_t)src[2] << 12);
      hdr = 3;
    }
    if (rs > (int64_t)sizeof(ctx.litBuf)) return -1;
    litLen = rs;
    if (type == 0) {
      if (hdr + rs > n) return -1;
      memcpy(ctx.litBuf, src + hdr, (size_t)rs);
      return hdr + rs;
    }
    if (hdr + 1 > n) return -1;
    memset(ctx.litBuf, src[hdr], (size_t)rs);
    return hdr + 1;
  }
  // Compressed (2) / Treeless (3)
  int64_t rs, cs, hdr;
  int nStreams;
  if (sizeFormat == 0) {  // 1 stream, 10-bit sizes
    if (n < 3) return -1;
    rs = (src[0] >> 4) | ((int64_t)(src[1] & 0x3F) << 4);
    cs = (src[1] >> 6) | ((int64_t)src[2] << 2);
    hdr = 3;
    nStreams = 1;
  } else if (sizeFormat == 1) {  // 4 streams, 10-bit
    if (n < 3) return -1;
    rs = (src[0] >> 4) | ((int64_t)(src[1] & 0x3F) << 4);
    cs = (src[1] >> 6) | ((int64_t)src[2] << 2);
    hdr = 3;
    nStreams = 4;
  } else if (sizeFormat == 2) {  // 4 streams, 14-bit
    if (n < 4) return -1;
    rs = (src[0] >> 4) | ((int64_t)src[1] << 4) | ((int64_t)(src[2] & 3) << 12);
    cs = (src[2] >> 2) | ((int64_t)src[3] << 6);
    hdr = 4;
    nStreams = 4;
  } else {  // 4 streams, 18-bit
    if (n < 5) return -1;
    rs = (src[0] >> 4) | ((int64_t)src[1] << 4) | ((int64_t)(src[2] & 0x3F) << 12);
    cs = (src[2] >> 6) | ((int64_t)src[3] << 2) | ((int64_t)src[4] << 10);
    hdr = 5;
    nStreams = 4;
  }
  if (rs > (int64_t)sizeof(ctx.litBuf) || hdr + cs > n) return -1;
  litLen = rs;
  const uint8_t* p = src + hdr;
  int64_t rem = cs;

  if (type == 2) {  // huffman description present
    if (rem < 1) return -1;
    uint8_t hb = p[0];
    uint8_t weights[256];
    int nw;
    if (hb >= 128) {  // direct 4-bit weights
      nw = hb - 127;
      int64_t wb = (nw + 1) / 2;
      if (1 + wb > rem) return -1;
      for (int i = 0; i < nw; i++) {
        uint8_t b = p[1 + i / 2];
        weights[i] = (i & 1) ? (b & 0xF) : (b >> 4);
      }
      p += 1 + wb;
      rem -= 1 + wb;
    } else {  // FSE-compressed weights
      if (1 + hb > rem) return -1;
      nw = fse_decompress(p + 1, hb, weights, 255);
      if (nw < 0) return -1;
      p += 1 + hb;
      rem -= 1 + hb;
    }
    if (!huf_build(ctx.huf, weights, nw)) return -1;
    ctx.haveHuf = true;
  } else if (!ctx.haveHuf) {
    return -1;  // treeless without a previous table
  }

  if (nStreams == 1) {
    if (!huf_stream(ctx.huf, p, rem, ctx.litBuf, rs)) return -1;
  } else {
    if (rem < 6) return -1;
    int64_t s1 = p[0] | ((int64_t)p[1] << 8);
    int64_t s2 = p[2] | ((int64_t)p[3] << 8);
    int64_t s3 = p[4] | ((int64_t)p[5] << 8);
    int64_t s4 = rem - 6 - s1 - s2 - s3;
    if (s4 < 0) return -1;
    int64_t o123 = (rs + 3) / 4;
    int64_t o4 = rs - 3 * o123;
    if (o4 < 0) return -1;
    const uint8_t* q = p + 6;
    if (!huf_stream(ctx.huf, q, s1, ctx.litBuf, o123)) return -1;
    if (!huf_stream(ctx.huf, q + s1, s2, ctx.litBuf + o123, o123)) return -1;
    if (!huf_stream(ctx.huf, q + s1 + s2, s3, ctx.litBuf + 2 * o123, o123))
      return -1;
    if (!huf_stream(ctx.huf, q + s1 + s2 + s3, s4, ctx.litBuf + 3 * o123, o4))
      return -1;
  }
  return hdr + cs;
}

// read one sequence table per its 2-bit mode. Returns bytes consumed, -1 err.
LSZ_HD inline int64_t read_seq_table(Ctx& ctx, FseTable& t, bool& have,
                                     int mode, const uint8_t* src, int64_t n,
                                     void (*predef)(int16_t*, int&, int&),
                                     int maxSymLimit) {
  if (mode == 0) {  // predefined
    int16_t c[64];
    int maxSym, tlog;
    predef(c, maxSym, tlog);
    if (!fse_build(t, c, maxSym, tlog)) return -1;
    have = true;
    return 0;
  }
  if (mode == 1) {  // RLE: 1 byte symbol
    if (n < 1) return -1;
    if (src[0] > maxSymLimit) return -1;
    fse_build_rle(t, src[0]);
    have = true;
    return 1;
  }
  if (mode == 2) {  // FSE table description
    BitFwd br{src, n};
    int16_t c[64];
    int tlog;
    int maxSym = fse_read_ncount(br, c, maxSymLimit, tlog);
    if (maxSym < 0) return -1;
    if (!fse_build(t, c, maxSym, tlog)) return -1;
    have = true;
    return br.bytes_consumed();
  }
  // repeat
  return have ? 0 : -1;
}

// decode one compressed block into dst (history = dst window start..pos)
LSZ_HD inline int64_t decode_block(Ctx& ctx, const uint8_t* src, int64_t n,
                                   uint8_t* dstBase, int64_t pos,
                                   int64_t dstCap) {
  int64_t litLen;
  int64_t c = decode_literals(ctx, src, n, litLen);
  if (c < 0) return -1;
  const uint8_t* p = src + c;
  int64_t rem = n - c;

  // sequences header
  if (rem < 1) return -1;
  int64_t nSeq;
  if (p[0] < 128) {
    nSeq = p[0];
    p += 1; rem -= 1;
  } else if (p[0] < 255) {
    if (rem < 2) return -1;
    nSeq = ((int64_t)(p[0] - 128) << 8) + p[1];
    p += 2; rem -= 2;
  } else {
    if (rem < 3) return -1;
    nSeq = p[1] + ((int64_t)p[2] << 8) + 0x7F00;
    p += 3; rem -= 3;
  }

  if (nSeq == 0) {
    if (pos + litLen > dstCap) return -1;
    memcpy(dstBase + pos, ctx.litBuf, (size_t)litLen);
    return litLen;
  }

  if (rem < 1) return -1;
  int modes = p[0];
  p += 1; rem -= 1;
  int llMode = (modes >> 6) & 3, ofMode = (modes >> 4) & 3,
      mlMode = (modes >> 2) & 3;

  int64_t used;
  used = read_seq_table(ctx, ctx.ll, ctx.haveLl, llMode, p, rem, predef_ll, 35);
  if (used < 0) return -1;
  p += used; rem -= used;
  used = read_seq_table(ctx, ctx.of, ctx.haveOf, ofMode, p, rem, predef_of, 31);
  if (used < 0) return -1;
  p += used; rem -= used;
  used = read_seq_table(ctx, ctx.ml, ctx.haveMl, mlMode, p, rem, predef_ml, 52);
  if (used < 0) return -1;
  p += used; rem -= used;

  // execute sequences from the backward bitstream
  BitBwd br;
  if (!br.init(p, rem)) return -1;
  FseState sLl, sOf, sMl;
  sLl.init(ctx.ll, br);
  sOf.init(ctx.of, br);
  sMl.init(ctx.ml, br);

  int64_t litPos = 0;
  int64_t out = pos;
  for (int64_t s = 0; s < nSeq; s++) {
    int ofCode = sOf.symbol(ctx.of);
    int mlCode = sMl.symbol(ctx.ml);
    int llCode = sLl.symbol(ctx.ll);

    uint32_t ofValue = (ofCode ? (1u << ofCode) : 1u) + br.read(ofCode);
    CodeExtra mle = ml_extra(mlCode);
    uint32_t matchLen = mle.base + br.read(mle.bits);
    CodeExtra lle = ll_extra(llCode);
    uint32_t litLenSeq = lle.base + br.read(lle.bits);

    // repeat-offset resolution
    uint32_t offset;
    if (ofValue > 3) {
      offset = ofValue - 3;
      ctx.rep[2] = ctx.rep[1];
      ctx.rep[1] = ctx.rep[0];
      ctx.rep[0] = offset;
    } else {
      uint32_t idx = ofValue - 1 + (litLenSeq == 0 ? 1 : 0);
      if (idx == 0) {
        offset = ctx.rep[0];
      } else {
        uint32_t tmp = (idx == 3) ? ctx.rep[0] - 1 : ctx.rep[idx];
        if (tmp == 0) tmp = 1;  // corner case per libzstd
        if (idx != 1) ctx.rep[2] = ctx.rep[1];
        ctx.rep[1] = ctx.rep[0];
        ctx.rep[0] = tmp;
        offset = tmp;
      }
    }

    // copy literals
    if (litPos + litLenSeq > litLen || out + litLenSeq > dstCap) return -1;
    memcpy(dstBase + out, ctx.litBuf + litPos, litLenSeq);
    litPos += litLenSeq;
    out += litLenSeq;
    // copy match (may overlap)
    if ((int64_t)offset > out || out + matchLen > dstCap) return -1;
    {
      const uint8_t* from = dstBase + out - offset;
      uint8_t* to = dstBase + out;
      if (offset >= matchLen) {
        memcpy(to, from, matchLen);
      } else {
        for (uint32_t i = 0; i < matchLen; i++) to[i] = from[i];
      }
      out += matchLen;
    }

    if (s + 1 < nSeq) {  // last sequence: no state update
      sLl.update(ctx.ll, br);
      sMl.update(ctx.ml, br);
      sOf.update(ctx.of, br);
      if (br.overflow) return -1;
    }
  }
  // trailing literals
  int64_t tail = litLen - litPos;
  if (tail < 0 || out + tail > dstCap) return -1;
  memcpy(dstBase + out, ctx.litBuf + litPos, (size_t)tail);
  out += tail;
  return out - pos;
}

// decode a full zstd frame sequence. Returns decompressed size or -1.
LSZ_HD inline int64_t decode(const uint8_t* src, int64_t n, uint8_t* dst,
                             int64_t dstCap, Ctx* ctx) {
  int64_t pos = 0;        // output position
  int64_t ip = 0;
  while (ip + 4 <= n) {
    uint32_t magic;
    memcpy(&magic, src + ip, 4);
    if ((magic & 0xFFFFFFF0u) == 0x184D2A50u) {  // skippable frame
      if (ip + 8 > n) return -1;
      uint32_t sz;
      memcpy(&sz, src + ip + 4, 4);
      ip += 8 + sz;
      continue;
    }
    if (magic != kMagic) return -1;
    ip += 4;
    if (ip >= n) return -1;
    uint8_t fhd = src[ip++];
    int fcsFlag = fhd >> 6;
    bool singleSeg = (fhd >> 5) & 1;
    bool checksum = (fhd >> 2) & 1;
    int dictFlag = fhd & 3;
    if (!singleSeg) {
      if (ip >= n) return -1;
      ip++;  // window descriptor (we size by dstCap)
    }
    static const int dictLen[4] = {0, 1, 2, 4};
    ip += dictLen[dictFlag];
    int fcsLen = 0;
    if (fcsFlag == 0) fcsLen = singleSeg ? 1 : 0;
    else if (fcsFlag == 1) fcsLen = 2;
    else if (fcsFlag == 2) fcsLen = 4;
    else fcsLen = 8;
    ip += fcsLen;  // content size informative only
    if (ip > n) return -1;

    // reset inter-block context per frame
    ctx->haveLl = ctx->haveOf = ctx->haveMl = false;
    ctx->haveHuf = false;
    ctx->rep[0] = 1; ctx->rep[1] = 4; ctx->rep[2] = 8;

    bool last = false;
    while (!last) {
      if (ip + 3 > n) return -1;
      uint32_t bh = src[ip] | ((uint32_t)src[ip + 1] << 8) |
                    ((uint32_t)src[ip + 2] << 16);
      ip += 3;
      last = bh & 1;
      int btype = (bh >> 1) & 3;
      int64_t bsize = bh >> 3;
      if (btype == 0) {  // raw
        if (ip + bsize > n || pos + bsize > dstCap) return -1;
        memcpy(dst + pos, src + ip, (size_t)bsize);
        ip += bsize;
        pos += bsize;
      } else if (btype == 1) {  // RLE
        if (ip + 1 > n || pos + bsize > dstCap) return -1;
        memset(dst + pos, src[ip], (size_t)bsize);
        ip += 1;
        pos += bsize;
      } else if (btype == 2) {
        if (ip + bsize > n) return -1;
        int64_t outb = decode_block(*ctx, src + ip, bsize, dst, pos, dstCap);
        if (outb < 0) return -1;
        ip += bsize;
        pos += outb;
      } else {
        return -1;
      }
    }
    if (checksum) ip += 4;
  }
  return pos;
}

}  // namespace lszstd
