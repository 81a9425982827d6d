// GPU zstd page decompressor (gfx950): one wave64 workgroup per parquet
// page, grid-stride over pages. Decode logic is shared with the host
// reference implementation (csrc/cpp/zstd_dec.h, differential-tested
// against libzstd); this file adds the CDNA4 execution strategy:
//
// - every lane runs the *sequential* control flow redundantly in
//   lockstep (same bytes, same results — no broadcasts, no divergence);
// - byte copies (raw blocks, literals, matches) fan out across all 64
//   lanes; overlapping matches use period replication
//   (dst[out+j] = dst[out-offset + j%offset], reads strictly below out);
// - FSE/huffman decode tables live in LDS (~10.5 KB — wave-coherent, no
//   fences); the 128 KB literals buffer is per-workgroup global scratch;
// - 4-stream huffman literals decode on 4 lanes concurrently;
// - cross-lane visibility of cooperative copies uses the proven snappy
//   kernel pattern: volatile (GLC) reads of wave-written regions +
//   s_waitcnt(0) after each cooperative segment.
//
// Throughput comes from pages in flight (thousands per scan unit), not
// from single-page speed — decompression scales with the GPU instead of
// the cgroup-capped host CPU quota.

#include <hip/hip_runtime.h>

#include "../cpp/zstd_dec.h"

namespace lsz_gpu {

using namespace lszstd;

#define LANES 64
static const int kLitBufCap = 1 << 17;  // 128 KB literals per block

__device__ inline void waitcnt0() { __builtin_amdgcn_s_waitcnt(0); }

// order LDS ops only (lgkmcnt(0); vmcnt=63, expcnt=7 untouched): the
// window read/write shuffle must not drain in-flight global stores —
// a full __syncthreads here costs ~1000 cycles per sequence (measured,
// profiles/r01_gpu_zstd.md)
__device__ inline void lds_sync() { __builtin_amdgcn_s_waitcnt(0xC07F); }

// unaligned-capable wide element copy: 8 bytes per lane per iteration
// (gfx950 flat loads support byte-aligned addresses), byte tail
template <bool VOL>
__device__ inline void wcopy_wide(uint8_t* dst, const uint8_t* src,
                                  int64_t len, int lane) {
  int64_t n8 = len >> 3;
  for (int64_t j = lane; j < n8; j += LANES) {
    uint64_t w;
    if (VOL)
      w = *(const volatile uint64_t*)(src + 8 * j);
    else
      __builtin_memcpy(&w, src + 8 * j, 8);
    *(uint64_t*)(dst + 8 * j) = w;
  }
  for (int64_t j = (n8 << 3) + lane; j < len; j += LANES)
    dst[j] = VOL ? ((volatile const uint8_t*)src)[j] : src[j];
}

// cooperative copy from the literals scratch (written by this wave
// earlier — volatile read to bypass stale L1)
__device__ inline void wcopy_from_lit(uint8_t* dst, const uint8_t* lit,
                                      uint32_t len, int lane) {
  wcopy_wide<true>(dst, lit, len, lane);
  waitcnt0();
}

// cooperative (possibly overlapping) match copy inside dst
__device__ inline void wcopy_match(uint8_t* base, int64_t out, uint32_t offset,
                                   uint32_t len, int lane) {
  for (uint32_t j = lane; j < len; j += LANES) {
    uint32_t sj = (offset >= len) ? j : (j % offset);
    base[out + j] = ((volatile const uint8_t*)base)[out - offset + sj];
  }
  waitcnt0();
}

// cooperative copy from the (read-only) compressed input
__device__ inline void wcopy_src(uint8_t* dst, const uint8_t* src,
                                 int64_t len, int lane) {
  wcopy_wide<false>(dst, src, len, lane);
  waitcnt0();
}

__device__ inline void wfill(uint8_t* dst, uint8_t v, int64_t len, int lane) {
  uint64_t w = 0x0101010101010101ull * v;
  int64_t n8 = len >> 3;
  for (int64_t j = lane; j < n8; j += LANES) *(uint64_t*)(dst + 8 * j) = w;
  for (int64_t j = (n8 << 3) + lane; j < len; j += LANES) dst[j] = v;
  waitcnt0();
}

struct LdsCtx {
  FseTable ll, of, ml;
  HufTable huf;
  int haveLl, haveOf, haveMl, haveHuf;
  int64_t scratch_i64;     // lane0 -> wave value handoff
  int16_t counts[64];      // ncount output (seq tables)
  uint8_t weights[256];    // huffman weights
  int scratch_i32;
  // lane0 table-build workspaces (LDS, not per-lane scratch memory —
  // dynamically-indexed locals would otherwise force a multi-KB private
  // segment that craters wave residency)
  FseTable wt;             // huffman-weight FSE table
  int16_t wcounts[256];
  uint8_t symTab[1 << kMaxTableLog];
  uint16_t symNext[256];
  // rolling window of the last 32 output bytes (win[31] most recent).
  // Matches with offset <= 32 read ONLY this window: no global reads,
  // no fences — the serial cost per sequence collapses to LDS ops.
  uint8_t win[32];
};

// slide `src[0..n)` into the 32-byte window (volatile read: src may be
// wave-written global memory that has been fenced)
__device__ inline void win_append_from(LdsCtx& c, const uint8_t* src,
                                       int64_t n, int lane) {
  if (n <= 0) return;
  uint8_t nb = 0;
  if (lane < 32) {
    int64_t k = (int64_t)lane + n;
    nb = (k < 32) ? c.win[k] : ((volatile const uint8_t*)src)[k - 32];
  }
  __syncthreads();
  if (lane < 32) c.win[lane] = nb;
  __syncthreads();
}

__device__ inline void win_fill(LdsCtx& c, uint8_t v, int lane) {
  if (lane < 32) c.win[lane] = v;
  __syncthreads();
}

// decode the literals section. All lanes in lockstep; table builds by
// lane 0 into LDS; stream decode on up to 4 lanes. Returns bytes
// consumed from src or -1; litLen receives the regenerated length.
__device__ inline int64_t dev_literals(LdsCtx& c, uint8_t* lit,
                                       const uint8_t* src, int64_t n,
                                       int64_t& litLen, int lane) {
  if (n < 1) return -1;
  int type = src[0] & 3;
  int sizeFormat = (src[0] >> 2) & 3;
  if (type == 0 || type == 1) {  // Raw / RLE
    int64_t rs, hdr;
    if ((sizeFormat & 1) == 0) {
      rs = src[0] >> 3;
      hdr = 1;
    } else if (sizeFormat == 1) {
      if (n < 2) return -1;
      rs = (src[0] >> 4) | ((int64_t)src[1] << 4);
      hdr = 2;
    } else {
      if (n < 3) return -1;
      rs = (src[0] >> 4) | ((int64_t)src[1] << 4) | ((int64_t)src[2] << 12);
      hdr = 3;
    }
    if (rs > kLitBufCap) return -1;
    litLen = rs;
    if (type == 0) {
      if (hdr + rs > n) return -1;
      wcopy_src(lit, src + hdr, rs, lane);
      return hdr + rs;
    }
    if (hdr + 1 > n) return -1;
    wfill(lit, src[hdr], rs, lane);
    return hdr + 1;
  }
  // Compressed (2) / Treeless (3)
  int64_t rs, cs, hdr;
  int nStreams;
  if (sizeFormat == 0) {
    if (n < 3) return -1;
    rs = (src[0] >> 4) | ((int64_t)(src[1] & 0x3F) << 4);
    cs = (src[1] >> 6) | ((int64_t)src[2] << 2);
    hdr = 3;
    nStreams = 1;
  } else if (sizeFormat == 1) {
    if (n < 3) return -1;
    rs = (src[0] >> 4) | ((int64_t)(src[1] & 0x3F) << 4);
    cs = (src[1] >> 6) | ((int64_t)src[2] << 2);
    hdr = 3;
    nStreams = 4;
  } else if (sizeFormat == 2) {
    if (n < 4) return -1;
    rs = (src[0] >> 4) | ((int64_t)src[1] << 4) | ((int64_t)(src[2] & 3) << 12);
    cs = (src[2] >> 2) | ((int64_t)src[3] << 6);
    hdr = 4;
    nStreams = 4;
  } else {
    if (n < 5) return -1;
    rs = (src[0] >> 4) | ((int64_t)src[1] << 4) | ((int64_t)(src[2] & 0x3F) << 12);
    cs = (src[2] >> 6) | ((int64_t)src[3] << 2) | ((int64_t)src[4] << 10);
    hdr = 5;
    nStreams = 4;
  }
  if (rs > kLitBufCap || hdr + cs > n) return -1;
  litLen = rs;
  const uint8_t* p = src + hdr;
  int64_t rem = cs;

  if (type == 2) {
    if (rem < 1) return -1;
    uint8_t hb = p[0];
    // weights parse + table build on lane 0 only (keeps the 256-byte
    // weights array and FSE decode state out of per-lane scratch)
    if (lane == 0) {
      int nw = -1;
      int64_t consumed = -1;
      if (hb >= 128) {
        nw = hb - 127;
        int64_t wb = (nw + 1) / 2;
        if (1 + wb <= rem) {
          for (int i = 0; i < nw; i++) {
            uint8_t b = p[1 + i / 2];
            c.weights[i] = (i & 1) ? (b & 0xF) : (b >> 4);
          }
          consumed = 1 + wb;
        }
      } else if (1 + hb <= rem) {
        nw = fse_decompress_ws(p + 1, hb, c.weights, 255, c.wt, c.wcounts,
                               c.symTab, c.symNext);
        consumed = 1 + hb;
      }
      bool ok = consumed >= 0 && nw >= 0 && huf_build(c.huf, c.weights, nw);
      c.haveHuf = ok ? 1 : -1;
      c.scratch_i64 = consumed;
    }
    __syncthreads();
    if (c.haveHuf < 0) return -1;
    p += c.scratch_i64;
    rem -= c.scratch_i64;
    __syncthreads();
  } else if (c.haveHuf != 1) {
    return -1;
  }

  bool ok = true;
  if (nStreams == 1) {
    if (lane == 0) ok = huf_stream(c.huf, p, rem, lit, rs);
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
    if (lane < 4) {
      // one shared decode instance: lanes 0-3 each take a stream
      int64_t starts[4] = {0, s1, s1 + s2, s1 + s2 + s3};
      int64_t lens[4] = {s1, s2, s3, s4};
      ok = huf_stream(c.huf, q + starts[lane], lens[lane], lit + lane * o123,
                      lane == 3 ? o4 : o123);
    }
  }
  waitcnt0();
  __syncthreads();
  if (__ballot(!ok)) return -1;
  return hdr + cs;
}

__device__ inline int64_t dev_read_seq_table(
    LdsCtx& c, FseTable& t, int& have, int mode, const uint8_t* src, int64_t n,
    int which, int maxSymLimit, int lane) {  // which: 0=LL 1=OF 2=ML
  if (mode == 0) {
    if (lane == 0) {
      int maxSym, tlog;
      if (which == 0) predef_ll(c.counts, maxSym, tlog);
      else if (which == 1) predef_of(c.counts, maxSym, tlog);
      else predef_ml(c.counts, maxSym, tlog);
      have = fse_build(t, c.counts, maxSym, tlog, c.symTab, c.symNext) ? 1 : -1;
    }
    __syncthreads();
    return have < 0 ? -1 : 0;
  }
  if (mode == 1) {
    if (n < 1 || src[0] > maxSymLimit) return -1;
    if (lane == 0) {
      fse_build_rle(t, src[0]);
      have = 1;
    }
    __syncthreads();
    return 1;
  }
  if (mode == 2) {
    if (lane == 0) {
      BitFwd br{src, n};
      int tlog;
      int maxSym = fse_read_ncount(br, c.counts, maxSymLimit, tlog);
      bool ok = maxSym >= 0 &&
                fse_build(t, c.counts, maxSym, tlog, c.symTab, c.symNext);
      have = ok ? 1 : -1;
      c.scratch_i64 = br.bytes_consumed();
    }
    __syncthreads();
    return have < 0 ? -1 : c.scratch_i64;
  }
  return have == 1 ? 0 : -1;
}

// one compressed block; every lane in lockstep
__device__ inline int64_t dev_block(LdsCtx& c, uint8_t* lit, uint32_t* rep,
                                    const uint8_t* src, int64_t n,
                                    uint8_t* dstBase, int64_t pos,
                                    int64_t dstCap, int lane) {
  int64_t litLen;
  int64_t consumed = dev_literals(c, lit, src, n, litLen, lane);
  if (consumed < 0) return -1;
  const uint8_t* p = src + consumed;
  int64_t rem = n - consumed;

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
    wcopy_from_lit(dstBase + pos, lit, (uint32_t)litLen, lane);
    win_append_from(c, lit, litLen, lane);
    return litLen;
  }

  if (rem < 1) return -1;
  int modes = p[0];
  p += 1; rem -= 1;
  int llMode = (modes >> 6) & 3, ofMode = (modes >> 4) & 3,
      mlMode = (modes >> 2) & 3;

  int64_t used;
  used = dev_read_seq_table(c, c.ll, c.haveLl, llMode, p, rem, 0, 35, lane);
  if (used < 0) return -1;
  p += used; rem -= used;
  used = dev_read_seq_table(c, c.of, c.haveOf, ofMode, p, rem, 1, 31, lane);
  if (used < 0) return -1;
  p += used; rem -= used;
  used = dev_read_seq_table(c, c.ml, c.haveMl, mlMode, p, rem, 2, 52, lane);
  if (used < 0) return -1;
  p += used; rem -= used;

  BitBwd br;
  if (!br.init(p, rem)) return -1;
  FseState sLl, sOf, sMl;
  sLl.init(c.ll, br);
  sOf.init(c.of, br);
  sMl.init(c.ml, br);

  // W = everything below this output position is store-fence'd (visible
  // to plain/GLC reads); advance lazily, fencing only when a large-offset
  // match must read not-yet-fenced bytes
  int64_t W = pos;
  int64_t litPos = 0;
  int64_t out = pos;
  for (int64_t s = 0; s < nSeq; s++) {
    int ofCode = sOf.symbol(c.of);
    int mlCode = sMl.symbol(c.ml);
    int llCode = sLl.symbol(c.ll);

    uint32_t ofValue = (ofCode ? (1u << ofCode) : 1u) + br.read(ofCode);
    CodeExtra mle = ml_extra(mlCode);
    uint32_t matchLen = mle.base + br.read(mle.bits);
    CodeExtra lle = ll_extra(llCode);
    uint32_t litLenSeq = lle.base + br.read(lle.bits);

    uint32_t offset;
    if (ofValue > 3) {
      offset = ofValue - 3;
      rep[2] = rep[1];
      rep[1] = rep[0];
      rep[0] = offset;
    } else {
      uint32_t idx = ofValue - 1 + (litLenSeq == 0 ? 1 : 0);
      if (idx == 0) {
        offset = rep[0];
      } else {
        uint32_t tmp = (idx == 3) ? rep[0] - 1 : rep[idx];
        if (tmp == 0) tmp = 1;
        if (idx != 1) rep[2] = rep[1];
        rep[1] = rep[0];
        rep[0] = tmp;
        offset = tmp;
      }
    }

    // ---- literals: litBuf reads are hazard-free (fenced after the
    // literals phase); dst writes are fire-and-forget ----
    if (litPos + litLenSeq > litLen || out + litLenSeq > dstCap) return -1;
    if (litLenSeq) {
      wcopy_wide<true>(dstBase + out, lit + litPos, litLenSeq, lane);
      // window <- last 32 bytes of (window ++ literals)
      uint8_t nb = 0;
      if (lane < 32) {
        int64_t k = (int64_t)lane + litLenSeq;   // index into win ++ lits
        nb = (k < 32) ? c.win[k]
                      : ((volatile const uint8_t*)lit)[litPos + (k - 32)];
      }
      lds_sync();
      if (lane < 32) c.win[lane] = nb;
      lds_sync();
      litPos += litLenSeq;
      out += litLenSeq;
    }

    // ---- match ----
    if ((int64_t)offset > out || out + matchLen > dstCap) return -1;
    if (offset <= 32) {
      // source is entirely inside the register window: no global reads
      for (uint32_t j = lane; j < matchLen; j += LANES)
        dstBase[out + j] = c.win[(32 - offset) + (j % offset)];
      uint8_t nb = 0;
      if (lane < 32) {
        int64_t k = (int64_t)lane + matchLen;
        nb = (k < 32) ? c.win[k]
                      : c.win[(32 - offset) + (uint32_t)((k - 32) % offset)];
      }
      lds_sync();
      if (lane < 32) c.win[lane] = nb;
      lds_sync();
    } else {
      // large offset: source bytes live below `out`; fence once if they
      // reach into the unfenced region
      uint32_t span = offset < matchLen ? offset : matchLen;
      if (out - (int64_t)offset + (int64_t)span > W) {
        waitcnt0();
        W = out;
      }
      for (uint32_t j = lane; j < matchLen; j += LANES)
        dstBase[out + j] =
            ((volatile const uint8_t*)dstBase)[out - offset + (j % offset)];
      uint8_t nb = 0;
      if (lane < 32) {
        int64_t k = (int64_t)lane + matchLen;
        nb = (k < 32)
                 ? c.win[k]
                 : ((volatile const uint8_t*)
                        dstBase)[out - offset + (uint32_t)((k - 32) % offset)];
      }
      lds_sync();
      if (lane < 32) c.win[lane] = nb;
      lds_sync();
    }
    out += matchLen;

    if (s + 1 < nSeq) {
      sLl.update(c.ll, br);
      sMl.update(c.ml, br);
      sOf.update(c.of, br);
      if (br.overflow) return -1;
    }
  }
  int64_t tail = litLen - litPos;
  if (tail < 0 || out + tail > dstCap) return -1;
  wcopy_from_lit(dstBase + out, lit + litPos, (uint32_t)tail, lane);
  win_append_from(c, lit + litPos, tail, lane);
  out += tail;
  return out - pos;
}

__device__ inline int64_t dev_frame(LdsCtx& c, uint8_t* lit, const uint8_t* src,
                                    int64_t n, uint8_t* dst, int64_t dstCap,
                                    int lane) {
  int64_t pos = 0, ip = 0;
  uint32_t rep[3];
  while (ip + 4 <= n) {
    uint32_t magic;
    memcpy(&magic, src + ip, 4);
    if ((magic & 0xFFFFFFF0u) == 0x184D2A50u) {
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
      ip++;
    }
    const int dictLen[4] = {0, 1, 2, 4};
    ip += dictLen[dictFlag];
    int fcsLen = (fcsFlag == 0) ? (singleSeg ? 1 : 0)
                                : (fcsFlag == 1 ? 2 : (fcsFlag == 2 ? 4 : 8));
    ip += fcsLen;
    if (ip > n) return -1;

    if (lane == 0) { c.haveLl = c.haveOf = c.haveMl = 0; c.haveHuf = 0; }
    if (lane < 32) c.win[lane] = 0;
    __syncthreads();
    rep[0] = 1; rep[1] = 4; rep[2] = 8;

    bool last = false;
    while (!last) {
      if (ip + 3 > n) return -1;
      uint32_t bh = src[ip] | ((uint32_t)src[ip + 1] << 8) |
                    ((uint32_t)src[ip + 2] << 16);
      ip += 3;
      last = bh & 1;
      int btype = (bh >> 1) & 3;
      int64_t bsize = bh >> 3;
      if (btype == 0) {
        if (ip + bsize > n || pos + bsize > dstCap) return -1;
        wcopy_src(dst + pos, src + ip, bsize, lane);
        win_append_from(c, src + ip, bsize, lane);
        ip += bsize;
        pos += bsize;
      } else if (btype == 1) {
        if (ip + 1 > n || pos + bsize > dstCap) return -1;
        wfill(dst + pos, src[ip], bsize, lane);
        if (bsize >= 32) {
          win_fill(c, src[ip], lane);
        } else {
          uint8_t v = src[ip];
          uint8_t nb = 0;
          if (lane < 32) {
            int64_t k = (int64_t)lane + bsize;
            nb = (k < 32) ? c.win[k] : v;
          }
          __syncthreads();
          if (lane < 32) c.win[lane] = nb;
          __syncthreads();
        }
        ip += 1;
        pos += bsize;
      } else if (btype == 2) {
        if (ip + bsize > n) return -1;
        int64_t outb = dev_block(c, lit, rep, src + ip, bsize, dst, pos,
                                 dstCap, lane);
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

__global__ void __launch_bounds__(LANES, 4) zstd_pages_kernel(
    const uint8_t* __restrict__ src, const int64_t* __restrict__ jobs,
    int64_t njobs, uint8_t* __restrict__ dst, uint8_t* __restrict__ scratch,
    int32_t* __restrict__ status) {
  __shared__ LdsCtx c;
  int lane = threadIdx.x;
  uint8_t* lit = scratch + (int64_t)blockIdx.x * kLitBufCap;
  for (int64_t p = blockIdx.x; p < njobs; p += gridDim.x) {
    const int64_t* j = jobs + 4 * p;
    int64_t r = dev_frame(c, lit, src + j[0], j[1], dst + j[2], j[3], lane);
    if (lane == 0) status[p] = (r == j[3]) ? 0 : 1;
    __syncthreads();
  }
}

}  // namespace lsz_gpu

namespace lakesoul {

void launch_zstd_decompress(const uint8_t* src, const int64_t* jobs,
                            int64_t njobs, uint8_t* dst, uint8_t* scratch,
                            int64_t nblocks, int32_t* status,
                            hipStream_t stream) {
  if (njobs == 0) return;
  dim3 grid((uint32_t)nblocks), block(LANES);
  hipLaunchKernelGGL(lsz_gpu::zstd_pages_kernel, grid, block, 0, stream, src,
                     jobs, njobs, dst, scratch, status);
}

int64_t lsz_gpu_litbuf_bytes() { return lsz_gpu::kLitBufCap; }

}  // namespace lakesoul
