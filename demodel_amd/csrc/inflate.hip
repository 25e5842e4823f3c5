// CDNA4 DEFLATE (RFC 1951) inflate kernel, v2.
//
// Design (lessons measured on the zstd kernel, profiles/zstd_words_pmc):
//
// * wave-per-stream, but ALL 64 lanes run the bit-serial decode
//   REDUNDANTLY in lockstep on register state — no cross-lane
//   publication, no barriers anywhere in the symbol loop.  LDS table
//   reads broadcast; input reads coalesce to one address.
// * a 32 KiB LDS output window — exactly DEFLATE's maximum match
//   distance, so EVERY back-reference is an LDS read (wave-internal
//   lgkm ordering); output bytes stream to HBM fire-and-forget and are
//   never read back.
// * 10-bit primary Huffman LUTs (litlen + dist) with a canonical
//   bit-by-bit fallback for longer codes; 4-byte batched bit refills.
// * launch_bounds grants the register budget (LDS caps occupancy).
//
// Aggregate throughput comes from many streams (gzip members, pages,
// bodies); within a stream DEFLATE remains serially entropy-coded
// (SURVEY.md §7 hard part (a)).

#include <hip/hip_runtime.h>

namespace {

enum {
  INF_OK = 0,
  INF_ERR_FORMAT = -2,
  INF_ERR_OVERFLOW = -3,
  INF_ERR_UNDERRUN = -4,
};

struct __align__(16) InflateDesc {
  uint64_t src;
  uint64_t src_len;
  uint64_t dst;
  uint64_t dst_cap;
  uint64_t written;   // out
  int64_t status;     // out
  uint64_t consumed;  // out
  uint64_t _pad1;
};

struct BitReader {
  const uint8_t* p;
  uint64_t len;
  uint64_t pos;       // next byte to load into the buffer
  uint64_t buf;       // bit buffer (LSB-first)
  int nbits;

  __device__ void init(const uint8_t* s, uint64_t n) {
    p = s; len = n; pos = 0; buf = 0; nbits = 0;
  }
  __device__ void fill() {
    while (nbits <= 32 && pos + 4 <= len) {
      uint64_t v = (uint64_t)p[pos] | ((uint64_t)p[pos + 1] << 8) |
                   ((uint64_t)p[pos + 2] << 16) |
                   ((uint64_t)p[pos + 3] << 24);
      buf |= v << nbits;
      nbits += 32;
      pos += 4;
    }
    while (nbits <= 56) {
      uint64_t byte = pos < len ? p[pos] : 0;  // zero-pad past EOF
      buf |= byte << nbits;
      nbits += 8;
      ++pos;
    }
  }
  __device__ uint32_t peek(int n) {
    if (nbits < n) fill();
    return (uint32_t)(buf & ((1ull << n) - 1));
  }
  __device__ void drop(int n) {
    buf >>= n;
    nbits -= n;
  }
  __device__ uint32_t bits(int n) {
    if (nbits < n) fill();
    uint32_t v = (uint32_t)(buf & ((1ull << n) - 1));
    buf >>= n;
    nbits -= n;
    return v;
  }
  __device__ uint32_t bit1() { return bits(1); }
  __device__ void align_byte() {
    int d = nbits & 7;
    buf >>= d;
    nbits -= d;
  }
  __device__ uint64_t byte_pos() const {
    return pos - (uint64_t)(nbits >> 3);
  }
  __device__ bool overran() const { return byte_pos() > len; }
};

// canonical fallback tables (also the source for LUT construction)
struct HuffTable {
  uint16_t count[16];
  uint16_t offset[16];
  uint32_t first[16];
  uint16_t sym[288];
};

// all-lane redundant build: every lane computes/writes the same values
// (same-address LDS stores are benign)
__device__ bool huff_build(HuffTable* t, const uint8_t* lens, int n) {
  for (int i = 0; i < 16; ++i) t->count[i] = 0;
  for (int i = 0; i < n; ++i) t->count[lens[i]]++;
  t->count[0] = 0;
  uint32_t code = 0;
  int total = 0;
  int left = 1;
  for (int l = 1; l < 16; ++l) {
    code = (code + t->count[l - 1]) << 1;
    t->first[l] = code;
    t->offset[l] = total;
    total += t->count[l];
    left = (left << 1) - t->count[l];
    if (left < 0) return false;
  }
  uint16_t next[16];
  for (int l = 0; l < 16; ++l) next[l] = t->offset[l];
  for (int i = 0; i < n; ++i)
    if (lens[i]) t->sym[next[lens[i]]++] = (uint16_t)i;
  return total > 0;
}

__device__ int huff_decode_slow(BitReader* br, const HuffTable* t) {
  uint32_t code = 0;
  for (int l = 1; l < 16; ++l) {
    code |= br->bit1();
    if (t->count[l] && code - t->first[l] < (uint32_t)t->count[l])
      return t->sym[t->offset[l] + (code - t->first[l])];
    code <<= 1;
  }
  return -1;
}

#define LUT_BITS 10
#define LUT_SIZE (1 << LUT_BITS)

// Build a primary LUT over the next LUT_BITS raw (LSB-first) bits.
// entry = (sym << 5) | code_len; 0 => fall back to bit-by-bit.
// Lanes cooperate: symbol loop is redundant, replica fill is striped.
__device__ void lut_build(uint16_t* lut, const HuffTable* t,
                          const uint8_t* lens, int n, int lane) {
  for (int i = lane; i < LUT_SIZE; i += 64) lut[i] = 0;
  // codes per canonical order: walk lengths
  uint32_t code = 0;
  for (int l = 1; l <= LUT_BITS; ++l) {
    code = (code + t->count[l - 1]) << 1;
    for (int k = 0; k < t->count[l]; ++k) {
      uint32_t c = code + k;
      uint16_t sym = t->sym[t->offset[l] + k];
      // bit-reverse the l-bit code (stream is LSB-first)
      uint32_t rev = __brev(c) >> (32 - l);
      uint16_t entry = (uint16_t)((sym << 5) | l);
      int reps = 1 << (LUT_BITS - l);
      for (int r = lane; r < reps; r += 64)
        lut[rev | (r << l)] = entry;
    }
  }
  (void)lens;
  (void)n;
}

__device__ __forceinline__ int huff_decode_lut(BitReader* br,
                                               const uint16_t* lut,
                                               const HuffTable* t) {
  uint32_t v = br->peek(LUT_BITS);
  uint16_t e = lut[v];
  if (e) {
    br->drop(e & 31);
    return e >> 5;
  }
  return huff_decode_slow(br, t);  // code longer than LUT_BITS
}

__constant__ uint16_t kLenBase[29] = {
    3, 4, 5, 6, 7, 8, 9, 10, 11, 13, 15, 17, 19, 23, 27, 31, 35, 43, 51,
    59, 67, 83, 99, 115, 131, 163, 195, 227, 258};
__constant__ uint8_t kLenExtra[29] = {
    0, 0, 0, 0, 0, 0, 0, 0, 1, 1, 1, 1, 2, 2, 2, 2, 3, 3, 3, 3, 4, 4, 4,
    4, 5, 5, 5, 5, 0};
__constant__ uint32_t kDistBase[30] = {
    1, 2, 3, 4, 5, 7, 9, 13, 17, 25, 33, 49, 65, 97, 129, 193, 257, 385,
    513, 769, 1025, 1537, 2049, 3073, 4097, 6145, 8193, 12289, 16385,
    24577};
__constant__ uint8_t kDistExtra[30] = {
    0, 0, 0, 0, 1, 1, 2, 2, 3, 3, 4, 4, 5, 5, 6, 6, 7, 7, 8, 8, 9, 9, 10,
    10, 11, 11, 12, 12, 13, 13};
__constant__ uint8_t kClOrder[19] = {
    16, 17, 18, 0, 8, 7, 9, 6, 10, 5, 11, 4, 12, 3, 13, 2, 14, 1, 15};

#define DWIN 32768          // DEFLATE max distance == window size
#define DWMASK (DWIN - 1)

#define LGKM0() asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory")

__global__ void __launch_bounds__(64, 2)
inflate_kernel(InflateDesc* __restrict__ descs, int n_streams) {
  __shared__ struct {
    HuffTable litlen_t;
    HuffTable dist_t;
    uint16_t litlen_lut[LUT_SIZE];
    uint16_t dist_lut[LUT_SIZE];
    uint8_t lens_buf[320];
    uint8_t win[DWIN];
  } sh;
  int lane = threadIdx.x;

  for (int s = blockIdx.x; s < n_streams; s += gridDim.x) {
    InflateDesc* d = &descs[s];
    const uint8_t* src = (const uint8_t*)d->src;
    uint8_t* out = (uint8_t*)d->dst;

    // every lane holds identical decoder state in registers
    BitReader br;
    br.init(src, d->src_len);
    uint64_t pos = 0;
    int64_t status = INF_OK;
    bool done = false;

    while (!done) {
      if (br.overran()) { status = INF_ERR_UNDERRUN; break; }
      uint32_t bfinal = br.bit1();
      uint32_t btype = br.bits(2);
      if (btype == 0) {                                   // stored
        br.align_byte();
        uint32_t blen = br.bits(16);
        uint32_t nlen = br.bits(16);
        if ((blen ^ 0xFFFFu) != nlen) { status = INF_ERR_FORMAT; break; }
        uint64_t so = br.byte_pos();
        if (so + blen > br.len) { status = INF_ERR_UNDERRUN; break; }
        if (pos + blen > d->dst_cap) { status = INF_ERR_OVERFLOW; break; }
        for (uint32_t i = lane; i < blen; i += 64) {
          uint8_t v = src[so + i];
          out[pos + i] = v;
          sh.win[(pos + i) & DWMASK] = v;
        }
        LGKM0();
        pos += blen;
        br.buf = 0;
        br.nbits = 0;
        br.pos = so + blen;
        if (bfinal) done = true;
        continue;
      }
      if (btype == 3) { status = INF_ERR_FORMAT; break; }

      // ---- table setup (redundant build; striped LUT fill) ---------
      bool okt = true;
      if (btype == 1) {
        for (int i = 0; i < 288; ++i)
          sh.lens_buf[i] = i < 144 ? 8 : i < 256 ? 9 : i < 280 ? 7 : 8;
        okt = huff_build(&sh.litlen_t, sh.lens_buf, 288);
        for (int i = 0; i < 30; ++i) sh.lens_buf[i] = 5;
        okt = okt && huff_build(&sh.dist_t, sh.lens_buf, 30);
      } else {
        int hlit = (int)br.bits(5) + 257;
        int hdist = (int)br.bits(5) + 1;
        int hclen = (int)br.bits(4) + 4;
        uint8_t cl_lens[19];
        for (int i = 0; i < 19; ++i) cl_lens[i] = 0;
        for (int i = 0; i < hclen; ++i)
          cl_lens[kClOrder[i]] = (uint8_t)br.bits(3);
        if (!huff_build(&sh.dist_t, cl_lens, 19)) {  // dist as CL scratch
          status = INF_ERR_FORMAT;
          break;
        }
        int total = hlit + hdist;
        int i = 0;
        while (i < total) {
          int symc = huff_decode_slow(&br, &sh.dist_t);
          if (symc < 0 || br.overran()) { okt = false; break; }
          if (symc < 16) {
            sh.lens_buf[i++] = (uint8_t)symc;
          } else if (symc == 16) {
            if (i == 0) { okt = false; break; }
            int rep = 3 + (int)br.bits(2);
            uint8_t v = sh.lens_buf[i - 1];
            while (rep-- > 0 && i < total) sh.lens_buf[i++] = v;
          } else if (symc == 17) {
            int rep = 3 + (int)br.bits(3);
            while (rep-- > 0 && i < total) sh.lens_buf[i++] = 0;
          } else {
            int rep = 11 + (int)br.bits(7);
            while (rep-- > 0 && i < total) sh.lens_buf[i++] = 0;
          }
        }
        okt = okt && i == total &&
              huff_build(&sh.litlen_t, sh.lens_buf, hlit) &&
              huff_build(&sh.dist_t, sh.lens_buf + hlit, hdist);
      }
      if (!okt) { status = INF_ERR_FORMAT; break; }
      lut_build(sh.litlen_lut, &sh.litlen_t, sh.lens_buf, 288, lane);
      lut_build(sh.dist_lut, &sh.dist_t, sh.lens_buf, 30, lane);
      LGKM0();

      // ---- symbol loop (all lanes lockstep, zero barriers) ---------
      // Literal batching: the decode is redundant-lockstep, so every
      // lane sees every literal; lane (n & 63) parks literal n in a
      // register and 64 accumulate into ONE coalesced 64 B store
      // (round-1 wrote each literal via lane 0 with 63 lanes idle —
      // the first wall on literal-heavy payloads).  Pending bytes are
      // flushed before any match (matches read the window) and at
      // block end.
      uint64_t lit_base = 0;
      uint32_t lit_n = 0;
      uint8_t lit_pend = 0;
      auto flush_lits = [&]() {
        if (lit_n) {
          if (lane < (int)lit_n) {
            out[lit_base + lane] = lit_pend;
            sh.win[(lit_base + lane) & DWMASK] = lit_pend;
          }
          lit_n = 0;
        }
      };
      while (true) {
        int sym = huff_decode_lut(&br, sh.litlen_lut, &sh.litlen_t);
        if (sym < 0 || br.overran()) {
          status = br.overran() ? INF_ERR_UNDERRUN : INF_ERR_FORMAT;
          break;
        }
        if (sym < 256) {
          if (pos >= d->dst_cap) { status = INF_ERR_OVERFLOW; break; }
          if (lit_n == 0) lit_base = pos;
          if (lane == (int)lit_n) lit_pend = (uint8_t)sym;
          ++lit_n;
          ++pos;
          if (lit_n == 64) flush_lits();
          continue;
        }
        if (sym == 256) break;  // end of block
        sym -= 257;
        if (sym >= 29) { status = INF_ERR_FORMAT; break; }
        uint32_t mlen = kLenBase[sym] + br.bits(kLenExtra[sym]);
        int dsym = huff_decode_lut(&br, sh.dist_lut, &sh.dist_t);
        if (dsym < 0 || dsym >= 30) { status = INF_ERR_FORMAT; break; }
        uint32_t dist = kDistBase[dsym] + br.bits(kDistExtra[dsym]);
        if (dist > pos) { status = INF_ERR_FORMAT; break; }
        if (pos + mlen > d->dst_cap) { status = INF_ERR_OVERFLOW; break; }
        flush_lits();  // the match may read the just-written window
        // every source byte is within DWIN -> always LDS
        LGKM0();
        if (dist >= mlen) {
          for (uint32_t k = lane; k < mlen; k += 64) {
            uint8_t v = sh.win[(pos + k - dist) & DWMASK];
            out[pos + k] = v;
            sh.win[(pos + k) & DWMASK] = v;
          }
        } else {
          uint32_t copied = 0;
          while (copied < mlen) {
            uint32_t n = min(dist, mlen - copied);
            for (uint32_t k = lane; k < n; k += 64) {
              uint8_t v = sh.win[(pos + copied + k - dist) & DWMASK];
              out[pos + copied + k] = v;
              sh.win[(pos + copied + k) & DWMASK] = v;
            }
            LGKM0();
            copied += n;
          }
        }
        pos += mlen;
      }
      flush_lits();  // trailing literals of the block
      if (status != INF_OK) break;
      if (bfinal) done = true;
    }

    if (lane == 0) {
      d->written = pos;
      d->status = status;
      d->consumed = br.byte_pos();
    }
    __syncthreads();
  }
}

}  // namespace

extern "C" void launch_inflate_streams(const uint64_t* desc, int n_streams,
                                       int* /*status_unused*/,
                                       hipStream_t stream) {
  if (n_streams <= 0) return;
  int blocks = n_streams < 4096 ? n_streams : 4096;
  hipLaunchKernelGGL(inflate_kernel, dim3(blocks), dim3(64), 0, stream,
                     (InflateDesc*)desc, n_streams);
}
