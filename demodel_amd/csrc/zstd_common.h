// Shared zstd decode primitives: compiled into both the CDNA4 kernel
// (csrc/zstd_kernel.hip) and a host reference decoder (csrc/zstd_host.cpp)
// so the entropy/format layer is testable and debuggable on CPU against
// the real encoder (kernels vs plain CPU references).
#pragma once

#include <stdint.h>
#include <string.h>

#if defined(__HIPCC__)
#define ZFN __host__ __device__
#define ZCONST static const
#define ZALIGN(n) __align__(n)
#else
#define ZFN inline
#define ZCONST static const
#define ZALIGN(n) alignas(n)
#include <algorithm>
using std::min;
#endif

#if !defined(__HIPCC__)
#include <stdio.h>
#include <stdlib.h>
#define ZCD(...)                                        \
  do {                                                  \
    if (getenv("ZSTD_DEBUG")) fprintf(stderr, __VA_ARGS__); \
  } while (0)
#else
#define ZCD(...)
#endif

ZFN int zclz(uint32_t x) {
#if defined(__HIP_DEVICE_COMPILE__)
  return __clz(x);
#else
  return x == 0 ? 32 : __builtin_clz(x);
#endif
}

namespace zstd_core {


enum {
  Z_OK = 0,
  Z_ERR_MAGIC = -1,
  Z_ERR_FORMAT = -2,
  Z_ERR_OVERFLOW = -3,
  Z_ERR_UNDERRUN = -4,
  Z_ERR_DICT = -5,
};

struct ZALIGN(16) ZstdDesc {
  uint64_t src;
  uint64_t src_len;
  uint64_t dst;
  uint64_t dst_cap;
  uint64_t written;   // out
  int64_t status;     // out
  uint64_t consumed;  // out
  uint64_t ws;        // workspace (>= 144 KiB)
};

// ---------------- backward bit stream (huff0 / FSE) ---------------------

struct BBits {
  const uint8_t* base;  // start of stream
  int64_t byte;         // next byte index to pull (moving down)
  uint64_t cont;        // bit container
  int nbits;            // valid bits in container
  bool fail;

  ZFN void init(const uint8_t* p, int64_t len) {
    base = p;
    fail = false;
    cont = 0;
    nbits = 0;
    if (len <= 0) { fail = true; byte = -1; return; }
    byte = len - 1;
    uint8_t last = p[byte];
    if (last == 0) { fail = true; return; }
    // load last byte minus sentinel bit
    int high = 31 - zclz((uint32_t)last);
    cont = (uint64_t)(last & ((1u << high) - 1));
    nbits = high;
    --byte;
    refill();
  }
  ZFN void refill() {
    // batched reload: 4 bytes per step (backward stream = byteswapped
    // little-endian load) instead of one global byte per iteration
    while (nbits <= 32 && byte >= 3) {
      // four independent byte loads (explicit: unaligned wide loads
      // misbehave on device)
      uint32_t v = ((uint32_t)base[byte] << 24) |
                   ((uint32_t)base[byte - 1] << 16) |
                   ((uint32_t)base[byte - 2] << 8) |
                   (uint32_t)base[byte - 3];
      cont = (cont << 32) | v;
      nbits += 32;
      byte -= 4;
    }
    while (nbits <= 56 && byte >= 0) {
      cont = (cont << 8) | base[byte];
      nbits += 8;
      --byte;
    }
  }
  // read n bits (MSB-first of the remaining stream)
  ZFN uint32_t get(int n) {
    if (n == 0) return 0;
    if (nbits < n) {
      refill();
      if (nbits < n) {
        // zstd permits reading past the start only as zero-padding of
        // final state updates; treat as zeros but flag excessive use
        int deficit = n - nbits;
        uint32_t v = (uint32_t)(cont << deficit);
        v &= (n < 32) ? ((1u << n) - 1) : 0xFFFFFFFFu;
        nbits = 0;
        cont = 0;
        fail = true;
        return v;
      }
    }
    nbits -= n;
    uint32_t v = (uint32_t)((cont >> nbits) & ((n < 32)
                                                   ? ((1ull << n) - 1)
                                                   : 0xFFFFFFFFull));
    return v;
  }
  ZFN uint32_t peek(int n) {
    if (nbits < n) refill();
    if (nbits >= n) return (uint32_t)((cont >> (nbits - n))
                                      & ((1ull << n) - 1));
    int deficit = n - nbits;
    return (uint32_t)((cont << deficit) & ((1ull << n) - 1));
  }
  ZFN void skip(int n) {
    nbits -= n;
    if (nbits < 0) { nbits = 0; }
    if (nbits < 25) refill();
  }
  ZFN bool exhausted() const { return byte < 0 && nbits == 0; }
};

// ---------------- forward little-endian byte reader ---------------------

struct FReader {
  const uint8_t* p;
  uint64_t len;
  uint64_t pos;
  bool fail;

  ZFN void init(const uint8_t* s, uint64_t n) {
    p = s; len = n; pos = 0; fail = false;
  }
  ZFN uint8_t u8() {
    if (pos >= len) { fail = true; return 0; }
    return p[pos++];
  }
  ZFN uint32_t u16() { uint32_t a = u8(); return a | (u8() << 8); }
  ZFN uint32_t u24() { uint32_t a = u16(); return a | (u8() << 16); }
  ZFN uint32_t u32() { uint32_t a = u16(); return a | (u16() << 16); }
  ZFN uint64_t u64v() {
    uint64_t a = u32();
    return a | ((uint64_t)u32() << 32);
  }
  ZFN bool need(uint64_t n) const { return pos + n <= len; }
};

// ---------------- FSE decode table --------------------------------------

struct FseEntry {
  uint8_t sym;
  uint8_t nbits;
  uint16_t base;
};

struct FseTable {
  FseEntry e[512];
  int log;  // accuracy log (table size = 1 << log)
};

// Build a decoding table from normalized counts (-1 => "less than one").
ZFN bool fse_build(FseTable* t, const int16_t* norm, int n_sym,
                          int log) {
  int size = 1 << log;
  if (log > 9) return false;
  uint8_t syms[512];
  int high = size - 1;
  // low-probability symbols get the top slots
  for (int s = 0; s < n_sym; ++s)
    if (norm[s] == -1) syms[high--] = (uint8_t)s;
  int step = (size >> 1) + (size >> 3) + 3;
  int pos = 0;
  for (int s = 0; s < n_sym; ++s) {
    for (int i = 0; i < norm[s]; ++i) {
      syms[pos] = (uint8_t)s;
      pos = (pos + step) & (size - 1);
      while (pos > high) pos = (pos + step) & (size - 1);
    }
  }
  if (pos != 0) { ZCD("fse_build: spread pos=%d\n", pos); return false; }
  // per-symbol next-state counters
  uint16_t next[256];
  if (n_sym > 256) return false;
  for (int s = 0; s < n_sym; ++s)
    next[s] = (uint16_t)(norm[s] == -1 ? 1 : (norm[s] < 0 ? 0 : norm[s]));
  for (int i = 0; i < size; ++i) {
    uint8_t s = syms[i];
    uint16_t x = next[s]++;
    int nb = log - (31 - zclz((uint32_t)x));
    t->e[i].sym = s;
    t->e[i].nbits = (uint8_t)nb;
    t->e[i].base = (uint16_t)((x << nb) - size);
  }
  t->log = log;
  return true;
}

// Read an FSE table description (forward bitstream, LSB-first nibbles).
// Returns consumed bytes, or -1 on error.  max_log limits accuracy.
ZFN int fse_read_ncount(int16_t* norm, int* n_sym_out, int* log_out,
                               const uint8_t* src, uint64_t src_len,
                               int max_sym, int max_log) {
  if (src_len < 1) return -1;
  for (int s = 0; s <= max_sym; ++s) norm[s] = 0;  // zero-run skips
  // forward LSB-first bit reader
  uint64_t bitpos = 0;
  auto getbits = [&](int n) -> uint32_t {
    uint64_t byte = bitpos >> 3;
    if (byte + 8 <= src_len) {
      uint64_t v;
      __builtin_memcpy(&v, src + byte, 8);
      uint32_t r = (uint32_t)((v >> (bitpos & 7)) & ((1ull << n) - 1));
      bitpos += n;
      return r;
    }
    uint64_t v = 0;
    for (int i = 0; i < 8 && byte + i < src_len; ++i)
      v |= (uint64_t)src[byte + i] << (8 * i);
    uint32_t r = (uint32_t)((v >> (bitpos & 7)) & ((1ull << n) - 1));
    bitpos += n;
    return r;
  };
  auto peekbits = [&](int n) -> uint32_t {
    uint64_t byte = bitpos >> 3;
    uint64_t v = 0;
    if (byte + 8 <= src_len) {
      __builtin_memcpy(&v, src + byte, 8);
    } else {
      for (int i = 0; i < 8 && byte + i < src_len; ++i)
        v |= (uint64_t)src[byte + i] << (8 * i);
    }
    return (uint32_t)((v >> (bitpos & 7)) & ((1ull << n) - 1));
  };
  // canonical FSE_readNCount
  int log = (int)getbits(4) + 5;
  ZCD("ncount: log=%d max_log=%d len=%llu\n", log, max_log,
      (unsigned long long)src_len);
  if (log > max_log) return -1;
  int remaining = (1 << log) + 1;
  int threshold = 1 << log;
  int nb = log + 1;
  int sym = 0;
  bool prev_zero = false;
  while (remaining > 1 && sym <= max_sym) {
    if (prev_zero) {
      while (peekbits(16) == 0xFFFF) {
        sym += 24;
        bitpos += 16;
        if (sym > max_sym) return -1;
      }
      while (peekbits(2) == 3) {
        sym += 3;
        bitpos += 2;
        if (sym > max_sym) return -1;
      }
      sym += (int)getbits(2);
      if (sym > max_sym) return -1;
      prev_zero = false;
      continue;
    }
    uint32_t bits2 = peekbits(nb);
    int max = (2 * threshold - 1) - remaining;
    int count;
    if ((int)(bits2 & (threshold - 1)) < max) {
      count = (int)(bits2 & (threshold - 1));
      bitpos += nb - 1;
    } else {
      count = (int)(bits2 & (2 * threshold - 1));
      if (count >= threshold) count -= max;
      bitpos += nb;
    }
    count--;  // -1 encodes "less than one"
    remaining -= count < 0 ? -count : count;
    ZCD("  sym %d count %d remaining %d\n", sym, count, remaining - (count < 0 ? -count : count) + (count < 0 ? -count : count));
    norm[sym++] = (int16_t)count;
    prev_zero = (count == 0);
    while (remaining < threshold) {
      --nb;
      threshold >>= 1;
    }
    if (remaining < 1) return -1;
  }
  if (remaining != 1) { ZCD("ncount: remaining=%d != 1 at sym %d\n", remaining, sym); return -1; }
  if (((bitpos + 7) >> 3) > src_len) { ZCD("ncount: overread\n"); return -1; }
  for (int s = sym; s <= max_sym; ++s) norm[s] = 0;
  *n_sym_out = sym;
  *log_out = log;
  return (int)((bitpos + 7) >> 3);
}

// ---------------- predefined sequence tables -----------------------------

ZCONST int16_t kLLDefault[36] = {
    4, 3, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 2, 1, 1, 1, 2, 2, 2, 2, 2, 2, 2,
    2, 2, 3, 2, 1, 1, 1, 1, 1, -1, -1, -1, -1};
ZCONST int16_t kMLDefault[53] = {
    1, 4, 3, 2, 2, 2, 2, 2, 2, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1,
    1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1,
    -1, -1, -1, -1, -1, -1, -1};
ZCONST int16_t kOFDefault[29] = {
    1, 1, 1, 1, 1, 1, 2, 2, 2, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1, 1,
    1, -1, -1, -1, -1, -1};

ZCONST uint32_t kLLBase[36] = {
    0, 1, 2, 3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15, 16, 18, 20, 22,
    24, 28, 32, 40, 48, 64, 128, 256, 512, 1024, 2048, 4096, 8192, 16384,
    32768, 65536};
ZCONST uint8_t kLLExtra[36] = {
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 1, 1, 1, 1, 2, 2, 3,
    3, 4, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15, 16};
ZCONST uint32_t kMLBase[53] = {
    3, 4, 5, 6, 7, 8, 9, 10, 11, 12, 13, 14, 15, 16, 17, 18, 19, 20, 21,
    22, 23, 24, 25, 26, 27, 28, 29, 30, 31, 32, 33, 34, 35, 37, 39, 41,
    43, 47, 51, 59, 67, 83, 99, 131, 259, 515, 1027, 2051, 4099, 8195,
    16387, 32771, 65539};
ZCONST uint8_t kMLExtra[53] = {
    0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0, 0,
    0, 0, 0, 0, 0, 0, 0, 0, 0, 1, 1, 1, 1, 2, 2, 3, 3, 4, 4, 5, 7, 8, 9,
    10, 11, 12, 13, 14, 15, 16};

// ---------------- Huffman (huff0) ---------------------------------------

struct HuffState {
  uint16_t lut[2048];  // (sym << 4) | nbits
  int log;             // table log (maxBits), <= 11
};

// Build LUT from weights[0..n-1] (last weight already derived).
ZFN bool huf_build(HuffState* h, const uint8_t* weights, int n) {
  uint32_t rank_count[13] = {0};
  uint32_t total = 0;
  int max_w = 0;
  for (int s = 0; s < n; ++s) {
    if (weights[s] > 12) return false;
    rank_count[weights[s]]++;
    if (weights[s]) {
      total += 1u << (weights[s] - 1);
      if (weights[s] > max_w) max_w = weights[s];
    }
  }
  if (total == 0) return false;
  int log = 32 - zclz(total - 1);  // ceil log2? total must be 2^log
  if ((1u << log) != total) return false;
  if (log > 11) return false;
  // rank start positions: weight 1 (longest codes) first
  uint32_t rank_start[14];
  uint32_t next = 0;
  for (int w = 1; w <= max_w; ++w) {
    rank_start[w] = next;
    next += rank_count[w] << (w - 1);
  }
  if (next != total) return false;
  for (int s = 0; s < n; ++s) {
    int w = weights[s];
    if (!w) continue;
    uint32_t len = 1u << (w - 1);
    uint32_t start = rank_start[w];
    uint16_t entry = (uint16_t)((s << 4) | (log + 1 - w));
    for (uint32_t u = 0; u < len; ++u) h->lut[start + u] = entry;
    rank_start[w] += len;
  }
  h->log = log;
  return true;
}

// Decode huffman weights section -> weights[] (including derived last).
// Returns number of symbols, or -1.
ZFN int huf_read_weights(uint8_t* weights, const uint8_t* src,
                                uint64_t src_len, uint64_t* consumed,
                                FseTable* scratch) {
  if (src_len < 1) return -1;
  uint8_t hdr = src[0];
  int n;
  if (hdr >= 128) {
    // direct: n = hdr - 127 weights, 4 bits each
    n = hdr - 127;
    uint64_t bytes = ((uint64_t)n + 1) / 2;
    if (1 + bytes > src_len) return -1;
    for (int i = 0; i < n; ++i) {
      uint8_t b = src[1 + i / 2];
      weights[i] = (i & 1) ? (b & 0xF) : (b >> 4);
    }
    *consumed = 1 + bytes;
  } else {
    // FSE-compressed weights
    uint64_t csize = hdr;
    if (1 + csize > src_len) return -1;
    const uint8_t* w_src = src + 1;
    int16_t norm[256];
    int nsym, log;
    int hdr_bytes = fse_read_ncount(norm, &nsym, &log, w_src, csize, 255,
                                    6);
    if (hdr_bytes < 0 || (uint64_t)hdr_bytes >= csize) return -1;
    if (!fse_build(scratch, norm, nsym, log)) return -1;
    BBits bb;
    bb.init(w_src + hdr_bytes, (int64_t)(csize - hdr_bytes));
    if (bb.fail) return -1;
    uint32_t st0 = bb.get(log);
    uint32_t st1 = bb.get(log);
    n = 0;
    // two interleaved states decode weights; when a state update crosses
    // the stream start (bb.fail), the other state flushes one final
    // symbol and decoding stops (FSE_decompress tail semantics)
    while (n < 254) {
      weights[n++] = scratch->e[st0].sym;
      st0 = scratch->e[st0].base + bb.get(scratch->e[st0].nbits);
      if (bb.fail) { weights[n++] = scratch->e[st1].sym; break; }
      weights[n++] = scratch->e[st1].sym;
      st1 = scratch->e[st1].base + bb.get(scratch->e[st1].nbits);
      if (bb.fail) { weights[n++] = scratch->e[st0].sym; break; }
    }
    *consumed = 1 + csize;
  }
  // derive the final weight
  uint32_t total = 0;
  for (int i = 0; i < n; ++i)
    if (weights[i]) total += 1u << (weights[i] - 1);
  if (total == 0) return -1;
  int log = 32 - zclz(total);  // floor log2(total) + 1
  uint32_t next_pow = 1u << log;
  uint32_t rest = next_pow - total;
  // rest must be a power of 2
  if (rest == 0 || (rest & (rest - 1)) != 0) return -1;
  int last_w = (31 - zclz(rest)) + 1;
  weights[n++] = (uint8_t)last_w;
  return n;
}

// ---------------- shared per-stream decoder state ------------------------

enum { ZOP_NONE = 0, ZOP_COPY = 1, ZOP_FILL = 2, ZOP_SEQS = 3,
       ZOP_DONE = 4, ZOP_ERR = 5, ZOP_LITS = 6 };

struct SeqRec {
  uint32_t ll, ml;
  uint32_t off;
};

struct ZShared {
  FseTable ll_t, ml_t, of_t;      // sequence tables (persist across blocks)
  FseTable scratch_t;             // weight/temp table
  HuffState huf;                  // literal table (persists for treeless)
  int have_huf;
  int ll_mode_repeat_ok, ml_ok, of_ok;  // table validity for repeat mode
  uint32_t rep[3];
  uint64_t pos;                   // output position
  uint64_t frame_end_src;         // end of current frame in src
  int64_t status;
  int op;
  // op params
  uint64_t a, b, c;               // generic: src/dst/len
  uint8_t fillv;
  // sequence-execution batch (decoded by lane 0 into ws, executed by wave)
  uint32_t n_seqs;
  uint64_t lit_ptr;               // ws literals
  uint64_t lit_len;
  uint64_t seq_ptr;               // ws seq records
  // huffman literal decode params (4 lanes)
  uint64_t hsrc[4];
  uint64_t hsrc_len[4];
  uint64_t hdst[4];
  uint64_t hdst_len[4];
  int hstreams;
  int herr;
};

// one sequence-table init from mode bits; returns consumed or -1
ZFN int seq_table_init(FseTable* t, int* repeat_ok, int mode,
                              const uint8_t* src, uint64_t len,
                              const int16_t* defaults, int n_def,
                              int def_log, int max_sym, int max_log,
                              FReader* fr_rle) {
  if (mode == 0) {  // predefined
    int16_t norm[64];
    for (int i = 0; i < n_def; ++i) norm[i] = defaults[i];
    if (!fse_build(t, norm, n_def, def_log)) return -1;
    *repeat_ok = 1;
    return 0;
  }
  if (mode == 1) {  // RLE: 1 byte symbol, table log 0
    if (len < 1) return -1;
    uint8_t sym = src[0];
    if (sym > max_sym) return -1;
    t->log = 0;
    t->e[0].sym = sym;
    t->e[0].nbits = 0;
    t->e[0].base = 0;
    *repeat_ok = 1;
    return 1;
  }
  if (mode == 2) {  // FSE description
    int16_t norm[64];
    int nsym, log;
    int used = fse_read_ncount(norm, &nsym, &log, src, len, max_sym,
                               max_log);
    if (used < 0) return -1;
    if (!fse_build(t, norm, nsym, log)) return -1;
    *repeat_ok = 1;
    return used;
  }
  // repeat
  if (!*repeat_ok) return -1;
  return 0;
}


}  // namespace zstd_core
