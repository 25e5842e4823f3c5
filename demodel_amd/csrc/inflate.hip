// CDNA4 DEFLATE (RFC 1951) inflate kernel.
//
// Cached proxy bodies keep their original Content-Encoding (gzip —
// reference CONTRIBUTING.md:116), and dataset streams carry
// deflate/gzip-compressed members; this kernel decompresses them without
// a host round-trip.
//
// DEFLATE is serially entropy-coded, so a single stream has no internal
// parallelism (SURVEY.md §7 hard part (a)); the design is therefore
// wave-per-stream: lane 0 owns the bit reader and Huffman decode, and the
// whole wave64 executes long match copies / stored-block copies in
// parallel.  Aggregate throughput comes from decompressing many streams
// (gzip members, parquet pages, per-file bodies) concurrently — the
// honest decomposition for this format.
//
// Back-references read the already-written output straight from HBM/L2;
// no LDS window.  Huffman decode is canonical count/offset bit-by-bit
// with tables in LDS.

#include <hip/hip_runtime.h>

namespace {

enum {
  INF_OK = 0,
  INF_ERR_FORMAT = -2,
  INF_ERR_OVERFLOW = -3,
  INF_ERR_UNDERRUN = -4,
};

struct __align__(16) InflateDesc {
  uint64_t src;       // device ptr to raw DEFLATE stream
  uint64_t src_len;
  uint64_t dst;       // device ptr to output
  uint64_t dst_cap;
  uint64_t written;   // out
  int64_t status;     // out
  uint64_t consumed;  // out: input bytes consumed (multi-member scans)
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
    while (nbits <= 56) {
      uint64_t byte = pos < len ? p[pos] : 0;  // zero-pad past EOF
      buf |= byte << nbits;
      nbits += 8;
      ++pos;
    }
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
    int drop = nbits & 7;
    buf >>= drop;
    nbits -= drop;
  }
  __device__ uint64_t byte_pos() const {
    return pos - (uint64_t)(nbits >> 3);
  }
  __device__ bool overran() const { return byte_pos() > len; }
};

struct HuffTable {
  uint16_t count[16];
  uint16_t offset[16];
  uint32_t first[16];
  uint16_t sym[288];
};

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

__device__ int huff_decode(BitReader* br, const HuffTable* t) {
  uint32_t code = 0;
  for (int l = 1; l < 16; ++l) {
    code |= br->bit1();
    if (t->count[l] && code - t->first[l] < (uint32_t)t->count[l])
      return t->sym[t->offset[l] + (code - t->first[l])];
    code <<= 1;
  }
  return -1;
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

// op kinds handed from lane 0 to the wave
enum { OP_NONE = 0, OP_MATCH = 1, OP_STORED = 2, OP_DONE = 3, OP_ERR = 4 };

struct Shared {
  HuffTable litlen;
  HuffTable dist;
  BitReader br;
  uint8_t lens_buf[320];
  uint64_t pos;        // output position
  int stage;           // 0 = at block boundary, 1 = inside huffman block
  int bfinal;
  int op;
  uint64_t op_pos;     // op-specific: match dst / stored dst
  uint64_t op_src;     // stored: src byte offset
  uint32_t op_len;
  uint32_t op_dist;
  int64_t status;
};

// lane-0 decode step: runs until a wave op is required; returns op kind.
__device__ int decode_until_op(Shared* sh, const uint8_t* src,
                               uint8_t* out, const InflateDesc* d) {
  BitReader br = sh->br;
  uint64_t pos = sh->pos;
  int op = OP_NONE;

  while (op == OP_NONE) {
    if (br.overran()) { sh->status = INF_ERR_UNDERRUN; op = OP_ERR; break; }
    if (sh->stage == 0) {
      sh->bfinal = (int)br.bit1();
      uint32_t btype = br.bits(2);
      if (btype == 0) {
        br.align_byte();
        uint32_t len = br.bits(16);
        uint32_t nlen = br.bits(16);
        if ((len ^ 0xFFFFu) != nlen) {
          sh->status = INF_ERR_FORMAT; op = OP_ERR; break;
        }
        uint64_t so = br.byte_pos();
        if (so + len > br.len) {
          sh->status = INF_ERR_UNDERRUN; op = OP_ERR; break;
        }
        if (pos + len > d->dst_cap) {
          sh->status = INF_ERR_OVERFLOW; op = OP_ERR; break;
        }
        br.buf = 0; br.nbits = 0; br.pos = so + len;
        sh->op_pos = pos; sh->op_src = so; sh->op_len = len;
        pos += len;
        op = OP_STORED;  // stage stays 0; bfinal checked after the copy
        break;
      }
      if (btype == 3) { sh->status = INF_ERR_FORMAT; op = OP_ERR; break; }
      bool okt = true;
      if (btype == 1) {
        for (int i = 0; i < 288; ++i)
          sh->lens_buf[i] = i < 144 ? 8 : i < 256 ? 9 : i < 280 ? 7 : 8;
        okt = huff_build(&sh->litlen, sh->lens_buf, 288);
        for (int i = 0; i < 30; ++i) sh->lens_buf[i] = 5;
        okt = okt && huff_build(&sh->dist, sh->lens_buf, 30);
      } else {
        int hlit = (int)br.bits(5) + 257;
        int hdist = (int)br.bits(5) + 1;
        int hclen = (int)br.bits(4) + 4;
        uint8_t cl_lens[19];
        for (int i = 0; i < 19; ++i) cl_lens[i] = 0;
        for (int i = 0; i < hclen; ++i)
          cl_lens[kClOrder[i]] = (uint8_t)br.bits(3);
        if (!huff_build(&sh->dist, cl_lens, 19)) {  // dist as CL scratch
          sh->status = INF_ERR_FORMAT; op = OP_ERR; break;
        }
        int total = hlit + hdist;
        int i = 0;
        while (i < total) {
          int symc = huff_decode(&br, &sh->dist);
          if (symc < 0 || br.overran()) { okt = false; break; }
          if (symc < 16) {
            sh->lens_buf[i++] = (uint8_t)symc;
          } else if (symc == 16) {
            if (i == 0) { okt = false; break; }
            int rep = 3 + (int)br.bits(2);
            uint8_t v = sh->lens_buf[i - 1];
            while (rep-- > 0 && i < total) sh->lens_buf[i++] = v;
          } else if (symc == 17) {
            int rep = 3 + (int)br.bits(3);
            while (rep-- > 0 && i < total) sh->lens_buf[i++] = 0;
          } else {
            int rep = 11 + (int)br.bits(7);
            while (rep-- > 0 && i < total) sh->lens_buf[i++] = 0;
          }
        }
        okt = okt && i == total &&
              huff_build(&sh->litlen, sh->lens_buf, hlit) &&
              huff_build(&sh->dist, sh->lens_buf + hlit, hdist);
      }
      if (!okt) { sh->status = INF_ERR_FORMAT; op = OP_ERR; break; }
      sh->stage = 1;
      continue;
    }

    // stage 1: symbol loop
    while (true) {
      int sym = huff_decode(&br, &sh->litlen);
      if (sym < 0 || br.overran()) {
        sh->status = br.overran() ? INF_ERR_UNDERRUN : INF_ERR_FORMAT;
        op = OP_ERR;
        break;
      }
      if (sym < 256) {
        if (pos >= d->dst_cap) {
          sh->status = INF_ERR_OVERFLOW; op = OP_ERR; break;
        }
        out[pos++] = (uint8_t)sym;
        continue;
      }
      if (sym == 256) {
        sh->stage = 0;
        if (sh->bfinal) { sh->status = INF_OK; op = OP_DONE; }
        break;  // back to block-boundary handling (or done)
      }
      sym -= 257;
      if (sym >= 29) { sh->status = INF_ERR_FORMAT; op = OP_ERR; break; }
      uint32_t mlen = kLenBase[sym] + br.bits(kLenExtra[sym]);
      int dsym = huff_decode(&br, &sh->dist);
      if (dsym < 0 || dsym >= 30) {
        sh->status = INF_ERR_FORMAT; op = OP_ERR; break;
      }
      uint32_t dist = kDistBase[dsym] + br.bits(kDistExtra[dsym]);
      if (dist > pos) { sh->status = INF_ERR_FORMAT; op = OP_ERR; break; }
      if (pos + mlen > d->dst_cap) {
        sh->status = INF_ERR_OVERFLOW; op = OP_ERR; break;
      }
      if (mlen >= 64 && dist >= 16) {
        sh->op_pos = pos; sh->op_len = mlen; sh->op_dist = dist;
        pos += mlen;
        op = OP_MATCH;
        break;
      }
      for (uint32_t i = 0; i < mlen; ++i, ++pos)
        out[pos] = out[pos - dist];
    }
  }

  sh->br = br;
  sh->pos = pos;
  return op;
}

__global__ void __launch_bounds__(64)
inflate_kernel(InflateDesc* __restrict__ descs, int n_streams) {
  __shared__ Shared sh;
  int lane = threadIdx.x;

  for (int s = blockIdx.x; s < n_streams; s += gridDim.x) {
    InflateDesc* d = &descs[s];
    const uint8_t* src = (const uint8_t*)d->src;
    uint8_t* out = (uint8_t*)d->dst;

    if (lane == 0) {
      sh.br.init(src, d->src_len);
      sh.pos = 0;
      sh.stage = 0;
      sh.bfinal = 0;
      sh.status = INF_OK;
      sh.op = OP_NONE;
    }
    __syncthreads();

    while (true) {
      if (lane == 0) sh.op = decode_until_op(&sh, src, out, d);
      __syncthreads();
      int op = sh.op;
      if (op == OP_DONE || op == OP_ERR) break;
      if (op == OP_MATCH) {
        uint64_t p0 = sh.op_pos;
        uint32_t dist = sh.op_dist, len = sh.op_len;
        if (dist >= len) {
          for (uint32_t i = lane; i < len; i += 64)
            out[p0 + i] = out[p0 + i - dist];
        } else {
          uint64_t copied = 0;
          while (copied < len) {
            uint32_t n = (uint32_t)min((uint64_t)dist, len - copied);
            for (uint32_t i = lane; i < n; i += 64)
              out[p0 + copied + i] = out[p0 + copied + i - dist];
            __syncthreads();
            copied += n;
          }
        }
      } else if (op == OP_STORED) {
        for (uint64_t i = lane; i < sh.op_len; i += 64)
          out[sh.op_pos + i] = src[sh.op_src + i];
        __syncthreads();
        if (lane == 0 && sh.bfinal) {
          sh.status = INF_OK;
          sh.op = OP_DONE;
        }
        __syncthreads();
        if (sh.op == OP_DONE) break;
      }
      __syncthreads();
    }

    if (lane == 0) {
      d->written = sh.pos;
      d->status = sh.status;
      d->consumed = sh.br.byte_pos();
    }
    __syncthreads();
  }
}

}  // namespace

extern "C" void launch_inflate_streams(const uint64_t* desc, int n_streams,
                                       int* /*status_unused*/,
                                       hipStream_t stream) {
  if (n_streams <= 0) return;
  int blocks = n_streams < 2048 ? n_streams : 2048;
  hipLaunchKernelGGL(inflate_kernel, dim3(blocks), dim3(64), 0, stream,
                     (InflateDesc*)desc, n_streams);
}
