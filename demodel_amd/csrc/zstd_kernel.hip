// CDNA4 Zstandard (RFC 8878) frame decoder.
// Decode primitives live in zstd_common.h (shared with the host reference
// decoder in zstd_host.cpp); this file is the wave-per-frame GPU
// orchestration: lane 0 parses frames/blocks and decodes entropy streams,
// the wave64 executes literal/match/stored copies, and the 4-stream
// Huffman literals section decodes on 4 lanes concurrently.
//
// Two kernels share the same parse/sequence machinery:
//   zstd_kernel<ZWIN>     — one frame per wave (few big frames).
//   zstd_kernel_x2<ZWIN>  — TWO frames per wave, sequence decode
//     software-interleaved: the per-sequence FSE chain is latency-bound
//     (PMC: zero LDS conflicts, wave stalled on the state->table->bits
//     dependency chain), so interleaving two INDEPENDENT chains in one
//     instruction stream doubles the exploitable ILP.  The two frames'
//     4-stream Huffman sections also decode concurrently (8 lanes busy
//     instead of 4).  Used when many frames provide pairing.

#include <hip/hip_runtime.h>

#include <stdlib.h>
#include <string.h>

#include "zstd_common.h"

using namespace zstd_core;

namespace {

// LDS output window: match copies read recent output from LDS instead of
// global memory, which removes the per-sequence vmcnt(0)+barrier ordering
// that dominated literal/sequence-heavy payloads (measured: matchy frames
// decode at ~108 MB/s/wave while word-salad crawled at ~1.5 — the
// difference was per-sequence synchronization, not decode math).  Window
// slot p & (ZWIN-1) holds output byte p; every producer (literals,
// matches, raw/RLE blocks) maintains it.  Matches farther back than the
// window take a rare global-read path behind an explicit vmcnt drain.

// LDS same-wave ordering fence.  GCN/CDNA processes a wave's LDS
// instructions in issue order, so a ds_read after a may-alias ds_write
// needs NO architectural fence — but the asm memory clobber also stops
// the compiler from caching LDS-resident values (FSE table entries)
// across the call, which costs register pressure + reloads in the
// per-sequence loop.  FENCE=false relies on in-order LDS + compiler
// alias analysis; selectable at runtime for A/B (DEMODEL_ZSTD_MODE).
template <bool FENCE>
__device__ __forceinline__ void lds_fence() {
  if (FENCE) asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
}

// one sequence's copies: literal run from the (global) literal buffer,
// then the match via the LDS window (or the rare far-global path)
template <int ZWIN, bool FENCE>
__device__ __forceinline__ void exec_seq(
    uint8_t* __restrict__ out, uint8_t* win, uint64_t wfrom,
    const uint8_t* lit_src, uint64_t p0, uint32_t ll, uint32_t len,
    uint32_t dist, int lane) {
  constexpr int ZWMASK = ZWIN - 1;
  for (uint32_t k = lane; k < ll; k += 64) {
    uint8_t v = lit_src[k];
    out[p0 + k] = v;
    win[(p0 + k) & ZWMASK] = v;
  }
  uint64_t mp = p0 + ll;
  if (dist <= ZWIN - 128 && mp - dist >= wfrom) {
    lds_fence<FENCE>();
    if (dist >= len) {
      for (uint32_t k = lane; k < len; k += 64) {
        uint8_t v = win[(mp + k - dist) & ZWMASK];
        out[mp + k] = v;
        win[(mp + k) & ZWMASK] = v;
      }
    } else {
      uint64_t copied = 0;
      while (copied < len) {
        uint32_t n = (uint32_t)min((uint64_t)dist,
                                   (uint64_t)len - copied);
        for (uint32_t k = lane; k < n; k += 64) {
          uint8_t v = win[(mp + copied + k - dist) & ZWMASK];
          out[mp + copied + k] = v;
          win[(mp + copied + k) & ZWMASK] = v;
        }
        lds_fence<FENCE>();
        copied += n;
      }
    }
  } else {
    // far match: read old output from HBM; drain our stores first
    // (global stores have NO same-address ordering guarantee, unlike
    // LDS — this fence is load-bearing and stays in both modes)
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    for (uint32_t k = lane; k < len; k += 64) {
      uint8_t v = out[mp + k - dist];
      out[mp + k] = v;
      win[(mp + k) & ZWMASK] = v;
    }
  }
}

// ------------------- lane-0 frame/block parser ------------------------

struct ParseState {
  FReader fr;
  int stage;            // 0 = expect frame magic, 1 = expect block hdr
  int frame_checksum;
  int pending_last;     // a wave op for the frame's final block is out
};

// Parse until a wave op is staged in sh (sh.op != ZOP_NONE).  Runs on
// lane 0 only; identical to round-1's inline parse body.
__device__ void parse_until_op(ParseState& pc, ZShared& sh,
                               const ZstdDesc* d, uint8_t* ws) {
  FReader& fr = pc.fr;
  sh.op = ZOP_NONE;
  while (sh.op == ZOP_NONE) {
    if (pc.pending_last) {
      pc.pending_last = 0;
      if (pc.frame_checksum) fr.pos += 4;  // xxh64 low32, not verified
      pc.stage = 0;
    }
    if (pc.stage == 0) {
      if (fr.pos >= fr.len) { sh.op = ZOP_DONE; break; }
      uint32_t magic = fr.u32();
      if (fr.fail) { sh.status = Z_ERR_UNDERRUN; sh.op = ZOP_ERR; break; }
      if ((magic & 0xFFFFFFF0u) == 0x184D2A50u) {  // skippable
        uint32_t sz = fr.u32();
        if (!fr.need(sz)) { sh.status = Z_ERR_UNDERRUN; sh.op = ZOP_ERR; break; }
        fr.pos += sz;
        continue;
      }
      if (magic != 0xFD2FB528u) { sh.status = Z_ERR_MAGIC; sh.op = ZOP_ERR; break; }
      uint8_t fhd = fr.u8();
      int dict_flag = fhd & 3;
      pc.frame_checksum = (fhd >> 2) & 1;
      int single_seg = (fhd >> 5) & 1;
      int fcs_flag = (fhd >> 6) & 3;
      if (dict_flag) { sh.status = Z_ERR_DICT; sh.op = ZOP_ERR; break; }
      if (!single_seg) (void)fr.u8();  // window descriptor
      if (fcs_flag == 0) { if (single_seg) (void)fr.u8(); }
      else if (fcs_flag == 1) (void)fr.u16();
      else if (fcs_flag == 2) (void)fr.u32();
      else (void)fr.u64v();
      if (fr.fail) { sh.status = Z_ERR_UNDERRUN; sh.op = ZOP_ERR; break; }
      sh.rep[0] = 1; sh.rep[1] = 4; sh.rep[2] = 8;
      sh.have_huf = 0;
      sh.ll_mode_repeat_ok = sh.ml_ok = sh.of_ok = 0;
      pc.stage = 1;
      continue;
    }

    // ---- stage 1: one block ---------------------------------
    uint32_t bh = fr.u24();
    if (fr.fail) { sh.status = Z_ERR_UNDERRUN; sh.op = ZOP_ERR; break; }
    int last = bh & 1;
    int btype = (bh >> 1) & 3;
    uint32_t bsize = bh >> 3;
    if (btype == 0) {                                  // raw block
      if (!fr.need(bsize)) { sh.status = Z_ERR_UNDERRUN; sh.op = ZOP_ERR; break; }
      if (sh.pos + bsize > d->dst_cap) { sh.status = Z_ERR_OVERFLOW; sh.op = ZOP_ERR; break; }
      sh.a = (uint64_t)(fr.p + fr.pos);
      sh.b = sh.pos;
      sh.c = bsize;
      fr.pos += bsize;
      sh.pos += bsize;
      pc.pending_last = last;
      sh.op = ZOP_COPY;
      break;
    }
    if (btype == 1) {                                  // RLE block
      uint8_t v = fr.u8();
      if (fr.fail) { sh.status = Z_ERR_UNDERRUN; sh.op = ZOP_ERR; break; }
      if (sh.pos + bsize > d->dst_cap) { sh.status = Z_ERR_OVERFLOW; sh.op = ZOP_ERR; break; }
      sh.fillv = v;
      sh.b = sh.pos;
      sh.c = bsize;
      sh.pos += bsize;
      pc.pending_last = last;
      sh.op = ZOP_FILL;
      break;
    }
    if (btype == 3) { sh.status = Z_ERR_FORMAT; sh.op = ZOP_ERR; break; }

    // ---- compressed block -----------------------------------
    if (!fr.need(bsize)) { sh.status = Z_ERR_UNDERRUN; sh.op = ZOP_ERR; break; }
    const uint8_t* blk = fr.p + fr.pos;
    uint64_t blen = bsize;
    fr.pos += bsize;
    FReader br_;
    br_.init(blk, blen);
    uint8_t lh0 = br_.u8();
    int lit_type = lh0 & 3;
    int size_fmt = (lh0 >> 2) & 3;
    uint64_t regen = 0, csize = 0;
    int n_hstreams = 1;
    if (lit_type == 0 || lit_type == 1) {
      if (size_fmt == 0 || size_fmt == 2)
        regen = lh0 >> 3;
      else if (size_fmt == 1)
        regen = (lh0 >> 4) | ((uint64_t)br_.u8() << 4);
      else
        regen = (lh0 >> 4) | ((uint64_t)br_.u8() << 4)
                | ((uint64_t)br_.u8() << 12);
    } else {
      uint32_t b1, b2, b3, b4;
      if (size_fmt == 0) {
        n_hstreams = 1;
        b1 = br_.u8(); b2 = br_.u8();
        regen = (lh0 >> 4) | ((b1 & 0x3F) << 4);
        csize = (b1 >> 6) | (b2 << 2);
      } else if (size_fmt == 1) {
        n_hstreams = 4;
        b1 = br_.u8(); b2 = br_.u8();
        regen = (lh0 >> 4) | ((b1 & 0x3F) << 4);
        csize = (b1 >> 6) | (b2 << 2);
      } else if (size_fmt == 2) {
        n_hstreams = 4;
        b1 = br_.u8(); b2 = br_.u8(); b3 = br_.u8();
        regen = (lh0 >> 4) | (b1 << 4) | ((uint64_t)(b2 & 3) << 12);
        csize = (b2 >> 2) | (b3 << 6);
      } else {
        n_hstreams = 4;
        b1 = br_.u8(); b2 = br_.u8(); b3 = br_.u8(); b4 = br_.u8();
        regen = (lh0 >> 4) | (b1 << 4) | ((uint64_t)(b2 & 0x3F) << 12);
        csize = (b2 >> 6) | (b3 << 2) | ((uint64_t)b4 << 10);
      }
    }
    if (br_.fail || regen > (131 << 10)) { sh.status = Z_ERR_FORMAT; sh.op = ZOP_ERR; break; }
    sh.lit_len = regen;
    sh.lit_ptr = (uint64_t)ws;
    if (lit_type == 0) {
      if (!br_.need(regen)) { sh.status = Z_ERR_UNDERRUN; sh.op = ZOP_ERR; break; }
      sh.hstreams = 0;
      sh.hsrc[0] = (uint64_t)(br_.p + br_.pos);
      br_.pos += regen;
    } else if (lit_type == 1) {
      if (!br_.need(1)) { sh.status = Z_ERR_UNDERRUN; sh.op = ZOP_ERR; break; }
      sh.hstreams = -1;
      sh.fillv = br_.u8();
    } else {
      const uint8_t* hsec = br_.p + br_.pos;
      if (!br_.need(csize)) { sh.status = Z_ERR_UNDERRUN; sh.op = ZOP_ERR; break; }
      br_.pos += csize;
      uint64_t hoff = 0;
      if (lit_type == 2) {
        uint8_t wbuf[256];
        uint64_t wcons = 0;
        int nw = huf_read_weights(wbuf, hsec, csize, &wcons,
                                  &sh.scratch_t);
        if (nw < 0 || !huf_build(&sh.huf, wbuf, nw)) {
          sh.status = Z_ERR_FORMAT; sh.op = ZOP_ERR; break;
        }
        sh.have_huf = 1;
        hoff = wcons;
      } else if (!sh.have_huf) {
        sh.status = Z_ERR_FORMAT; sh.op = ZOP_ERR; break;
      }
      const uint8_t* hdata = hsec + hoff;
      uint64_t hlen = csize - hoff;
      if (n_hstreams == 1) {
        sh.hstreams = 1;
        sh.hsrc[0] = (uint64_t)hdata;
        sh.hsrc_len[0] = hlen;
        sh.hdst[0] = (uint64_t)ws;
        sh.hdst_len[0] = regen;
      } else {
        if (hlen < 6) { sh.status = Z_ERR_UNDERRUN; sh.op = ZOP_ERR; break; }
        uint32_t s1 = hdata[0] | (hdata[1] << 8);
        uint32_t s2 = hdata[2] | (hdata[3] << 8);
        uint32_t s3 = hdata[4] | (hdata[5] << 8);
        uint64_t rest = hlen - 6;
        if ((uint64_t)s1 + s2 + s3 > rest) { sh.status = Z_ERR_FORMAT; sh.op = ZOP_ERR; break; }
        uint64_t q = (regen + 3) / 4;
        sh.hstreams = 4;
        const uint8_t* p1 = hdata + 6;
        sh.hsrc[0] = (uint64_t)p1; sh.hsrc_len[0] = s1;
        sh.hsrc[1] = (uint64_t)(p1 + s1); sh.hsrc_len[1] = s2;
        sh.hsrc[2] = (uint64_t)(p1 + s1 + s2); sh.hsrc_len[2] = s3;
        sh.hsrc[3] = (uint64_t)(p1 + s1 + s2 + s3);
        sh.hsrc_len[3] = rest - s1 - s2 - s3;
        for (int k = 0; k < 4; ++k) {
          sh.hdst[k] = (uint64_t)(ws + (uint64_t)k * q);
          sh.hdst_len[k] = (k < 3) ? q : regen - 3 * q;
        }
      }
    }

    // sequences header
    uint32_t n_seq;
    uint8_t sb0 = br_.u8();
    if (br_.fail) { sh.status = Z_ERR_UNDERRUN; sh.op = ZOP_ERR; break; }
    if (sb0 < 128) {
      n_seq = sb0;
    } else if (sb0 < 255) {
      n_seq = ((uint32_t)(sb0 - 128) << 8) + br_.u8();
    } else {
      n_seq = (uint32_t)br_.u16() + 0x7F00;
    }
    sh.n_seqs = n_seq;
    if (n_seq > 0) {
      uint8_t modes = br_.u8();
      if (br_.fail || (modes & 3)) { sh.status = Z_ERR_FORMAT; sh.op = ZOP_ERR; break; }
      const uint8_t* tp = br_.p + br_.pos;
      uint64_t tleft = br_.len - br_.pos;
      int used;
      used = seq_table_init(&sh.ll_t, &sh.ll_mode_repeat_ok,
                            (modes >> 6) & 3, tp, tleft, kLLDefault,
                            36, 6, 35, 9, nullptr);
      if (used < 0) { sh.status = Z_ERR_FORMAT; sh.op = ZOP_ERR; break; }
      tp += used; tleft -= used;
      used = seq_table_init(&sh.of_t, &sh.of_ok, (modes >> 4) & 3,
                            tp, tleft, kOFDefault, 29, 5, 31, 8,
                            nullptr);
      if (used < 0) { sh.status = Z_ERR_FORMAT; sh.op = ZOP_ERR; break; }
      tp += used; tleft -= used;
      used = seq_table_init(&sh.ml_t, &sh.ml_ok, (modes >> 2) & 3,
                            tp, tleft, kMLDefault, 53, 6, 52, 9,
                            nullptr);
      if (used < 0) { sh.status = Z_ERR_FORMAT; sh.op = ZOP_ERR; break; }
      tp += used; tleft -= used;
      sh.a = (uint64_t)tp;   // FSE sequence bitstream
      sh.c = tleft;
    } else {
      sh.a = 0;
      sh.c = 0;
    }
    pc.pending_last = last;
    sh.op = ZOP_LITS;
    break;
  }
}

// ------------------- per-lane-replicated sequence decode ----------------
// All 64 lanes run the identical FSE decode in lockstep on their own
// register state (LDS table reads broadcast, bitstream reads coalesce to
// one address) — zero cross-lane publication, zero waits in the
// per-sequence loop; matches read the LDS window.

struct SeqCtx {
  BBits sq;
  uint32_t ll_state, ml_state, of_state;
  uint32_t rep0, rep1, rep2;
  uint64_t pos_r, lit_used;
  const uint8_t* lit_base;
  uint64_t wfrom;
  uint32_t c_ll, c_len, c_dist;   // pending (decoded, unexecuted) seq
  uint32_t n_ll, n_len, n_dist;   // freshly decoded seq
  uint32_t i, nseq;
  int err;
  bool have;
};

__device__ __forceinline__ void seq_init(SeqCtx& s, ZShared& sh,
                                         uint64_t wfrom) {
  s.nseq = sh.n_seqs;
  s.err = sh.herr;
  s.i = 0;
  s.have = false;
  s.sq.init((const uint8_t*)sh.a, (int64_t)sh.c);
  s.ll_state = s.of_state = s.ml_state = 0;
  s.rep0 = sh.rep[0]; s.rep1 = sh.rep[1]; s.rep2 = sh.rep[2];
  if (s.nseq) {
    if (s.sq.fail) s.err = 1;
    s.ll_state = s.sq.get(sh.ll_t.log);
    s.of_state = s.sq.get(sh.of_t.log);
    s.ml_state = s.sq.get(sh.ml_t.log);
  }
  s.pos_r = sh.pos;
  s.wfrom = wfrom;
  s.lit_used = 0;
  s.lit_base = (const uint8_t*)sh.lit_ptr;
  s.c_ll = s.c_len = s.c_dist = 0;
  s.n_ll = s.n_len = s.n_dist = 0;
}

// decode sequence s.i into n_ll/n_len/n_dist + advance FSE states
__device__ __forceinline__ void seq_decode(SeqCtx& s, ZShared& sh) {
  // one LDS dword per table entry: exec_seq's asm memory clobbers would
  // otherwise force re-loads of sym/base/nbits
  FseEntry oe = sh.of_t.e[s.of_state];
  FseEntry me = sh.ml_t.e[s.ml_state];
  FseEntry le = sh.ll_t.e[s.ll_state];
  uint8_t ofc = oe.sym, mlc = me.sym, llc = le.sym;
  if (ofc > 31 || mlc > 52 || llc > 35) { s.err = 1; return; }
  uint32_t ofv = (1u << ofc) + s.sq.get(ofc);
  // ml+ll extras in ONE bit read (<=32 bits): get(a)+get(b) composes as
  // get(a+b) with the first read in the high bits
  int mlx = kMLExtra[mlc], llx = kLLExtra[llc];
  uint32_t ex = s.sq.get(mlx + llx);
  s.n_len = kMLBase[mlc] + (ex >> llx);
  s.n_ll = kLLBase[llc]
           + (ex & ((llx < 32) ? ((1u << llx) - 1) : 0xFFFFFFFFu));
  if (ofv > 3) {
    s.n_dist = ofv - 3;
    s.rep2 = s.rep1; s.rep1 = s.rep0; s.rep0 = s.n_dist;
  } else {
    uint32_t idx = ofv + (s.n_ll == 0 ? 1 : 0);
    if (idx == 1) {
      s.n_dist = s.rep0;
    } else if (idx == 2) {
      s.n_dist = s.rep1;
      s.rep1 = s.rep0; s.rep0 = s.n_dist;
    } else if (idx == 3) {
      s.n_dist = s.rep2;
      s.rep2 = s.rep1; s.rep1 = s.rep0; s.rep0 = s.n_dist;
    } else {
      s.n_dist = s.rep0 - 1;
      if (s.n_dist == 0 || s.rep0 == 0) { s.err = 1; return; }
      s.rep2 = s.rep1; s.rep1 = s.rep0; s.rep0 = s.n_dist;
    }
  }
  if (s.i + 1 < s.nseq) {
    // the three state refreshes (<=9 bits each, ll|ml|of order) in ONE
    // bit read: one refill check instead of three
    int b_ll = le.nbits, b_ml = me.nbits, b_of = oe.nbits;
    uint32_t bits = s.sq.get(b_ll + b_ml + b_of);
    s.of_state = oe.base + (bits & ((1u << b_of) - 1));
    s.ml_state = me.base + ((bits >> b_of) & ((1u << b_ml) - 1));
    s.ll_state = le.base + (bits >> (b_of + b_ml));
  }
}

// bounds-check the freshly decoded seq against the post-exec position
// and stage it as the pending one
__device__ __forceinline__ void seq_commit(SeqCtx& s, ZShared& sh,
                                           uint64_t dst_cap) {
  if (s.lit_used + s.n_ll > sh.lit_len ||
      s.pos_r + s.n_ll + (uint64_t)s.n_len > dst_cap ||
      (uint64_t)s.n_dist > s.pos_r + s.n_ll) {
    s.err = 1;
    return;
  }
  s.c_ll = s.n_ll; s.c_len = s.n_len; s.c_dist = s.n_dist;
  s.have = true;
}

// decode one Huffman literal stream (runs on one lane)
__device__ __forceinline__ void huf_stream_decode(ZShared& sh, int k) {
  BBits bb;
  bb.init((const uint8_t*)sh.hsrc[k], (int64_t)sh.hsrc_len[k]);
  uint8_t* t = (uint8_t*)sh.hdst[k];
  uint64_t want = sh.hdst_len[k];
  int log = sh.huf.log;
  int bad = bb.fail && want > 0;
  for (uint64_t i = 0; i < want && !bad; ++i) {
    uint32_t v = bb.peek(log);
    uint16_t e = sh.huf.lut[v];
    bb.skip(e & 0xF);
    t[i] = (uint8_t)(e >> 4);
  }
  if (bad) sh.herr = 1;
}

// Window size trades per-wave speed (bigger window = fewer far matches)
// against occupancy (LDS-limited resident workgroups); the launcher picks
// 16 KiB when many frames provide parallelism, 64 KiB for few frames.
// second launch_bounds arg = waves/SIMD: LDS (window + tables) already
// caps residency at ~5 (16K) / 2 (64K) workgroups per CU, so granting the
// full register budget is free — without it the compiler spills the
// sequence loop into AGPRs/scratch (1626 v_accvgpr_read in the ISA).
template <int ZWIN, bool FENCE>
__global__ void __launch_bounds__(64, ZWIN <= 16 * 1024 ? 2 : 1)
zstd_kernel(ZstdDesc* __restrict__ descs, int n_streams) {
  constexpr int ZWMASK = ZWIN - 1;
  __shared__ ZShared sh;
  __shared__ int seq_err;
  __shared__ uint64_t win_from;   // output pos from which win[] is valid
  __shared__ uint8_t win[ZWIN];
  int lane = threadIdx.x;

  for (int sidx = blockIdx.x; sidx < n_streams; sidx += gridDim.x) {
    ZstdDesc* d = &descs[sidx];
    uint8_t* out = (uint8_t*)d->dst;
    uint8_t* ws = (uint8_t*)d->ws;

    ParseState pc;
    if (lane == 0) {
      pc.fr.init((const uint8_t*)d->src, d->src_len);
      pc.stage = 0;
      pc.frame_checksum = 0;
      pc.pending_last = 0;
      sh.pos = 0;
      sh.status = Z_OK;
      sh.op = ZOP_NONE;
      sh.herr = 0;
      win_from = 0;
    }
    __syncthreads();

    bool done = false;
    while (!done) {
      if (lane == 0) parse_until_op(pc, sh, d, ws);
      __syncthreads();

      int op = sh.op;
      if (op == ZOP_DONE || op == ZOP_ERR) {
        done = true;
      } else if (op == ZOP_COPY) {
        const uint8_t* s = (const uint8_t*)sh.a;
        for (uint64_t i = lane; i < sh.c; i += 64) out[sh.b + i] = s[i];
        if (lane == 0) win_from = sh.b + sh.c;  // window gap over block
      } else if (op == ZOP_FILL) {
        for (uint64_t i = lane; i < sh.c; i += 64) out[sh.b + i] = sh.fillv;
        if (lane == 0) win_from = sh.b + sh.c;
      } else if (op == ZOP_LITS) {
        // ---- 1) literals into ws ------------------------------------
        if (sh.hstreams == 0) {
          const uint8_t* s = (const uint8_t*)sh.hsrc[0];
          uint8_t* t = (uint8_t*)sh.lit_ptr;
          for (uint64_t i = lane; i < sh.lit_len; i += 64) t[i] = s[i];
        } else if (sh.hstreams == -1) {
          uint8_t* t = (uint8_t*)sh.lit_ptr;
          for (uint64_t i = lane; i < sh.lit_len; i += 64) t[i] = sh.fillv;
        } else if (lane < sh.hstreams) {
          huf_stream_decode(sh, lane);
        }
        __syncthreads();

        // ---- 2) sequences (software-pipelined: decode i+1 before the
        // copies of i, so decode latency hides under copy latency) -----
        SeqCtx s;
        seq_init(s, sh, win_from);
        for (; s.i < s.nseq && !s.err; ++s.i) {
          seq_decode(s, sh);
          if (s.err) break;
          if (s.have) {
            exec_seq<ZWIN, FENCE>(out, win, s.wfrom, s.lit_base + s.lit_used,
                           s.pos_r, s.c_ll, s.c_len, s.c_dist, lane);
            s.pos_r += s.c_ll + s.c_len;
            s.lit_used += s.c_ll;
          }
          seq_commit(s, sh, d->dst_cap);
        }
        if (!s.err && s.have) {
          exec_seq<ZWIN, FENCE>(out, win, s.wfrom, s.lit_base + s.lit_used,
                         s.pos_r, s.c_ll, s.c_len, s.c_dist, lane);
          s.pos_r += s.c_ll + s.c_len;
          s.lit_used += s.c_ll;
        }
        if (lane == 0) {
          seq_err = s.err;
          sh.rep[0] = s.rep0; sh.rep[1] = s.rep1; sh.rep[2] = s.rep2;
          if (!s.err) sh.pos = s.pos_r;
        }
        __syncthreads();
        // trailing literals
        if (lane == 0 && !seq_err) {
          uint64_t rest = sh.lit_len - s.lit_used;
          if (sh.pos + rest > d->dst_cap) {
            seq_err = 1;
          } else {
            sh.a = sh.lit_ptr + s.lit_used;
            sh.b = sh.pos;
            sh.c = rest;
            sh.pos += rest;
          }
        }
        __syncthreads();
        if (seq_err) {
          if (lane == 0) { sh.status = Z_ERR_FORMAT; sh.op = ZOP_ERR; }
          __syncthreads();
          done = true;
        } else {
          const uint8_t* sl = (const uint8_t*)sh.a;
          for (uint64_t k = lane; k < sh.c; k += 64) {
            uint8_t v = sl[k];
            out[sh.b + k] = v;
            win[(sh.b + k) & ZWMASK] = v;
          }
        }
      }
      __syncthreads();
    }

    if (lane == 0) {
      d->written = sh.pos;
      d->status = (sh.op == ZOP_ERR) ? sh.status : Z_OK;
      d->consumed = pc.fr.pos;
      sh.herr = 0;
    }
    __syncthreads();
  }
}

// ------------------- two frames per wave (ILP interleave) ---------------
// The per-sequence decode chain (FSE state -> LDS table entry -> bit
// reads -> next state) is the latency wall on literal/sequence-heavy
// payloads; executing TWO frames' chains from one instruction stream
// lets the scheduler overlap their latencies.  Each frame keeps its own
// ZShared + LDS window; copies for both frames issue back-to-back.
template <int ZWIN, bool FENCE>
__global__ void __launch_bounds__(64, 1)
zstd_kernel_x2(ZstdDesc* __restrict__ descs, int n_streams) {
  constexpr int ZWMASK = ZWIN - 1;
  __shared__ ZShared sh[2];
  __shared__ int seq_err2[2];
  __shared__ int fin[2];
  __shared__ uint64_t win_from[2];
  __shared__ uint8_t win[2][ZWIN];
  int lane = threadIdx.x;
  int n_pairs = (n_streams + 1) / 2;

  for (int pair = blockIdx.x; pair < n_pairs; pair += gridDim.x) {
    ZstdDesc* d[2];
    d[0] = &descs[pair * 2];
    bool hasB = pair * 2 + 1 < n_streams;
    d[1] = hasB ? &descs[pair * 2 + 1] : d[0];
    uint8_t* out[2] = {(uint8_t*)d[0]->dst, (uint8_t*)d[1]->dst};
    uint8_t* wsp[2] = {(uint8_t*)d[0]->ws, (uint8_t*)d[1]->ws};

    ParseState pc[2];
    if (lane == 0) {
      for (int f = 0; f < 2; ++f) {
        pc[f].fr.init((const uint8_t*)d[f]->src, d[f]->src_len);
        pc[f].stage = 0;
        pc[f].frame_checksum = 0;
        pc[f].pending_last = 0;
        sh[f].pos = 0;
        sh[f].status = Z_OK;
        sh[f].op = ZOP_NONE;
        sh[f].herr = 0;
        win_from[f] = 0;
      }
      fin[0] = 0;
      fin[1] = hasB ? 0 : 1;
      if (!hasB) sh[1].op = ZOP_DONE;
    }
    __syncthreads();

    while (!fin[0] || !fin[1]) {
      if (lane == 0) {
        for (int f = 0; f < 2; ++f)
          if (!fin[f] && sh[f].op == ZOP_NONE)
            parse_until_op(pc[f], sh[f], d[f], wsp[f]);
      }
      __syncthreads();

      int op[2] = {fin[0] ? ZOP_NONE : sh[0].op,
                   fin[1] ? ZOP_NONE : sh[1].op};

      // ---- simple ops (COPY/FILL), back-to-back per frame ----------
      for (int f = 0; f < 2; ++f) {
        if (op[f] == ZOP_COPY) {
          const uint8_t* s = (const uint8_t*)sh[f].a;
          for (uint64_t i = lane; i < sh[f].c; i += 64)
            out[f][sh[f].b + i] = s[i];
          if (lane == 0) win_from[f] = sh[f].b + sh[f].c;
        } else if (op[f] == ZOP_FILL) {
          for (uint64_t i = lane; i < sh[f].c; i += 64)
            out[f][sh[f].b + i] = sh[f].fillv;
          if (lane == 0) win_from[f] = sh[f].b + sh[f].c;
        }
      }

      bool litsA = op[0] == ZOP_LITS, litsB = op[1] == ZOP_LITS;
      if (litsA || litsB) {
        // ---- 1) literals: both frames concurrently -----------------
        // raw/RLE copies on half-waves; Huffman streams of frame f on
        // lanes [f*4, f*4+4) — 8 serial decoders running at once
        {
          int f = lane >> 5;
          int sub = lane & 31;
          int fop = (f == 0) ? (litsA ? 1 : 0) : (litsB ? 1 : 0);
          if (fop) {
            ZShared& s = sh[f];
            if (s.hstreams == 0) {
              const uint8_t* src = (const uint8_t*)s.hsrc[0];
              uint8_t* t = (uint8_t*)s.lit_ptr;
              for (uint64_t i = sub; i < s.lit_len; i += 32) t[i] = src[i];
            } else if (s.hstreams == -1) {
              uint8_t* t = (uint8_t*)s.lit_ptr;
              for (uint64_t i = sub; i < s.lit_len; i += 32)
                t[i] = s.fillv;
            }
          }
          int hf = lane >> 2, hs = lane & 3;
          if (hf < 2 && ((hf == 0 && litsA) || (hf == 1 && litsB))
              && sh[hf].hstreams > 0 && hs < sh[hf].hstreams)
            huf_stream_decode(sh[hf], hs);
        }
        __syncthreads();

        // ---- 2) sequences, two chains interleaved ------------------
        SeqCtx A, B;
        if (litsA) seq_init(A, sh[0], win_from[0]);
        if (litsB) seq_init(B, sh[1], win_from[1]);
        bool contA = litsA && A.nseq > 0 && !A.err;
        bool contB = litsB && B.nseq > 0 && !B.err;
        while (contA || contB) {
          if (contA) seq_decode(A, sh[0]);
          if (contB) seq_decode(B, sh[1]);
          if (contA && !A.err) {
            if (A.have) {
              exec_seq<ZWIN, FENCE>(out[0], win[0], A.wfrom,
                             A.lit_base + A.lit_used, A.pos_r,
                             A.c_ll, A.c_len, A.c_dist, lane);
              A.pos_r += A.c_ll + A.c_len;
              A.lit_used += A.c_ll;
            }
            seq_commit(A, sh[0], d[0]->dst_cap);
          }
          if (contB && !B.err) {
            if (B.have) {
              exec_seq<ZWIN, FENCE>(out[1], win[1], B.wfrom,
                             B.lit_base + B.lit_used, B.pos_r,
                             B.c_ll, B.c_len, B.c_dist, lane);
              B.pos_r += B.c_ll + B.c_len;
              B.lit_used += B.c_ll;
            }
            seq_commit(B, sh[1], d[1]->dst_cap);
          }
          if (contA) { ++A.i; contA = A.i < A.nseq && !A.err; }
          if (contB) { ++B.i; contB = B.i < B.nseq && !B.err; }
        }
        // drain the last pending sequence of each frame
        if (litsA && !A.err && A.have) {
          exec_seq<ZWIN, FENCE>(out[0], win[0], A.wfrom,
                         A.lit_base + A.lit_used, A.pos_r,
                         A.c_ll, A.c_len, A.c_dist, lane);
          A.pos_r += A.c_ll + A.c_len;
          A.lit_used += A.c_ll;
        }
        if (litsB && !B.err && B.have) {
          exec_seq<ZWIN, FENCE>(out[1], win[1], B.wfrom,
                         B.lit_base + B.lit_used, B.pos_r,
                         B.c_ll, B.c_len, B.c_dist, lane);
          B.pos_r += B.c_ll + B.c_len;
          B.lit_used += B.c_ll;
        }
        if (lane == 0) {
          if (litsA) {
            seq_err2[0] = A.err;
            sh[0].rep[0] = A.rep0; sh[0].rep[1] = A.rep1;
            sh[0].rep[2] = A.rep2;
            if (!A.err) sh[0].pos = A.pos_r;
          }
          if (litsB) {
            seq_err2[1] = B.err;
            sh[1].rep[0] = B.rep0; sh[1].rep[1] = B.rep1;
            sh[1].rep[2] = B.rep2;
            if (!B.err) sh[1].pos = B.pos_r;
          }
          // trailing literals staging
          if (litsA && !seq_err2[0]) {
            uint64_t rest = sh[0].lit_len - A.lit_used;
            if (sh[0].pos + rest > d[0]->dst_cap) seq_err2[0] = 1;
            else {
              sh[0].a = sh[0].lit_ptr + A.lit_used;
              sh[0].b = sh[0].pos;
              sh[0].c = rest;
              sh[0].pos += rest;
            }
          }
          if (litsB && !seq_err2[1]) {
            uint64_t rest = sh[1].lit_len - B.lit_used;
            if (sh[1].pos + rest > d[1]->dst_cap) seq_err2[1] = 1;
            else {
              sh[1].a = sh[1].lit_ptr + B.lit_used;
              sh[1].b = sh[1].pos;
              sh[1].c = rest;
              sh[1].pos += rest;
            }
          }
        }
        __syncthreads();
        // trailing literal copies (half-wave per frame) or error out
        {
          int f = lane >> 5;
          int sub = lane & 31;
          bool active = (f == 0) ? litsA : litsB;
          if (active && !seq_err2[f]) {
            const uint8_t* sl = (const uint8_t*)sh[f].a;
            for (uint64_t k = sub; k < sh[f].c; k += 32) {
              uint8_t v = sl[k];
              out[f][sh[f].b + k] = v;
              win[f][(sh[f].b + k) & ZWMASK] = v;
            }
          }
        }
        if (lane == 0) {
          if (litsA && seq_err2[0]) {
            sh[0].status = Z_ERR_FORMAT; sh[0].op = ZOP_ERR;
          }
          if (litsB && seq_err2[1]) {
            sh[1].status = Z_ERR_FORMAT; sh[1].op = ZOP_ERR;
          }
        }
        __syncthreads();
        op[0] = fin[0] ? ZOP_NONE : sh[0].op;
        op[1] = fin[1] ? ZOP_NONE : sh[1].op;
      }

      // ---- end of round: consume ops / mark finished frames --------
      if (lane == 0) {
        for (int f = 0; f < 2; ++f) {
          if (fin[f]) continue;
          if (op[f] == ZOP_DONE || op[f] == ZOP_ERR) fin[f] = 1;
          else sh[f].op = ZOP_NONE;
        }
      }
      __syncthreads();
    }

    if (lane == 0) {
      d[0]->written = sh[0].pos;
      d[0]->status = (sh[0].op == ZOP_ERR) ? sh[0].status : Z_OK;
      d[0]->consumed = pc[0].fr.pos;
      if (hasB) {
        d[1]->written = sh[1].pos;
        d[1]->status = (sh[1].op == ZOP_ERR) ? sh[1].status : Z_OK;
        d[1]->consumed = pc[1].fr.pos;
      }
      sh[0].herr = sh[1].herr = 0;
    }
    __syncthreads();
  }
}

}  // namespace

extern "C" void launch_zstd_frames(const uint64_t* desc, int n_frames,
                                   int* /*unused*/, hipStream_t stream,
                                   int window /*0=auto,16384,65536*/) {
  if (n_frames <= 0) return;
  // both windows are CORRECT for any match distance (exec_seq falls
  // back to a global read past the LDS window); the choice is pure
  // occupancy vs far-match speed.  Callers that run MANY CONCURRENT
  // small launches (per-shard dataset decode) pass 16384 explicitly:
  // LDS is shared chip-wide, so 64 KiB windows would cap residency at
  // ~2 workgroups/CU across all their launches combined.
  bool small = window == 16 * 1024 || (window == 0 && n_frames >= 768);
  // DEMODEL_ZSTD_MODE: "x1" (default) / "x2" (paired interleave),
  // optional "nf" suffix drops the LDS exec fences ("x1nf", "x2nf").
  // Measured on MI355X (profiles/zstd_x2_ab.md): x2 LOSES on word
  // salad (5.9 vs 8.9 GB/s) — the doubled live state parks in AGPRs
  // and occupancy halves — so x1 stays the default.
  const char* mode = getenv("DEMODEL_ZSTD_MODE");
  bool use_x2 = mode && strstr(mode, "x2");
  bool fence = !(mode && strstr(mode, "nf"));
  // "w8": 8 KiB window — LDS/WG drops 28.9->20.7 KB so CU residency
  // rises 5->7 workgroups; pays more far-match fallbacks.  A/B lever.
  bool w8 = (mode && strstr(mode, "w8")) || window == 8 * 1024;
  int blocks = n_frames < 4096 ? n_frames : 4096;
  if (w8 && small && !use_x2) {
    if (fence)
      hipLaunchKernelGGL((zstd_kernel<8 * 1024, true>), dim3(blocks),
                         dim3(64), 0, stream, (ZstdDesc*)desc, n_frames);
    else
      hipLaunchKernelGGL((zstd_kernel<8 * 1024, false>), dim3(blocks),
                         dim3(64), 0, stream, (ZstdDesc*)desc, n_frames);
  } else if (use_x2 && small && n_frames >= 2) {
    int pairs = (n_frames + 1) / 2;
    int pblocks = pairs < 4096 ? pairs : 4096;
    if (fence)
      hipLaunchKernelGGL((zstd_kernel_x2<16 * 1024, true>), dim3(pblocks),
                         dim3(64), 0, stream, (ZstdDesc*)desc, n_frames);
    else
      hipLaunchKernelGGL((zstd_kernel_x2<16 * 1024, false>), dim3(pblocks),
                         dim3(64), 0, stream, (ZstdDesc*)desc, n_frames);
  } else if (small) {
    if (fence)
      hipLaunchKernelGGL((zstd_kernel<16 * 1024, true>), dim3(blocks),
                         dim3(64), 0, stream, (ZstdDesc*)desc, n_frames);
    else
      hipLaunchKernelGGL((zstd_kernel<16 * 1024, false>), dim3(blocks),
                         dim3(64), 0, stream, (ZstdDesc*)desc, n_frames);
  } else {
    if (fence)
      hipLaunchKernelGGL((zstd_kernel<64 * 1024, true>), dim3(blocks),
                         dim3(64), 0, stream, (ZstdDesc*)desc, n_frames);
    else
      hipLaunchKernelGGL((zstd_kernel<64 * 1024, false>), dim3(blocks),
                         dim3(64), 0, stream, (ZstdDesc*)desc, n_frames);
  }
}
