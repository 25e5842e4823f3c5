// CDNA4 Zstandard (RFC 8878) frame decoder.
// Decode primitives live in zstd_common.h (shared with the host reference
// decoder in zstd_host.cpp); this file is the wave-per-frame GPU
// orchestration: lane 0 parses frames/blocks and decodes entropy streams,
// the wave64 executes literal/match/stored copies, and the 4-stream
// Huffman literals section decodes on 4 lanes concurrently.

#include <hip/hip_runtime.h>

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

// one sequence's copies: literal run from the (global) literal buffer,
// then the match via the LDS window (or the rare far-global path)
template <int ZWIN>
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
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
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
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        copied += n;
      }
    }
  } else {
    // far match: read old output from HBM; drain our stores first
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    for (uint32_t k = lane; k < len; k += 64) {
      uint8_t v = out[mp + k - dist];
      out[mp + k] = v;
      win[(mp + k) & ZWMASK] = v;
    }
  }
}

// Window size trades per-wave speed (bigger window = fewer far matches)
// against occupancy (LDS-limited resident workgroups); the launcher picks
// 16 KiB when many frames provide parallelism, 64 KiB for few frames.
// second launch_bounds arg = waves/SIMD: LDS (window + tables) already
// caps residency at ~5 (16K) / 2 (64K) workgroups per CU, so granting the
// full register budget is free — without it the compiler spills the
// sequence loop into AGPRs/scratch (1626 v_accvgpr_read in the ISA).
template <int ZWIN>
__global__ void __launch_bounds__(64, ZWIN <= 16 * 1024 ? 2 : 1)
zstd_kernel(ZstdDesc* __restrict__ descs, int n_streams) {
  constexpr int ZWMASK = ZWIN - 1;
  __shared__ ZShared sh;
  __shared__ SeqRec cur;
  __shared__ int seq_err;
  __shared__ uint64_t win_from;   // output pos from which win[] is valid
  __shared__ uint8_t win[ZWIN];
  int lane = threadIdx.x;

  for (int sidx = blockIdx.x; sidx < n_streams; sidx += gridDim.x) {
    ZstdDesc* d = &descs[sidx];
    const uint8_t* src = (const uint8_t*)d->src;
    uint8_t* out = (uint8_t*)d->dst;
    uint8_t* ws = (uint8_t*)d->ws;
    uint8_t* lit_ws = ws;                    // 140 KiB literal buffer
    (void)lit_ws;

    // lane-0 persistent parse state (registers)
    FReader fr;
    int stage = 0;          // 0 = expect frame magic, 1 = expect block hdr
    int frame_checksum = 0;
    int pending_last = 0;   // a wave op for the frame's final block is out

    if (lane == 0) {
      fr.init(src, d->src_len);
      sh.pos = 0;
      sh.status = Z_OK;
      sh.op = ZOP_NONE;
      sh.herr = 0;
      win_from = 0;
    }
    __syncthreads();

    bool done = false;
    while (!done) {
      // ---------------- lane 0: parse until a wave op ------------------
      if (lane == 0) {
        sh.op = ZOP_NONE;
        while (sh.op == ZOP_NONE) {
          if (pending_last) {
            pending_last = 0;
            if (frame_checksum) fr.pos += 4;  // xxh64 low32, not verified
            stage = 0;
          }
          if (stage == 0) {
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
            frame_checksum = (fhd >> 2) & 1;
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
            stage = 1;
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
            pending_last = last;
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
            pending_last = last;
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
          pending_last = last;
          sh.op = ZOP_LITS;
          break;
        }
      }
      __syncthreads();

      int op = sh.op;
      if (op == ZOP_DONE || op == ZOP_ERR) {
        done = true;
      } else if (op == ZOP_COPY) {
        const uint8_t* s = (const uint8_t*)sh.a;
        for (uint64_t i = lane; i < sh.c; i += 64) out[sh.b + i] = s[i];
        if (lane == 0) win_from = sh.b + sh.c;  // window gap over this block
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
          BBits bb;
          bb.init((const uint8_t*)sh.hsrc[lane],
                  (int64_t)sh.hsrc_len[lane]);
          uint8_t* t = (uint8_t*)sh.hdst[lane];
          uint64_t want = sh.hdst_len[lane];
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
        __syncthreads();

        // ---- 2) sequences: every lane decodes redundantly -----------
        // All 64 lanes run the identical FSE decode in lockstep on their
        // own register state (LDS table reads broadcast, bitstream reads
        // coalesce to one address) — zero cross-lane publication, zero
        // waits in the per-sequence loop; matches read the LDS window.
        __syncthreads();
        uint32_t nseq = sh.n_seqs;
        int err = sh.herr;
        BBits sq;
        sq.init((const uint8_t*)sh.a, (int64_t)sh.c);
        uint32_t ll_state = 0, of_state = 0, ml_state = 0;
        uint32_t rep0 = sh.rep[0], rep1 = sh.rep[1], rep2 = sh.rep[2];
        if (nseq) {
          if (sq.fail) err = 1;
          ll_state = sq.get(sh.ll_t.log);
          of_state = sq.get(sh.of_t.log);
          ml_state = sq.get(sh.ml_t.log);
        }
        uint64_t pos_r = sh.pos;       // per-lane replicated position
        uint64_t wfrom = win_from;     // uniform: only changes between ops
        uint64_t lit_used = 0;
        const uint8_t* lit_base = (const uint8_t*)sh.lit_ptr;
        // software-pipelined: decode sequence i+1 BEFORE executing the
        // copies of sequence i, so the decode's LDS/bit latency chain
        // overlaps the literal run's global load latency.
        uint32_t c_ll = 0, c_len = 0, c_dist = 0;
        bool have = false;
        for (uint32_t i = 0; i < nseq && !err; ++i) {
          // one LDS dword per table entry: exec_seq's asm memory
          // clobbers would otherwise force re-loads of sym/base/nbits
          FseEntry oe = sh.of_t.e[of_state];
          FseEntry me = sh.ml_t.e[ml_state];
          FseEntry le = sh.ll_t.e[ll_state];
          uint8_t ofc = oe.sym, mlc = me.sym, llc = le.sym;
          if (ofc > 31 || mlc > 52 || llc > 35) { err = 1; break; }
          uint32_t ofv = (1u << ofc) + sq.get(ofc);
          // ml+ll extras in ONE bit read (<=32 bits): get(a)+get(b)
          // composes as get(a+b) with the first read in the high bits
          int mlx = kMLExtra[mlc], llx = kLLExtra[llc];
          uint32_t ex = sq.get(mlx + llx);
          uint32_t n_len = kMLBase[mlc] + (ex >> llx);
          uint32_t n_ll = kLLBase[llc]
                          + (ex & ((llx < 32) ? ((1u << llx) - 1)
                                              : 0xFFFFFFFFu));
          uint32_t n_dist;
          if (ofv > 3) {
            n_dist = ofv - 3;
            rep2 = rep1; rep1 = rep0; rep0 = n_dist;
          } else {
            uint32_t idx = ofv + (n_ll == 0 ? 1 : 0);
            if (idx == 1) {
              n_dist = rep0;
            } else if (idx == 2) {
              n_dist = rep1;
              rep1 = rep0; rep0 = n_dist;
            } else if (idx == 3) {
              n_dist = rep2;
              rep2 = rep1; rep1 = rep0; rep0 = n_dist;
            } else {
              n_dist = rep0 - 1;
              if (n_dist == 0 || rep0 == 0) { err = 1; break; }
              rep2 = rep1; rep1 = rep0; rep0 = n_dist;
            }
          }
          if (i + 1 < nseq) {
            // the three state refreshes (<=9 bits each, ll|ml|of order)
            // in ONE bit read: one refill check instead of three
            int b_ll = le.nbits, b_ml = me.nbits, b_of = oe.nbits;
            uint32_t bits = sq.get(b_ll + b_ml + b_of);
            of_state = oe.base + (bits & ((1u << b_of) - 1));
            ml_state = me.base + ((bits >> b_of) & ((1u << b_ml) - 1));
            ll_state = le.base + (bits >> (b_of + b_ml));
          }
          // execute the PREVIOUS sequence while this decode's loads land
          if (have) {
            exec_seq<ZWIN>(out, win, wfrom, lit_base + lit_used, pos_r,
                           c_ll, c_len, c_dist, lane);
            pos_r += c_ll + c_len;
            lit_used += c_ll;
          }
          // bounds for the NEW sequence (against post-exec position)
          if (lit_used + n_ll > sh.lit_len ||
              pos_r + n_ll + (uint64_t)n_len > d->dst_cap ||
              (uint64_t)n_dist > pos_r + n_ll) {
            err = 1;
            break;
          }
          c_ll = n_ll; c_len = n_len; c_dist = n_dist;
          have = true;
        }
        // drain the last decoded sequence
        if (!err && have) {
          exec_seq<ZWIN>(out, win, wfrom, lit_base + lit_used, pos_r,
                         c_ll, c_len, c_dist, lane);
          pos_r += c_ll + c_len;
          lit_used += c_ll;
        }
        if (lane == 0) {
          seq_err = err;
          sh.rep[0] = rep0; sh.rep[1] = rep1; sh.rep[2] = rep2;
        }
        if (lane == 0 && !err) sh.pos = pos_r;
        __syncthreads();
        // trailing literals
        if (lane == 0 && !seq_err) {
          uint64_t rest = sh.lit_len - lit_used;
          if (sh.pos + rest > d->dst_cap) {
            seq_err = 1;
          } else {
            sh.a = sh.lit_ptr + lit_used;
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
          const uint8_t* s = (const uint8_t*)sh.a;
          for (uint64_t k = lane; k < sh.c; k += 64) {
            uint8_t v = s[k];
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
      d->consumed = fr.pos;
      sh.herr = 0;
    }
    __syncthreads();
  }
}

}  // namespace

extern "C" void launch_zstd_frames(const uint64_t* desc, int n_frames,
                                   int* /*unused*/, hipStream_t stream,
                                   int window /*0=auto,16384,65536*/) {
  if (n_frames <= 0) return;
  int blocks = n_frames < 4096 ? n_frames : 4096;
  // both windows are CORRECT for any match distance (exec_seq falls
  // back to a global read past the LDS window); the choice is pure
  // occupancy vs far-match speed.  Callers that run MANY CONCURRENT
  // small launches (per-shard dataset decode) pass 16384 explicitly:
  // LDS is shared chip-wide, so 64 KiB windows would cap residency at
  // ~2 workgroups/CU across all their launches combined.
  bool small = window == 16 * 1024 || (window == 0 && n_frames >= 768);
  if (small) {
    hipLaunchKernelGGL(zstd_kernel<16 * 1024>, dim3(blocks), dim3(64), 0,
                       stream, (ZstdDesc*)desc, n_frames);
  } else {
    hipLaunchKernelGGL(zstd_kernel<64 * 1024>, dim3(blocks), dim3(64), 0,
                       stream, (ZstdDesc*)desc, n_frames);
  }
}
