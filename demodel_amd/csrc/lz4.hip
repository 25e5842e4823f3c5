// CDNA4 LZ4 raw-block decompressor — wave-per-stream, LDS output window.
//
// Parquet's LZ4 / LZ4_RAW page codecs (what pyarrow writes for
// compression="lz4") are the raw LZ4 BLOCK format (lz4_Block_format.md):
//   sequence: token byte
//     high nibble = literal length (15 => +255-extension bytes)
//     literals follow
//     u16 LE match offset (1..65535)
//     low nibble = match length - 4 (15 => +255-extension bytes)
//   the final sequence is literals-only (stream ends after them).
//
// Execution shape shared with snappy.hip/zstd_kernel.hip: all 64 lanes
// run the byte-serial decode REDUNDANTLY in lockstep on register state
// (uniform control flow; token/length loads coalesce to one address),
// the wave executes each literal/match copy cooperatively, and a
// 16 KiB LDS window serves near back-references (offsets are <= 64 KiB
// by format; farther-than-window ones take the global-read path behind
// a vmcnt drain).

#include <hip/hip_runtime.h>

namespace {

enum {
  LZ4_OK = 0,
  LZ4_ERR_FORMAT = -2,
  LZ4_ERR_OVERFLOW = -3,
  LZ4_ERR_UNDERRUN = -4,
};

struct __align__(16) Lz4Desc {
  uint64_t src;
  uint64_t src_len;
  uint64_t dst;
  uint64_t dst_cap;
  uint64_t written;   // out
  int64_t status;     // out
  uint64_t consumed;  // out
  uint64_t _pad1;
};

constexpr int LWIN = 16 * 1024;
constexpr int LWMASK = LWIN - 1;

__global__ void __launch_bounds__(64, 2)
lz4_kernel(Lz4Desc* __restrict__ descs, int n_streams) {
  __shared__ uint8_t win[LWIN];
  int lane = threadIdx.x;

  for (int sidx = blockIdx.x; sidx < n_streams; sidx += gridDim.x) {
    Lz4Desc* d = &descs[sidx];
    const uint8_t* src = (const uint8_t*)d->src;
    uint8_t* out = (uint8_t*)d->dst;
    uint64_t slen = d->src_len;
    uint64_t cap = d->dst_cap;

    // every lane holds the same decode state (redundant lockstep)
    uint64_t sp = 0;
    uint64_t op = 0;
    int err = 0;

    while (!err && sp < slen) {
      uint8_t token = src[sp++];
      // ---- literals --------------------------------------------------
      uint64_t lit = token >> 4;
      if (lit == 15) {
        while (true) {
          if (sp >= slen) { err = LZ4_ERR_UNDERRUN; break; }
          uint8_t b = src[sp++];
          lit += b;
          if (b != 255) break;
          if (lit > (1ull << 40)) { err = LZ4_ERR_FORMAT; break; }
        }
        if (err) break;
      }
      if (sp + lit > slen) { err = LZ4_ERR_UNDERRUN; break; }
      if (op + lit > cap) { err = LZ4_ERR_OVERFLOW; break; }
      for (uint64_t k = lane; k < lit; k += 64) {
        uint8_t v = src[sp + k];
        out[op + k] = v;
        win[(op + k) & LWMASK] = v;
      }
      sp += lit;
      op += lit;
      if (sp >= slen) break;  // last sequence: literals only

      // ---- match -----------------------------------------------------
      if (sp + 2 > slen) { err = LZ4_ERR_UNDERRUN; break; }
      uint64_t dist = (uint64_t)src[sp] | ((uint64_t)src[sp + 1] << 8);
      sp += 2;
      uint64_t mlen = (token & 0xF);
      if (mlen == 15) {
        while (true) {
          if (sp >= slen) { err = LZ4_ERR_UNDERRUN; break; }
          uint8_t b = src[sp++];
          mlen += b;
          if (b != 255) break;
          if (mlen > (1ull << 40)) { err = LZ4_ERR_FORMAT; break; }
        }
        if (err) break;
      }
      mlen += 4;  // minmatch
      if (dist == 0 || dist > op) { err = LZ4_ERR_FORMAT; break; }
      if (op + mlen > cap) { err = LZ4_ERR_OVERFLOW; break; }
      // every output byte passes through win[], so slot p & LWMASK is
      // valid for p in [op-LWIN, op)
      if (dist <= LWIN - 128) {
        asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        if (dist >= mlen) {
          for (uint64_t k = lane; k < mlen; k += 64) {
            uint8_t v = win[(op + k - dist) & LWMASK];
            out[op + k] = v;
            win[(op + k) & LWMASK] = v;
          }
        } else {
          uint64_t copied = 0;
          while (copied < mlen) {
            uint64_t n = dist < mlen - copied ? dist : mlen - copied;
            for (uint64_t k = lane; k < n; k += 64) {
              uint8_t v = win[(op + copied + k - dist) & LWMASK];
              out[op + copied + k] = v;
              win[(op + copied + k) & LWMASK] = v;
            }
            asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
            copied += n;
          }
        }
      } else {
        // far match: read old output from HBM; drain our stores first
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        if (dist >= mlen) {
          for (uint64_t k = lane; k < mlen; k += 64) {
            uint8_t v = out[op + k - dist];
            out[op + k] = v;
            win[(op + k) & LWMASK] = v;
          }
        } else {
          uint64_t copied = 0;
          while (copied < mlen) {
            uint64_t n = dist < mlen - copied ? dist : mlen - copied;
            for (uint64_t k = lane; k < n; k += 64) {
              uint8_t v = out[op + copied + k - dist];
              out[op + copied + k] = v;
              win[(op + copied + k) & LWMASK] = v;
            }
            asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
            copied += n;
          }
        }
      }
      op += mlen;
    }

    if (lane == 0) {
      d->written = op;
      d->status = err;
      d->consumed = sp;
    }
    __syncthreads();
  }
}

}  // namespace

extern "C" void launch_lz4_streams(const uint64_t* desc, int n_streams,
                                   hipStream_t stream) {
  if (n_streams <= 0) return;
  int blocks = n_streams < 4096 ? n_streams : 4096;
  hipLaunchKernelGGL(lz4_kernel, dim3(blocks), dim3(64), 0, stream,
                     (Lz4Desc*)desc, n_streams);
}
