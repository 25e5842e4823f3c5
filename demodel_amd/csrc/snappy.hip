// CDNA4 Snappy decompressor — wave-per-stream, LDS output window.
//
// Snappy is parquet's DEFAULT page codec (real HF datasets like c4-en
// ship snappy pages), so the GPU page-decompression path needs it next
// to zstd (csrc/zstd_kernel.hip).  The format (google/snappy
// format_description.txt) is byte-oriented LZ77 with no entropy stage:
//   preamble: uncompressed length as varint
//   elements: tag byte, low 2 bits select
//     00 literal    len = (tag>>2)+1, or 60..63 => next 1..4 LE bytes +1
//     01 copy1      len = ((tag>>2)&7)+4,  offset = ((tag>>5)<<8)|byte
//     10 copy2      len = (tag>>2)+1,      offset = u16 LE
//     11 copy4      len = (tag>>2)+1,      offset = u32 LE
//
// Same execution shape as the zstd kernel: all 64 lanes run the decode
// redundantly in lockstep on register state (uniform control flow, the
// tag/byte loads coalesce to one address), the wave executes each
// literal/copy cooperatively, and a 16 KiB LDS window serves near
// back-references (snappy encoders emit offsets < 64 KiB; farther ones
// take the global-read path behind a vmcnt drain).

#include <hip/hip_runtime.h>

namespace {

enum {
  SNAP_OK = 0,
  SNAP_ERR_FORMAT = -2,
  SNAP_ERR_OVERFLOW = -3,
  SNAP_ERR_UNDERRUN = -4,
};

struct __align__(16) SnappyDesc {
  uint64_t src;
  uint64_t src_len;
  uint64_t dst;
  uint64_t dst_cap;
  uint64_t written;   // out
  int64_t status;     // out
  uint64_t consumed;  // out
  uint64_t _pad1;
};

constexpr int SWIN = 16 * 1024;
constexpr int SWMASK = SWIN - 1;

__global__ void __launch_bounds__(64, 2)
snappy_kernel(SnappyDesc* __restrict__ descs, int n_streams) {
  __shared__ uint8_t win[SWIN];
  int lane = threadIdx.x;

  for (int sidx = blockIdx.x; sidx < n_streams; sidx += gridDim.x) {
    SnappyDesc* d = &descs[sidx];
    const uint8_t* src = (const uint8_t*)d->src;
    uint8_t* out = (uint8_t*)d->dst;
    uint64_t slen = d->src_len;

    // every lane holds the same decode state (redundant lockstep)
    uint64_t sp = 0;   // src position
    uint64_t op = 0;   // out position
    int err = 0;

    // preamble varint: uncompressed length
    uint64_t ulen = 0;
    int shift = 0;
    while (true) {
      if (sp >= slen || shift > 35) { err = SNAP_ERR_FORMAT; break; }
      uint8_t b = src[sp++];
      ulen |= (uint64_t)(b & 0x7F) << shift;
      shift += 7;
      if (!(b & 0x80)) break;
    }
    if (!err && ulen > d->dst_cap) err = SNAP_ERR_OVERFLOW;

    while (!err && sp < slen) {
      uint8_t tag = src[sp++];
      uint32_t kind = tag & 3;
      if (kind == 0) {
        // ---- literal -------------------------------------------------
        uint64_t len = (tag >> 2) + 1;
        if (len > 60) {
          uint32_t nb = (uint32_t)len - 60;     // 1..4 length bytes
          if (sp + nb > slen) { err = SNAP_ERR_UNDERRUN; break; }
          uint64_t v = 0;
          for (uint32_t k = 0; k < nb; ++k)
            v |= (uint64_t)src[sp + k] << (8 * k);
          sp += nb;
          len = v + 1;
        }
        if (sp + len > slen) { err = SNAP_ERR_UNDERRUN; break; }
        if (op + len > ulen) { err = SNAP_ERR_OVERFLOW; break; }
        for (uint64_t k = lane; k < len; k += 64) {
          uint8_t v = src[sp + k];
          out[op + k] = v;
          win[(op + k) & SWMASK] = v;
        }
        sp += len;
        op += len;
      } else {
        // ---- copy ----------------------------------------------------
        uint64_t len, dist;
        if (kind == 1) {
          if (sp >= slen) { err = SNAP_ERR_UNDERRUN; break; }
          len = ((tag >> 2) & 7) + 4;
          dist = ((uint64_t)(tag >> 5) << 8) | src[sp];
          sp += 1;
        } else if (kind == 2) {
          if (sp + 2 > slen) { err = SNAP_ERR_UNDERRUN; break; }
          len = (tag >> 2) + 1;
          dist = (uint64_t)src[sp] | ((uint64_t)src[sp + 1] << 8);
          sp += 2;
        } else {
          if (sp + 4 > slen) { err = SNAP_ERR_UNDERRUN; break; }
          len = (tag >> 2) + 1;
          dist = (uint64_t)src[sp] | ((uint64_t)src[sp + 1] << 8) |
                 ((uint64_t)src[sp + 2] << 16) |
                 ((uint64_t)src[sp + 3] << 24);
          sp += 4;
        }
        if (dist == 0 || dist > op) { err = SNAP_ERR_FORMAT; break; }
        if (op + len > ulen) { err = SNAP_ERR_OVERFLOW; break; }
        // every output byte passes through win[], so slot p & SWMASK is
        // valid for p in [op-SWIN, op): the LDS path needs only
        // dist <= SWIN (margin for the same-iteration write ordering)
        if (dist <= SWIN - 128) {
          // near match via the LDS window
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
          if (dist >= len) {
            for (uint64_t k = lane; k < len; k += 64) {
              uint8_t v = win[(op + k - dist) & SWMASK];
              out[op + k] = v;
              win[(op + k) & SWMASK] = v;
            }
          } else {
            uint64_t copied = 0;
            while (copied < len) {
              uint64_t n = dist < len - copied ? dist : len - copied;
              for (uint64_t k = lane; k < n; k += 64) {
                uint8_t v = win[(op + copied + k - dist) & SWMASK];
                out[op + copied + k] = v;
                win[(op + copied + k) & SWMASK] = v;
              }
              asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
              copied += n;
            }
          }
        } else {
          // far match: read old output from HBM; drain our stores first
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
          if (dist >= len) {
            for (uint64_t k = lane; k < len; k += 64) {
              uint8_t v = out[op + k - dist];
              out[op + k] = v;
              win[(op + k) & SWMASK] = v;
            }
          } else {
            uint64_t copied = 0;
            while (copied < len) {
              uint64_t n = dist < len - copied ? dist : len - copied;
              for (uint64_t k = lane; k < n; k += 64) {
                uint8_t v = out[op + copied + k - dist];
                out[op + copied + k] = v;
                win[(op + copied + k) & SWMASK] = v;
              }
              asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
              copied += n;
            }
          }
        }
        op += len;
      }
    }
    if (!err && op != ulen) err = SNAP_ERR_UNDERRUN;

    if (lane == 0) {
      d->written = op;
      d->status = err;
      d->consumed = sp;
    }
    __syncthreads();
  }
}

}  // namespace

extern "C" void launch_snappy_streams(const uint64_t* desc, int n_streams,
                                      hipStream_t stream) {
  if (n_streams <= 0) return;
  int blocks = n_streams < 4096 ? n_streams : 4096;
  hipLaunchKernelGGL(snappy_kernel, dim3(blocks), dim3(64), 0, stream,
                     (SnappyDesc*)desc, n_streams);
}
