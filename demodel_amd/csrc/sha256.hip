// CDNA4 (gfx950) SHA-256 kernels for blob verification.
//
// Two shapes for two jobs (the digest-as-identity
// semantics come from the reference's Ollama manifest capture,
// CONTRIBUTING.md:133-149):
//
// 1. sha256_batch — lane-per-chunk: every thread owns one independent
//    chunk and grinds its own chain.  SHA-256 is strictly sequential in
//    its 64-byte blocks, so single-chain GPU hashing is latency-bound;
//    aggregate throughput comes from thousands of concurrent chains.
//    This is the engine's fast verify path: blobs carry a sidecar of
//    per-64KiB-chunk digests (cache/store.py), so a landed HBM blob is
//    re-verified chunk-parallel at memory-bandwidth-class rates.
//
// 2. sha256_chain — the exact whole-blob digest (what upstream etags and
//    Ollama layer digests are).  One sequential chain; the kernel exists
//    so the full verify can run on-device without a D2H of the blob, and
//    is pipelined per landed chunk.  It is honest about physics: a single
//    chain has no parallelism, so the fast path above is the default.
//
// Wave64 notes: no cross-lane traffic at all in the batch kernel — each
// lane is an independent hash engine, the ideal CDNA shape for this op.

#include <hip/hip_runtime.h>

namespace {

__constant__ uint32_t K256[64] = {
    0x428a2f98, 0x71374491, 0xb5c0fbcf, 0xe9b5dba5, 0x3956c25b, 0x59f111f1,
    0x923f82a4, 0xab1c5ed5, 0xd807aa98, 0x12835b01, 0x243185be, 0x550c7dc3,
    0x72be5d74, 0x80deb1fe, 0x9bdc06a7, 0xc19bf174, 0xe49b69c1, 0xefbe4786,
    0x0fc19dc6, 0x240ca1cc, 0x2de92c6f, 0x4a7484aa, 0x5cb0a9dc, 0x76f988da,
    0x983e5152, 0xa831c66d, 0xb00327c8, 0xbf597fc7, 0xc6e00bf3, 0xd5a79147,
    0x06ca6351, 0x14292967, 0x27b70a85, 0x2e1b2138, 0x4d2c6dfc, 0x53380d13,
    0x650a7354, 0x766a0abb, 0x81c2c92e, 0x92722c85, 0xa2bfe8a1, 0xa81a664b,
    0xc24b8b70, 0xc76c51a3, 0xd192e819, 0xd6990624, 0xf40e3585, 0x106aa070,
    0x19a4c116, 0x1e376c08, 0x2748774c, 0x34b0bcb5, 0x391c0cb3, 0x4ed8aa4a,
    0x5b9cca4f, 0x682e6ff3, 0x748f82ee, 0x78a5636f, 0x84c87814, 0x8cc70208,
    0x90befffa, 0xa4506ceb, 0xbef9a3f7, 0xc67178f2};

__device__ __forceinline__ uint32_t rotr(uint32_t x, int n) {
  return __builtin_rotateright32(x, n);
}

__device__ __forceinline__ uint32_t bswap32(uint32_t v) {
  return __builtin_bswap32(v);
}

// one compression round set over a prepared 16-word schedule seed
__device__ void sha256_compress(uint32_t state[8], const uint32_t block[16]) {
  uint32_t w[16];
#pragma unroll
  for (int i = 0; i < 16; ++i) w[i] = block[i];
  uint32_t a = state[0], b = state[1], c = state[2], d = state[3];
  uint32_t e = state[4], f = state[5], g = state[6], h = state[7];
#pragma unroll
  for (int t = 0; t < 64; ++t) {
    uint32_t wt;
    if (t < 16) {
      wt = w[t];
    } else {
      uint32_t w15 = w[(t - 15) & 15], w2 = w[(t - 2) & 15];
      uint32_t s0 = rotr(w15, 7) ^ rotr(w15, 18) ^ (w15 >> 3);
      uint32_t s1 = rotr(w2, 17) ^ rotr(w2, 19) ^ (w2 >> 10);
      wt = w[t & 15] = w[t & 15] + s0 + w[(t - 7) & 15] + s1;
    }
    uint32_t S1 = rotr(e, 6) ^ rotr(e, 11) ^ rotr(e, 25);
    uint32_t ch = (e & f) ^ (~e & g);
    uint32_t t1 = h + S1 + ch + K256[t] + wt;
    uint32_t S0 = rotr(a, 2) ^ rotr(a, 13) ^ rotr(a, 22);
    uint32_t maj = (a & b) ^ (a & c) ^ (b & c);
    uint32_t t2 = S0 + maj;
    h = g; g = f; f = e; e = d + t1;
    d = c; c = b; b = a; a = t1 + t2;
  }
  state[0] += a; state[1] += b; state[2] += c; state[3] += d;
  state[4] += e; state[5] += f; state[6] += g; state[7] += h;
}

__device__ __forceinline__ void sha256_init_state(uint32_t s[8]) {
  s[0] = 0x6a09e667; s[1] = 0xbb67ae85; s[2] = 0x3c6ef372;
  s[3] = 0xa54ff53a; s[4] = 0x510e527f; s[5] = 0x9b05688c;
  s[6] = 0x1f83d9ab; s[7] = 0x5be0cd19;
}

// full SHA-256 of [p, p+len); p need not be aligned (u32 fast path when it
// is — chunk starts are 4B-aligned whenever chunk_bytes % 4 == 0).
__device__ void sha256_bytes(const uint8_t* p, uint64_t len,
                             uint32_t out[8]) {
  uint32_t s[8];
  sha256_init_state(s);
  uint64_t full = len / 64;
  const bool aligned = (((uintptr_t)p) & 3u) == 0;
  uint32_t blk[16];
  for (uint64_t b = 0; b < full; ++b) {
    const uint8_t* q = p + b * 64;
    if (aligned) {
      const uint32_t* q32 = (const uint32_t*)q;
#pragma unroll
      for (int i = 0; i < 16; ++i) blk[i] = bswap32(q32[i]);
    } else {
#pragma unroll
      for (int i = 0; i < 16; ++i)
        blk[i] = ((uint32_t)q[4 * i] << 24) | ((uint32_t)q[4 * i + 1] << 16) |
                 ((uint32_t)q[4 * i + 2] << 8) | (uint32_t)q[4 * i + 3];
    }
    sha256_compress(s, blk);
  }
  // tail + padding (1..2 final blocks)
  uint32_t rem = (uint32_t)(len - full * 64);
  uint8_t tail[128];
#pragma unroll 1
  for (uint32_t i = 0; i < rem; ++i) tail[i] = p[full * 64 + i];
  tail[rem] = 0x80;
  uint32_t pad_blocks = (rem + 1 + 8 <= 64) ? 1 : 2;
  uint32_t total = pad_blocks * 64;
  for (uint32_t i = rem + 1; i < total - 8; ++i) tail[i] = 0;
  uint64_t bits = len * 8;
#pragma unroll
  for (int i = 0; i < 8; ++i)
    tail[total - 8 + i] = (uint8_t)(bits >> (56 - 8 * i));
  for (uint32_t b = 0; b < pad_blocks; ++b) {
    const uint8_t* q = tail + b * 64;
#pragma unroll
    for (int i = 0; i < 16; ++i)
      blk[i] = ((uint32_t)q[4 * i] << 24) | ((uint32_t)q[4 * i + 1] << 16) |
               ((uint32_t)q[4 * i + 2] << 8) | (uint32_t)q[4 * i + 3];
    sha256_compress(s, blk);
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) out[i] = s[i];
}

__global__ void sha256_batch_kernel(const uint8_t* __restrict__ data,
                                    uint64_t nbytes, uint64_t chunk_bytes,
                                    uint32_t* __restrict__ out,
                                    int n_chunks) {
  int tid = blockIdx.x * blockDim.x + threadIdx.x;
  int stride = gridDim.x * blockDim.x;
  for (int c = tid; c < n_chunks; c += stride) {
    uint64_t off = (uint64_t)c * chunk_bytes;
    uint64_t len = min(chunk_bytes, nbytes - off);
    uint32_t d[8];
    sha256_bytes(data + off, len, d);
#pragma unroll
    for (int i = 0; i < 8; ++i) out[c * 8 + i] = d[i];
  }
}

// ---- whole-blob chain ---------------------------------------------------
// state layout (device): 8 x u32 running state.
//
// The chain is sequential by construction, but the message schedules
// (W[t] + K[t]) of DIFFERENT 64-byte blocks are independent: one wave
// precomputes 64 blocks' schedules in parallel into LDS (one block per
// lane, 16 KiB), then the rounds grind through the 64 blocks reading
// precomputed KW via LDS broadcast — every lane redundantly computes the
// same state (lockstep, divergence-free), so only the round chain's
// ~4-op critical path remains serial.  ~10x the naive one-lane chain.

__global__ void sha256_chain_init_kernel(uint32_t* state) {
  if (threadIdx.x == 0 && blockIdx.x == 0) sha256_init_state(state);
}

__global__ void __launch_bounds__(64, 1)
sha256_chain_update_kernel(uint32_t* __restrict__ state,
                           const uint8_t* __restrict__ data,
                           uint64_t nblocks) {
  __shared__ uint32_t kw[64][65];  // [block][round], +1 pad vs bank stride
  if (blockIdx.x != 0) return;
  int lane = threadIdx.x;
  uint32_t s[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) s[i] = state[i];
  const bool aligned = (((uintptr_t)data) & 3u) == 0;

  for (uint64_t base = 0; base < nblocks; base += 64) {
    int group = (int)min((uint64_t)64, nblocks - base);
    // phase A: each lane expands one block's schedule
    if (lane < group) {
      const uint8_t* q = data + (base + lane) * 64;
      uint32_t w[16];
      if (aligned) {
        const uint32_t* q32 = (const uint32_t*)q;
#pragma unroll
        for (int i = 0; i < 16; ++i) w[i] = bswap32(q32[i]);
      } else {
#pragma unroll
        for (int i = 0; i < 16; ++i)
          w[i] = ((uint32_t)q[4 * i] << 24) |
                 ((uint32_t)q[4 * i + 1] << 16) |
                 ((uint32_t)q[4 * i + 2] << 8) | (uint32_t)q[4 * i + 3];
      }
#pragma unroll
      for (int t = 0; t < 64; ++t) {
        uint32_t wt;
        if (t < 16) {
          wt = w[t];
        } else {
          uint32_t w15 = w[(t - 15) & 15], w2 = w[(t - 2) & 15];
          uint32_t s0 = rotr(w15, 7) ^ rotr(w15, 18) ^ (w15 >> 3);
          uint32_t s1 = rotr(w2, 17) ^ rotr(w2, 19) ^ (w2 >> 10);
          wt = w[t & 15] = w[t & 15] + s0 + w[(t - 7) & 15] + s1;
        }
        kw[lane][t] = wt + K256[t];
      }
    }
    __syncthreads();
    // phase B: all lanes redundantly run the rounds.  Stage the block's
    // whole schedule into registers first: 64 independent ds_reads issue
    // back-to-back (one latency total) instead of one blocking read per
    // round.
    for (int b = 0; b < group; ++b) {
      uint32_t kwreg[64];
#pragma unroll
      for (int t = 0; t < 64; ++t) kwreg[t] = kw[b][t];
      uint32_t a = s[0], bb = s[1], c = s[2], d = s[3];
      uint32_t e = s[4], f = s[5], g = s[6], h = s[7];
      // all lanes compute identical values, so the compiler scalarizes
      // the rounds onto the SALU — lengthening the dependence chain with
      // cross-unit hops.  Pin the two chain registers to VGPRs.
      asm("" : "+v"(a), "+v"(bb), "+v"(c), "+v"(d),
              "+v"(e), "+v"(f), "+v"(g), "+v"(h));
#pragma unroll
      for (int t = 0; t < 64; ++t) {
        uint32_t S1 = rotr(e, 6) ^ rotr(e, 11) ^ rotr(e, 25);
        uint32_t ch = (e & f) ^ (~e & g);
        uint32_t t1 = h + S1 + ch + kwreg[t];
        uint32_t S0 = rotr(a, 2) ^ rotr(a, 13) ^ rotr(a, 22);
        uint32_t maj = (a & bb) ^ (a & c) ^ (bb & c);
        uint32_t t2 = S0 + maj;
        h = g; g = f; f = e; e = d + t1;
        d = c; c = bb; bb = a; a = t1 + t2;
      }
      s[0] += a; s[1] += bb; s[2] += c; s[3] += d;
      s[4] += e; s[5] += f; s[6] += g; s[7] += h;
    }
    __syncthreads();
  }
  if (lane == 0) {
#pragma unroll
    for (int i = 0; i < 8; ++i) state[i] = s[i];
  }
}

}  // namespace

extern "C" void launch_sha256_batch(const void* data, size_t nbytes,
                                    size_t chunk_bytes, uint32_t* out,
                                    int n_chunks, hipStream_t stream) {
  if (n_chunks <= 0) return;
  int threads = 256;
  int blocks = (n_chunks + threads - 1) / threads;
  if (blocks > 4096) blocks = 4096;  // grid-stride beyond this
  hipLaunchKernelGGL(sha256_batch_kernel, dim3(blocks), dim3(threads), 0,
                     stream, (const uint8_t*)data, (uint64_t)nbytes,
                     (uint64_t)chunk_bytes, out, n_chunks);
}

extern "C" void launch_sha256_chain_init(uint32_t* state,
                                         hipStream_t stream) {
  hipLaunchKernelGGL(sha256_chain_init_kernel, dim3(1), dim3(64), 0, stream,
                     state);
}

extern "C" void launch_sha256_chain_update(uint32_t* state, const void* data,
                                           size_t nblocks,
                                           hipStream_t stream) {
  if (nblocks == 0) return;
  hipLaunchKernelGGL(sha256_chain_update_kernel, dim3(1), dim3(64), 0,
                     stream, state, (const uint8_t*)data,
                     (uint64_t)nblocks);
}
