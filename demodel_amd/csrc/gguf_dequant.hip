// CDNA4 GGUF dequantization kernels: GGML quant
// blocks -> bf16, run while a pulled Ollama blob (reference protocol:
// CONTRIBUTING.md:127-153, application/vnd.ollama.image.model layers)
// still sits in HBM.  Block layouts follow the public GGML formats:
//   q4_0:  18 B / 32 elems  (f16 d, 16 B nibbles; v = d*(q-8))
//   q8_0:  34 B / 32 elems  (f16 d, 32 int8;     v = d*q)
//   q4_K: 144 B / 256 elems (f16 d, f16 dmin, 12 B 6-bit scales, 128 B)
//   q6_K: 210 B / 256 elems (128 B ql, 64 B qh, 16 int8 scales, f16 d)
//
// Mapping: q4_0/q8_0 one lane per 32-elem block; q4_K/q6_K one wave64 per
// 256-elem superblock (4 elems/lane) — scales decoded per-lane from the
// packed bytes, no LDS needed, stores coalesced in 64 B/lane groups.

#include <hip/hip_runtime.h>

namespace {

__device__ __forceinline__ float f16_to_f32(const uint8_t* p) {
  uint16_t h = (uint16_t)p[0] | ((uint16_t)p[1] << 8);
  uint32_t sign = (uint32_t)(h & 0x8000u) << 16;
  uint32_t exp = (h >> 10) & 0x1F;
  uint32_t man = h & 0x3FF;
  uint32_t f;
  if (exp == 0) {
    if (man == 0) {
      f = sign;
    } else {
      // f16 subnormal: value = man * 2^-24; after k shifts bit10 is set
      // and value = (1 + frac) * 2^(-14-k) -> f32 biased exp = 113 - k
      int k = 0;
      while (!(man & 0x400)) { man <<= 1; ++k; }
      f = sign | ((uint32_t)(113 - k) << 23) | ((man & 0x3FF) << 13);
    }
  } else if (exp == 31) {
    f = sign | 0x7F800000u | (man << 13);
  } else {
    f = sign | ((exp - 15 + 127) << 23) | (man << 13);
  }
  return __uint_as_float(f);
}

__device__ __forceinline__ uint16_t f32_to_bf16(float v) {
  uint32_t x = __float_as_uint(v);
  if ((x & 0x7fffffffu) > 0x7f800000u) return (uint16_t)((x >> 16) | 0x40);
  uint32_t lsb = (x >> 16) & 1u;
  return (uint16_t)((x + 0x7fffu + lsb) >> 16);
}

// ---- q4_0: one lane per block ------------------------------------------

__global__ void dequant_q4_0(const uint8_t* __restrict__ src,
                             uint16_t* __restrict__ dst, int64_t n_blocks) {
  int64_t b0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t b = b0; b < n_blocks; b += stride) {
    const uint8_t* q = src + b * 18;
    float d = f16_to_f32(q);
    uint16_t* o = dst + b * 32;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      uint8_t byte = q[2 + j];
      o[j] = f32_to_bf16(d * (float)((int)(byte & 0xF) - 8));
      o[j + 16] = f32_to_bf16(d * (float)((int)(byte >> 4) - 8));
    }
  }
}

// ---- q8_0: one lane per block ------------------------------------------

__global__ void dequant_q8_0(const uint8_t* __restrict__ src,
                             uint16_t* __restrict__ dst, int64_t n_blocks) {
  int64_t b0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t b = b0; b < n_blocks; b += stride) {
    const uint8_t* q = src + b * 34;
    float d = f16_to_f32(q);
    uint16_t* o = dst + b * 32;
#pragma unroll
    for (int j = 0; j < 32; ++j)
      o[j] = f32_to_bf16(d * (float)(int8_t)q[2 + j]);
  }
}

// ---- q4_K: one wave per 256-elem superblock, 4 elems/lane ---------------
// scale/min unpack follows GGML get_scale_min_k4.

__device__ __forceinline__ void scale_min_k4(int j, const uint8_t* s,
                                             uint8_t* d, uint8_t* m) {
  if (j < 4) {
    *d = s[j] & 63;
    *m = s[j + 4] & 63;
  } else {
    *d = (s[j + 4] & 0xF) | ((s[j - 4] >> 6) << 4);
    *m = (s[j + 4] >> 4) | ((s[j] >> 6) << 4);
  }
}

__global__ void dequant_q4_K(const uint8_t* __restrict__ src,
                             uint16_t* __restrict__ dst, int64_t n_sblocks) {
  int64_t sb0 = (int64_t)blockIdx.x * (blockDim.x / 64) + (threadIdx.x / 64);
  int64_t stride = (int64_t)gridDim.x * (blockDim.x / 64);
  int lane = threadIdx.x & 63;
  for (int64_t sb = sb0; sb < n_sblocks; sb += stride) {
    const uint8_t* blk = src + sb * 144;
    float d = f16_to_f32(blk);
    float dmin = f16_to_f32(blk + 2);
    const uint8_t* scales = blk + 4;
    const uint8_t* qs = blk + 16;
    uint16_t* o = dst + sb * 256;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int j = r * 64 + lane;       // element index 0..255, lane-contiguous
      int sub = j >> 5;            // 32-elem sub-block 0..7
      uint8_t sc, mn;
      scale_min_k4(sub, scales, &sc, &mn);
      int l = j & 31;
      int pair = j >> 6;           // 64-elem pair index 0..3
      uint8_t byte = qs[pair * 32 + l];
      int nib = ((j >> 5) & 1) ? (byte >> 4) : (byte & 0xF);
      o[j] = f32_to_bf16(d * (float)sc * (float)nib -
                         dmin * (float)mn);
    }
  }
}

// ---- q6_K: one wave per superblock, 4 elems/lane ------------------------

__global__ void dequant_q6_K(const uint8_t* __restrict__ src,
                             uint16_t* __restrict__ dst, int64_t n_sblocks) {
  int64_t sb0 = (int64_t)blockIdx.x * (blockDim.x / 64) + (threadIdx.x / 64);
  int64_t stride = (int64_t)gridDim.x * (blockDim.x / 64);
  int lane = threadIdx.x & 63;
  for (int64_t sb = sb0; sb < n_sblocks; sb += stride) {
    const uint8_t* blk = src + sb * 210;
    const uint8_t* ql = blk;
    const uint8_t* qh = blk + 128;
    const int8_t* scales = (const int8_t*)(blk + 192);
    float d = f16_to_f32(blk + 208);
    uint16_t* o = dst + sb * 256;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int j = r * 64 + lane;       // element 0..255, lane-contiguous
      int n = j >> 7;              // 128-half 0..1
      int rr = j & 127;
      int half = rr >> 5;          // 0..3
      int l = rr & 31;
      int q;
      const uint8_t* qln = ql + n * 64;
      const uint8_t* qhn = qh + n * 32;
      if (half == 0)
        q = (int)((qln[l] & 0xF) | (((qhn[l] >> 0) & 3) << 4)) - 32;
      else if (half == 1)
        q = (int)((qln[32 + l] & 0xF) | (((qhn[l] >> 2) & 3) << 4)) - 32;
      else if (half == 2)
        q = (int)((qln[l] >> 4) | (((qhn[l] >> 4) & 3) << 4)) - 32;
      else
        q = (int)((qln[32 + l] >> 4) | (((qhn[l] >> 6) & 3) << 4)) - 32;
      int sc = scales[n * 8 + half * 2 + (l >> 4)];
      o[j] = f32_to_bf16(d * (float)sc * (float)q);
    }
  }
}

// ---- q4_1: one lane per 32-elem block (20 B: f16 d, f16 m, 16 B nibbles;
// v = d*q + m) -------------------------------------------------------------

__global__ void dequant_q4_1(const uint8_t* __restrict__ src,
                             uint16_t* __restrict__ dst, int64_t n_blocks) {
  int64_t b0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t b = b0; b < n_blocks; b += stride) {
    const uint8_t* q = src + b * 20;
    float d = f16_to_f32(q);
    float m = f16_to_f32(q + 2);
    uint16_t* o = dst + b * 32;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      uint8_t byte = q[4 + j];
      o[j] = f32_to_bf16(d * (float)(byte & 0xF) + m);
      o[j + 16] = f32_to_bf16(d * (float)(byte >> 4) + m);
    }
  }
}

// ---- q5_0 / q5_1: one lane per 32-elem block ----------------------------
// q5_0 (22 B): f16 d, u32 qh (5th bits), 16 B nibbles; v = d*(q-16)
// q5_1 (24 B): f16 d, f16 m, u32 qh, 16 B nibbles;     v = d*q + m

__global__ void dequant_q5_0(const uint8_t* __restrict__ src,
                             uint16_t* __restrict__ dst, int64_t n_blocks) {
  int64_t b0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t b = b0; b < n_blocks; b += stride) {
    const uint8_t* q = src + b * 22;
    float d = f16_to_f32(q);
    uint32_t qh = (uint32_t)q[2] | ((uint32_t)q[3] << 8) |
                  ((uint32_t)q[4] << 16) | ((uint32_t)q[5] << 24);
    uint16_t* o = dst + b * 32;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      uint8_t byte = q[6 + j];
      int x0 = (int)((byte & 0xF) | (((qh >> j) << 4) & 0x10)) - 16;
      int x1 = (int)((byte >> 4) | ((qh >> (j + 12)) & 0x10)) - 16;
      o[j] = f32_to_bf16(d * (float)x0);
      o[j + 16] = f32_to_bf16(d * (float)x1);
    }
  }
}

__global__ void dequant_q5_1(const uint8_t* __restrict__ src,
                             uint16_t* __restrict__ dst, int64_t n_blocks) {
  int64_t b0 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t b = b0; b < n_blocks; b += stride) {
    const uint8_t* q = src + b * 24;
    float d = f16_to_f32(q);
    float m = f16_to_f32(q + 2);
    uint32_t qh = (uint32_t)q[4] | ((uint32_t)q[5] << 8) |
                  ((uint32_t)q[6] << 16) | ((uint32_t)q[7] << 24);
    uint16_t* o = dst + b * 32;
#pragma unroll
    for (int j = 0; j < 16; ++j) {
      uint8_t byte = q[8 + j];
      int x0 = (int)((byte & 0xF) | (((qh >> j) << 4) & 0x10));
      int x1 = (int)((byte >> 4) | ((qh >> (j + 12)) & 0x10));
      o[j] = f32_to_bf16(d * (float)x0 + m);
      o[j + 16] = f32_to_bf16(d * (float)x1 + m);
    }
  }
}

// ---- q5_K: one wave per 256-elem superblock -----------------------------
// 176 B: f16 d, f16 dmin, 12 B 6-bit scales (q4_K packing), 32 B qh
// (5th bits, bit `sub` of qh[l]), 128 B nibbles.

__global__ void dequant_q5_K(const uint8_t* __restrict__ src,
                             uint16_t* __restrict__ dst, int64_t n_sblocks) {
  int64_t sb0 = (int64_t)blockIdx.x * (blockDim.x / 64) + (threadIdx.x / 64);
  int64_t stride = (int64_t)gridDim.x * (blockDim.x / 64);
  int lane = threadIdx.x & 63;
  for (int64_t sb = sb0; sb < n_sblocks; sb += stride) {
    const uint8_t* blk = src + sb * 176;
    float d = f16_to_f32(blk);
    float dmin = f16_to_f32(blk + 2);
    const uint8_t* scales = blk + 4;
    const uint8_t* qh = blk + 16;
    const uint8_t* qs = blk + 48;
    uint16_t* o = dst + sb * 256;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int j = r * 64 + lane;
      int sub = j >> 5;
      uint8_t sc, mn;
      scale_min_k4(sub, scales, &sc, &mn);
      int l = j & 31;
      int pair = j >> 6;
      uint8_t byte = qs[pair * 32 + l];
      int nib = (sub & 1) ? (byte >> 4) : (byte & 0xF);
      int q = nib + (((qh[l] >> sub) & 1) << 4);
      o[j] = f32_to_bf16(d * (float)sc * (float)q -
                         dmin * (float)mn);
    }
  }
}

// ---- q3_K: one wave per superblock --------------------------------------
// 110 B: 32 B hmask, 64 B 2-bit qs, 12 B packed 6-bit scales, f16 d.

__global__ void dequant_q3_K(const uint8_t* __restrict__ src,
                             uint16_t* __restrict__ dst, int64_t n_sblocks) {
  int64_t sb0 = (int64_t)blockIdx.x * (blockDim.x / 64) + (threadIdx.x / 64);
  int64_t stride = (int64_t)gridDim.x * (blockDim.x / 64);
  int lane = threadIdx.x & 63;
  for (int64_t sb = sb0; sb < n_sblocks; sb += stride) {
    const uint8_t* blk = src + sb * 110;
    const uint8_t* hmask = blk;
    const uint8_t* qs = blk + 32;
    const uint8_t* sp = blk + 96;
    float d = f16_to_f32(blk + 108);
    // 12 packed bytes -> 16 6-bit scales (ggml kmask unpack)
    uint32_t w0 = (uint32_t)sp[0] | ((uint32_t)sp[1] << 8) |
                  ((uint32_t)sp[2] << 16) | ((uint32_t)sp[3] << 24);
    uint32_t w1 = (uint32_t)sp[4] | ((uint32_t)sp[5] << 8) |
                  ((uint32_t)sp[6] << 16) | ((uint32_t)sp[7] << 24);
    uint32_t w2 = (uint32_t)sp[8] | ((uint32_t)sp[9] << 8) |
                  ((uint32_t)sp[10] << 16) | ((uint32_t)sp[11] << 24);
    uint32_t aux[4];
    aux[0] = (w0 & 0x0f0f0f0fu) | (((w2 >> 0) & 0x03030303u) << 4);
    aux[1] = (w1 & 0x0f0f0f0fu) | (((w2 >> 2) & 0x03030303u) << 4);
    aux[2] = ((w0 >> 4) & 0x0f0f0f0fu) | (((w2 >> 4) & 0x03030303u) << 4);
    aux[3] = ((w1 >> 4) & 0x0f0f0f0fu) | (((w2 >> 6) & 0x03030303u) << 4);
    uint16_t* o = dst + sb * 256;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int j = r * 64 + lane;
      int nh = j >> 7;             // 128-half
      int rr = j & 127;
      int jj = rr >> 5;            // shift group 0..3
      int l = rr & 31;
      int is = nh * 8 + jj * 2 + (l >> 4);
      int sc = (int)((aux[is >> 2] >> ((is & 3) * 8)) & 0xFF) - 32;
      int shift = jj * 2;
      int bit = nh * 4 + jj;
      int qv = (int)((qs[nh * 32 + l] >> shift) & 3) -
               (((hmask[l] >> bit) & 1) ? 0 : 4);
      o[j] = f32_to_bf16(d * (float)sc * (float)qv);
    }
  }
}

// ---- q2_K: one wave per superblock --------------------------------------
// 84 B: 16 B scales (lo nib = scale, hi nib = min), 64 B 2-bit qs,
// f16 d, f16 dmin.

__global__ void dequant_q2_K(const uint8_t* __restrict__ src,
                             uint16_t* __restrict__ dst, int64_t n_sblocks) {
  int64_t sb0 = (int64_t)blockIdx.x * (blockDim.x / 64) + (threadIdx.x / 64);
  int64_t stride = (int64_t)gridDim.x * (blockDim.x / 64);
  int lane = threadIdx.x & 63;
  for (int64_t sb = sb0; sb < n_sblocks; sb += stride) {
    const uint8_t* blk = src + sb * 84;
    const uint8_t* scales = blk;
    const uint8_t* qs = blk + 16;
    float d = f16_to_f32(blk + 80);
    float dmin = f16_to_f32(blk + 82);
    uint16_t* o = dst + sb * 256;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int j = r * 64 + lane;
      int nh = j >> 7;
      int rr = j & 127;
      int jj = rr >> 5;
      int l = rr & 31;
      uint8_t sc = scales[nh * 8 + jj * 2 + (l >> 4)];
      int qv = (qs[nh * 32 + l] >> (jj * 2)) & 3;
      o[j] = f32_to_bf16(d * (float)(sc & 0xF) * (float)qv -
                         dmin * (float)(sc >> 4));
    }
  }
}

}  // namespace

// qtype ids follow GGML: 2=q4_0, 3=q4_1, 6=q5_0, 7=q5_1, 8=q8_0, 10=q2_K,
// 11=q3_K, 12=q4_K, 13=q5_K, 14=q6_K
extern "C" void launch_gguf_dequant(int qtype, const void* src,
                                    uint16_t* dst, int64_t n_blocks,
                                    hipStream_t stream) {
  if (n_blocks <= 0) return;
  const uint8_t* s = (const uint8_t*)src;
  if (qtype == 2 || qtype == 3 || qtype == 6 || qtype == 7 ||
      qtype == 8) {
    int64_t want = (n_blocks + 255) / 256;
    int blocks = want > 8192 ? 8192 : (int)want;
    if (qtype == 2)
      hipLaunchKernelGGL(dequant_q4_0, dim3(blocks), dim3(256), 0, stream,
                         s, dst, n_blocks);
    else if (qtype == 3)
      hipLaunchKernelGGL(dequant_q4_1, dim3(blocks), dim3(256), 0, stream,
                         s, dst, n_blocks);
    else if (qtype == 6)
      hipLaunchKernelGGL(dequant_q5_0, dim3(blocks), dim3(256), 0, stream,
                         s, dst, n_blocks);
    else if (qtype == 7)
      hipLaunchKernelGGL(dequant_q5_1, dim3(blocks), dim3(256), 0, stream,
                         s, dst, n_blocks);
    else
      hipLaunchKernelGGL(dequant_q8_0, dim3(blocks), dim3(256), 0, stream,
                         s, dst, n_blocks);
    return;
  }
  if (qtype == 10 || qtype == 11 || qtype == 12 || qtype == 13 ||
      qtype == 14) {
    // 4 waves per 256-thread workgroup, one superblock per wave
    int64_t want = (n_blocks + 3) / 4;
    int blocks = want > 8192 ? 8192 : (int)want;
    if (qtype == 10)
      hipLaunchKernelGGL(dequant_q2_K, dim3(blocks), dim3(256), 0, stream,
                         s, dst, n_blocks);
    else if (qtype == 11)
      hipLaunchKernelGGL(dequant_q3_K, dim3(blocks), dim3(256), 0, stream,
                         s, dst, n_blocks);
    else if (qtype == 12)
      hipLaunchKernelGGL(dequant_q4_K, dim3(blocks), dim3(256), 0, stream,
                         s, dst, n_blocks);
    else if (qtype == 13)
      hipLaunchKernelGGL(dequant_q5_K, dim3(blocks), dim3(256), 0, stream,
                         s, dst, n_blocks);
    else
      hipLaunchKernelGGL(dequant_q6_K, dim3(blocks), dim3(256), 0, stream,
                         s, dst, n_blocks);
    return;
  }
  // unknown qtype: trap loudly rather than silently skip
  abort();
}
