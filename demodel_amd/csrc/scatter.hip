// CDNA4 tensor-scatter + cast kernels.
//
// A landed safetensors blob sits contiguously in HBM; loading it into a
// model means copying each tensor's byte range into that tensor's (possibly
// allocator-placed, possibly TP-sharded) destination storage.  Descriptors
// are (src_off, dst_ptr, len) u64 triples built on the host; one workgroup
// per <=1 MiB descriptor piece, 16-byte vectorized, coalesced on both
// sides.  HBM3E-bound by design: the kernel does nothing per byte but move
// it (and optionally cast).

#include <hip/hip_runtime.h>

namespace {

struct __align__(16) RangeDesc {
  uint64_t src_off;
  uint64_t dst_ptr;
  uint64_t len;
  uint64_t _pad;
};

__global__ void scatter_ranges_kernel(const uint8_t* __restrict__ src,
                                      const RangeDesc* __restrict__ desc,
                                      int n_desc) {
  for (int di = blockIdx.x; di < n_desc; di += gridDim.x) {
    RangeDesc d = desc[di];
    const uint8_t* s = src + d.src_off;
    uint8_t* t = (uint8_t*)d.dst_ptr;
    uint64_t len = d.len;
    // head: byte copy until 16B-aligned on the destination
    uint64_t head = (16 - ((uint64_t)t & 15)) & 15;
    if (head > len) head = len;
    for (uint64_t i = threadIdx.x; i < head; i += blockDim.x) t[i] = s[i];
    s += head; t += head; len -= head;
    // 16-byte vector body when the source is co-aligned; else u32/u8
    uint64_t v16 = len / 16;
    if ((((uint64_t)s) & 15) == 0) {
      const uint4* s4 = (const uint4*)s;
      uint4* t4 = (uint4*)t;
      for (uint64_t i = threadIdx.x; i < v16; i += blockDim.x)
        t4[i] = s4[i];
    } else if ((((uint64_t)s) & 3) == 0) {
      // src 4B-aligned: manual 16B gather via four u32 loads
      const uint32_t* s1 = (const uint32_t*)s;
      uint4* t4 = (uint4*)t;
      for (uint64_t i = threadIdx.x; i < v16; i += blockDim.x) {
        uint4 v;
        v.x = s1[i * 4 + 0]; v.y = s1[i * 4 + 1];
        v.z = s1[i * 4 + 2]; v.w = s1[i * 4 + 3];
        t4[i] = v;
      }
    } else {
      for (uint64_t i = threadIdx.x; i < v16 * 16; i += blockDim.x)
        t[i] = s[i];
    }
    // tail bytes
    for (uint64_t i = v16 * 16 + threadIdx.x; i < len; i += blockDim.x)
      t[i] = s[i];
  }
}

__device__ __forceinline__ uint16_t f32_to_bf16_rne(float f) {
  uint32_t x = __float_as_uint(f);
  if ((x & 0x7fffffffu) > 0x7f800000u) return (uint16_t)((x >> 16) | 0x40);
  uint32_t lsb = (x >> 16) & 1u;
  return (uint16_t)((x + 0x7fffu + lsb) >> 16);
}

__global__ void cast_f32_bf16_kernel(const float* __restrict__ src,
                                     uint16_t* __restrict__ dst,
                                     uint64_t n) {
  uint64_t i0 = ((uint64_t)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  uint64_t stride = (uint64_t)gridDim.x * blockDim.x * 4;
  for (uint64_t i = i0; i + 4 <= n; i += stride) {
    float4 v = *(const float4*)(src + i);
    ushort4 o;
    o.x = f32_to_bf16_rne(v.x); o.y = f32_to_bf16_rne(v.y);
    o.z = f32_to_bf16_rne(v.z); o.w = f32_to_bf16_rne(v.w);
    *(ushort4*)(dst + i) = o;
  }
  // ragged tail handled by the first threads
  uint64_t tail_start = (n / 4) * 4;
  uint64_t ti = tail_start + (uint64_t)(blockIdx.x * blockDim.x + threadIdx.x);
  if (blockIdx.x == 0 && ti < n) dst[ti] = f32_to_bf16_rne(src[ti]);
}

}  // namespace

extern "C" void launch_scatter_ranges(const void* src, const uint64_t* desc,
                                      int n_desc, hipStream_t stream) {
  if (n_desc <= 0) return;
  int blocks = n_desc < 8192 ? n_desc : 8192;
  hipLaunchKernelGGL(scatter_ranges_kernel, dim3(blocks), dim3(256), 0,
                     stream, (const uint8_t*)src, (const RangeDesc*)desc,
                     n_desc);
}

extern "C" void launch_cast_f32_to_bf16(const float* src, uint16_t* dst,
                                        size_t n, hipStream_t stream) {
  if (n == 0) return;
  uint64_t want = (n + 1023) / 1024;
  int blocks = want > 8192 ? 8192 : (int)want;
  hipLaunchKernelGGL(cast_f32_bf16_kernel, dim3(blocks), dim3(256), 0,
                     stream, src, dst, (uint64_t)n);
}
