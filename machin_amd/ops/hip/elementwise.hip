// gfx950 fused elementwise kernels for the learner data path.
//
// u8 -> bf16 dequantization with scale (Atari frame normalization):
// torch's .to(bf16).mul_() pair runs at ~160 GB/s on this path; this
// kernel reads 16 bytes/lane and writes 32 bytes/lane of packed bf16
// (guide §6 G13: vectorize ANY memory-bound kernel), one pass.
#include "common.h"
#include <hip/hip_bf16.h>

typedef uint32_t u32;
typedef uint4 u8x16;  // 16 bytes per lane

union bf16x2_u {
  u32 u;
  __hip_bfloat162 v;
};

__device__ __forceinline__ u32 pack2(float a, float b) {
  bf16x2_u r;
  r.v = __hip_bfloat162(__float2bfloat16(a), __float2bfloat16(b));
  return r.u;
}

__global__ void u8_to_bf16_scale_kernel(const u8x16* __restrict__ in,
                                        uint4* __restrict__ out0,
                                        uint4* __restrict__ out1,
                                        int64_t n16, float scale) {
  // each lane: 16 u8 in -> 16 bf16 out (two uint4 stores)
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n16;
       i += (int64_t)gridDim.x * blockDim.x) {
    u8x16 raw = in[i];
    const unsigned char* b = (const unsigned char*)&raw;
    uint4 lo, hi;
    lo.x = pack2(b[0] * scale, b[1] * scale);
    lo.y = pack2(b[2] * scale, b[3] * scale);
    lo.z = pack2(b[4] * scale, b[5] * scale);
    lo.w = pack2(b[6] * scale, b[7] * scale);
    hi.x = pack2(b[8] * scale, b[9] * scale);
    hi.y = pack2(b[10] * scale, b[11] * scale);
    hi.z = pack2(b[12] * scale, b[13] * scale);
    hi.w = pack2(b[14] * scale, b[15] * scale);
    out0[2 * i] = lo;
    out0[2 * i + 1] = hi;
  }
}

// scalar tail / unaligned fallback
__global__ void u8_to_bf16_scale_tail_kernel(
    const unsigned char* __restrict__ in, __hip_bfloat16* __restrict__ out,
    int64_t start, int64_t n, float scale) {
  for (int64_t i = start + blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    out[i] = __float2bfloat16((float)in[i] * scale);
  }
}

void u8_to_bf16_scale_launch(const unsigned char* in, void* out, int64_t n,
                             float scale, hipStream_t stream) {
  const int block = 256;
  int64_t n16 = n / 16;
  if (n16 > 0) {
    hipLaunchKernelGGL(u8_to_bf16_scale_kernel,
                       dim3(ma_grid(n16, block)), dim3(block), 0, stream,
                       (const u8x16*)in, (uint4*)out, nullptr, n16, scale);
  }
  if (n % 16) {
    hipLaunchKernelGGL(u8_to_bf16_scale_tail_kernel, dim3(1), dim3(block),
                       0, stream, in, (__hip_bfloat16*)out, n16 * 16, n,
                       scale);
  }
}
