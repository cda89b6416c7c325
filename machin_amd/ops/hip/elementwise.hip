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

// ---------------------------------------------------------------------
// fused categorical policy head (IMPALA/A2C loss block).
//
// Replaces the eager chain logits.float() -> log_softmax -> gather ->
// exp/mul/sum entropy (~6 kernels forward, ~8 backward over the
// [T*B, A] logits) with ONE kernel each way. A is small (Atari: 6),
// so the whole row lives in registers; one thread per row.
//
// forward:  taken_logp[i] = log_softmax(logits[i])[a_i]
//           entropy[i]    = -sum_a p log p
// backward: dlogits[i,a] = g_taken[i] * (1[a==a_i] - p_a)
//                        + g_ent[i] * (-p_a * (log p_a + H_i))
// ---------------------------------------------------------------------
#define PG_MAX_A 32

__global__ void pg_head_fwd_kernel(const __bf16* __restrict__ logits,
                                   const int64_t* __restrict__ actions,
                                   float* __restrict__ taken_logp,
                                   float* __restrict__ entropy,
                                   int64_t N, int A) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < N; i += (int64_t)gridDim.x * blockDim.x) {
    float z[PG_MAX_A];
    float zmax = -1e30f;
    for (int a = 0; a < A; ++a) {
      z[a] = (float)logits[i * A + a];
      zmax = fmaxf(zmax, z[a]);
    }
    float sum = 0.0f;
    for (int a = 0; a < A; ++a) {
      z[a] = __expf(z[a] - zmax);
      sum += z[a];
    }
    float inv = 1.0f / sum, logsum = __logf(sum);
    float H = 0.0f;
    for (int a = 0; a < A; ++a) {
      float p = z[a] * inv;
      float logp = __logf(z[a]) - logsum;  // = z_orig - zmax - logsum
      H -= p * logp;
    }
    int ai = (int)actions[i];
    taken_logp[i] = __logf(z[ai]) - logsum;
    entropy[i] = H;
  }
}

__global__ void pg_head_bwd_kernel(const __bf16* __restrict__ logits,
                                   const int64_t* __restrict__ actions,
                                   const float* __restrict__ g_taken,
                                   const float* __restrict__ g_ent,
                                   __bf16* __restrict__ dlogits,
                                   int64_t N, int A) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < N; i += (int64_t)gridDim.x * blockDim.x) {
    float z[PG_MAX_A];
    float zmax = -1e30f;
    for (int a = 0; a < A; ++a) {
      z[a] = (float)logits[i * A + a];
      zmax = fmaxf(zmax, z[a]);
    }
    float sum = 0.0f;
    for (int a = 0; a < A; ++a) {
      z[a] = __expf(z[a] - zmax);
      sum += z[a];
    }
    float inv = 1.0f / sum, logsum = __logf(sum);
    float H = 0.0f;
    for (int a = 0; a < A; ++a) {
      float p = z[a] * inv;
      H -= p * (__logf(z[a]) - logsum);
    }
    int ai = (int)actions[i];
    float gt = g_taken[i], ge = g_ent[i];
    for (int a = 0; a < A; ++a) {
      float p = z[a] * inv;
      float logp = __logf(z[a]) - logsum;
      float g = gt * ((a == ai ? 1.0f : 0.0f) - p)
                - ge * p * (logp + H);
      dlogits[i * A + a] = (__bf16)g;
    }
  }
}

void pg_head_fwd_launch(const void* logits, const int64_t* actions,
                        float* taken_logp, float* entropy, int64_t N,
                        int A, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(pg_head_fwd_kernel, dim3(ma_grid(N, block)),
                     dim3(block), 0, stream, (const __bf16*)logits,
                     actions, taken_logp, entropy, N, A);
}

void pg_head_bwd_launch(const void* logits, const int64_t* actions,
                        const float* g_taken, const float* g_ent,
                        void* dlogits, int64_t N, int A,
                        hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(pg_head_bwd_kernel, dim3(ma_grid(N, block)),
                     dim3(block), 0, stream, (const __bf16*)logits,
                     actions, g_taken, g_ent, (__bf16*)dlogits, N, A);
}
