// gfx950 fused weight-gradient kernel for the Atari stem conv
// (Conv2d(C_in=4, C_out=32, kernel 8x8, stride 4) on 84x84 frames).
//
// Why hand-written: the weight gradient of this layer is a skinny
// reduction GEMM dW[32,256] = dy^T[32,K] @ im2col(x)[K,256] with
// K = batch*400 (16.4M at bench batch 40960). MIOpen/CK pick kernels
// that run it at ~2% MFMA efficiency (2.5-5.4 ms/step measured,
// profiles/impala_bench_kernel_trace_r01.md). The op is HBM-bound in
// principle: dy (1.05 GB bf16) + x (1.16 GB u8, read ONCE as u8 and
// dequantized in-register) ≈ 2.2 GB -> ~0.4 ms at 6 TB/s.
//
// Design (guide §3, §5):
// - mfma_f32_16x16x32_bf16 tiles; output 32x256 = 2x16 tiles of
//   16x16; 4 waves per workgroup, each owns (M-tile, 8 N-tiles),
//   32 f32 accumulator registers per lane.
// - split-K: each workgroup reduces its K-slice and atomicAdd's the
//   32x256 fp32 partial (atomics are a rounding-free fp32 add; the
//   slab is tiny so contention is negligible — guide §6 G12).
// - x rows are assembled from the NHWC u8 frame tensor: one im2col
//   row = 8 segments of 32 contiguous bytes; 256 threads stage a
//   16-row chunk into LDS as bf16 (dequant scale fused into the
//   u8->bf16 conversion).
#include "common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

#define CONV1_CIN 4
#define CONV1_COUT 32
#define CONV1_KSZ 8
#define CONV1_STRIDE 4
#define CONV1_OHW 20          // output spatial (84-8)/4+1
#define CONV1_POS (CONV1_OHW * CONV1_OHW)       // 400 positions
#define CONV1_N (CONV1_KSZ * CONV1_KSZ * CONV1_CIN)  // 256 patch size
#define KC 32  // k-rows per chunk: one full MFMA K-step (16x16x32)

__device__ __forceinline__ __bf16 u8_bf16(unsigned char v, float scale) {
  return (__bf16)((float)v * scale);
}

// dy: [K, 32] bf16 row-major (K = B*400, NHWC conv output layout)
// frames: [B, 84, 84, 4] u8 (NHWC)
// out: [32, 256] fp32, PRE-ZEROED, atomicAdd target
__global__ __launch_bounds__(256)
void conv1_wrw_kernel(const __bf16* __restrict__ dy,
                      const unsigned char* __restrict__ frames,
                      float* __restrict__ out,
                      float* __restrict__ b_out, int64_t K,
                      int64_t k_per_wg, float scale) {
  __shared__ __bf16 lds[KC * CONV1_COUT + KC * CONV1_N];  // dy | x
  __bf16* s_dy = lds;                    // [KC][32]
  __bf16* s_x = lds + KC * CONV1_COUT;   // [KC][256]

  const int tid = threadIdx.x;
  const int wave = tid / MA_WAVE;  // 0..3
  const int lane = tid % MA_WAVE;
  const int mt = wave & 1;         // M-tile (0..1): rows mt*16..mt*16+15
  const int ng = wave >> 1;        // N-group (0..1): tiles ng*8..ng*8+7

  f32x4 acc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) acc[i] = (f32x4)(0.0f);
  float bias_acc = 0.0f;

  const int64_t k_begin = (int64_t)blockIdx.x * k_per_wg;
  const int64_t k_end = min(k_begin + k_per_wg, K);

  for (int64_t kc = k_begin; kc < k_end; kc += KC) {
    // ---- stage dy chunk: 32 rows x 32 bf16 (64B/row) --------------
    // threads 0..127: each loads 16B (8 bf16)
    if (tid < 128) {
      int row = tid >> 2;        // 0..31
      int seg = tid & 3;         // 0..3 (8 bf16 each)
      int64_t kk = kc + row;
      bf16x8 v;
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = (__bf16)0.0f;
      if (kk < k_end) {
        v = *(const bf16x8*)(dy + kk * CONV1_COUT + seg * 8);
      }
      *(bf16x8*)(s_dy + row * CONV1_COUT + seg * 8) = v;
    }
    // ---- stage x chunk: 32 rows x 256 bf16 from u8 patches --------
    // each im2col row: 8 segments of 32 u8; 256 segments total;
    // 256 threads -> each converts one full 32-byte segment
    {
      int row = tid >> 3;        // k-row 0..31
      int r = tid & 7;           // patch row 0..7
      int64_t kk = kc + row;
      bf16x8 q0, q1, q2, q3;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        q0[j] = (__bf16)0.0f;
        q1[j] = (__bf16)0.0f;
        q2[j] = (__bf16)0.0f;
        q3[j] = (__bf16)0.0f;
      }
      if (kk < k_end) {
        int64_t b = kk / CONV1_POS;
        int pos = (int)(kk % CONV1_POS);
        int oh = pos / CONV1_OHW, ow = pos % CONV1_OHW;
        const unsigned char* src =
            frames
            + ((b * 84 + (int64_t)oh * CONV1_STRIDE + r) * 84
               + (int64_t)ow * CONV1_STRIDE) * CONV1_CIN;
        uint4 raw0 = *(const uint4*)src;
        uint4 raw1 = *(const uint4*)(src + 16);
        unsigned int words[8] = {raw0.x, raw0.y, raw0.z, raw0.w,
                                 raw1.x, raw1.y, raw1.z, raw1.w};
#pragma unroll
        for (int w = 0; w < 2; ++w) {
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            q0[w * 4 + j] = u8_bf16((words[w] >> (8 * j)) & 0xFF, scale);
            q1[w * 4 + j] =
                u8_bf16((words[w + 2] >> (8 * j)) & 0xFF, scale);
            q2[w * 4 + j] =
                u8_bf16((words[w + 4] >> (8 * j)) & 0xFF, scale);
            q3[w * 4 + j] =
                u8_bf16((words[w + 6] >> (8 * j)) & 0xFF, scale);
          }
        }
      }
      bf16x8* dst = (bf16x8*)(s_x + row * CONV1_N + r * 32);
      dst[0] = q0;
      dst[1] = q1;
      dst[2] = q2;
      dst[3] = q3;
    }
    __syncthreads();

    // ---- fused bias: wave 0 lanes 0..31 sum dy columns (dy is
    // already in LDS; saves a separate 1 GB reduction pass) ---------
    if (wave == 0 && lane < CONV1_COUT) {
#pragma unroll
      for (int k = 0; k < KC; ++k) {
        bias_acc += (float)s_dy[k * CONV1_COUT + lane];
      }
    }

    // ---- MFMA: A[m,k]=dy[k, mt*16+m] (transposed read), B[k,n]=x --
    // A fragment (16x16x32): lane holds m=lane%16, k=(lane/16)*8+j
    bf16x8 a_frag;
    {
      int m = lane & 15;
      int k0 = (lane >> 4) * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        a_frag[j] = s_dy[(k0 + j) * CONV1_COUT + mt * 16 + m];
      }
    }
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
      int n0 = (ng * 8 + nt) * 16;
      bf16x8 b_frag;
      int n = lane & 15;
      int k0 = (lane >> 4) * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        b_frag[j] = s_x[(k0 + j) * CONV1_N + n0 + n];
      }
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_frag, b_frag, acc[nt], 0, 0, 0);
    }
    __syncthreads();
  }

  // ---- epilogue: atomicAdd partials (C/D map: col=lane&15,
  // row=(lane>>4)*4+reg) ------------------------------------------
  if (wave == 0 && lane < CONV1_COUT) {
    atomicAdd(&b_out[lane], bias_acc);
  }
#pragma unroll
  for (int nt = 0; nt < 8; ++nt) {
    int col = (ng * 8 + nt) * 16 + (lane & 15);
    int row_base = mt * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      atomicAdd(&out[(row_base + reg) * CONV1_N + col], acc[nt][reg]);
    }
  }
}

// reorder [32][r*32+c*4+ci] fp32 -> conv weight grad [32][4][8][8]
__global__ void conv1_wrw_reorder_kernel(const float* __restrict__ in,
                                         float* __restrict__ out) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= CONV1_COUT * CONV1_N) return;
  int co = i / CONV1_N;
  int rem = i % CONV1_N;
  int r = rem / 32;
  int c = (rem % 32) / CONV1_CIN;
  int ci = rem % CONV1_CIN;
  out[((co * CONV1_CIN + ci) * CONV1_KSZ + r) * CONV1_KSZ + c] =
      in[i];
}

__global__ void conv1_wrw_v2_kernel(const __bf16*, const unsigned char*,
                                    float*, float*, int64_t, int64_t,
                                    float);
__global__ void conv1_fwd_v2_kernel(const unsigned char*, const __bf16*,
                                    const float*, __bf16*, int64_t, float);

void conv1_wrw_launch(const void* dy, const unsigned char* frames,
                      float* scratch, float* grad_w, float* grad_b,
                      int64_t K, float scale, hipStream_t stream) {
  HIP_CHECK(hipMemsetAsync(scratch, 0,
                           CONV1_COUT * CONV1_N * sizeof(float), stream));
  HIP_CHECK(hipMemsetAsync(grad_b, 0, CONV1_COUT * sizeof(float),
                           stream));
  // split-K: target ~2048 workgroups (8 XCDs x 32 CUs x 8 blocks)
  int64_t target_wg = 2048;
  if (const char* e = getenv("MACHIN_CONV1_WG")) target_wg = atol(e);
  int64_t k_per_wg = (K + target_wg - 1) / target_wg;
  k_per_wg = ((k_per_wg + KC - 1) / KC) * KC;
  if (k_per_wg < KC) k_per_wg = KC;
  int grid = (int)((K + k_per_wg - 1) / k_per_wg);
  const bool v1 = getenv("MACHIN_CONV1_V1") != nullptr;
  if (v1) {
    hipLaunchKernelGGL(conv1_wrw_kernel, dim3(grid), dim3(256), 0, stream,
                       (const __bf16*)dy, frames, scratch, grad_b, K,
                       k_per_wg, scale);
  } else {
    hipLaunchKernelGGL(conv1_wrw_v2_kernel, dim3(grid), dim3(256), 0,
                       stream, (const __bf16*)dy, frames, scratch, grad_b,
                       K, k_per_wg, scale);
  }
  HIP_CHECK(hipGetLastError());
  hipLaunchKernelGGL(conv1_wrw_reorder_kernel,
                     dim3((CONV1_COUT * CONV1_N + 255) / 256), dim3(256),
                     0, stream, scratch, grad_w);
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------
// layout probe: one mfma_f32_16x16x32_bf16 tile, D = A[16,32] @ B[32,16]
// with the fragment maps assumed above. Used by tests to pin the
// lane->element mapping before trusting the conv kernel.
// ---------------------------------------------------------------------
__global__ void mfma_probe_kernel(const __bf16* __restrict__ A,
                                  const __bf16* __restrict__ B,
                                  float* __restrict__ D) {
  int lane = threadIdx.x;
  bf16x8 a_frag, b_frag;
  int m = lane & 15;
  int k0 = (lane >> 4) * 8;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a_frag[j] = A[m * 32 + k0 + j];       // A[m][k] row-major
    b_frag[j] = B[(k0 + j) * 16 + (lane & 15)];  // B[k][n] row-major
  }
  f32x4 acc = (f32x4)(0.0f);
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_frag, b_frag, acc, 0, 0,
                                                0);
  int col = lane & 15;
  int row_base = (lane >> 4) * 4;
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    D[(row_base + reg) * 16 + col] = acc[reg];
  }
}

void mfma_probe_launch(const void* A, const void* B, float* D,
                       hipStream_t stream) {
  hipLaunchKernelGGL(mfma_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const __bf16*)A, (const __bf16*)B, D);
}

// ---------------------------------------------------------------------
// fused forward for the same stem conv: out[K,32] = im2col(x)[K,256]
// @ W[256,32] + bias, x read directly as u8 (dequant fused).
// Per workgroup: 64 output rows x all 32 channels; K-loop over the 8
// patch rows (32 patch values each, one contiguous 32-byte u8
// segment per (k-row, patch-row) — same addressing as the wrw
// kernel). W (256x32 bf16, 16 KB) is staged to LDS once.
// ---------------------------------------------------------------------
#define FWD_ROWS 64  // k-rows (output pixels) per workgroup

__global__ __launch_bounds__(256)
void conv1_fwd_kernel(const unsigned char* __restrict__ frames,
                      const __bf16* __restrict__ weight,  // [256][32]
                      const float* __restrict__ bias,     // [32]
                      __bf16* __restrict__ out,           // [K][32]
                      int64_t K, float scale) {
  __shared__ __bf16 s_x[FWD_ROWS * 32];   // one patch-row chunk
  __shared__ __bf16 s_w[CONV1_N * 32];    // full weight, staged once

  const int tid = threadIdx.x;
  const int wave = tid / MA_WAVE;  // 0..3
  const int lane = tid % MA_WAVE;
  // wave -> (2 M-tiles of 16 rows) x (2 N-tiles of 16 cols): wave w
  // owns M-tile pair row (w&1) and N-tile (w>>1)
  const int wm = wave & 1;   // 0..1 -> rows wm*32 .. wm*32+31 (2 tiles)
  const int wn = wave >> 1;  // 0..1 -> cols wn*16 .. wn*16+15

  // stage the whole weight [256][32]
  for (int i = tid; i < CONV1_N * 32 / 8; i += 256) {
    ((bf16x8*)s_w)[i] = ((const bf16x8*)weight)[i];
  }

  const int64_t row0 = (int64_t)blockIdx.x * FWD_ROWS;
  f32x4 acc[2];  // two M-tiles (16x16 each) per wave
  acc[0] = (f32x4)(0.0f);
  acc[1] = (f32x4)(0.0f);
  __syncthreads();

  for (int pr = 0; pr < CONV1_KSZ; ++pr) {  // patch rows = K chunks
    // stage 64 k-rows x 32 patch values (u8 segment -> bf16)
    {
      int krow = tid >> 2;          // 0..63
      int quarter = tid & 3;        // 8 values each
      int64_t kk = row0 + krow;
      bf16x8 q;
#pragma unroll
      for (int j = 0; j < 8; ++j) q[j] = (__bf16)0.0f;
      if (kk < K) {
        int64_t b = kk / CONV1_POS;
        int pos = (int)(kk % CONV1_POS);
        int oh = pos / CONV1_OHW, ow = pos % CONV1_OHW;
        const unsigned char* src =
            frames
            + ((b * 84 + (int64_t)oh * CONV1_STRIDE + pr) * 84
               + (int64_t)ow * CONV1_STRIDE) * CONV1_CIN
            + quarter * 8;
        uint2 raw = *(const uint2*)src;
        unsigned int words[2] = {raw.x, raw.y};
#pragma unroll
        for (int w = 0; w < 2; ++w) {
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            q[w * 4 + j] = u8_bf16((words[w] >> (8 * j)) & 0xFF, scale);
          }
        }
      }
      *(bf16x8*)(s_x + krow * 32 + quarter * 8) = q;
    }
    __syncthreads();

    // MFMA over this chunk's K=32 (patch cols x channels)
#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
      int m = wm * 32 + mt * 16 + (lane & 15);
      int k0 = (lane >> 4) * 8;
      // A[m][k]: contiguous along k -> one vector read
      bf16x8 a_frag = *(bf16x8*)(s_x + m * 32 + k0);
      bf16x8 b_frag;
      int n = wn * 16 + (lane & 15);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        b_frag[j] = s_w[(pr * 32 + k0 + j) * 32 + n];
      }
      acc[mt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_frag, b_frag, acc[mt], 0, 0, 0);
    }
    __syncthreads();
  }

  // epilogue: bias add, bf16 store (C/D map: col=lane&15,
  // row=(lane>>4)*4+reg)
#pragma unroll
  for (int mt = 0; mt < 2; ++mt) {
    int col = wn * 16 + (lane & 15);
    float b = bias != nullptr ? bias[col] : 0.0f;
    int row_base = wm * 32 + mt * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int64_t kk = row0 + row_base + reg;
      if (kk < K) {
        out[kk * 32 + col] = (__bf16)(acc[mt][reg] + b);
      }
    }
  }
}

void conv1_fwd_launch(const unsigned char* frames, const void* weight,
                      const float* bias, void* out, int64_t K,
                      float scale, hipStream_t stream) {
  int64_t grid = (K + FWD_ROWS - 1) / FWD_ROWS;
  const bool v1 = getenv("MACHIN_CONV1_V1") != nullptr;
  if (v1) {
    hipLaunchKernelGGL(conv1_fwd_kernel, dim3((unsigned)grid), dim3(256),
                       0, stream, frames, (const __bf16*)weight, bias,
                       (__bf16*)out, K, scale);
  } else {
    hipLaunchKernelGGL(conv1_fwd_v2_kernel, dim3((unsigned)grid),
                       dim3(256), 0, stream, frames,
                       (const __bf16*)weight, bias, (__bf16*)out, K,
                       scale);
  }
  HIP_CHECK(hipGetLastError());
}

// =====================================================================
// v2 kernels: tiled LDS images + ds_read_b64_tr_b16 fragment loads.
//
// v1 profile (round-1 VERDICT weak #6): the MFMA operand loads were
// per-element ds_read_u16 (72 narrow reads per wave per K-chunk) —
// LDS-instruction-bound at 1.80 ms wrw / 1.46 ms fwd vs a ~0.4 ms
// HBM bound. gfx950's transpose-read fetches 4 bf16 per lane with a
// 32 B stride (lane l passes base + (l&15)*2; elem j comes from
// base + j*32 B), so storing x/dy in contiguous [4][16] bf16 tiles
// (tile = 4 k-rows x 16 cols, 128 B) turns one 16x16x32 B-fragment
// into TWO ds instructions (k 8-rows = two stacked tiles, second via
// offset immediate). Writes stay wide (16-element rows are 32 B
// contiguous). Per wave per chunk: 18 LDS reads vs v1's 72.
//
// Tiled image layout (x): [n_tile 16][k_tile 8][4][16] bf16
//   elem (k, c) -> nt=c>>4, kt=k>>2: addr = nt*512 + kt*64
//                                         + (k&3)*16 + (c&15)
// dy image: [m_tile 2][k_tile 8][4][16] with the same inner tiles.
// =====================================================================
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

__device__ __forceinline__ bf16x4 ds_tr16(const __bf16* addr) {
  bf16x4 out;
  asm volatile("ds_read_b64_tr_b16 %0, %1"
               : "=v"(out)
               : "v"((unsigned)(uintptr_t)addr));
  return out;
}

__device__ __forceinline__ bf16x4 ds_tr16_off128(const __bf16* addr) {
  bf16x4 out;
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:128"
               : "=v"(out)
               : "v"((unsigned)(uintptr_t)addr));
  return out;
}

__device__ __forceinline__ void pack8(bf16x8& dst, bf16x4 lo, bf16x4 hi) {
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    dst[j] = lo[j];
    dst[4 + j] = hi[j];
  }
}

__global__ __launch_bounds__(256)
void conv1_wrw_v2_kernel(const __bf16* __restrict__ dy,
                         const unsigned char* __restrict__ frames,
                         float* __restrict__ out,
                         float* __restrict__ b_out, int64_t K,
                         int64_t k_per_wg, float scale) {
  __shared__ __bf16 s_dy[2 * 8 * 64];   // [mt][kt][4][16]
  __shared__ __bf16 s_x[16 * 8 * 64];   // [nt][kt][4][16]

  const int tid = threadIdx.x;
  const int wave = tid / MA_WAVE;
  const int lane = tid % MA_WAVE;
  const int mt = wave & 1;
  const int ng = wave >> 1;

  f32x4 acc[8];
#pragma unroll
  for (int i = 0; i < 8; ++i) acc[i] = (f32x4)(0.0f);
  float bias_acc = 0.0f;

  // per-lane fragment base addresses (constant across the K loop):
  // group g = lane>>4 covers k-tiles 2g, 2g+1 -> byte offset g*256
  const __bf16* a_base = s_dy + mt * 512 + (lane >> 4) * 128 + (lane & 15);
  const __bf16* b_base = s_x + ng * 8 * 512 + (lane >> 4) * 128
                         + (lane & 15);

  const int64_t k_begin = (int64_t)blockIdx.x * k_per_wg;
  const int64_t k_end = min(k_begin + k_per_wg, K);

  for (int64_t kc = k_begin; kc < k_end; kc += KC) {
    // ---- stage dy chunk into [mt][kt][4][16] tiles ----------------
    if (tid < 128) {
      int row = tid >> 2;        // k-row 0..31
      int seg = tid & 3;         // 8 cols each
      int64_t kk = kc + row;
      bf16x8 v;
#pragma unroll
      for (int j = 0; j < 8; ++j) v[j] = (__bf16)0.0f;
      float bsum = 0.0f;
      if (kk < k_end) {
        v = *(const bf16x8*)(dy + kk * CONV1_COUT + seg * 8);
      }
      __bf16* dst = s_dy + (seg >> 1) * 512 + (row >> 2) * 64
                    + (row & 3) * 16 + (seg & 1) * 8;
      *(bf16x8*)dst = v;
      (void)bsum;
    }
    // ---- stage x chunk into [nt][kt][4][16] tiles -----------------
    // thread (krow = tid&31, r = tid>>3? ) -> mapping: krow low bits
    // keeps the 8-lane write groups on distinct banks
    {
      int krow = tid & 31;
      int r = tid >> 5;          // patch row 0..7
      int64_t kk = kc + krow;
      bf16x8 q0, q1, q2, q3;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        q0[j] = (__bf16)0.0f;
        q1[j] = (__bf16)0.0f;
        q2[j] = (__bf16)0.0f;
        q3[j] = (__bf16)0.0f;
      }
      if (kk < k_end) {
        int64_t b = kk / CONV1_POS;
        int pos = (int)(kk % CONV1_POS);
        int oh = pos / CONV1_OHW, ow = pos % CONV1_OHW;
        const unsigned char* src =
            frames
            + ((b * 84 + (int64_t)oh * CONV1_STRIDE + r) * 84
               + (int64_t)ow * CONV1_STRIDE) * CONV1_CIN;
        uint4 raw0 = *(const uint4*)src;
        uint4 raw1 = *(const uint4*)(src + 16);
        unsigned int words[8] = {raw0.x, raw0.y, raw0.z, raw0.w,
                                 raw1.x, raw1.y, raw1.z, raw1.w};
#pragma unroll
        for (int w = 0; w < 2; ++w) {
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            q0[w * 4 + j] = u8_bf16((words[w] >> (8 * j)) & 0xFF, scale);
            q1[w * 4 + j] =
                u8_bf16((words[w + 2] >> (8 * j)) & 0xFF, scale);
            q2[w * 4 + j] =
                u8_bf16((words[w + 4] >> (8 * j)) & 0xFF, scale);
            q3[w * 4 + j] =
                u8_bf16((words[w + 6] >> (8 * j)) & 0xFF, scale);
          }
        }
      }
      // cols r*32 .. r*32+31 span n-tiles 2r and 2r+1; each 16-col
      // half is one contiguous tile row (32 B = 2 vector writes)
      __bf16* base = s_x + (krow >> 2) * 64 + (krow & 3) * 16;
      bf16x8* t0 = (bf16x8*)(base + (r * 2) * 512);
      bf16x8* t1 = (bf16x8*)(base + (r * 2 + 1) * 512);
      t0[0] = q0;
      t0[1] = q1;
      t1[0] = q2;
      t1[1] = q3;
    }
    __syncthreads();

    // ---- fragments via transpose-reads ----------------------------
    bf16x8 a_frag;
    pack8(a_frag, ds_tr16(a_base), ds_tr16_off128(a_base));
    // fused bias from the fragment already in registers: lane group g
    // of an ng==0 wave holds dy[k=8g..8g+7][col mt*16+(lane&15)]
    if (ng == 0) {
#pragma unroll
      for (int j = 0; j < 8; ++j) bias_acc += (float)a_frag[j];
    }
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
      const __bf16* bb = b_base + nt * 512;
      bf16x8 b_frag;
      pack8(b_frag, ds_tr16(bb), ds_tr16_off128(bb));
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_frag, b_frag, acc[nt], 0, 0, 0);
    }
    __syncthreads();
  }

  if (ng == 0) {
    // reduce the 4 lane groups' partial column sums, then one
    // atomic per column from lanes 0..15
    bias_acc += __shfl_down(bias_acc, 32, 64);
    bias_acc += __shfl_down(bias_acc, 16, 64);
    if (lane < 16) atomicAdd(&b_out[mt * 16 + lane], bias_acc);
  }
#pragma unroll
  for (int nt = 0; nt < 8; ++nt) {
    int col = (ng * 8 + nt) * 16 + (lane & 15);
    int row_base = mt * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      atomicAdd(&out[(row_base + reg) * CONV1_N + col], acc[nt][reg]);
    }
  }
}

// fwd v2: weight staged ONCE into [nt 2][kt 64][4][16] tiles; the
// B fragment becomes 2 transpose-reads per (pr) chunk, hoisted out of
// the mt loop (v1 re-read it per element AND per m-tile).
__global__ __launch_bounds__(256)
void conv1_fwd_v2_kernel(const unsigned char* __restrict__ frames,
                         const __bf16* __restrict__ weight,  // [256][32]
                         const float* __restrict__ bias,
                         __bf16* __restrict__ out, int64_t K,
                         float scale) {
  __shared__ __bf16 s_x[FWD_ROWS * 32];
  __shared__ __bf16 s_w[2 * 64 * 64];  // [nt][kt][4][16]

  const int tid = threadIdx.x;
  const int wave = tid / MA_WAVE;
  const int lane = tid % MA_WAVE;
  const int wm = wave & 1;
  const int wn = wave >> 1;

  // stage weight into tiles: thread tid owns row k=tid (32 cols)
  {
    int k = tid;
    bf16x8 r0 = ((const bf16x8*)(weight + k * 32))[0];
    bf16x8 r1 = ((const bf16x8*)(weight + k * 32))[1];
    bf16x8 r2 = ((const bf16x8*)(weight + k * 32))[2];
    bf16x8 r3 = ((const bf16x8*)(weight + k * 32))[3];
    __bf16* base = s_w + (k >> 2) * 64 + (k & 3) * 16;
    ((bf16x8*)base)[0] = r0;
    ((bf16x8*)base)[1] = r1;
    ((bf16x8*)(base + 64 * 64))[0] = r2;
    ((bf16x8*)(base + 64 * 64))[1] = r3;
  }

  const int64_t row0 = (int64_t)blockIdx.x * FWD_ROWS;
  f32x4 acc[2];
  acc[0] = (f32x4)(0.0f);
  acc[1] = (f32x4)(0.0f);
  const __bf16* w_base = s_w + wn * 64 * 64 + (lane >> 4) * 128
                         + (lane & 15);
  __syncthreads();

  for (int pr = 0; pr < CONV1_KSZ; ++pr) {
    {
      int krow = tid >> 2;
      int quarter = tid & 3;
      int64_t kk = row0 + krow;
      bf16x8 q;
#pragma unroll
      for (int j = 0; j < 8; ++j) q[j] = (__bf16)0.0f;
      if (kk < K) {
        int64_t b = kk / CONV1_POS;
        int pos = (int)(kk % CONV1_POS);
        int oh = pos / CONV1_OHW, ow = pos % CONV1_OHW;
        const unsigned char* src =
            frames
            + ((b * 84 + (int64_t)oh * CONV1_STRIDE + pr) * 84
               + (int64_t)ow * CONV1_STRIDE) * CONV1_CIN
            + quarter * 8;
        uint2 raw = *(const uint2*)src;
        unsigned int words[2] = {raw.x, raw.y};
#pragma unroll
        for (int w = 0; w < 2; ++w) {
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            q[w * 4 + j] = u8_bf16((words[w] >> (8 * j)) & 0xFF, scale);
          }
        }
      }
      *(bf16x8*)(s_x + krow * 32 + quarter * 8) = q;
    }
    __syncthreads();

    // B fragment for this pr chunk: k-tiles pr*8 + {2g, 2g+1}
    const __bf16* wb = w_base + pr * 8 * 64;
    bf16x8 b_frag;
    pack8(b_frag, ds_tr16(wb), ds_tr16_off128(wb));
#pragma unroll
    for (int mt2 = 0; mt2 < 2; ++mt2) {
      int m = wm * 32 + mt2 * 16 + (lane & 15);
      int k0 = (lane >> 4) * 8;
      bf16x8 a_frag = *(bf16x8*)(s_x + m * 32 + k0);
      acc[mt2] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_frag, b_frag, acc[mt2], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int mt2 = 0; mt2 < 2; ++mt2) {
    int col = wn * 16 + (lane & 15);
    float b = bias != nullptr ? bias[col] : 0.0f;
    int row_base = wm * 32 + mt2 * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      int64_t kk = row0 + row_base + reg;
      if (kk < K) {
        out[kk * 32 + col] = (__bf16)(acc[mt2][reg] + b);
      }
    }
  }
}
