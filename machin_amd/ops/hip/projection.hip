// gfx950 categorical (C51) distribution projection.
//
// Reference semantics: machin/frame/algorithms/rainbow.py:221-301
// (python index_add_ scatter). Here: one wavefront per batch row, the
// projected row accumulated in LDS with LDS atomics (no global
// atomics, guide §6 G12), then written out coalesced. All waves run a
// uniform iteration count so __syncthreads() is safe.
#include "common.h"

#define PROJ_MAX_ATOMS 256
#define PROJ_WAVES_PER_BLOCK 4

__global__ void categorical_projection_kernel(
    const float* __restrict__ next_dist, const float* __restrict__ rew,
    const float* __restrict__ nd, float* __restrict__ out, int64_t B,
    int64_t A, float gamma, float v_min, float v_max, float delta_z,
    int64_t iters) {
  __shared__ float acc[PROJ_WAVES_PER_BLOCK][PROJ_MAX_ATOMS];
  const int wave = threadIdx.x / MA_WAVE;
  const int lane = threadIdx.x % MA_WAVE;
  const int64_t row_stride = (int64_t)gridDim.x * PROJ_WAVES_PER_BLOCK;

  for (int64_t it = 0; it < iters; ++it) {
    int64_t row = it * row_stride + blockIdx.x * PROJ_WAVES_PER_BLOCK + wave;
    bool active = row < B;
    if (active) {
      for (int64_t a = lane; a < A; a += MA_WAVE) acc[wave][a] = 0.0f;
    }
    __syncthreads();
    if (active) {
      float r = rew[row];
      float ndr = nd[row];
      for (int64_t a = lane; a < A; a += MA_WAVE) {
        float z = v_min + delta_z * (float)a;
        float tz = r + gamma * ndr * z;
        tz = fminf(fmaxf(tz, v_min), v_max);
        float pos = (tz - v_min) / delta_z;
        int64_t lo = (int64_t)floorf(pos);
        int64_t hi = (int64_t)ceilf(pos);
        float p = next_dist[row * A + a];
        if (lo == hi) {
          atomicAdd(&acc[wave][lo], p);
        } else {
          atomicAdd(&acc[wave][lo], p * ((float)hi - pos));
          atomicAdd(&acc[wave][hi], p * (pos - (float)lo));
        }
      }
    }
    __syncthreads();
    if (active) {
      for (int64_t a = lane; a < A; a += MA_WAVE) {
        out[row * A + a] = acc[wave][a];
      }
    }
    __syncthreads();
  }
}

void categorical_projection_launch(const float* next_dist, const float* rew,
                                   const float* nd, float* out, int64_t B,
                                   int64_t A, float gamma, float v_min,
                                   float v_max, hipStream_t stream) {
  float delta_z = (v_max - v_min) / (float)(A - 1);
  const int block = MA_WAVE * PROJ_WAVES_PER_BLOCK;
  int64_t row_groups = (B + PROJ_WAVES_PER_BLOCK - 1) / PROJ_WAVES_PER_BLOCK;
  int grid = ma_grid(row_groups, 1);
  int64_t rows_per_pass = (int64_t)grid * PROJ_WAVES_PER_BLOCK;
  int64_t iters = (B + rows_per_pass - 1) / rows_per_pass;
  hipLaunchKernelGGL(categorical_projection_kernel, dim3(grid), dim3(block),
                     0, stream, next_dist, rew, nd, out, B, A, gamma, v_min,
                     v_max, delta_z, iters);
}
