// gfx950 fused distribution kernels: Gaussian / tanh-squashed Gaussian
// sample + log-prob in one pass, on-device noise generation (Philox).
//
// Reference semantics being fused (several eager torch ops each):
//   machin/frame/algorithms/sac.py policy sampling (tanh-Gaussian)
//   machin/frame/noise/generator.py:33-194 (Normal/OU generators)
//
// Layout [B, D]: one wavefront per batch row, lanes stride over D,
// log-prob reduced across the wave with shfl_xor (64-lane tree).
#include "common.h"

#define LOG_SQRT_2PI 0.91893853320467274178f  // 0.5*log(2*pi)

__device__ __forceinline__ float ma_wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    v += __shfl_xor(v, off, MA_WAVE);
  }
  return v;
}

// one normal sample per (index) from philox counter
__device__ __forceinline__ float ma_normal1(uint64_t seed, uint64_t subseq,
                                            uint64_t offset) {
  uint32_t r[4];
  ma_philox4(seed, subseq, offset, r);
  float n1, n2;
  ma_box_muller(ma_u32_to_uniform(r[0]), ma_u32_to_uniform(r[1]), &n1, &n2);
  return n1;
}

__global__ void gaussian_sample_logprob_kernel(
    const float* __restrict__ mu, const float* __restrict__ log_std,
    float* __restrict__ act, float* __restrict__ logp, int64_t B, int64_t D,
    uint64_t seed, uint64_t offset, int tanh_squash, float epsilon) {
  const int wave = threadIdx.x / MA_WAVE;
  const int lane = threadIdx.x % MA_WAVE;
  const int waves_per_block = blockDim.x / MA_WAVE;
  for (int64_t row = blockIdx.x * waves_per_block + wave; row < B;
       row += (int64_t)gridDim.x * waves_per_block) {
    float lp = 0.0f;
    for (int64_t d = lane; d < D; d += MA_WAVE) {
      int64_t k = row * D + d;
      float m = mu[k];
      float ls = log_std[k];
      float std = __expf(ls);
      float eps = ma_normal1(seed, (uint64_t)k, offset);
      float u = m + eps * std;
      float term = -0.5f * eps * eps - ls - LOG_SQRT_2PI;
      if (tanh_squash) {
        float a = tanhf(u);
        term -= logf(1.0f - a * a + epsilon);
        act[k] = a;
      } else {
        act[k] = u;
      }
      lp += term;
    }
    lp = ma_wave_sum(lp);
    if (lane == 0) logp[row] = lp;
  }
}

// log-prob of GIVEN actions under N(mu, std) (optionally atanh first
// for tanh-squashed policies re-evaluating stored actions).
__global__ void gaussian_logprob_kernel(const float* __restrict__ mu,
                                        const float* __restrict__ log_std,
                                        const float* __restrict__ act,
                                        float* __restrict__ logp, int64_t B,
                                        int64_t D, int tanh_squash,
                                        float epsilon) {
  const int wave = threadIdx.x / MA_WAVE;
  const int lane = threadIdx.x % MA_WAVE;
  const int waves_per_block = blockDim.x / MA_WAVE;
  for (int64_t row = blockIdx.x * waves_per_block + wave; row < B;
       row += (int64_t)gridDim.x * waves_per_block) {
    float lp = 0.0f;
    for (int64_t d = lane; d < D; d += MA_WAVE) {
      int64_t k = row * D + d;
      float a = act[k];
      float u = a;
      float sq_term = 0.0f;
      if (tanh_squash) {
        float c = fminf(fmaxf(a, -1.0f + 1e-6f), 1.0f - 1e-6f);
        u = 0.5f * (logf(1.0f + c) - logf(1.0f - c));  // atanh
        sq_term = logf(1.0f - c * c + epsilon);
      }
      float ls = log_std[k];
      float z = (u - mu[k]) * __expf(-ls);
      lp += -0.5f * z * z - ls - LOG_SQRT_2PI - sq_term;
    }
    lp = ma_wave_sum(lp);
    if (lane == 0) logp[row] = lp;
  }
}

__global__ void normal_noise_kernel(float* __restrict__ x, int64_t n,
                                    float mean, float std, uint64_t seed,
                                    uint64_t offset, int add) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float v = mean + std * ma_normal1(seed, (uint64_t)i, offset);
    x[i] = add ? (x[i] + v) : v;
  }
}

// Ornstein-Uhlenbeck step: x += theta*(mu - x)*dt + sigma*sqrt(dt)*N(0,1)
__global__ void ou_update_kernel(float* __restrict__ x, int64_t n, float mu,
                                 float theta, float sigma, float dt,
                                 uint64_t seed, uint64_t offset) {
  float sdt = sqrtf(dt);
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    float v = x[i];
    v += theta * (mu - v) * dt + sigma * sdt * ma_normal1(seed, (uint64_t)i, offset);
    x[i] = v;
  }
}

void gaussian_sample_logprob_launch(const float* mu, const float* log_std,
                                    float* act, float* logp, int64_t B,
                                    int64_t D, uint64_t seed, uint64_t offset,
                                    int tanh_squash, float epsilon,
                                    hipStream_t stream) {
  const int block = 256;
  const int wpb = block / MA_WAVE;
  int grid = ma_grid((B + wpb - 1) / wpb, 1);
  hipLaunchKernelGGL(gaussian_sample_logprob_kernel, dim3(grid), dim3(block),
                     0, stream, mu, log_std, act, logp, B, D, seed, offset,
                     tanh_squash, epsilon);
}

void gaussian_logprob_launch(const float* mu, const float* log_std,
                             const float* act, float* logp, int64_t B,
                             int64_t D, int tanh_squash, float epsilon,
                             hipStream_t stream) {
  const int block = 256;
  const int wpb = block / MA_WAVE;
  int grid = ma_grid((B + wpb - 1) / wpb, 1);
  hipLaunchKernelGGL(gaussian_logprob_kernel, dim3(grid), dim3(block), 0,
                     stream, mu, log_std, act, logp, B, D, tanh_squash,
                     epsilon);
}

void normal_noise_launch(float* x, int64_t n, float mean, float std,
                         uint64_t seed, uint64_t offset, int add,
                         hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(normal_noise_kernel, dim3(ma_grid(n, block)),
                     dim3(block), 0, stream, x, n, mean, std, seed, offset,
                     add);
}

void ou_update_launch(float* x, int64_t n, float mu, float theta, float sigma,
                      float dt, uint64_t seed, uint64_t offset,
                      hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(ou_update_kernel, dim3(ma_grid(n, block)), dim3(block),
                     0, stream, x, n, mu, theta, sigma, dt, seed, offset);
}
