// gfx950 reverse-time scan kernels: discounted returns, GAE, V-trace.
//
// Reference semantics being replaced (python loops):
//   machin/frame/algorithms/a2c.py:269-326   (GAE / discounted return)
//   machin/frame/algorithms/impala.py:317-371 (V-trace)
//
// Layout: [T, B] row-major, so at fixed t consecutive b are contiguous
// -> each thread owns one batch column b and the per-step loads of a
// wave are fully coalesced (guide §2). T is small (rollout length),
// B*num_cols fills the chip; per-thread sequential recursion over T is
// the natural CDNA formulation (no cross-lane dependency at all).
#include "common.h"

__global__ void discounted_returns_kernel(const float* __restrict__ rew,
                                          const float* __restrict__ nd,
                                          const float* __restrict__ bootstrap,
                                          float* __restrict__ out, int64_t T,
                                          int64_t B, float gamma) {
  for (int64_t b = blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += (int64_t)gridDim.x * blockDim.x) {
    float r = bootstrap[b];
    for (int64_t t = T - 1; t >= 0; --t) {
      int64_t k = t * B + b;
      r = rew[k] + gamma * nd[k] * r;
      out[k] = r;
    }
  }
}

__global__ void gae_kernel(const float* __restrict__ rew,
                           const float* __restrict__ val,
                           const float* __restrict__ next_val,
                           const float* __restrict__ nd,
                           float* __restrict__ out, int64_t T, int64_t B,
                           float gamma, float lam) {
  for (int64_t b = blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += (int64_t)gridDim.x * blockDim.x) {
    float acc = 0.0f;
    for (int64_t t = T - 1; t >= 0; --t) {
      int64_t k = t * B + b;
      float ndk = nd[k];
      float delta = rew[k] + gamma * ndk * next_val[k] - val[k];
      acc = delta + gamma * lam * ndk * acc;
      out[k] = acc;
    }
  }
}

// V-trace: one backward pass producing both vs and pg advantages.
// carry vs[t+1] while walking t = T-1 .. 0.
__global__ void vtrace_kernel(const float* __restrict__ blp,
                              const float* __restrict__ tlp,
                              const float* __restrict__ rew,
                              const float* __restrict__ val,
                              const float* __restrict__ bootstrap,
                              const float* __restrict__ nd,
                              float* __restrict__ vs_out,
                              float* __restrict__ pg_adv, int64_t T,
                              int64_t B, float gamma, float rho_clip,
                              float c_clip, float pg_rho_clip) {
  for (int64_t b = blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += (int64_t)gridDim.x * blockDim.x) {
    float acc = 0.0f;             // vs[t] - V[t] carry
    float vs_next = bootstrap[b]; // vs[t+1]
    float v_next = bootstrap[b];  // V[t+1]
    for (int64_t t = T - 1; t >= 0; --t) {
      int64_t k = t * B + b;
      float rho = __expf(tlp[k] - blp[k]);
      float ndk = nd[k];
      float vk = val[k];
      float td = rew[k] + gamma * ndk * v_next - vk;
      float delta = fminf(rho, rho_clip) * td;
      acc = delta + gamma * ndk * fminf(rho, c_clip) * acc;
      float vs_t = acc + vk;
      pg_adv[k] =
          fminf(rho, pg_rho_clip) * (rew[k] + gamma * ndk * vs_next - vk);
      vs_out[k] = vs_t;
      vs_next = vs_t;
      v_next = vk;
    }
  }
}

void discounted_returns_launch(const float* rew, const float* nd,
                               const float* bootstrap, float* out, int64_t T,
                               int64_t B, float gamma, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(discounted_returns_kernel, dim3(ma_grid(B, block)),
                     dim3(block), 0, stream, rew, nd, bootstrap, out, T, B,
                     gamma);
}

void gae_launch(const float* rew, const float* val, const float* next_val,
                const float* nd, float* out, int64_t T, int64_t B, float gamma,
                float lam, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(gae_kernel, dim3(ma_grid(B, block)), dim3(block), 0,
                     stream, rew, val, next_val, nd, out, T, B, gamma, lam);
}

void vtrace_launch(const float* blp, const float* tlp, const float* rew,
                   const float* val, const float* bootstrap, const float* nd,
                   float* vs_out, float* pg_adv, int64_t T, int64_t B,
                   float gamma, float rho_clip, float c_clip,
                   float pg_rho_clip, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(vtrace_kernel, dim3(ma_grid(B, block)), dim3(block), 0,
                     stream, blp, tlp, rew, val, bootstrap, nd, vs_out,
                     pg_adv, T, B, gamma, rho_clip, c_clip, pg_rho_clip);
}

// n-step truncated return: G_t = sum_{k<n} gamma^k r_{t+k} * prod_{j<k}
// alive_{t+j}. One thread per (t, b) cell; reads are coalesced across b
// at each k offset (layout [T, B]); n is tiny (2-5) so the inner loop
// is register-resident. Replaces the reference python double loop
// (machin/frame/algorithms/rainbow.py:179-189).
__global__ void nstep_returns_kernel(const float* __restrict__ rew,
                                     const float* __restrict__ alive,
                                     float* __restrict__ out, int64_t T,
                                     int64_t B, float gamma, int n) {
  int64_t total = T * B;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (int64_t)gridDim.x * blockDim.x) {
    int64_t t = i / B;
    int64_t b = i - t * B;
    float g = 0.0f, factor = 1.0f, a = 1.0f;
    for (int k = 0; k < n && t + k < T; ++k) {
      int64_t kk = (t + k) * B + b;
      g += factor * a * rew[kk];
      a *= alive[kk];
      factor *= gamma;
    }
    out[i] = g;
  }
}

void nstep_returns_launch(const float* rew, const float* alive, float* out,
                          int64_t T, int64_t B, float gamma, int n,
                          hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(nstep_returns_kernel, dim3(ma_grid(T * B, block)),
                     dim3(block), 0, stream, rew, alive, out, T, B, gamma,
                     n);
}

// batch-major V-trace: identical math to vtrace_kernel over [B, T]
// row-major inputs (each batch row contiguous). Lets the e2e pipeline
// train straight from segment-major HBM pools with ZERO-copy frame
// gathers (the [T, B] layout forced a 1.2 GB transpose per step).
// Tiny tensors (5 x T*B floats, L2-resident), so the per-lane-
// contiguous access pattern is immaterial.
__global__ void vtrace_bt_kernel(const float* __restrict__ blp,
                                 const float* __restrict__ tlp,
                                 const float* __restrict__ rew,
                                 const float* __restrict__ val,
                                 const float* __restrict__ bootstrap,
                                 const float* __restrict__ nd,
                                 float* __restrict__ vs_out,
                                 float* __restrict__ pg_adv, int64_t T,
                                 int64_t B, float gamma, float rho_clip,
                                 float c_clip, float pg_rho_clip) {
  for (int64_t b = blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += (int64_t)gridDim.x * blockDim.x) {
    float acc = 0.0f;
    float vs_next = bootstrap[b];
    float v_next = bootstrap[b];
    const int64_t row = b * T;
    for (int64_t t = T - 1; t >= 0; --t) {
      int64_t k = row + t;
      float rho = __expf(tlp[k] - blp[k]);
      float ndk = nd[k];
      float vk = val[k];
      float td = rew[k] + gamma * ndk * v_next - vk;
      float delta = fminf(rho, rho_clip) * td;
      acc = delta + gamma * ndk * fminf(rho, c_clip) * acc;
      float vs_t = acc + vk;
      pg_adv[k] =
          fminf(rho, pg_rho_clip) * (rew[k] + gamma * ndk * vs_next - vk);
      vs_out[k] = vs_t;
      vs_next = vs_t;
      v_next = vk;
    }
  }
}

void vtrace_bt_launch(const float* blp, const float* tlp, const float* rew,
                      const float* val, const float* bootstrap,
                      const float* nd, float* vs_out, float* pg_adv,
                      int64_t T, int64_t B, float gamma, float rho_clip,
                      float c_clip, float pg_rho_clip,
                      hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(vtrace_bt_kernel, dim3(ma_grid(B, block)),
                     dim3(block), 0, stream, blp, tlp, rew, val,
                     bootstrap, nd, vs_out, pg_adv, T, B, gamma,
                     rho_clip, c_clip, pg_rho_clip);
}
