// Shared helpers for machin_amd gfx950 kernels.
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>
#include <cstdlib>

#define MA_WAVE 64  // CDNA4 wavefront width (not 32)

// Grid sizing for memory-bound kernels (guide §6 G11): cap blocks at
// ~8 per CU x 256 CUs and grid-stride the rest.
static inline int ma_grid(int64_t total, int block) {
  int64_t blocks = (total + block - 1) / block;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

#define HIP_CHECK(expr)                                                    \
  do {                                                                     \
    hipError_t _e = (expr);                                                \
    if (_e != hipSuccess) {                                                \
      throw std::runtime_error(std::string("HIP error: ") +                \
                               hipGetErrorString(_e));                     \
    }                                                                      \
  } while (0)

// ---------------------------------------------------------------------
// Philox 4x32-10 counter-based RNG (public algorithm, Salmon et al.).
// One call produces 4 uint32 from (seed, subsequence, offset).
// ---------------------------------------------------------------------
struct ma_philox_state {
  uint32_t ctr[4];
  uint32_t key[2];
};

__device__ __forceinline__ void ma_philox_init(ma_philox_state& s,
                                               uint64_t seed,
                                               uint64_t subsequence,
                                               uint64_t offset) {
  s.key[0] = (uint32_t)(seed);
  s.key[1] = (uint32_t)(seed >> 32);
  s.ctr[0] = (uint32_t)(offset);
  s.ctr[1] = (uint32_t)(offset >> 32);
  s.ctr[2] = (uint32_t)(subsequence);
  s.ctr[3] = (uint32_t)(subsequence >> 32);
}

__device__ __forceinline__ uint32_t ma_mulhilo(uint32_t a, uint32_t b,
                                               uint32_t* hi) {
  uint64_t p = (uint64_t)a * (uint64_t)b;
  *hi = (uint32_t)(p >> 32);
  return (uint32_t)p;
}

__device__ __forceinline__ void ma_philox_round(ma_philox_state& s) {
  constexpr uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  uint32_t hi0, hi1;
  uint32_t lo0 = ma_mulhilo(M0, s.ctr[0], &hi0);
  uint32_t lo1 = ma_mulhilo(M1, s.ctr[2], &hi1);
  uint32_t c0 = hi1 ^ s.ctr[1] ^ s.key[0];
  uint32_t c1 = lo1;
  uint32_t c2 = hi0 ^ s.ctr[3] ^ s.key[1];
  uint32_t c3 = lo0;
  s.ctr[0] = c0; s.ctr[1] = c1; s.ctr[2] = c2; s.ctr[3] = c3;
  constexpr uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
  s.key[0] += W0; s.key[1] += W1;
}

__device__ __forceinline__ void ma_philox4(uint64_t seed, uint64_t subseq,
                                           uint64_t offset, uint32_t out[4]) {
  ma_philox_state s;
  ma_philox_init(s, seed, subseq, offset);
#pragma unroll
  for (int i = 0; i < 10; ++i) ma_philox_round(s);
  out[0] = s.ctr[0]; out[1] = s.ctr[1]; out[2] = s.ctr[2]; out[3] = s.ctr[3];
}

// uniform in (0, 1]
__device__ __forceinline__ float ma_u32_to_uniform(uint32_t x) {
  return (float)(x >> 8) * (1.0f / 16777216.0f) + (1.0f / 33554432.0f);
}

// two normals from two uniforms (Box-Muller)
__device__ __forceinline__ void ma_box_muller(float u1, float u2, float* n1,
                                              float* n2) {
  float r = sqrtf(-2.0f * logf(u1));
  float s, c;
  __sincosf(6.28318530717958647692f * u2, &s, &c);
  *n1 = r * c;
  *n2 = r * s;
}
