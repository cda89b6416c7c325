// gfx950 sum-tree kernels for prioritized replay.
//
// Replaces the numpy WeightTree hot loops of the reference
// (machin/frame/buffers/prioritized_buffer.py:96-231) with device
// kernels over the same heap layout used by
// machin_amd/frame/buffers/prioritized_buffer.py: one float32 array of
// length 2*capacity, root at index 1, leaves at [capacity, 2*capacity).
//
// Design notes (guide refs):
// - update: scatter leaves, then one kernel per level recomputing
//   parent = left + right for each touched path. Duplicate parents
//   recompute the same value -> benign, no atomics needed.
// - sample: each query walks root->leaf. The top TOP_LEVELS of the
//   tree (16 KiB) are staged in LDS per block (guide §2: LDS to cut
//   repeated global reads; the top levels are read by EVERY query).
// - build: bottom-up level sweep, coalesced adds.
#include "common.h"

// ---------------------------------------------------------------------
// scatter new leaf weights
// ---------------------------------------------------------------------
__global__ void sumtree_scatter_kernel(float* __restrict__ tree,
                                       const int64_t* __restrict__ idx,
                                       const float* __restrict__ w,
                                       int64_t m, int64_t cap) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < m;
       i += (int64_t)gridDim.x * blockDim.x) {
    tree[cap + idx[i]] = w[i];
  }
}

// recompute the ancestors of touched leaves at one level.
// level shift s: parent node of leaf idx is (cap + idx) >> s.
__global__ void sumtree_repair_kernel(float* __restrict__ tree,
                                      const int64_t* __restrict__ idx,
                                      int64_t m, int64_t cap, int shift) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < m;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t p = (cap + idx[i]) >> shift;
    tree[p] = tree[2 * p] + tree[2 * p + 1];
  }
}

// full level recompute for build: parents in [lo, lo+n)
__global__ void sumtree_level_kernel(float* __restrict__ tree, int64_t lo,
                                     int64_t n) {
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t p = lo + i;
    tree[p] = tree[2 * p] + tree[2 * p + 1];
  }
}

// ---------------------------------------------------------------------
// stratified sampling walk. Top levels staged in LDS.
// ---------------------------------------------------------------------
#define SUMTREE_LDS_NODES 4096  // nodes [1, 4096): top 12 levels, 16 KiB

__global__ void sumtree_sample_kernel(const float* __restrict__ tree,
                                      const float* __restrict__ u,
                                      int64_t* __restrict__ out,
                                      int64_t n, int64_t cap, int depth,
                                      int64_t size) {
  __shared__ float top[SUMTREE_LDS_NODES];
  // cooperative stage of the top of the tree (coalesced float reads)
  int64_t stage = min((int64_t)SUMTREE_LDS_NODES, 2 * cap);
  for (int64_t i = threadIdx.x; i < stage; i += blockDim.x) {
    top[i] = tree[i];
  }
  __syncthreads();

  for (int64_t q = blockIdx.x * blockDim.x + threadIdx.x; q < n;
       q += (int64_t)gridDim.x * blockDim.x) {
    float w = u[q];
    int64_t node = 1;
    for (int d = 0; d < depth; ++d) {
      int64_t left = node << 1;
      float lw = (left + 1 < stage) ? top[left] : tree[left];
      // branchless step: go right iff w > left subtree weight
      bool right = w > lw;
      w = right ? (w - lw) : w;
      node = left + (right ? 1 : 0);
    }
    int64_t leaf = node - cap;
    if (leaf >= size) leaf = size - 1;
    if (leaf < 0) leaf = 0;
    out[q] = leaf;
  }
}

// ---------------------------------------------------------------------
// host-side drivers (called from bindings.cpp)
// ---------------------------------------------------------------------
void sumtree_update_launch(float* tree, const int64_t* idx, const float* w,
                           int64_t m, int64_t cap, int depth,
                           hipStream_t stream) {
  if (m == 0) return;
  const int block = 256;
  int grid = ma_grid(m, block);
  hipLaunchKernelGGL(sumtree_scatter_kernel, dim3(grid), dim3(block), 0,
                     stream, tree, idx, w, m, cap);
  for (int shift = 1; shift <= depth; ++shift) {
    hipLaunchKernelGGL(sumtree_repair_kernel, dim3(grid), dim3(block), 0,
                       stream, tree, idx, m, cap, shift);
  }
}

void sumtree_build_launch(float* tree, int64_t cap, hipStream_t stream) {
  const int block = 256;
  for (int64_t n = cap >> 1, lo = cap >> 1; n >= 1; n >>= 1, lo >>= 1) {
    hipLaunchKernelGGL(sumtree_level_kernel, dim3(ma_grid(n, block)),
                       dim3(block), 0, stream, tree, lo, n);
  }
}

void sumtree_sample_launch(const float* tree, const float* u, int64_t* out,
                           int64_t n, int64_t cap, int depth, int64_t size,
                           hipStream_t stream) {
  if (n == 0) return;
  const int block = 256;
  hipLaunchKernelGGL(sumtree_sample_kernel, dim3(ma_grid(n, block)),
                     dim3(block), 0, stream, tree, u, out, n, cap, depth,
                     size);
}
