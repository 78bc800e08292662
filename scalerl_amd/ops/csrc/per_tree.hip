// Device-resident prioritized-replay sum tree (SURVEY.md §2.2 flagship
// native component).  Reimplements the semantics of the reference's
// array-backed segment tree (data/segment_tree.py:7-197) and
// PrioritizedReplayBuffer (data/replay_buffer.py:276-381) with the tree
// living in HBM and updated/sampled by kernels — no host round-trips.
//
// Layout: tree = float[2*M], M = capacity rounded to a power of two.
// Leaves tree[M + i] hold p_i^alpha; internal node j = tree[2j] + tree[2j+1];
// tree[1] = total mass.  Concurrent batched updates are safe including
// duplicate indices: atomicExch on the leaf yields per-thread deltas that
// telescope, and ancestor updates are atomicAdd of those deltas.

#include "common.h"

extern "C" __global__ void per_update_kernel(
    float* __restrict__ tree, long M,
    const long* __restrict__ idx, const float* __restrict__ prio, int B) {
  for (int k = blockIdx.x * blockDim.x + threadIdx.x; k < B;
       k += gridDim.x * blockDim.x) {
    long node = M + idx[k];
    const float old = atomicExch(&tree[node], prio[k]);
    const float delta = prio[k] - old;
    for (node >>= 1; node >= 1; node >>= 1)
      atomicAdd(&tree[node], delta);
  }
}

extern "C" int per_update(float* tree, long M, const long* idx,
                          const float* prio, long B, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(per_update_kernel, dim3(grid_1d(B, block)), dim3(block),
                     0, stream, tree, M, idx, prio, (int)B);
  CHECK_LAUNCH();
  return 0;
}

// Stratified proportional sampling (replay_buffer.py:353-368): sample k
// draws u_k = (k + xi_k)/B * total, then root-to-leaf prefix-sum descent
// (segment_tree.py:139-166).
extern "C" __global__ void per_sample_kernel(
    const float* __restrict__ tree, long M, long size,
    const float* __restrict__ uniforms,  // [B] in [0,1)
    int B, long* __restrict__ idx_out, float* __restrict__ prio_out) {
  const float total = tree[1];
  for (int k = blockIdx.x * blockDim.x + threadIdx.x; k < B;
       k += gridDim.x * blockDim.x) {
    float u = (k + uniforms[k]) / (float)B * total;
    long node = 1;
    while (node < M) {
      const long left = node << 1;
      const float lv = tree[left];
      if (u <= lv) {
        node = left;
      } else {
        u -= lv;
        node = left + 1;
      }
    }
    long i = node - M;
    if (i >= size) i = size - 1;  // guard fp edge past the last live leaf
    idx_out[k] = i;
    prio_out[k] = tree[M + i];
  }
}

extern "C" int per_sample(const float* tree, long M, long size,
                          const float* uniforms, long B, long* idx_out,
                          float* prio_out, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(per_sample_kernel, dim3(grid_1d(B, block)), dim3(block),
                     0, stream, tree, M, size, uniforms, (int)B, idx_out,
                     prio_out);
  CHECK_LAUNCH();
  return 0;
}

// Min over live leaves (for the max-IS-weight normalizer).  The reference
// keeps a second min-tree; at 8 TB/s a flat reduce over <= capacity floats
// is microseconds, so one kernel replaces the whole MinSegmentTree.
// Priorities are >= 0, so integer atomicMin on the float bit pattern works.
extern "C" __global__ void leaf_min_kernel(const float* __restrict__ leaves,
                                           long size,
                                           unsigned int* __restrict__ out) {
  float m = 3.4e38f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < size;
       i += (long)gridDim.x * blockDim.x)
    m = fminf(m, leaves[i]);
  // wave reduce
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    m = fminf(m, __shfl_down(m, off, WAVE));
  if ((threadIdx.x & (WAVE - 1)) == 0)
    atomicMin(out, __float_as_uint(m));
}

extern "C" int per_leaf_min(const float* tree, long M, long size,
                            unsigned int* out_bits, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(leaf_min_kernel, dim3(grid_1d(size, block)), dim3(block),
                     0, stream, tree + M, size, out_bits);
  CHECK_LAUNCH();
  return 0;
}
