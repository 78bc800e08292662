// Shared helpers for ScaleRL-MI355X HIP kernels (gfx950 / CDNA4).
//
// Conventions:
//  - every exported entry point is extern "C", returns 0 on success or a
//    hipError_t, and takes its hipStream_t last;
//  - tensors arrive as raw contiguous device pointers (fp32 unless noted);
//  - wave width is 64 (CDNA), hard-coded per the CDNA4 programming guide.
#pragma once

#include <hip/hip_runtime.h>

#define WAVE 64

#define CHECK_LAUNCH()                                                        \
  do {                                                                        \
    hipError_t err_ = hipPeekAtLastError();                                   \
    if (err_ != hipSuccess) return (int)err_;                                 \
  } while (0)

__device__ __forceinline__ float wave_reduce_sum(float v) {
  #pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    v += __shfl_down(v, off, WAVE);
  return v;  // valid in lane 0 of the wave
}

// Block-wide sum into a single float, returned valid in thread 0.
// Requires blockDim.x <= 1024 (<= 16 waves); `scratch` needs >= 16 floats.
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  int nwaves = (blockDim.x + WAVE - 1) / WAVE;
  v = (threadIdx.x < nwaves) ? scratch[threadIdx.x] : 0.0f;
  if (wid == 0) {
    #pragma unroll
    for (int off = 8; off > 0; off >>= 1)
      v += __shfl_down(v, off, WAVE);
  }
  return v;
}

__device__ __forceinline__ float sigmoidf_(float x) {
  return 1.0f / (1.0f + __expf(-x));
}

static inline int grid_1d(long n, int block) {
  long g = (n + block - 1) / block;
  // >> 256 CUs x a few blocks each is plenty; cap to keep grid-stride loops.
  if (g > 32768) g = 32768;
  return (int)g;
}
