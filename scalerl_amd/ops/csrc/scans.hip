// Backward-scan family: GAE(lambda), discounted returns, n-step folds.
//
// Reference semantics (reimplemented):
//   GAE        — config only in the reference (rl_args.py:338-340); standard
//                adv_t = delta_t + gamma*lambda*(1-d_t)*adv_{t+1}.
//   returns    — A3C TD fold (parallel_a3c.py:274-276), hpc fold
//                (hpc/generation.py:142-147): R_t = r_t + gamma*(1-d_t)*R_{t+1}.
//   n-step     — MultiStepReplayBuffer insert-time fold
//                (data/replay_buffer.py:230-273): over a window of n steps,
//                stop at the first done.
//
// All scans are sequential in T and parallel over the batch column: one
// thread per column, grid-stride.  T*B is tiny (<= a few MB) — the win is
// replacing a Python loop of eager ops with one launch.

#include "common.h"

extern "C" __global__ void gae_kernel(
    const float* __restrict__ rewards,    // [T,B]
    const float* __restrict__ values,     // [T,B]
    const float* __restrict__ bootstrap,  // [B]
    const float* __restrict__ discounts,  // [T,B] gamma*(1-done_{t+1})
    float lam, int T, int B,
    float* __restrict__ advantages,       // [T,B] out
    float* __restrict__ returns) {        // [T,B] out (adv + V)
  for (int b = blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += gridDim.x * blockDim.x) {
    float adv = 0.f;
    for (int t = T - 1; t >= 0; --t) {
      const long row = (long)t * B + b;
      const float v_tp1 = (t + 1 < T) ? values[row + B] : bootstrap[b];
      const float delta = rewards[row] + discounts[row] * v_tp1 - values[row];
      adv = delta + discounts[row] * lam * adv;
      advantages[row] = adv;
      returns[row] = adv + values[row];
    }
  }
}

extern "C" int gae_scan(const float* rewards, const float* values,
                        const float* bootstrap, const float* discounts,
                        float lam, long T, long B, float* advantages,
                        float* returns, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(gae_kernel, dim3(grid_1d(B, block)), dim3(block), 0,
                     stream, rewards, values, bootstrap, discounts, lam,
                     (int)T, (int)B, advantages, returns);
  CHECK_LAUNCH();
  return 0;
}

extern "C" __global__ void discounted_returns_kernel(
    const float* __restrict__ rewards,    // [T,B]
    const float* __restrict__ discounts,  // [T,B]
    const float* __restrict__ bootstrap,  // [B]
    int T, int B,
    float* __restrict__ returns) {        // [T,B] out
  for (int b = blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += gridDim.x * blockDim.x) {
    float acc = bootstrap ? bootstrap[b] : 0.f;
    for (int t = T - 1; t >= 0; --t) {
      const long row = (long)t * B + b;
      acc = rewards[row] + discounts[row] * acc;
      returns[row] = acc;
    }
  }
}

extern "C" int discounted_returns(const float* rewards, const float* discounts,
                                  const float* bootstrap, long T, long B,
                                  float* returns, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(discounted_returns_kernel, dim3(grid_1d(B, block)),
                     dim3(block), 0, stream, rewards, discounts, bootstrap,
                     (int)T, (int)B, returns);
  CHECK_LAUNCH();
  return 0;
}

// n-step fold over a contiguous chunk of transitions per column:
//   folded_r[t] = sum_{k=0..n-1} gamma^k r[t+k], stopping at the first done;
//   next_idx[t] = index of the state that bootstraps (t+m, m = steps used);
//   folded_d[t] = 1 if a done occurred inside the window.
// Used at replay-ingest time (device-resident replay).
extern "C" __global__ void nstep_fold_kernel(
    const float* __restrict__ rewards,  // [T,B]
    const float* __restrict__ dones,    // [T,B] (0/1)
    float gamma, int n, int T, int B,
    float* __restrict__ folded_r,       // [T,B] out
    float* __restrict__ folded_d,       // [T,B] out
    int* __restrict__ steps_used) {     // [T,B] out (m in 1..n)
  const long total = (long)T * B;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const int t = (int)(i / B);
    const int b = (int)(i % B);
    float acc = 0.f, g = 1.f, done = 0.f;
    int m = 0;
    for (int k = 0; k < n && t + k < T; ++k) {
      const long row = (long)(t + k) * B + b;
      acc += g * rewards[row];
      g *= gamma;
      m = k + 1;
      if (dones[row] != 0.f) { done = 1.f; break; }
    }
    folded_r[i] = acc;
    folded_d[i] = done;
    steps_used[i] = m;
  }
}

extern "C" int nstep_fold(const float* rewards, const float* dones,
                          float gamma, long n, long T, long B,
                          float* folded_r, float* folded_d, int* steps_used,
                          hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(nstep_fold_kernel, dim3(grid_1d(T * B, block)),
                     dim3(block), 0, stream, rewards, dones, (float)gamma,
                     (int)n, (int)T, (int)B, folded_r, folded_d, steps_used);
  CHECK_LAUNCH();
  return 0;
}
