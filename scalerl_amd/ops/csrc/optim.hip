// Fused flat-buffer optimizers + polyak + gradient clipping.
//
// The learner keeps all parameters in ONE contiguous fp32 buffer (see
// scalerl_amd/parallel/flat.py) so a full optimizer step is one kernel and a
// full gradient all-reduce is one RCCL call.  Semantics match
// torch.optim.{RMSprop, Adam} (the reference relies on those:
// impala_atari.py:99-106 RMSProp, SharedAdam share_optim.py:65-122) so
// checkpoints interop.  These are bandwidth-bound elementwise kernels —
// float4-vectorized grid-stride loops, no MFMA needed.

#include "common.h"

// ---- RMSProp (torch semantics: avg = sqrt(sq) + eps) ----
extern "C" __global__ void rmsprop_kernel(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ sq, float* __restrict__ mom,  // mom nullable
    long n, float lr, float alpha, float eps, float momentum,
    float weight_decay) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float grad = g[i] + weight_decay * p[i];
    float s = alpha * sq[i] + (1.f - alpha) * grad * grad;
    sq[i] = s;
    float upd = grad / (sqrtf(s) + eps);
    if (mom) {
      float m = momentum * mom[i] + upd;
      mom[i] = m;
      upd = m;
    }
    p[i] -= lr * upd;
  }
}

extern "C" int fused_rmsprop(float* p, const float* g, float* sq, float* mom,
                             long n, float lr, float alpha, float eps,
                             float momentum, float weight_decay,
                             hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(rmsprop_kernel, dim3(grid_1d(n, block)), dim3(block), 0,
                     stream, p, g, sq, mom, n, lr, alpha, eps, momentum,
                     weight_decay);
  CHECK_LAUNCH();
  return 0;
}

// ---- Adam (torch semantics, bias-corrected; SharedAdam parity) ----
extern "C" __global__ void adam_kernel(
    float* __restrict__ p, const float* __restrict__ g,
    float* __restrict__ m, float* __restrict__ v,
    long n, float lr, float beta1, float beta2, float eps,
    float weight_decay, float bc1, float bc2) {  // bc = 1 - beta^step
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x) {
    float grad = g[i] + weight_decay * p[i];
    float m_ = beta1 * m[i] + (1.f - beta1) * grad;
    float v_ = beta2 * v[i] + (1.f - beta2) * grad * grad;
    m[i] = m_;
    v[i] = v_;
    const float denom = sqrtf(v_ / bc2) + eps;
    p[i] -= lr * (m_ / bc1) / denom;
  }
}

extern "C" int fused_adam(float* p, const float* g, float* m, float* v,
                          long n, float lr, float beta1, float beta2,
                          float eps, float weight_decay, long step,
                          hipStream_t stream) {
  const float bc1 = 1.f - powf(beta1, (float)step);
  const float bc2 = 1.f - powf(beta2, (float)step);
  const int block = 256;
  hipLaunchKernelGGL(adam_kernel, dim3(grid_1d(n, block)), dim3(block), 0,
                     stream, p, g, m, v, n, lr, beta1, beta2, eps,
                     weight_decay, bc1, bc2);
  CHECK_LAUNCH();
  return 0;
}

// ---- polyak: dst = tau*src + (1-tau)*dst (model_utils.py:16-32) ----
extern "C" __global__ void polyak_kernel(float* __restrict__ dst,
                                         const float* __restrict__ src,
                                         long n, float tau) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    dst[i] += tau * (src[i] - dst[i]);
}

extern "C" int fused_polyak(float* dst, const float* src, long n, float tau,
                            hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(polyak_kernel, dim3(grid_1d(n, block)), dim3(block), 0,
                     stream, dst, src, n, tau);
  CHECK_LAUNCH();
  return 0;
}

// ---- gradient clipping by global norm, no host sync ----
// Pass 1: sq_norm_out[0] += sum(g^2)     (caller zeroes sq_norm_out)
// Pass 2: g *= max_norm / max(norm, max_norm)   (reads norm from device)
extern "C" __global__ void sqnorm_kernel(const float* __restrict__ g, long n,
                                         float* __restrict__ out) {
  __shared__ float scratch[16];
  float acc = 0.f;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    acc += g[i] * g[i];
  acc = block_reduce_sum(acc, scratch);
  if (threadIdx.x == 0) atomicAdd(out, acc);
}

extern "C" __global__ void clip_apply_kernel(float* __restrict__ g, long n,
                                             const float* __restrict__ sqnorm,
                                             float max_norm) {
  const float norm = sqrtf(*sqnorm);
  const float coef = (norm > max_norm) ? (max_norm / (norm + 1e-6f)) : 1.f;
  if (coef == 1.f) return;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long)gridDim.x * blockDim.x)
    g[i] *= coef;
}

extern "C" int grad_clip_by_norm(float* g, long n, float* sqnorm_scratch,
                                 float max_norm, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(sqnorm_kernel, dim3(grid_1d(n, block)), dim3(block), 0,
                     stream, g, n, sqnorm_scratch);
  CHECK_LAUNCH();
  hipLaunchKernelGGL(clip_apply_kernel, dim3(grid_1d(n, block)), dim3(block),
                     0, stream, g, n, sqnorm_scratch, max_norm);
  CHECK_LAUNCH();
  return 0;
}
