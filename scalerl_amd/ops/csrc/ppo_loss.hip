// Fused PPO clip loss: policy surrogate + value MSE + entropy bonus +
// analytic gradients w.r.t. logits and values, one launch.
//
// Semantics match the composed torch ops in runtime/ppo.py (PPO clip per
// Schulman et al. 2017; DD-PPO config 5):
//   ratio   = exp(logp(a) - old_logp)
//   s1      = ratio * adv;  s2 = clamp(ratio, 1-eps, 1+eps) * adv
//   pg      = -mean(min(s1, s2))
//   v_loss  = mean((v - ret)^2)
//   ent     = mean(H(pi))
//   total   = pg + vcoef * v_loss - ecoef * ent
// Gradient through min() follows torch: the s1 branch when s1 <= s2, else
// the s2 branch (zero when its clamp is saturated).

#include "common.h"

#define MAX_A 32

extern "C" __global__ void __launch_bounds__(256)
ppo_fused_loss_kernel(const float* __restrict__ logits,    // [N,A]
                      const long* __restrict__ actions,    // [N]
                      const float* __restrict__ old_logp,  // [N]
                      const float* __restrict__ adv,       // [N]
                      const float* __restrict__ returns,   // [N]
                      const float* __restrict__ values,    // [N]
                      float clip_eps, float vcoef, float ecoef,
                      long N, int A,
                      float* __restrict__ grad_logits,     // [N,A] out
                      float* __restrict__ grad_values,     // [N] out
                      float* __restrict__ loss_out) {      // [3] out: pg,v,ent
  __shared__ float scratch[16];
  float pg_acc = 0.f, v_acc = 0.f, ent_acc = 0.f;
  const float invN = 1.0f / (float)N;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < N;
       i += (long)gridDim.x * blockDim.x) {
    const float* row = logits + i * A;
    float* grow = grad_logits + i * A;
    const long a = actions[i];
    float mx = -1e30f;
    for (int j = 0; j < A; ++j) mx = fmaxf(mx, row[j]);
    float sum = 0.f;
    for (int j = 0; j < A; ++j) sum += __expf(row[j] - mx);
    const float lse = mx + __logf(sum);

    float p[MAX_A];
    float plogp = 0.f;
    for (int j = 0; j < A; ++j) {
      const float lp = row[j] - lse;
      p[j] = __expf(lp);
      plogp += p[j] * lp;
    }
    ent_acc += -plogp;

    const float logp_a = row[a] - lse;
    const float ratio = __expf(logp_a - old_logp[i]);
    const float rc = fminf(fmaxf(ratio, 1.f - clip_eps), 1.f + clip_eps);
    const float s1 = ratio * adv[i];
    const float s2 = rc * adv[i];
    pg_acc += -fminf(s1, s2);
    // d(-min)/d logp_a: s1 branch -> -adv*ratio; s2 branch -> -adv*ratio
    // only when the clamp is NOT saturated (then s2 == s1 anyway), else 0
    float dmin_dlogpa = 0.f;
    if (s1 <= s2) dmin_dlogpa = -s1;            // d s1/d logp_a = s1
    else if (ratio > 1.f - clip_eps && ratio < 1.f + clip_eps)
      dmin_dlogpa = -s1;
    // chain into logits: d logp_a / d z_j = onehot - p_j; plus entropy term
    for (int j = 0; j < A; ++j) {
      const float onehot = (j == (int)a) ? 1.f : 0.f;
      const float g_pg = dmin_dlogpa * (onehot - p[j]);
      const float lp = row[j] - lse;
      const float g_ent = p[j] * (lp - plogp);  // d(sum p logp)/dz_j
      grow[j] = (g_pg + ecoef * g_ent) * invN;
    }
    const float vd = values[i] - returns[i];
    v_acc += vd * vd;
    grad_values[i] = vcoef * 2.f * vd * invN;
  }
  float t;
  t = block_reduce_sum(pg_acc, scratch);
  if (threadIdx.x == 0) atomicAdd(&loss_out[0], t * invN);
  __syncthreads();
  t = block_reduce_sum(v_acc, scratch);
  if (threadIdx.x == 0) atomicAdd(&loss_out[1], t * invN);
  __syncthreads();
  t = block_reduce_sum(ent_acc, scratch);
  if (threadIdx.x == 0) atomicAdd(&loss_out[2], t * invN);
}

extern "C" int ppo_fused_loss(const float* logits, const long* actions,
                              const float* old_logp, const float* adv,
                              const float* returns, const float* values,
                              float clip_eps, float vcoef, float ecoef,
                              long N, long A, float* grad_logits,
                              float* grad_values, float* loss_out,
                              hipStream_t stream) {
  if (A > MAX_A) return -2;
  const int block = 256;
  hipLaunchKernelGGL(ppo_fused_loss_kernel, dim3(grid_1d(N, block)),
                     dim3(block), 0, stream, logits, actions, old_logp, adv,
                     returns, values, clip_eps, vcoef, ecoef, N, (int)A,
                     grad_logits, grad_values, loss_out);
  CHECK_LAUNCH();
  return 0;
}
