// Fused IMPALA learner math: V-trace targets + policy-gradient / baseline /
// entropy losses + analytic input gradients, in ONE kernel launch.
//
// Reference semantics (reimplemented, not copied):
//   scalerl/algorithms/impala/vtrace.py:78-172  (from_importance_weights)
//   scalerl/algorithms/impala/vtrace.py:31-75   (from_logits / log-prob)
//   scalerl/algorithms/impala/loss_fn.py:1-23   (pg / baseline / entropy sums)
//
// The reference runs this as a Python loop over T plus ~15 eager CUDA ops;
// here the whole thing is a single kernel: one workgroup per batch column b,
// phase 1 computes per-step log-softmax statistics and V-trace deltas in
// parallel over t, phase 2 runs the O(T) backward scan (LDS-resident),
// phase 3 emits loss partial sums and the analytic gradients w.r.t. the
// learner logits and baseline values.
//
// Losses are SUMS over [T,B] (reference uses torch.sum), reported raw
// (without their cost coefficients); gradients INCLUDE the coefficients so
// d(total)/d(input) back-propagates in one pass:
//   total = pg + baseline_cost * baseline + entropy_cost * entropy.

#include "common.h"

#define MAX_A 32  // Atari action sets are <= 18

extern "C" __global__ void __launch_bounds__(256)
impala_fused_loss_kernel(
    const float* __restrict__ behavior_logits,  // [T,B,A]
    const float* __restrict__ target_logits,    // [T,B,A]
    const long* __restrict__ actions,           // [T,B]
    const float* __restrict__ rewards,          // [T,B]
    const float* __restrict__ discounts,        // [T,B]  gamma*(1-done_{t+1})
    const float* __restrict__ values,           // [T,B]
    const float* __restrict__ bootstrap,        // [B]
    float clip_rho, float clip_c, float clip_pg_rho,
    float baseline_cost, float entropy_cost,
    int T, int B, int A,
    float* __restrict__ grad_logits,            // [T,B,A] out
    float* __restrict__ grad_values,            // [T,B]   out
    float* __restrict__ loss_out,               // [3] out (atomic): pg, baseline, entropy
    float* __restrict__ vs_out) {               // [T,B] out or nullptr
  extern __shared__ float smem[];
  float* s_rho   = smem;          // raw importance ratio rho_t
  float* s_lse   = smem + T;      // log-sum-exp of target logits row
  float* s_delta = smem + 2 * T;  // V-trace delta, then reused as acc (vs - V)
  float* s_pgadv = smem + 3 * T;  // clipped pg advantage
  float* s_red   = smem + 4 * T;  // 16 floats of reduction scratch

  const int b = blockIdx.x;
  if (b >= B) return;

  // ---- phase 1: per-step statistics (parallel over t) ----
  for (int t = threadIdx.x; t < T; t += blockDim.x) {
    const float* tl = target_logits + ((long)t * B + b) * A;
    const float* bl = behavior_logits + ((long)t * B + b) * A;
    const long a = actions[(long)t * B + b];
    float tmax = -1e30f, bmax = -1e30f;
    for (int i = 0; i < A; ++i) {
      tmax = fmaxf(tmax, tl[i]);
      bmax = fmaxf(bmax, bl[i]);
    }
    float tsum = 0.f, bsum = 0.f;
    for (int i = 0; i < A; ++i) {
      tsum += __expf(tl[i] - tmax);
      bsum += __expf(bl[i] - bmax);
    }
    const float lse_t = tmax + __logf(tsum);
    const float lse_b = bmax + __logf(bsum);
    const float log_rho = (tl[a] - lse_t) - (bl[a] - lse_b);
    const float rho = __expf(log_rho);
    s_rho[t] = rho;
    s_lse[t] = lse_t;
    const float v_t = values[(long)t * B + b];
    const float v_tp1 = (t + 1 < T) ? values[(long)(t + 1) * B + b] : bootstrap[b];
    const float r_t = rewards[(long)t * B + b];
    const float g_t = discounts[(long)t * B + b];
    s_delta[t] = fminf(rho, clip_rho) * (r_t + g_t * v_tp1 - v_t);
  }
  __syncthreads();

  // ---- phase 2: sequential backward scan (thread 0; T is O(100)) ----
  if (threadIdx.x == 0) {
    float acc = 0.f;
    for (int t = T - 1; t >= 0; --t) {
      const float c_t = fminf(s_rho[t], clip_c);  // lambda = 1, c_bar = 1
      acc = s_delta[t] + discounts[(long)t * B + b] * c_t * acc;
      s_delta[t] = acc;  // now holds vs_t - V_t
    }
    // pg advantage needs vs_{t+1}: do it in the same pass (forward).
    for (int t = 0; t < T; ++t) {
      const float v_tp1 = (t + 1 < T) ? values[(long)(t + 1) * B + b] : bootstrap[b];
      const float vs_tp1 = (t + 1 < T) ? (s_delta[t + 1] + v_tp1) : bootstrap[b];
      const float v_t = values[(long)t * B + b];
      const float r_t = rewards[(long)t * B + b];
      const float g_t = discounts[(long)t * B + b];
      s_pgadv[t] = fminf(s_rho[t], clip_pg_rho) * (r_t + g_t * vs_tp1 - v_t);
    }
  }
  __syncthreads();

  // ---- phase 3: losses + gradients (parallel over t) ----
  float pg_sum = 0.f, base_sum = 0.f, ent_sum = 0.f;
  for (int t = threadIdx.x; t < T; t += blockDim.x) {
    const long row = (long)t * B + b;
    const float* tl = target_logits + row * A;
    float* gl = grad_logits + row * A;
    const long a = actions[row];
    const float lse_t = s_lse[t];
    const float adv = s_pgadv[t];
    const float vs_minus_v = s_delta[t];
    const float v_t = values[row];

    if (vs_out) vs_out[row] = vs_minus_v + v_t;
    grad_values[row] = baseline_cost * (-vs_minus_v);  // d 0.5(V-vs)^2 / dV
    base_sum += 0.5f * vs_minus_v * vs_minus_v;

    float p[MAX_A];
    float plogp = 0.f;
    for (int i = 0; i < A; ++i) {
      const float lp = tl[i] - lse_t;
      p[i] = __expf(lp);
      plogp += p[i] * lp;
    }
    ent_sum += plogp;                      // sum p log p (negative entropy)
    pg_sum += -(tl[a] - lse_t) * adv;      // cross-entropy * advantage
    for (int i = 0; i < A; ++i) {
      const float onehot = (i == (int)a) ? 1.0f : 0.0f;
      const float g_pg = adv * (p[i] - onehot);
      const float g_ent = p[i] * ((tl[i] - lse_t) - plogp);
      gl[i] = g_pg + entropy_cost * g_ent;
    }
  }
  __syncthreads();  // s_red overlaps s_rho users above

  float tot;
  tot = block_reduce_sum(pg_sum, s_red);
  if (threadIdx.x == 0) atomicAdd(&loss_out[0], tot);
  __syncthreads();
  tot = block_reduce_sum(base_sum, s_red);
  if (threadIdx.x == 0) atomicAdd(&loss_out[1], tot);
  __syncthreads();
  tot = block_reduce_sum(ent_sum, s_red);
  if (threadIdx.x == 0) atomicAdd(&loss_out[2], tot);
}

extern "C" int impala_fused_loss(
    const float* behavior_logits, const float* target_logits,
    const long* actions, const float* rewards, const float* discounts,
    const float* values, const float* bootstrap,
    float clip_rho, float clip_c, float clip_pg_rho,
    float baseline_cost, float entropy_cost,
    long T, long B, long A,
    float* grad_logits, float* grad_values, float* loss_out, float* vs_out,
    hipStream_t stream) {
  if (A > MAX_A) return -2;
  const size_t smem = (4 * (size_t)T + 16) * sizeof(float);
  if (smem > 160 * 1024) return -3;
  hipLaunchKernelGGL(impala_fused_loss_kernel, dim3((int)B), dim3(256), smem,
                     stream, behavior_logits, target_logits, actions, rewards,
                     discounts, values, bootstrap, clip_rho, clip_c,
                     clip_pg_rho, baseline_cost, entropy_cost, (int)T, (int)B,
                     (int)A, grad_logits, grad_values, loss_out, vs_out);
  CHECK_LAUNCH();
  return 0;
}

// Standalone V-trace (no losses): used by tests and by algorithms that
// need vs / pg_advantages as tensors (reference vtrace.py:78-172 surface).
extern "C" __global__ void __launch_bounds__(256)
vtrace_kernel(const float* __restrict__ log_rhos,   // [T,B]
              const float* __restrict__ discounts,  // [T,B]
              const float* __restrict__ rewards,    // [T,B]
              const float* __restrict__ values,     // [T,B]
              const float* __restrict__ bootstrap,  // [B]
              float clip_rho, float clip_c, float clip_pg_rho,
              int T, int B,
              float* __restrict__ vs,               // [T,B] out
              float* __restrict__ pg_adv) {         // [T,B] out
  // One thread per batch column: the scan is sequential in T anyway and B
  // columns give the parallelism (grid-stride for large B).
  for (int b = blockIdx.x * blockDim.x + threadIdx.x; b < B;
       b += gridDim.x * blockDim.x) {
    float acc = 0.f;
    for (int t = T - 1; t >= 0; --t) {
      const long row = (long)t * B + b;
      const float rho = __expf(log_rhos[row]);
      const float v_t = values[row];
      const float v_tp1 = (t + 1 < T) ? values[row + B] : bootstrap[b];
      const float delta =
          fminf(rho, clip_rho) * (rewards[row] + discounts[row] * v_tp1 - v_t);
      acc = delta + discounts[row] * fminf(rho, clip_c) * acc;
      vs[row] = acc + v_t;
    }
    for (int t = 0; t < T; ++t) {
      const long row = (long)t * B + b;
      const float vs_tp1 = (t + 1 < T) ? vs[row + B] : bootstrap[b];
      const float rho = __expf(log_rhos[row]);
      pg_adv[row] = fminf(rho, clip_pg_rho) *
                    (rewards[row] + discounts[row] * vs_tp1 - values[row]);
    }
  }
}

extern "C" int vtrace_from_log_rhos(
    const float* log_rhos, const float* discounts, const float* rewards,
    const float* values, const float* bootstrap,
    float clip_rho, float clip_c, float clip_pg_rho,
    long T, long B, float* vs, float* pg_adv, hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(vtrace_kernel, dim3(grid_1d(B, block)), dim3(block), 0,
                     stream, log_rhos, discounts, rewards, values, bootstrap,
                     clip_rho, clip_c, clip_pg_rho, (int)T, (int)B, vs, pg_adv);
  CHECK_LAUNCH();
  return 0;
}
