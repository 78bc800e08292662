// Fused DQN TD loss: double/vanilla DQN target + Huber/MSE + PER IS-weights
// + |TD| priorities + analytic gradient w.r.t. the online Q row — one launch.
//
// Reference semantics (reimplemented):
//   dqn_agent.py:155-171  (double-DQN target, MSE),
//   apex/worker.py:134-161 (PER IS weights (1/(N*P))^beta / max_w, priority
//                           update from |TD|).
//
// IS weights are computed in-kernel from raw priorities + (p_total, p_min)
// device scalars so sampling->loss needs no host round-trip:
//   P_i = prio_i / p_total;  w_i = (N * P_i)^-beta / (N * p_min/p_total)^-beta.
// `prios` may be nullptr -> uniform weights (plain DQN).

#include "common.h"

extern "C" __global__ void __launch_bounds__(256)
td_loss_kernel(const float* __restrict__ q,            // [B,A] online Q(s)
               const float* __restrict__ q_next_online,// [B,A] online Q(s') (double) or null
               const float* __restrict__ q_next_target,// [B,A] target Q(s')
               const long* __restrict__ actions,       // [B]
               const float* __restrict__ rewards,      // [B] (n-step folded)
               const float* __restrict__ discounts,    // [B] gamma^m*(1-d)
               const float* __restrict__ prios,        // [B] or null
               const float* __restrict__ p_total,      // [1] or null
               const float* __restrict__ p_min,        // [1] or null
               float beta, long replay_size,
               int B, int A, int huber, float huber_delta,
               float* __restrict__ grad_q,             // [B,A] out (zeroed by caller)
               float* __restrict__ td_abs,             // [B] out (new priorities)
               float* __restrict__ loss_out) {         // [1] out (atomic, mean)
  __shared__ float scratch[16];
  float loss_acc = 0.f;
  for (int i = blockIdx.x * blockDim.x + threadIdx.x; i < B;
       i += gridDim.x * blockDim.x) {
    const float* qrow = q + (long)i * A;
    const long a = actions[i];
    // target: r + gamma * Q_target(s', a*)
    int astar = 0;
    if (q_next_online) {  // double DQN: argmax over online net
      float best = -1e30f;
      for (int j = 0; j < A; ++j) {
        const float v = q_next_online[(long)i * A + j];
        if (v > best) { best = v; astar = j; }
      }
    } else {  // vanilla: argmax over target net
      float best = -1e30f;
      for (int j = 0; j < A; ++j) {
        const float v = q_next_target[(long)i * A + j];
        if (v > best) { best = v; astar = j; }
      }
    }
    const float target = rewards[i] + discounts[i] * q_next_target[(long)i * A + astar];
    const float td = qrow[a] - target;
    td_abs[i] = fabsf(td);

    float w = 1.f;
    if (prios) {
      const float pt = *p_total;
      const float pm = *p_min;
      const float w_i = __powf((float)replay_size * (prios[i] / pt), -beta);
      const float w_max = __powf((float)replay_size * (pm / pt), -beta);
      w = w_i / w_max;
    }
    float l, dldtd;
    if (huber) {
      const float atd = fabsf(td);
      if (atd <= huber_delta) { l = 0.5f * td * td; dldtd = td; }
      else { l = huber_delta * (atd - 0.5f * huber_delta); dldtd = huber_delta * ((td > 0) ? 1.f : -1.f); }
    } else {
      l = td * td;       // MSE as the reference writes it (dqn_agent.py:171)
      dldtd = 2.f * td;
    }
    loss_acc += w * l / (float)B;
    grad_q[(long)i * A + a] = w * dldtd / (float)B;
  }
  const float tot = block_reduce_sum(loss_acc, scratch);
  if (threadIdx.x == 0) atomicAdd(loss_out, tot);
}

extern "C" int fused_td_loss(
    const float* q, const float* q_next_online, const float* q_next_target,
    const long* actions, const float* rewards, const float* discounts,
    const float* prios, const float* p_total, const float* p_min,
    float beta, long replay_size, long B, long A, int huber,
    float huber_delta, float* grad_q, float* td_abs, float* loss_out,
    hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(td_loss_kernel, dim3(grid_1d(B, block)), dim3(block), 0,
                     stream, q, q_next_online, q_next_target, actions, rewards,
                     discounts, prios, p_total, p_min, beta, replay_size,
                     (int)B, (int)A, huber, huber_delta, grad_q, td_abs,
                     loss_out);
  CHECK_LAUNCH();
  return 0;
}
