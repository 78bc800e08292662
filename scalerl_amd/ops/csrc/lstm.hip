// LSTM cell pointwise kernels (forward + backward).
//
// Reference semantics: AtariNet's per-step done-masked LSTM unroll
// (algorithms/utils/atari_model.py:109-120) — the reference steps
// nn.LSTM in a Python loop; here the per-step GEMMs run through
// rocBLAS/hipBLASLt (torch.mm on flat weights) and ALL pointwise gate math
// is one fused kernel per direction.  Gate order follows torch.nn.LSTM:
// [input, forget, cell(g), output] chunks of the 4H axis.
//
// Forward overwrites the preactivation buffer with the ACTIVATED gates so
// backward needs no recompute and no extra memory.
// Done-masking (state *= notdone) is applied by the caller between steps —
// it must precede the recurrent GEMM, so it cannot be fused here.

#include "common.h"

extern "C" __global__ void lstm_pointwise_fwd_kernel(
    float* __restrict__ gates,          // [B,4H] in: preact, out: activated
    const float* __restrict__ c_prev,   // [B,H]
    float* __restrict__ h_out,          // [B,H]
    float* __restrict__ c_out,          // [B,H]
    long B, long H) {
  const long total = B * H;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long b = i / H, j = i % H;
    float* grow = gates + b * 4 * H;
    const float ig = sigmoidf_(grow[j]);
    const float fg = sigmoidf_(grow[H + j]);
    const float gg = tanhf(grow[2 * H + j]);
    const float og = sigmoidf_(grow[3 * H + j]);
    const float c = fg * c_prev[i] + ig * gg;
    grow[j] = ig;
    grow[H + j] = fg;
    grow[2 * H + j] = gg;
    grow[3 * H + j] = og;
    c_out[i] = c;
    h_out[i] = og * tanhf(c);
  }
}

extern "C" int lstm_pointwise_fwd(float* gates, const float* c_prev,
                                  float* h_out, float* c_out, long B, long H,
                                  hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(lstm_pointwise_fwd_kernel, dim3(grid_1d(B * H, block)),
                     dim3(block), 0, stream, gates, c_prev, h_out, c_out, B, H);
  CHECK_LAUNCH();
  return 0;
}

extern "C" __global__ void lstm_pointwise_bwd_kernel(
    const float* __restrict__ gates,   // [B,4H] activated (from fwd)
    const float* __restrict__ c_prev,  // [B,H]
    const float* __restrict__ c_out,   // [B,H]
    const float* __restrict__ dh,      // [B,H]
    const float* __restrict__ dc_in,   // [B,H] carry (may be null)
    float* __restrict__ dgates,        // [B,4H] out: preact grads
    float* __restrict__ dc_prev,       // [B,H] out
    long B, long H) {
  const long total = B * H;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long b = i / H, j = i % H;
    const float* grow = gates + b * 4 * H;
    float* dgrow = dgates + b * 4 * H;
    const float ig = grow[j];
    const float fg = grow[H + j];
    const float gg = grow[2 * H + j];
    const float og = grow[3 * H + j];
    const float tc = tanhf(c_out[i]);
    const float dhi = dh[i];
    const float dc = (dc_in ? dc_in[i] : 0.f) + dhi * og * (1.f - tc * tc);
    dgrow[j] = dc * gg * ig * (1.f - ig);
    dgrow[H + j] = dc * c_prev[i] * fg * (1.f - fg);
    dgrow[2 * H + j] = dc * ig * (1.f - gg * gg);
    dgrow[3 * H + j] = dhi * tc * og * (1.f - og);
    dc_prev[i] = dc * fg;
  }
}

extern "C" int lstm_pointwise_bwd(const float* gates, const float* c_prev,
                                  const float* c_out, const float* dh,
                                  const float* dc_in, float* dgates,
                                  float* dc_prev, long B, long H,
                                  hipStream_t stream) {
  const int block = 256;
  hipLaunchKernelGGL(lstm_pointwise_bwd_kernel, dim3(grid_1d(B * H, block)),
                     dim3(block), 0, stream, gates, c_prev, c_out, dh, dc_in,
                     dgates, dc_prev, B, H);
  CHECK_LAUNCH();
  return 0;
}
