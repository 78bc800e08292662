// Whole-sequence masked-LSTM forward/backward driven from C++.
//
// The Python per-step loop costs ~20 ms of enqueue per learner iteration
// at B=128 (measured — profiles/README.md): 162 steps x {addmm +
// pointwise} x Python/launch overhead.  This moves the T-loop into C++:
// per step one rocBLAS SGEMM (recurrent h @ W_hh^T) + one fused kernel
// (mask + gate add + activations), all enqueued from native code on the
// caller's stream.  Numerics identical to the Python path (fp32 GEMM,
// fp32 pointwise, same masking order), so the same oracle tests apply.
//
// Layouts (row-major):
//   xg      [T,B,4H]  x @ W_ih^T + b  (in); overwritten with ACTIVATED
//                     gates (backward consumes them)
//   w_hh    [4H,H]
//   notdone [T,B]
//   h, c    [B,H]     in: initial state; out: final state
//   hs      [T,B,H]   per-step outputs
//   hs_in   [T,B,H]   masked h entering each step (for dW_hh)
//   cs_in   [T,B,H]   masked c entering each step
//   cs_out  [T,B,H]   cell state after each step

#include "common.h"

#include <rocblas/rocblas.h>

static rocblas_handle g_blas = nullptr;

static int ensure_blas(hipStream_t stream) {
  if (g_blas == nullptr) {
    if (rocblas_create_handle(&g_blas) != rocblas_status_success) return -10;
  }
  if (rocblas_set_stream(g_blas, stream) != rocblas_status_success)
    return -11;
  return 0;
}

// h,c ← h*nd, c*nd; record the masked values for backward.
extern "C" __global__ void lstm_mask_kernel(
    float* __restrict__ h, float* __restrict__ c,
    const float* __restrict__ notdone,  // [B]
    float* __restrict__ hs_in, float* __restrict__ cs_in, long B, long H) {
  const long total = B * H;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const float nd = notdone[i / H];
    const float hv = h[i] * nd;
    const float cv = c[i] * nd;
    h[i] = hv;
    c[i] = cv;
    hs_in[i] = hv;
    cs_in[i] = cv;
  }
}

// gates_act = act(gemm_out + xg); h,c updated; hs/cs_out recorded.
extern "C" __global__ void lstm_step_pointwise_kernel(
    const float* __restrict__ gemm_out,  // [B,4H] h @ W_hh^T
    float* __restrict__ xg,              // [B,4H] in: x-preact, out: activated
    float* __restrict__ h,               // [B,H] in: masked prev, out: new
    float* __restrict__ c,               // [B,H] in: masked prev, out: new
    float* __restrict__ hs,              // [B,H] out
    float* __restrict__ cs_out,          // [B,H] out
    long B, long H) {
  const long total = B * H;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long b = i / H, j = i % H;
    const long row = b * 4 * H;
    const float ig = sigmoidf_(xg[row + j] + gemm_out[row + j]);
    const float fg = sigmoidf_(xg[row + H + j] + gemm_out[row + H + j]);
    const float gg = tanhf(xg[row + 2 * H + j] + gemm_out[row + 2 * H + j]);
    const float og = sigmoidf_(xg[row + 3 * H + j] + gemm_out[row + 3 * H + j]);
    const float cv = fg * c[i] + ig * gg;
    const float hv = og * tanhf(cv);
    xg[row + j] = ig;
    xg[row + H + j] = fg;
    xg[row + 2 * H + j] = gg;
    xg[row + 3 * H + j] = og;
    c[i] = cv;
    h[i] = hv;
    hs[i] = hv;
    cs_out[i] = cv;
  }
}

// dh_total = d_hs[t] + dh_carry; produce preact dgates + dc_prev; then the
// caller GEMMs dh_carry = dgates @ W_hh and this kernel's next call masks.
extern "C" __global__ void lstm_step_pointwise_bwd_kernel(
    const float* __restrict__ gates_act,  // [B,4H]
    const float* __restrict__ cs_in,      // [B,H]
    const float* __restrict__ cs_out,     // [B,H]
    const float* __restrict__ d_hs_t,     // [B,H] upstream at this step
    const float* __restrict__ dh_carry,   // [B,H] recurrent carry (or null)
    const float* __restrict__ dc_carry,   // [B,H] (or null)
    float* __restrict__ dgates,           // [B,4H] out (preact grads)
    float* __restrict__ dc_prev,          // [B,H] out (pre-mask)
    long B, long H) {
  const long total = B * H;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const long b = i / H, j = i % H;
    const long row = b * 4 * H;
    const float ig = gates_act[row + j];
    const float fg = gates_act[row + H + j];
    const float gg = gates_act[row + 2 * H + j];
    const float og = gates_act[row + 3 * H + j];
    const float tc = tanhf(cs_out[i]);
    const float dh = d_hs_t[i] + (dh_carry ? dh_carry[i] : 0.f);
    const float dc = (dc_carry ? dc_carry[i] : 0.f) + dh * og * (1.f - tc * tc);
    dgates[row + j] = dc * gg * ig * (1.f - ig);
    dgates[row + H + j] = dc * cs_in[i] * fg * (1.f - fg);
    dgates[row + 2 * H + j] = dc * ig * (1.f - gg * gg);
    dgates[row + 3 * H + j] = dh * tc * og * (1.f - og);
    dc_prev[i] = dc * fg;
  }
}

extern "C" __global__ void lstm_mask_carry_kernel(
    float* __restrict__ dh_carry, float* __restrict__ dc_carry,
    const float* __restrict__ notdone, long B, long H) {
  const long total = B * H;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    const float nd = notdone[i / H];
    dh_carry[i] *= nd;
    dc_carry[i] *= nd;
  }
}

// Row-major helper: C[m,n] = A[m,k] @ B[n,k]^T  (beta = 0)
static int sgemm_nt(long m, long n, long k, const float* A, const float* Bm,
                    float* C) {
  const float one = 1.f, zero = 0.f;
  // column-major view: C_cm[n,m] = B_cm[k,n]^T x A_cm[k,m]
  return (rocblas_sgemm(g_blas, rocblas_operation_transpose,
                        rocblas_operation_none, (rocblas_int)n,
                        (rocblas_int)m, (rocblas_int)k, &one, Bm,
                        (rocblas_int)k, A, (rocblas_int)k, &zero, C,
                        (rocblas_int)n) == rocblas_status_success) ? 0 : -12;
}

// Row-major helper: C[m,n] = A[m,k] @ B[k,n]  (beta = 0)
static int sgemm_nn(long m, long n, long k, const float* A, const float* Bm,
                    float* C) {
  const float one = 1.f, zero = 0.f;
  // column-major view: C_cm[n,m] = B_cm[n,k] x A_cm[k,m]
  return (rocblas_sgemm(g_blas, rocblas_operation_none,
                        rocblas_operation_none, (rocblas_int)n,
                        (rocblas_int)m, (rocblas_int)k, &one, Bm,
                        (rocblas_int)n, A, (rocblas_int)k, &zero, C,
                        (rocblas_int)n) == rocblas_status_success) ? 0 : -12;
}

extern "C" int masked_lstm_seq_fwd(
    float* xg, const float* w_hh, const float* notdone, float* h, float* c,
    float* hs, float* hs_in, float* cs_in, float* cs_out, float* gemm_tmp,
    long T, long B, long H, hipStream_t stream) {
  int rc = ensure_blas(stream);
  if (rc) return rc;
  const int block = 256;
  const int grid = grid_1d(B * H, block);
  for (long t = 0; t < T; ++t) {
    const long off = t * B * H;
    hipLaunchKernelGGL(lstm_mask_kernel, dim3(grid), dim3(block), 0, stream,
                       h, c, notdone + t * B, hs_in + off, cs_in + off, B, H);
    // gemm_tmp[B,4H] = h[B,H] @ w_hh[4H,H]^T
    rc = sgemm_nt(B, 4 * H, H, h, w_hh, gemm_tmp);
    if (rc) return rc;
    hipLaunchKernelGGL(lstm_step_pointwise_kernel, dim3(grid), dim3(block),
                       0, stream, gemm_tmp, xg + t * B * 4 * H, h, c,
                       hs + off, cs_out + off, B, H);
  }
  CHECK_LAUNCH();
  return 0;
}

extern "C" int masked_lstm_seq_bwd(
    const float* gates_act,   // [T,B,4H] (activated, from fwd)
    const float* cs_in, const float* cs_out,  // [T,B,H]
    const float* d_hs,        // [T,B,H] upstream
    const float* notdone,     // [T,B]
    const float* w_hh,        // [4H,H]
    float* dgates_all,        // [T,B,4H] out
    float* dh_carry,          // [B,H] in: d h_T; out: d h_0 (post-mask)
    float* dc_carry,          // [B,H] in: d c_T; out: d c_0 (post-mask)
    float* dc_prev_tmp,       // [B,H] scratch
    long T, long B, long H, hipStream_t stream) {
  int rc = ensure_blas(stream);
  if (rc) return rc;
  const int block = 256;
  const int grid = grid_1d(B * H, block);
  for (long t = T - 1; t >= 0; --t) {
    const long off = t * B * H;
    float* dgates = dgates_all + t * B * 4 * H;
    hipLaunchKernelGGL(lstm_step_pointwise_bwd_kernel, dim3(grid),
                       dim3(block), 0, stream, gates_act + t * B * 4 * H,
                       cs_in + off, cs_out + off, d_hs + off, dh_carry,
                       dc_carry, dgates, dc_prev_tmp, B, H);
    // dh_carry[B,H] = dgates[B,4H] @ w_hh[4H,H]   (pointwise consumed the
    // old carry above, so the overwrite is safe)
    rc = sgemm_nn(B, H, 4 * H, dgates, w_hh, dh_carry);
    if (rc) return rc;
    // grad of the forward `state *= notdone[t]`: mask the new dh carry and
    // the fresh dc_prev, which becomes the next (earlier) step's dc carry
    hipLaunchKernelGGL(lstm_mask_carry_kernel, dim3(grid), dim3(block), 0,
                       stream, dh_carry, dc_prev_tmp, notdone + t * B, B, H);
    if (hipMemcpyAsync(dc_carry, dc_prev_tmp, B * H * sizeof(float),
                       hipMemcpyDeviceToDevice, stream) != hipSuccess)
      return -13;
  }
  CHECK_LAUNCH();
  return 0;
}
