// EXPERIMENTAL: hand-written MFMA implicit-GEMM forward convolutions for
// the three fixed Atari encoder shapes (SURVEY.md §7 hard-parts list;
// replaces MIOpen, whose find is box-dependent — profiles/README.md).
//
//   conv1: [N,4,84,84] u8 -> /255 -> conv 8x8 s4 -> ReLU -> [N,32,20,20]
//   conv2: [N,32,20,20]  -> conv 4x4 s2 -> ReLU -> [N,64,9,9]
//   conv3: [N,64,9,9]    -> conv 3x3 s1 -> ReLU -> [N,64,7,7]
//
// GEMM view: M = N*OH*OW output pixels, Ncol = K_out, Kdim = C*KH*KW.
// bf16 inputs/weights, fp32 accumulation via v_mfma_f32_16x16x32_bf16.
//
// Fragment maps for mfma_f32_16x16x32_bf16 (cdna4_isa.md §10 family):
//   A (16x32): lane l, elem j -> row = l & 15, k = (l >> 4) * 8 + j
//   B (32x16): lane l, elem j -> col = l & 15, k = (l >> 4) * 8 + j
//   C/D:       lane l, reg r  -> col = l & 15, row = (l >> 4) * 4 + r
// These constants are validated on-device by `mfma_selftest` below (run
// by the gpu test before any conv test) — if the map is wrong, ONE place
// to fix.
//
// Correctness-first structure (the guide's ladder step-0 shape): gather
// A-fragments straight from global (L2-cached; input reuse across the
// KH*KW window makes the working set cache-resident at these sizes),
// weights staged through LDS once per block.  Tuning (swizzled LDS
// staging, glds, XCD remap) is round-2 work on top of measured PMC data.

#include "common.h"

typedef __bf16 bf16_t;
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef bf16_t bf16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float bf16_to_f32(bf16_t v) { return (float)v; }

// ---- self-test: D = A*B for one 16x16x32 tile, plain layouts ----
extern "C" __global__ void mfma_selftest_kernel(
    const float* __restrict__ A,   // [16][32] row-major
    const float* __restrict__ B,   // [32][16] row-major
    float* __restrict__ D) {       // [16][16] row-major
  const int lane = threadIdx.x & (WAVE - 1);
  bf16x8 a, b;
  #pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int ar = lane & 15, ak = (lane >> 4) * 8 + j;
    const int bc = lane & 15, bk = (lane >> 4) * 8 + j;
    a[j] = (bf16_t)A[ar * 32 + ak];
    b[j] = (bf16_t)B[bk * 16 + bc];
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int col = lane & 15, row = (lane >> 4) * 4 + r;
    D[row * 16 + col] = acc[r];
  }
}

extern "C" int mfma_selftest(const float* A, const float* B, float* D,
                             hipStream_t stream) {
  hipLaunchKernelGGL(mfma_selftest_kernel, dim3(1), dim3(64), 0, stream,
                     A, B, D);
  CHECK_LAUNCH();
  return 0;
}

// Forward kernels live in conv_fwd.hip (v2, LDS-staged).  The v1 generic
// gather-per-lane template below is retained ONLY as the shape reference
// for the backward kernels' masks; it is compiled out.
#if 0
template <int C, int KH, int KW, int STRIDE, int IH, int IW, int OH, int OW,
          int KOUT, bool IN_U8, typename in_t>
__global__ __launch_bounds__(256) void conv_fwd_kernel(
    const in_t* __restrict__ input,     // [N, C, IH, IW]
    const bf16_t* __restrict__ weight,  // [KOUT, C, KH, KW]
    const float* __restrict__ bias,     // [KOUT] or nullptr
    bf16_t* __restrict__ output,        // [N, KOUT, OH, OW]
    int batch, int relu) {
  constexpr int KDIM = C * KH * KW;
  constexpr int KTILES = (KDIM + 31) / 32;
  const int M = batch * OH * OW;

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wm = wave >> 1;          // 0..1: M sub-tile
  const int wn = wave & 1;           // 0..1: N sub-tile
  const int m_base = blockIdx.x * 32 + wm * 16;
  const int n_base = blockIdx.y * 32 + wn * 16;
  // NOTE: no early return before __syncthreads (barrier divergence hangs
  // the CU); out-of-range tiles run with masked stores instead.

  // stage the weight panel for this block's 32 output channels in LDS:
  // layout [32][KDIM] bf16 (KDIM <= 576 -> <= 36 KB)
  __shared__ bf16_t w_lds[32][KDIM];
  const int ch0 = blockIdx.y * 32;
  for (int idx = threadIdx.x; idx < 32 * KDIM; idx += blockDim.x) {
    const int ch = idx / KDIM, k = idx % KDIM;
    w_lds[ch][k] = (ch0 + ch < KOUT) ? weight[(long)(ch0 + ch) * KDIM + k]
                                     : (bf16_t)0.f;
  }
  __syncthreads();

  // decode this lane's A row (one output pixel) once
  const int arow = m_base + (lane & 15);
  int n_img = 0, oy = 0, ox = 0;
  bool row_ok = arow < M;
  if (row_ok) {
    n_img = arow / (OH * OW);
    const int rem = arow % (OH * OW);
    oy = rem / OW;
    ox = rem % OW;
  }
  const long in_img_off = (long)n_img * C * IH * IW;

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int kt = 0; kt < KTILES; ++kt) {
    bf16x8 a, b;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int k = kt * 32 + (lane >> 4) * 8 + j;
      float av = 0.f;
      if (row_ok && k < KDIM) {
        const int c = k / (KH * KW);
        const int kr = k % (KH * KW);
        const int ky = kr / KW, kx = kr % KW;
        const int iy = oy * STRIDE + ky, ix = ox * STRIDE + kx;
        const in_t raw = input[in_img_off + ((long)c * IH + iy) * IW + ix];
        av = IN_U8 ? (float)raw * (1.0f / 255.0f) : (float)raw;
      }
      a[j] = (bf16_t)av;
      b[j] = (k < KDIM) ? w_lds[wn * 16 + (lane & 15)][k] : (bf16_t)0.f;
    }
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }

  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int col = n_base + (lane & 15);
    const int row = m_base + (lane >> 4) * 4 + r;
    if (row < M && col < KOUT) {
      const int ni = row / (OH * OW);
      const int rem = row % (OH * OW);
      float v = acc[r] + (bias ? bias[col] : 0.f);
      if (relu) v = fmaxf(v, 0.f);
      output[(((long)ni * KOUT + col) * OH * OW) + rem] = (bf16_t)v;
    }
  }
}

#define DEF_CONV(NAME, C_, KH_, KW_, S_, IH_, IW_, OH_, OW_, KO_, U8, T)      \
  extern "C" int NAME(const void* in, const void* w, const float* bias,       \
                      void* out, long batch, int relu, hipStream_t stream) {  \
    const long M = batch * OH_ * OW_;                                         \
    dim3 grid((unsigned)((M + 31) / 32), (KO_ + 31) / 32);                    \
    hipLaunchKernelGGL(                                                       \
        (conv_fwd_kernel<C_, KH_, KW_, S_, IH_, IW_, OH_, OW_, KO_, U8, T>),  \
        grid, dim3(256), 0, stream, (const T*)in, (const bf16_t*)w, bias,     \
        (bf16_t*)out, (int)batch, relu);                                      \
    CHECK_LAUNCH();                                                           \
    return 0;                                                                 \
  }

#endif  // v1 forward (superseded by conv_fwd.hip)

// Backward kernels live in conv_bwd.hip (v2).  v1 retained for
// reference only, compiled out.
#if 0
// ---- backward: weight gradient ---------------------------------------
// dw[k_out, kdim] = sum_pixels dy[pixel, k_out] * im2col(x)[pixel, kdim]
// GEMM: rows = K_OUT, cols = KDIM, reduce over M = N*OH*OW pixels.
// The natural grid ((KOUT/16) x (KDIM/16)) underfills 256 CUs, so the
// pixel axis is split across blockIdx.z and partial tiles atomicAdd into
// the fp32 dw buffer (zeroed by the caller).
template <int C, int KH, int KW, int STRIDE, int IH, int IW, int OH, int OW,
          int KOUT, bool IN_U8, typename in_t>
__global__ __launch_bounds__(64) void conv_wgrad_kernel(
    const in_t* __restrict__ input,    // [N, C, IH, IW]
    const bf16_t* __restrict__ dout,   // [N, KOUT, OH, OW]
    float* __restrict__ dweight,       // [KOUT, C*KH*KW] fp32 (atomic)
    int batch, int split) {
  constexpr int KDIM = C * KH * KW;
  const int lane = threadIdx.x & (WAVE - 1);
  const int M = batch * OH * OW;
  const int per_split = (M + split - 1) / split;
  const int p_begin = blockIdx.z * per_split;
  const int p_end = min(p_begin + per_split, M);

  const int kout_base = blockIdx.x * 16;
  const int kdim_base = blockIdx.y * 16;
  const int arow = kout_base + (lane & 15);   // k_out
  const int bcol = kdim_base + (lane & 15);   // kdim
  int bc = 0, bky = 0, bkx = 0;
  if (bcol < KDIM) {
    bc = bcol / (KH * KW);
    const int r = bcol % (KH * KW);
    bky = r / KW;
    bkx = r % KW;
  }

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int p0 = p_begin; p0 < p_end; p0 += 32) {
    bf16x8 a, b;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int p = p0 + (lane >> 4) * 8 + j;
      float av = 0.f, bv = 0.f;
      if (p < p_end) {
        const int n = p / (OH * OW);
        const int rem = p % (OH * OW);
        const int oy = rem / OW, ox = rem % OW;
        if (arow < KOUT)
          av = (float)dout[(((long)n * KOUT + arow) * OH + oy) * OW + ox];
        if (bcol < KDIM) {
          const in_t raw = input[(((long)n * C + bc) * IH + oy * STRIDE + bky)
                                 * IW + ox * STRIDE + bkx];
          bv = IN_U8 ? (float)raw * (1.0f / 255.0f) : (float)raw;
        }
      }
      a[j] = (bf16_t)av;
      b[j] = (bf16_t)bv;
    }
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int col = kdim_base + (lane & 15);
    const int row = kout_base + (lane >> 4) * 4 + r;
    if (row < KOUT && col < KDIM)
      atomicAdd(&dweight[(long)row * KDIM + col], acc[r]);
  }
}

// ---- backward: input gradient ----------------------------------------
// dx[pixel(n,iy,ix), c] = sum_{k_out,ky,kx valid} dy[n,k_out,oy,ox] *
//                         w[k_out, c, ky, kx],  oy=(iy-ky)/S exact.
// GEMM rows = input pixels, cols = C, reduce dim = KOUT*KH*KW with
// validity-masked dy gathers.
template <int C, int KH, int KW, int STRIDE, int IH, int IW, int OH, int OW,
          int KOUT>
__global__ __launch_bounds__(64) void conv_dgrad_kernel(
    const bf16_t* __restrict__ dout,    // [N, KOUT, OH, OW]
    const bf16_t* __restrict__ weight,  // [KOUT, C, KH, KW]
    bf16_t* __restrict__ dinput,        // [N, C, IH, IW]
    int batch) {
  constexpr int RDIM = KOUT * KH * KW;
  constexpr int RTILES = (RDIM + 31) / 32;
  const int M = batch * IH * IW;
  const int lane = threadIdx.x & (WAVE - 1);
  const int m_base = blockIdx.x * 16;
  const int c_base = blockIdx.y * 16;

  const int arow = m_base + (lane & 15);
  int n = 0, iy = 0, ix = 0;
  const bool row_ok = arow < M;
  if (row_ok) {
    n = arow / (IH * IW);
    const int rem = arow % (IH * IW);
    iy = rem / IW;
    ix = rem % IW;
  }

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int rt = 0; rt < RTILES; ++rt) {
    bf16x8 a, b;
    #pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int k = rt * 32 + (lane >> 4) * 8 + j;
      float av = 0.f, bv = 0.f;
      if (k < RDIM) {
        const int kout = k / (KH * KW);
        const int r = k % (KH * KW);
        const int ky = r / KW, kx = r % KW;
        if (row_ok) {
          const int ty = iy - ky, tx = ix - kx;
          if (ty >= 0 && tx >= 0 && ty % STRIDE == 0 && tx % STRIDE == 0) {
            const int oy = ty / STRIDE, ox = tx / STRIDE;
            if (oy < OH && ox < OW)
              av = (float)dout[(((long)n * KOUT + kout) * OH + oy) * OW + ox];
          }
        }
        const int bcol = c_base + (lane & 15);
        if (bcol < C)
          bv = (float)weight[(((long)kout * C + bcol) * KH + ky) * KW + kx];
      }
      a[j] = (bf16_t)av;
      b[j] = (bf16_t)bv;
    }
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }
  #pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int col = c_base + (lane & 15);
    const int row = m_base + (lane >> 4) * 4 + r;
    if (row < M && col < C) {
      const int ni = row / (IH * IW);
      const int rem = row % (IH * IW);
      dinput[((long)ni * C + col) * IH * IW + rem] = (bf16_t)acc[r];
    }
  }
}

#define DEF_WGRAD(NAME, C_, KH_, KW_, S_, IH_, IW_, OH_, OW_, KO_, U8, T)     \
  extern "C" int NAME(const void* in, const void* dout, float* dw,            \
                      long batch, long split, hipStream_t stream) {           \
    dim3 grid((KO_ + 15) / 16, (C_ * KH_ * KW_ + 15) / 16, (unsigned)split);  \
    hipLaunchKernelGGL(                                                       \
        (conv_wgrad_kernel<C_, KH_, KW_, S_, IH_, IW_, OH_, OW_, KO_, U8, T>),\
        grid, dim3(64), 0, stream, (const T*)in, (const bf16_t*)dout, dw,     \
        (int)batch, (int)split);                                              \
    CHECK_LAUNCH();                                                           \
    return 0;                                                                 \
  }

#define DEF_DGRAD(NAME, C_, KH_, KW_, S_, IH_, IW_, OH_, OW_, KO_)            \
  extern "C" int NAME(const void* dout, const void* w, void* din,             \
                      long batch, hipStream_t stream) {                       \
    const long M = batch * IH_ * IW_;                                         \
    dim3 grid((unsigned)((M + 15) / 16), (C_ + 15) / 16);                     \
    hipLaunchKernelGGL((conv_dgrad_kernel<C_, KH_, KW_, S_, IH_, IW_, OH_,    \
                                          OW_, KO_>),                         \
                       grid, dim3(64), 0, stream, (const bf16_t*)dout,        \
                       (const bf16_t*)w, (bf16_t*)din, (int)batch);           \
    CHECK_LAUNCH();                                                           \
    return 0;                                                                 \
  }

DEF_WGRAD(atari_conv1_wgrad_u8, 4, 8, 8, 4, 84, 84, 20, 20, 32, true,
          unsigned char)
DEF_WGRAD(atari_conv2_wgrad, 32, 4, 4, 2, 20, 20, 9, 9, 64, false, bf16_t)
DEF_WGRAD(atari_conv3_wgrad, 64, 3, 3, 1, 9, 9, 7, 7, 64, false, bf16_t)
DEF_DGRAD(atari_conv2_dgrad, 32, 4, 4, 2, 20, 20, 9, 9, 64)
DEF_DGRAD(atari_conv3_dgrad, 64, 3, 3, 1, 9, 9, 7, 7, 64)

#endif  // v1 backward (superseded by conv_bwd.hip)
