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

// ---- generic implicit-GEMM forward conv (compile-time shape) ----
// Block: 4 waves in a 2(M)x2(N) arrangement -> 32x32 output tile.
// IN_U8: input is uint8, normalized by /255 on load (conv1).
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

DEF_CONV(atari_conv1_fwd_u8, 4, 8, 8, 4, 84, 84, 20, 20, 32, true,
         unsigned char)
DEF_CONV(atari_conv1_fwd_bf16, 4, 8, 8, 4, 84, 84, 20, 20, 32, false, bf16_t)
DEF_CONV(atari_conv2_fwd, 32, 4, 4, 2, 20, 20, 9, 9, 64, false, bf16_t)
DEF_CONV(atari_conv3_fwd, 64, 3, 3, 1, 9, 9, 7, 7, 64, false, bf16_t)
