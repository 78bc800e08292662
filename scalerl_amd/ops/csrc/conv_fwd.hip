// Hand-written MFMA implicit-GEMM forward convolutions for the three fixed
// Atari encoder shapes (SURVEY.md §7 hard-parts; replaces MIOpen whose find
// is box-dependent — profiles/README.md r1 finding 1).
//
//   conv1: [N,4,84,84] u8 -> /255 -> conv 8x8 s4 -> +bias, ReLU -> [N,32,20,20]
//   conv2: [N,32,20,20]  -> conv 4x4 s2 -> +bias, ReLU -> [N,64,9,9]
//   conv3: [N,64,9,9]    -> conv 3x3 s1 -> +bias, ReLU -> [N,64,7,7]
//
// v2 design (r2; v1's gather-per-lane step-0 structure measured 3x slower
// than tuned MIOpen):
//  - one workgroup per image (conv2/3: images) / per half-image (conv1),
//    input plane staged ONCE into LDS as bf16 (u8 normalize fused on stage);
//  - A fragments read CONTIGUOUS kernel-window rows from LDS:
//    conv1 j-span = one full 8-px window row (2x ds_read_b64),
//    conv2 j-span = two 4-px window rows   (4x ds_read_b32),
//    conv3 scalar (least work of the three);
//  - B fragments read 16B-contiguous weight rows straight from global
//    (L1/L2-resident: every block reuses the same panel);
//  - C tiles scatter into an LDS output plane; one coalesced vectorized
//    writeback per block applies bias+ReLU (the MFMA C/D col-major lane map
//    would otherwise scatter 2B stores across output planes);
//  - grid = N (20736 at the bench batch) x 1-2 >> 256 CUs.
//
// Fragment maps for v_mfma_f32_16x16x32_bf16 (validated on-device by
// mfma_selftest in conv_atari.hip):
//   A (16x32): lane l, elem j -> row = l & 15, k = (l >> 4) * 8 + j
//   B (32x16): lane l, elem j -> col = l & 15, k = (l >> 4) * 8 + j
//   C/D:       lane l, reg r  -> col = l & 15, row = (l >> 4) * 4 + r

#include "common.h"

typedef __bf16 bf16_t;
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef bf16_t bf16x8 __attribute__((ext_vector_type(8)));
typedef bf16_t bf16x4 __attribute__((ext_vector_type(4)));
typedef bf16_t bf16x2 __attribute__((ext_vector_type(2)));

// ---------------------------------------------------------------- conv1 --
// Block: 256 thr (4 waves), half an image's output rows (OYB=10 of 20).
// LDS: img [4][44][96] bf16 (33.8 KB) + out [32][224] f32-free bf16 tile.
template <bool IN_U8, typename in_t>
__global__ __launch_bounds__(256) void conv1_fwd_v2(
    const in_t* __restrict__ input,     // [N, 4, 84, 84]
    const bf16_t* __restrict__ weight,  // [32, 256]  (c*64 + ky*8 + kx)
    const float* __restrict__ bias,     // [32] or nullptr
    bf16_t* __restrict__ output,        // [N, 32, 20, 20]
    int batch, int relu) {
  constexpr int C = 4, IH = 84, IW = 84, OW = 20;
  constexpr int OYB = 10;            // output rows per block
  constexpr int IYB = 44;            // input rows needed: (OYB-1)*4 + 8
  constexpr int PITCH = 96;          // padded LDS row (84 px)
  constexpr int MPX = OYB * OW;      // 200 output px per block
  constexpr int KDIM = 256;

  __shared__ bf16_t img[C * IYB * PITCH];      // 33792 B
  __shared__ bf16_t out_lds[32 * MPX];         // 12800 B

  const int n = blockIdx.y;
  const int ob = blockIdx.x;                   // 0..1: output-row block
  const int oy0 = ob * OYB;
  const int iy0 = oy0 * 4;

  // ---- stage input rows iy0..iy0+43 (u8 -> /255 bf16, or bf16 copy) ----
  // one (c,row) strip = 84 px; threads sweep (c*IYB + row) * 84 elements
  for (int idx = threadIdx.x; idx < C * IYB * IW; idx += blockDim.x) {
    const int c = idx / (IYB * IW);
    const int rem = idx % (IYB * IW);
    const int row = rem / IW, x = rem % IW;
    const in_t raw = input[(((long)n * C + c) * IH + iy0 + row) * IW + x];
    img[(c * IYB + row) * PITCH + x] =
        (bf16_t)(IN_U8 ? (float)raw * (1.0f / 255.0f) : (float)raw);
  }
  __syncthreads();

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & (WAVE - 1);
  const int g = lane >> 4;                     // k-subgroup 0..3
  const int lr = lane & 15;

  // tiles: 13 m-tiles (208 rows, masked past 200) x 2 n-tiles
  constexpr int MT = (MPX + 15) / 16, NT = 2, TILES = MT * NT;
  for (int t = wave; t < TILES; t += 4) {
    const int mt = t >> 1, nt = t & 1;
    const int px = mt * 16 + lr;               // local output pixel
    const bool ok = px < MPX;
    const int oy_l = px / OW, ox = px % OW;
    const int ch = nt * 16 + lr;               // output channel (A? no: B col)
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    #pragma unroll
    for (int kt = 0; kt < KDIM / 32; ++kt) {
      const int k0 = kt * 32 + g * 8;          // j spans k0..k0+7 = one
      const int c = k0 >> 6;                   // full 8-px window row
      const int ky = (k0 & 63) >> 3;
      bf16x8 a;
      if (ok) {
        const int off = ((c * IYB + oy_l * 4 + ky) * PITCH + ox * 4);
        const bf16x4* p = (const bf16x4*)&img[off];   // 8B-aligned
        bf16x4 lo = p[0], hi = p[1];
        a[0] = lo[0]; a[1] = lo[1]; a[2] = lo[2]; a[3] = lo[3];
        a[4] = hi[0]; a[5] = hi[1]; a[6] = hi[2]; a[7] = hi[3];
      } else {
        a = (bf16x8)(bf16_t)0.f;
      }
      const bf16x8 b = *(const bf16x8*)&weight[(long)ch * KDIM + k0];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = mt * 16 + g * 4 + r;     // pixel
      if (row < MPX)
        out_lds[(nt * 16 + lr) * MPX + row] = (bf16_t)acc[r];
    }
  }
  __syncthreads();

  // ---- coalesced writeback with fused bias+ReLU ----
  for (int idx = threadIdx.x; idx < 32 * MPX; idx += blockDim.x) {
    const int ch = idx / MPX, px = idx % MPX;
    float v = (float)out_lds[ch * MPX + px] + (bias ? bias[ch] : 0.f);
    if (relu) v = fmaxf(v, 0.f);
    output[(((long)n * 32 + ch) * 400) + oy0 * OW + px] = (bf16_t)v;
  }
}

// ---------------------------------------------------------------- conv2 --
// Block: 256 thr, one whole image (81 output px x 64 ch).
// LDS: img [32][20][24] bf16 (30.7 KB) + out [64][81].
__global__ __launch_bounds__(256) void conv2_fwd_v2(
    const bf16_t* __restrict__ input,   // [N, 32, 20, 20]
    const bf16_t* __restrict__ weight,  // [64, 512]  (c*16 + ky*4 + kx)
    const float* __restrict__ bias,     // [64] or nullptr
    bf16_t* __restrict__ output,        // [N, 64, 9, 9]
    int batch, int relu) {
  constexpr int C = 32, IH = 20, IW = 20, OW = 9, PITCH = 24;
  constexpr int MPX = 81, KDIM = 512, KOUT = 64;

  __shared__ bf16_t img[C * IH * PITCH];       // 30720 B
  __shared__ bf16_t out_lds[KOUT * MPX];       // 10368 B

  const int n = blockIdx.x;
  for (int idx = threadIdx.x; idx < C * IH * IW; idx += blockDim.x) {
    const int c = idx / (IH * IW);
    const int rem = idx % (IH * IW);
    img[(c * IH + rem / IW) * PITCH + rem % IW] =
        input[((long)n * C * IH * IW) + idx];
  }
  __syncthreads();

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & (WAVE - 1);
  const int g = lane >> 4, lr = lane & 15;

  constexpr int MT = (MPX + 15) / 16, NT = KOUT / 16, TILES = MT * NT;  // 24
  for (int t = wave; t < TILES; t += 4) {
    const int mt = t / NT, nt = t % NT;
    const int px = mt * 16 + lr;
    const bool ok = px < MPX;
    const int oy = px / OW, ox = px % OW;
    const int ch = nt * 16 + lr;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    #pragma unroll 4
    for (int kt = 0; kt < KDIM / 32; ++kt) {
      const int k0 = kt * 32 + g * 8;          // spans 2 window rows of 4
      const int c = k0 >> 4;
      const int ky = (k0 & 15) >> 2;           // 0 or 2; rows ky, ky+1
      bf16x8 a;
      if (ok) {
        const int base = (c * IH + oy * 2 + ky) * PITCH + ox * 2;
        const bf16x2* p0 = (const bf16x2*)&img[base];
        const bf16x2* p1 = (const bf16x2*)&img[base + PITCH];
        bf16x2 a0 = p0[0], a1 = p0[1], b0 = p1[0], b1 = p1[1];
        a[0] = a0[0]; a[1] = a0[1]; a[2] = a1[0]; a[3] = a1[1];
        a[4] = b0[0]; a[5] = b0[1]; a[6] = b1[0]; a[7] = b1[1];
      } else {
        a = (bf16x8)(bf16_t)0.f;
      }
      const bf16x8 b = *(const bf16x8*)&weight[(long)ch * KDIM + k0];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = mt * 16 + g * 4 + r;
      if (row < MPX)
        out_lds[(nt * 16 + lr) * MPX + row] = (bf16_t)acc[r];
    }
  }
  __syncthreads();

  for (int idx = threadIdx.x; idx < KOUT * MPX; idx += blockDim.x) {
    const int ch = idx / MPX, px = idx % MPX;
    float v = (float)out_lds[idx] + (bias ? bias[ch] : 0.f);
    if (relu) v = fmaxf(v, 0.f);
    output[((long)n * KOUT + ch) * MPX + px] = (bf16_t)v;
  }
}

// ---------------------------------------------------------------- conv3 --
// Block: 256 thr, TWO images (2x49 px x 64 ch).  Window area 9 is coprime
// with the 8-elem j-span, so A gathers stay scalar LDS reads (conv3 is the
// smallest of the three).
__global__ __launch_bounds__(256) void conv3_fwd_v2(
    const bf16_t* __restrict__ input,   // [N, 64, 9, 9]
    const bf16_t* __restrict__ weight,  // [64, 576]  (c*9 + ky*3 + kx)
    const float* __restrict__ bias,     // [64] or nullptr
    bf16_t* __restrict__ output,        // [N, 64, 7, 7]
    int batch, int relu) {
  constexpr int C = 64, IH = 9, IW = 9, OW = 7, PITCH = 12;
  constexpr int PXI = 49, KDIM = 576, KOUT = 64, IMGS = 2;
  constexpr int MPX = PXI * IMGS;              // 98 rows (2 images)

  __shared__ bf16_t img[IMGS * C * IH * PITCH];  // 27648 B
  __shared__ bf16_t out_lds[KOUT * MPX];         // 12544 B

  const int n0 = blockIdx.x * IMGS;
  const int n_here = min(IMGS, batch - n0);
  for (int idx = threadIdx.x; idx < n_here * C * IH * IW;
       idx += blockDim.x) {
    const int i = idx / (C * IH * IW);
    const int rem = idx % (C * IH * IW);
    const int c = rem / (IH * IW), rr = rem % (IH * IW);
    img[((i * C + c) * IH + rr / IW) * PITCH + rr % IW] =
        input[((long)(n0 + i) * C * IH * IW) + rem];
  }
  __syncthreads();

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & (WAVE - 1);
  const int g = lane >> 4, lr = lane & 15;

  constexpr int MT = (MPX + 15) / 16, NT = KOUT / 16, TILES = MT * NT;  // 28
  for (int t = wave; t < TILES; t += 4) {
    const int mt = t / NT, nt = t % NT;
    const int row = mt * 16 + lr;
    const int i = row / PXI, px = row % PXI;
    const bool ok = row < PXI * n_here;
    const int oy = px / OW, ox = px % OW;
    const int ch = nt * 16 + lr;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int kt = 0; kt < KDIM / 32; ++kt) {
      const int k0 = kt * 32 + g * 8;
      bf16x8 a;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int k = k0 + j;
        const int c = k / 9, r = k % 9;
        a[j] = ok ? img[((i * C + c) * IH + oy + r / 3) * PITCH + ox + r % 3]
                  : (bf16_t)0.f;
      }
      const bf16x8 b = *(const bf16x8*)&weight[(long)ch * KDIM + k0];
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int rw = mt * 16 + g * 4 + r;
      if (rw < MPX)
        out_lds[(nt * 16 + lr) * MPX + rw] = (bf16_t)acc[r];
    }
  }
  __syncthreads();

  for (int idx = threadIdx.x; idx < n_here * KOUT * PXI;
       idx += blockDim.x) {
    const int i = idx / (KOUT * PXI);
    const int rem = idx % (KOUT * PXI);
    const int ch = rem / PXI, px = rem % PXI;
    float v = (float)out_lds[ch * MPX + i * PXI + px] +
              (bias ? bias[ch] : 0.f);
    if (relu) v = fmaxf(v, 0.f);
    output[((long)(n0 + i) * KOUT + ch) * PXI + px] = (bf16_t)v;
  }
}

// ---- exported entry points (same ABI as v1) -----------------------------
extern "C" int atari_conv1_fwd_u8(const void* in, const void* w,
                                  const float* bias, void* out, long batch,
                                  int relu, hipStream_t stream) {
  hipLaunchKernelGGL((conv1_fwd_v2<true, unsigned char>),
                     dim3(2, (unsigned)batch), dim3(256), 0, stream,
                     (const unsigned char*)in, (const bf16_t*)w, bias,
                     (bf16_t*)out, (int)batch, relu);
  CHECK_LAUNCH();
  return 0;
}

extern "C" int atari_conv1_fwd_bf16(const void* in, const void* w,
                                    const float* bias, void* out, long batch,
                                    int relu, hipStream_t stream) {
  hipLaunchKernelGGL((conv1_fwd_v2<false, bf16_t>),
                     dim3(2, (unsigned)batch), dim3(256), 0, stream,
                     (const bf16_t*)in, (const bf16_t*)w, bias, (bf16_t*)out,
                     (int)batch, relu);
  CHECK_LAUNCH();
  return 0;
}

extern "C" int atari_conv2_fwd(const void* in, const void* w,
                               const float* bias, void* out, long batch,
                               int relu, hipStream_t stream) {
  hipLaunchKernelGGL(conv2_fwd_v2, dim3((unsigned)batch), dim3(256), 0,
                     stream, (const bf16_t*)in, (const bf16_t*)w, bias,
                     (bf16_t*)out, (int)batch, relu);
  CHECK_LAUNCH();
  return 0;
}

extern "C" int atari_conv3_fwd(const void* in, const void* w,
                               const float* bias, void* out, long batch,
                               int relu, hipStream_t stream) {
  hipLaunchKernelGGL(conv3_fwd_v2, dim3((unsigned)((batch + 1) / 2)),
                     dim3(256), 0, stream, (const bf16_t*)in,
                     (const bf16_t*)w, bias, (bf16_t*)out, (int)batch, relu);
  CHECK_LAUNCH();
  return 0;
}

// ------------------------------------------------- fwd v3 (panel) ------
// conv2/conv3 forward with the im2col panel staged in LDS (K-quartered to
// fit the 64 KB static cap), so the A side becomes contiguous vector
// reads like the B side (v2's conv2 A-loads are 4x ds_read_b32 due to
// ox-parity misalignment; conv3's are scalar).  conv1 fwd already beats
// MIOpen in v2 form and is left alone.  EXPERIMENTAL until
// hardware-validated (r3).
//   NQ: K quarters; KQP = KDIM/NQ (multiple of 32)
//   PADM: M rows padded to a multiple of 16 (garbage rows masked at store)
template <int C, int KH, int KW, int STRIDE, int IH, int IW, int OH, int OW,
          int KOUT, int NQ, int TPW>
__global__ __launch_bounds__(256) void convN_fwd_v3(
    const bf16_t* __restrict__ input,   // [N, C, IH, IW]
    const bf16_t* __restrict__ weight,  // [KOUT, KDIM]
    const float* __restrict__ bias,     // [KOUT] or nullptr
    bf16_t* __restrict__ output,        // [N, KOUT, OH, OW]
    int batch, int relu) {
  constexpr int KDIM = C * KH * KW;
  constexpr int KQP = KDIM / NQ;
  constexpr int MPX = OH * OW;
  constexpr int MT = (MPX + 15) / 16;
  constexpr int PADM = MT * 16;
  constexpr int NT = KOUT / 16;

  __shared__ bf16_t panel[PADM * KQP];
  __shared__ bf16_t out_lds[KOUT * MPX];

  const int n = blockIdx.x;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & (WAVE - 1);
  const int g = lane >> 4, lr = lane & 15;

  f32x4 acc[TPW];
  #pragma unroll
  for (int i = 0; i < TPW; ++i) acc[i] = (f32x4){0.f, 0.f, 0.f, 0.f};

  for (int qq = 0; qq < NQ; ++qq) {
    __syncthreads();  // previous quarter's readers done
    for (int idx = threadIdx.x; idx < MPX * KQP; idx += blockDim.x) {
      const int px = idx / KQP, kq = idx % KQP;
      const int k = qq * KQP + kq;
      const int c = k / (KH * KW);
      const int r = k % (KH * KW);
      const int ky = r / KW, kx = r % KW;
      const int oy = px / OW, ox = px % OW;
      panel[px * KQP + kq] =
          input[(((long)n * C + c) * IH + oy * STRIDE + ky) * IW +
                ox * STRIDE + kx];
    }
    __syncthreads();

    #pragma unroll
    for (int ti = 0; ti < TPW; ++ti) {
      const int t = wave + ti * 4;
      const int mt = t / NT, nt = t % NT;
      const int px = mt * 16 + lr;              // may be a padded row:
      const int ch = nt * 16 + lr;              // garbage masked at store
      #pragma unroll 4
      for (int kt = 0; kt < KQP / 32; ++kt) {
        const int k0 = kt * 32 + g * 8;
        const bf16x8 a = (px < MPX)
            ? *(const bf16x8*)&panel[px * KQP + k0] : (bf16x8)(bf16_t)0.f;
        const bf16x8 b =
            *(const bf16x8*)&weight[(long)ch * KDIM + qq * KQP + k0];
        acc[ti] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[ti],
                                                          0, 0, 0);
      }
    }
  }
  #pragma unroll
  for (int ti = 0; ti < TPW; ++ti) {
    const int t = wave + ti * 4;
    const int mt = t / NT, nt = t % NT;
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = mt * 16 + g * 4 + r;
      if (row < MPX)
        out_lds[(nt * 16 + lr) * MPX + row] = (bf16_t)acc[ti][r];
    }
  }
  __syncthreads();
  for (int idx = threadIdx.x; idx < KOUT * MPX; idx += blockDim.x) {
    const int ch = idx / MPX;
    float v = (float)out_lds[idx] + (bias ? bias[ch] : 0.f);
    if (relu) v = fmaxf(v, 0.f);
    output[((long)n * KOUT) * MPX + idx] = (bf16_t)v;
  }
}

extern "C" int atari_conv2_fwd_v3(const void* in, const void* w,
                                  const float* bias, void* out, long batch,
                                  int relu, hipStream_t stream) {
  // 6 mt x 4 nt = 24 tiles -> TPW 6; 2 K-quarters (panel 96x256 = 49 KB)
  hipLaunchKernelGGL((convN_fwd_v3<32, 4, 4, 2, 20, 20, 9, 9, 64, 2, 6>),
                     dim3((unsigned)batch), dim3(256), 0, stream,
                     (const bf16_t*)in, (const bf16_t*)w, bias,
                     (bf16_t*)out, (int)batch, relu);
  CHECK_LAUNCH();
  return 0;
}

extern "C" int atari_conv3_fwd_v3(const void* in, const void* w,
                                  const float* bias, void* out, long batch,
                                  int relu, hipStream_t stream) {
  // 4 mt x 4 nt = 16 tiles -> TPW 4; 2 K-quarters (panel 64x288 = 36.9 KB)
  hipLaunchKernelGGL((convN_fwd_v3<64, 3, 3, 1, 9, 9, 7, 7, 64, 2, 4>),
                     dim3((unsigned)batch), dim3(256), 0, stream,
                     (const bf16_t*)in, (const bf16_t*)w, bias,
                     (bf16_t*)out, (int)batch, relu);
  CHECK_LAUNCH();
  return 0;
}
