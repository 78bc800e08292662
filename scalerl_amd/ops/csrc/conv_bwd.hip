// Backward (wgrad + dgrad) MFMA kernels for the Atari encoder convs — v2,
// same LDS-staging design as conv_fwd.hip (the v1 gather-per-lane versions
// measured far behind MIOpen and conv3 dgrad failed its oracle on HW).
//
// wgrad: dw[ko][k] = sum_px dy[px,ko] * im2col(x)[px,k]
//   - each block owns the FULL dw tile-set (conv1) or a KDIM quarter
//     (conv2/3), loops over a strided subset of images with x and dy staged
//     in LDS, accumulates in registers, atomicAdds once at the end;
//   - A (dy) fragments are 16B-contiguous pixel runs (1 ds_read_b128);
//     B (im2col) fragments are scalar LDS gathers.
// dgrad: dx[ipx][c] = sum_{ko,ky,kx valid} dy[n,ko,oy,ox] * w[ko,c,ky,kx]
//   - per-image blocks stage dy in LDS (masked scalar gathers), weights
//     come from global (L1-resident), dx staged through LDS for one
//     coalesced vectorized writeback.
//
// Fragment maps as in conv_fwd.hip (validated by mfma_selftest).

#include "common.h"

typedef __bf16 bf16_t;
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef bf16_t bf16x8 __attribute__((ext_vector_type(8)));

// ---------------------------------------------------------------- wgrad --

// conv2/conv3 wgrad share a template: block owns a KDIM quarter.
//  CONVID 2: x [N,32,20,20], dy [N,64,9,9],  dw [64,512], quarter = 8 ch
//  CONVID 3: x [N,64,9,9],   dy [N,64,7,7],  dw [64,576], quarter = 16 ch
template <int C, int KH, int KW, int STRIDE, int IH, int IW, int OH, int OW,
          int KOUT, int CQ, int NTQ, int TPW, bool IN_U8 = false,
          typename in_t = bf16_t>
__global__ __launch_bounds__(256) void convN_wgrad_v2(
    const in_t* __restrict__ input, const bf16_t* __restrict__ dout,
    float* __restrict__ dweight, int batch) {
  constexpr int KDIM = C * KH * KW;
  constexpr int KQ = CQ * KH * KW;          // k-columns per quarter
  constexpr int IPITCH = ((IW + 3) & ~3);
  constexpr int MPX = OH * OW;
  constexpr int PXP = ((MPX + 31) & ~31);   // pitch AND k-loop bound
  constexpr int MT = KOUT / 16;

  __shared__ bf16_t img[CQ * IH * IPITCH];
  __shared__ bf16_t dy[KOUT * PXP];

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & (WAVE - 1);
  const int g = lane >> 4, lr = lane & 15;
  const int q = blockIdx.y;                 // KDIM quarter
  const int c0 = q * CQ;

  f32x4 acc[TPW];
  #pragma unroll
  for (int i = 0; i < TPW; ++i) acc[i] = (f32x4){0.f, 0.f, 0.f, 0.f};

  for (int n = blockIdx.x; n < batch; n += gridDim.x) {
    __syncthreads();
    for (int idx = threadIdx.x; idx < CQ * IH * IW; idx += blockDim.x) {
      const int c = idx / (IH * IW);
      const int rem = idx % (IH * IW);
      const in_t raw = input[(((long)n * C + c0 + c) * IH * IW) + rem];
      img[(c * IH + rem / IW) * IPITCH + rem % IW] =
          (bf16_t)(IN_U8 ? (float)raw * (1.0f / 255.0f) : (float)raw);
    }
    for (int idx = threadIdx.x; idx < KOUT * PXP; idx += blockDim.x) {
      const int ko = idx / PXP, px = idx % PXP;
      dy[idx] = (px < MPX)
          ? dout[((long)n * KOUT + ko) * MPX + px] : (bf16_t)0.f;
    }
    __syncthreads();

    #pragma unroll
    for (int ti = 0; ti < TPW; ++ti) {
      const int t = wave + ti * 4;
      const int mt = t / NTQ, nt = t % NTQ;
      const int ko = mt * 16 + lr;
      const int kq = nt * 16 + lr;          // column within quarter
      const int c = kq / (KH * KW);
      const int r = kq % (KH * KW);
      const int ky = r / KW, kx = r % KW;
      #pragma unroll 2
      for (int pt = 0; pt < PXP / 32; ++pt) {
        const int p0 = pt * 32 + g * 8;
        const bf16x8 a = *(const bf16x8*)&dy[ko * PXP + p0];
        bf16x8 b;
        #pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int p = p0 + j;
          const int oy = p / OW, ox = p % OW;
          bf16_t v = (bf16_t)0.f;
          if (p < MPX)
            v = img[(c * IH + oy * STRIDE + ky) * IPITCH + ox * STRIDE + kx];
          b[j] = v;
        }
        acc[ti] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[ti],
                                                          0, 0, 0);
      }
    }
  }
  #pragma unroll
  for (int ti = 0; ti < TPW; ++ti) {
    const int t = wave + ti * 4;
    const int mt = t / NTQ, nt = t % NTQ;
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int col = q * KQ + nt * 16 + lr;
      const int row = mt * 16 + g * 4 + r;
      atomicAdd(&dweight[(long)row * KDIM + col], acc[ti][r]);
    }
  }
}

// ---------------------------------------------------------------- dgrad --
// Template over shape; IMGS images per block.  M = IMGS*IH*IW input px,
// N = C, K = KOUT*KH*KW with validity-masked dy gathers from LDS.
template <int C, int KH, int KW, int STRIDE, int IH, int IW, int OH, int OW,
          int KOUT, int IMGS>
__global__ __launch_bounds__(256) void convN_dgrad_v2(
    const bf16_t* __restrict__ dout,    // [N, KOUT, OH, OW]
    const bf16_t* __restrict__ weight,  // [KOUT, C, KH, KW]
    bf16_t* __restrict__ dinput,        // [N, C, IH, IW]
    int batch) {
  constexpr int RDIM = KOUT * KH * KW;
  constexpr int OPX = OH * OW;
  constexpr int OPP = ((OPX + 7) & ~7);
  constexpr int IPX = IH * IW;
  constexpr int MPX = IMGS * IPX;
  constexpr int MT = (MPX + 15) / 16, NT = C / 16;

  __shared__ bf16_t dy[IMGS * KOUT * OPP];
  __shared__ bf16_t dx[C * MPX];

  const int n0 = blockIdx.x * IMGS;
  const int n_here = min(IMGS, batch - n0);
  for (int idx = threadIdx.x; idx < n_here * KOUT * OPX;
       idx += blockDim.x) {
    const int i = idx / (KOUT * OPX);
    const int rem = idx % (KOUT * OPX);
    const int ko = rem / OPX, px = rem % OPX;
    dy[(i * KOUT + ko) * OPP + px] =
        dout[((long)(n0 + i) * KOUT + ko) * OPX + px];
  }
  __syncthreads();

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & (WAVE - 1);
  const int g = lane >> 4, lr = lane & 15;

  for (int t = wave; t < MT * NT; t += 4) {
    const int mt = t / NT, nt = t % NT;
    const int row = mt * 16 + lr;
    const int i = row / IPX, px = row % IPX;
    const bool ok = row < IPX * n_here;
    const int iy = px / IW, ix = px % IW;
    const int c = nt * 16 + lr;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    for (int kt = 0; kt < (RDIM + 31) / 32; ++kt) {
      const int k0 = kt * 32 + g * 8;
      bf16x8 a, b;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int k = k0 + j;
        bf16_t av = (bf16_t)0.f, bv = (bf16_t)0.f;
        if (k < RDIM) {
          const int ko = k / (KH * KW);
          const int r = k % (KH * KW);
          const int ky = r / KW, kx = r % KW;
          if (ok) {
            const int ty = iy - ky, tx = ix - kx;
            if (ty >= 0 && tx >= 0 && ty % STRIDE == 0 &&
                tx % STRIDE == 0) {
              const int oy = ty / STRIDE, ox = tx / STRIDE;
              if (oy < OH && ox < OW)
                av = dy[(i * KOUT + ko) * OPP + oy * OW + ox];
            }
          }
          bv = weight[(((long)ko * C + c) * KH + ky) * KW + kx];
        }
        a[j] = av;
        b[j] = bv;
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int rw = mt * 16 + g * 4 + r;
      if (rw < MPX)
        dx[(nt * 16 + lr) * MPX + rw] = (bf16_t)acc[r];
    }
  }
  __syncthreads();

  for (int idx = threadIdx.x; idx < n_here * C * IPX; idx += blockDim.x) {
    const int i = idx / (C * IPX);
    const int rem = idx % (C * IPX);
    const int c = rem / IPX, px = rem % IPX;
    dinput[((long)(n0 + i) * C + c) * IPX + px] =
        dx[c * MPX + i * IPX + px];
  }
}

// ---- exported entry points (same ABI as v1; `split` = grid-size hint) ---
static inline int wgrad_grid(long units) {
  long g = units < 768 ? units : 768;
  return (int)(g < 1 ? 1 : g);
}

extern "C" int atari_conv1_wgrad_u8(const void* in, const void* dout,
                                    float* dw, long batch, long split,
                                    hipStream_t stream) {
  (void)split;
  // 2 mt x 4 nt = 8 tiles -> 2 per wave; quarter = 1 input channel
  hipLaunchKernelGGL(
      (convN_wgrad_v2<4, 8, 8, 4, 84, 84, 20, 20, 32, 1, 4, 2, true,
                      unsigned char>),
      dim3(wgrad_grid(batch), 4), dim3(256), 0, stream,
      (const unsigned char*)in, (const bf16_t*)dout, dw, (int)batch);
  CHECK_LAUNCH();
  return 0;
}

extern "C" int atari_conv2_wgrad(const void* in, const void* dout, float* dw,
                                 long batch, long split, hipStream_t stream) {
  (void)split;
  // 4 mt x 4 nt = 16 tiles -> 4 per wave; quarter = 4 input channels
  hipLaunchKernelGGL(
      (convN_wgrad_v2<32, 4, 4, 2, 20, 20, 9, 9, 64, 4, 4, 4>),
      dim3(wgrad_grid(batch), 8), dim3(256), 0, stream, (const bf16_t*)in,
      (const bf16_t*)dout, dw, (int)batch);
  CHECK_LAUNCH();
  return 0;
}

extern "C" int atari_conv3_wgrad(const void* in, const void* dout, float* dw,
                                 long batch, long split, hipStream_t stream) {
  (void)split;
  // 4 mt x 9 nt = 36 tiles -> 9 per wave; quarter = 16 input channels
  hipLaunchKernelGGL(
      (convN_wgrad_v2<64, 3, 3, 1, 9, 9, 7, 7, 64, 16, 9, 9>),
      dim3(wgrad_grid(batch), 4), dim3(256), 0, stream, (const bf16_t*)in,
      (const bf16_t*)dout, dw, (int)batch);
  CHECK_LAUNCH();
  return 0;
}

extern "C" int atari_conv2_dgrad(const void* dout, const void* w, void* din,
                                 long batch, hipStream_t stream) {
  hipLaunchKernelGGL((convN_dgrad_v2<32, 4, 4, 2, 20, 20, 9, 9, 64, 1>),
                     dim3((unsigned)batch), dim3(256), 0, stream,
                     (const bf16_t*)dout, (const bf16_t*)w, (bf16_t*)din,
                     (int)batch);
  CHECK_LAUNCH();
  return 0;
}

extern "C" int atari_conv3_dgrad(const void* dout, const void* w, void* din,
                                 long batch, hipStream_t stream) {
  hipLaunchKernelGGL((convN_dgrad_v2<64, 3, 3, 1, 9, 9, 7, 7, 64, 2>),
                     dim3((unsigned)((batch + 1) / 2)), dim3(256), 0, stream,
                     (const bf16_t*)dout, (const bf16_t*)w, (bf16_t*)din,
                     (int)batch);
  CHECK_LAUNCH();
  return 0;
}

// ------------------------------------------------ dgrad v3 (stride-2) --
// Parity-decomposed input gradient for conv2 (4x4 stride 2): the masked
// v2 formulation wastes 1/stride^2 = 4x of its MFMA work on taps whose
// (iy-ky) % 2 != 0 (measured 27.9 ms = 45.8% of native-mode GPU time,
// profiles/r2_micro_kernel_stats_native_conv.csv).  Here each of the 4
// input-parity classes (py, px) reduces only over its OWN 2x2 sub-kernel:
//   dx[n,c,2u+py,2v+px] = sum_{ko,a,b} dy[n,ko,u-a,v-b] * w[ko,c,py+2a,px+2b]
// K = KOUT*4 = 256 (vs 1024 masked), boundary masks only.
// EXPERIMENTAL until hardware-validated (SCALERL_EXPERIMENTAL gpu test).
__global__ __launch_bounds__(256) void conv2_dgrad_v3(
    const bf16_t* __restrict__ dout,    // [N, 64, 9, 9]
    const bf16_t* __restrict__ weight,  // [64, 32, 4, 4]
    bf16_t* __restrict__ dinput,        // [N, 32, 20, 20]
    int batch) {
  constexpr int C = 32, KOUT = 64, OH = 9, OW = 9, IH = 20, IW = 20;
  constexpr int OPX = OH * OW, OPP = 88, IPX = IH * IW;
  constexpr int HP = 10;                 // per-parity rows/cols (20/2)
  constexpr int MP = HP * HP;            // 100 px per parity class
  constexpr int KP = KOUT * 4;           // 256
  constexpr int MT = (MP + 15) / 16;     // 7
  constexpr int NT = C / 16;             // 2

  __shared__ bf16_t dy[KOUT * OPP];      // 11264 B
  __shared__ bf16_t dx[C * IPX];         // 25600 B

  const int n = blockIdx.x;
  for (int idx = threadIdx.x; idx < KOUT * OPX; idx += blockDim.x) {
    const int ko = idx / OPX, px = idx % OPX;
    dy[ko * OPP + px] = dout[((long)n * KOUT + ko) * OPX + px];
  }
  __syncthreads();

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & (WAVE - 1);
  const int g = lane >> 4, lr = lane & 15;

  // 4 parity classes x 7 m-tiles x 2 n-tiles = 56 tiles
  for (int t = wave; t < 4 * MT * NT; t += 4) {
    const int par = t / (MT * NT);
    const int rem = t % (MT * NT);
    const int mt = rem / NT, nt = rem % NT;
    const int py = par >> 1, px_par = par & 1;
    const int row = mt * 16 + lr;        // parity-local pixel
    const bool ok = row < MP;
    const int u = row / HP, v = row % HP;
    const int c = nt * 16 + lr;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    #pragma unroll
    for (int kt = 0; kt < KP / 32; ++kt) {
      const int k0 = kt * 32 + g * 8;
      bf16x8 a, b;
      #pragma unroll
      for (int j = 0; j < 8; ++j) {
        const int k = k0 + j;            // = ko*4 + a*2 + b
        const int ko = k >> 2, ab = k & 3;
        const int aa = ab >> 1, bb = ab & 1;
        bf16_t av = (bf16_t)0.f;
        if (ok) {
          const int oy = u - aa, ox = v - bb;
          if (oy >= 0 && oy < OH && ox >= 0 && ox < OW)
            av = dy[ko * OPP + oy * OW + ox];
        }
        a[j] = av;
        b[j] = weight[(((long)ko * C + c) * 4 + py + 2 * aa) * 4 +
                      px_par + 2 * bb];
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int rw = mt * 16 + g * 4 + r;
      if (rw < MP) {
        const int uu = rw / HP, vv = rw % HP;
        dx[(nt * 16 + lr) * IPX + (2 * uu + py) * IW + 2 * vv + px_par] =
            (bf16_t)acc[r];
      }
    }
  }
  __syncthreads();

  for (int idx = threadIdx.x; idx < C * IPX; idx += blockDim.x)
    dinput[((long)n * C) * IPX + idx] = dx[idx];
}

extern "C" int atari_conv2_dgrad_v3(const void* dout, const void* w,
                                    void* din, long batch,
                                    hipStream_t stream) {
  hipLaunchKernelGGL(conv2_dgrad_v3, dim3((unsigned)batch), dim3(256), 0,
                     stream, (const bf16_t*)dout, (const bf16_t*)w,
                     (bf16_t*)din, (int)batch);
  CHECK_LAUNCH();
  return 0;
}

// ------------------------------------------------ wgrad v3 (panel) -----
// The v2 wgrad's B side re-gathers im2col values scalar-by-scalar for
// every tile x ktile (r2c4 A/B: conv1 4.07 vs MIOpen ~1.24 ms).  v3
// builds the quarter's im2col panel [KQ x PXC] in LDS ONCE per
// (image, pixel-chunk), making BOTH MFMA operands contiguous vector
// reads.  EXPERIMENTAL until hardware-validated (r3).
//   PCH: pixel chunks per image (keeps LDS under the 64 KB static cap)
//   PXC: padded chunk length (multiple of 32; zero-filled tail)
template <int C, int KH, int KW, int STRIDE, int IH, int IW, int OH, int OW,
          int KOUT, int CQ, int NTQ, int TPW, int PCH, int PXC,
          bool IN_U8 = false, typename in_t = bf16_t>
__global__ __launch_bounds__(256) void convN_wgrad_v3(
    const in_t* __restrict__ input, const bf16_t* __restrict__ dout,
    float* __restrict__ dweight, int batch) {
  constexpr int KDIM = C * KH * KW;
  constexpr int KQ = CQ * KH * KW;
  constexpr int MPX = OH * OW;
  constexpr int CHUNK = (MPX + PCH - 1) / PCH;   // real px per chunk
  constexpr int MT = KOUT / 16;

  __shared__ bf16_t panel[KQ * PXC];
  __shared__ bf16_t dy[KOUT * PXC];

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & (WAVE - 1);
  const int g = lane >> 4, lr = lane & 15;
  const int q = blockIdx.y;
  const int c0 = q * CQ;

  f32x4 acc[TPW];
  #pragma unroll
  for (int i = 0; i < TPW; ++i) acc[i] = (f32x4){0.f, 0.f, 0.f, 0.f};

  const int units = batch * PCH;
  for (int u = blockIdx.x; u < units; u += gridDim.x) {
    const int n = u / PCH, ch = u % PCH;
    const int p0g = ch * CHUNK;
    const int nreal = min(CHUNK, MPX - p0g);
    __syncthreads();
    for (int idx = threadIdx.x; idx < KOUT * PXC; idx += blockDim.x) {
      const int ko = idx / PXC, p = idx % PXC;
      dy[idx] = (p < nreal)
          ? dout[((long)n * KOUT + ko) * MPX + p0g + p] : (bf16_t)0.f;
    }
    for (int idx = threadIdx.x; idx < KQ * PXC; idx += blockDim.x) {
      const int kq = idx / PXC, p = idx % PXC;
      bf16_t v = (bf16_t)0.f;
      if (p < nreal) {
        const int c = kq / (KH * KW);
        const int r = kq % (KH * KW);
        const int ky = r / KW, kx = r % KW;
        const int pg = p0g + p;
        const int oy = pg / OW, ox = pg % OW;
        const in_t raw = input[(((long)n * C + c0 + c) * IH +
                                oy * STRIDE + ky) * IW + ox * STRIDE + kx];
        v = (bf16_t)(IN_U8 ? (float)raw * (1.0f / 255.0f) : (float)raw);
      }
      panel[idx] = v;
    }
    __syncthreads();

    #pragma unroll
    for (int ti = 0; ti < TPW; ++ti) {
      const int t = wave + ti * 4;
      const int mt = t / NTQ, nt = t % NTQ;
      const int ko = mt * 16 + lr;
      const int kq = nt * 16 + lr;
      #pragma unroll 2
      for (int pt = 0; pt < PXC / 32; ++pt) {
        const int p0 = pt * 32 + g * 8;
        const bf16x8 a = *(const bf16x8*)&dy[ko * PXC + p0];
        const bf16x8 b = *(const bf16x8*)&panel[kq * PXC + p0];
        acc[ti] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[ti],
                                                          0, 0, 0);
      }
    }
  }
  #pragma unroll
  for (int ti = 0; ti < TPW; ++ti) {
    const int t = wave + ti * 4;
    const int mt = t / NTQ, nt = t % NTQ;
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int col = q * KQ + nt * 16 + lr;
      const int row = mt * 16 + g * 4 + r;
      atomicAdd(&dweight[(long)row * KDIM + col], acc[ti][r]);
    }
  }
}

extern "C" int atari_conv1_wgrad_v3(const void* in, const void* dout,
                                    float* dw, long batch,
                                    hipStream_t stream) {
  // 2 chunks of 200 px (PXC 224); quarter = 1 channel; 2mt x 4nt, TPW 2
  hipLaunchKernelGGL(
      (convN_wgrad_v3<4, 8, 8, 4, 84, 84, 20, 20, 32, 1, 4, 2, 2, 224,
                      true, unsigned char>),
      dim3(wgrad_grid(2 * batch), 4), dim3(256), 0, stream,
      (const unsigned char*)in, (const bf16_t*)dout, dw, (int)batch);
  CHECK_LAUNCH();
  return 0;
}

extern "C" int atari_conv2_wgrad_v3(const void* in, const void* dout,
                                    float* dw, long batch,
                                    hipStream_t stream) {
  // 1 chunk of 81 px (PXC 96); quarter = 4 ch; 4mt x 4nt, TPW 4
  hipLaunchKernelGGL(
      (convN_wgrad_v3<32, 4, 4, 2, 20, 20, 9, 9, 64, 4, 4, 4, 1, 96>),
      dim3(wgrad_grid(batch), 8), dim3(256), 0, stream, (const bf16_t*)in,
      (const bf16_t*)dout, dw, (int)batch);
  CHECK_LAUNCH();
  return 0;
}

extern "C" int atari_conv3_wgrad_v3(const void* in, const void* dout,
                                    float* dw, long batch,
                                    hipStream_t stream) {
  // 1 chunk of 49 px (PXC 64); quarter = 16 ch; 4mt x 9nt, TPW 9
  hipLaunchKernelGGL(
      (convN_wgrad_v3<64, 3, 3, 1, 9, 9, 7, 7, 64, 16, 9, 9, 1, 64>),
      dim3(wgrad_grid(batch), 4), dim3(256), 0, stream, (const bf16_t*)in,
      (const bf16_t*)dout, dw, (int)batch);
  CHECK_LAUNCH();
  return 0;
}
