// MFMA implicit-GEMM convolution for the DenseNet bottleneck layers
// (gfx950 / CDNA4), NHWC layout, bf16 operands, f32 accumulation, with the
// pre-activation BN + ReLU fused into the input load stage.
//
// GEMM view (stride-1 "same" conv, NHWC):
//   out[p, k] = sum_{(r,s), c} relu(x[p + D(r,s), c] * scale_c + shift_c)
//               * w[k, r, s, c]
// M = B*H*W spatial positions, N = K_out (<= 96 here: bottleneck convs are
// C->96 (1x1) and 96->24 (3x3)), K-dim = R*S*C.
//
// Tiling: 256-thread workgroups (4 waves), BM=128 positions x BN=K_out,
// BK=32 k-dim chunk = one v_mfma_f32_16x16x32_bf16 step; wave w owns rows
// [w*32, w*32+32) => 2 x (BN/16) accumulator fragments.  A (input) and B
// (weight) tiles are LDS-staged; BN/ReLU applied on the way in.
// Correctness-first: single-buffered, two barriers per K-chunk.  See
// profiles/ for measured utilization.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64
#define DEVC __device__ __forceinline__

typedef __hip_bfloat16 bf16_t;
typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;

DEVC float bf2f(bf16_t v) { return __bfloat162float(v); }
DEVC bf16_t f2bf(float v) { return __float2bfloat16(v); }

// ---------------------------------------------------------------------
// Fused BN-ReLU-conv forward.
//   x     [B, H, W, C]   bf16 (NHWC physical)
//   w     [K, R, S, C]   bf16 (channels_last physical)
//   scale [C], shift [C] f32  (BN folded: scale=gamma*rsqrt(var+eps),
//                              shift=beta-mean*scale; identity: 1/0)
//   out   [B, H, W, K]   bf16
// Grid: (ceil(BHW/BM), 1); 256 threads.  KOUT <= 96, C % 8 == 0.
// ---------------------------------------------------------------------

#define BM 128

// BKC = kdim chunk staged per barrier pair.  32 for arbitrary C (1x1
// bottleneck convs, C=48..684); 96 for the 3x3 convs (C always 96 there):
// 3x fewer barriers, 12 MFMAs per wave between barriers instead of 4.
template <int BKC>
__device__ __forceinline__
void bnrelu_conv_mfma_body(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ w,
    const float* __restrict__ scale, const float* __restrict__ shift,
    bf16_t* __restrict__ out,
    int Bn, int H, int W, int C, int KOUT, int R, int S, int relu_on) {
  __shared__ bf16_t lA[BM][BKC];
  __shared__ bf16_t lB[96][BKC];
  __shared__ float lscale[BKC], lshift[BKC];

  const int HW = H * W;
  const long M = (long)Bn * HW;
  const long m0 = (long)blockIdx.x * BM;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wv = tid >> 6;               // wave id 0..3
  const int PR = R / 2, PS = S / 2;      // "same" padding

  const int NFRAG = (KOUT + 15) / 16;    // <= 6
  f32x4 acc[2][6];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 6; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int r = 0; r < R; ++r) {
    for (int s = 0; s < S; ++s) {
      for (int c0 = 0; c0 < C; c0 += BKC) {
        // ---- stage scale/shift for this channel chunk
        if (tid < BKC) {
          const int c = c0 + tid;
          lscale[tid] = c < C ? scale[c] : 0.f;
          lshift[tid] = c < C ? shift[c] : 0.f;
        }
        __syncthreads();
        // ---- stage A: BM positions x BKC channels; each thread stages
        // its share of the BM*BKC/8 16-byte (8 x bf16) pieces
        constexpr int PIECES_PER_ROW = BKC / 8;
        constexpr int A_PIECES = BM * PIECES_PER_ROW;
#pragma unroll
        for (int i = 0; i < A_PIECES / 256; ++i) {
          const int flat = tid * (A_PIECES / 256) + i;
          const int arow = flat / PIECES_PER_ROW;
          const int apc = flat % PIECES_PER_ROW;
          const long p = m0 + arow;
          // Conv zero-padding applies AFTER the BN+ReLU (the reference
          // composition pads the normalized tensor): out-of-bounds taps
          // contribute exactly 0, never relu(shift).
          bool inb = false;
          long base = 0;
          if (p < M) {
            const int b = (int)(p / HW);
            const int rem = (int)(p % HW);
            const int y = rem / W + r - PR;
            const int xw = rem % W + s - PS;
            if (y >= 0 && y < H && xw >= 0 && xw < W) {
              inb = true;
              base = (((long)b * H + y) * W + xw) * C + c0 + apc * 8;
            }
          }
          if (inb && c0 + apc * 8 + 8 <= C) {
            // fast path: aligned 16-byte load + BN/ReLU
            const bf16x8 v8 = *(const bf16x8*)(x + base);
#pragma unroll
            for (int e = 0; e < 8; ++e) {
              const int cc = apc * 8 + e;
              float v = bf2f(((const bf16_t*)&v8)[e]) * lscale[cc] + lshift[cc];
              if (relu_on) v = fmaxf(v, 0.f);
              lA[arow][cc] = f2bf(v);
            }
          } else {
#pragma unroll
            for (int e = 0; e < 8; ++e) {
              const int cc = apc * 8 + e;
              const int c = c0 + cc;
              float v = 0.f;
              if (inb && c < C) {
                v = bf2f(x[base + e]) * lscale[cc] + lshift[cc];
                if (relu_on) v = fmaxf(v, 0.f);
              }
              lA[arow][cc] = f2bf(v);
            }
          }
        }
        // ---- stage B: KOUT rows x BKC channels of tap (r,s)
        // thread t stages pieces of w[k, r, s, c0..c0+31]
        for (int flat = tid; flat < NFRAG * 16 * PIECES_PER_ROW;
             flat += blockDim.x) {
          const int k = flat / PIECES_PER_ROW;
          const int bpc = flat % PIECES_PER_ROW;
          if (k < 96) {
#pragma unroll
            for (int e = 0; e < 8; ++e) {
              const int c = c0 + bpc * 8 + e;
              float v = 0.f;
              if (k < KOUT && c < C)
                v = bf2f(w[(((long)k * R + r) * S + s) * C + c]);
              lB[k][bpc * 8 + e] = f2bf(v);
            }
          }
        }
        __syncthreads();
        // ---- MFMA: wave wv covers rows [wv*32, wv*32+32)
#pragma unroll
        for (int ks = 0; ks < BKC / 32; ++ks) {
#pragma unroll
          for (int mf = 0; mf < 2; ++mf) {
            const int arow = wv * 32 + mf * 16 + (lane & 15);
            const int kk = ks * 32 + (lane >> 4) * 8;
            bf16x8 afrag = *(const bf16x8*)&lA[arow][kk];
#pragma unroll
            for (int nf = 0; nf < 6; ++nf) {
              if (nf >= NFRAG) break;
              const int brow = nf * 16 + (lane & 15);
              bf16x8 bfrag = *(const bf16x8*)&lB[brow][kk];
              acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  afrag, bfrag, acc[mf][nf], 0, 0, 0);
            }
          }
        }
        __syncthreads();
      }
    }
  }

  // ---- epilogue: C/D layout col=lane&15 (out channel), row=(lane>>4)*4+reg
#pragma unroll
  for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 6; ++nf) {
      if (nf >= NFRAG) break;
      const int k = nf * 16 + (lane & 15);
      if (k >= KOUT) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int mrow = wv * 32 + mf * 16 + (lane >> 4) * 4 + reg;
        const long p = m0 + mrow;
        if (p < M) out[p * KOUT + k] = f2bf(acc[mf][nf][reg]);
      }
    }
  }
}

extern "C" __global__ __launch_bounds__(256, 2)
void bnrelu_conv_mfma_fwd(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ w,
    const float* __restrict__ scale, const float* __restrict__ shift,
    bf16_t* __restrict__ out,
    int Bn, int H, int W, int C, int KOUT, int R, int S, int relu_on) {
  bnrelu_conv_mfma_body<32>(x, w, scale, shift, out, Bn, H, W, C, KOUT,
                            R, S, relu_on);
}

extern "C" __global__ __launch_bounds__(256, 2)
void bnrelu_conv_mfma_fwd_bk96(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ w,
    const float* __restrict__ scale, const float* __restrict__ shift,
    bf16_t* __restrict__ out,
    int Bn, int H, int W, int C, int KOUT, int R, int S, int relu_on) {
  bnrelu_conv_mfma_body<96>(x, w, scale, shift, out, Bn, H, W, C, KOUT,
                            R, S, relu_on);
}

// ---------------------------------------------------------------------
// FP8 (OCP e4m3) variant — BASELINE config 5.  x and w are pre-quantized
// e4m3 bytes; their per-tensor dequant scales are folded into the BN
// affine (x) and the output scale (w), so the kernel math is:
//   out[p,k] = out_scale * sum relu8(x_q[p+D,c]*scale_c + shift_c) * w_q[k,..]
// where relu8 re-quantizes the normalized activation to e4m3 for the MFMA
// (v_mfma_f32_16x16x32_fp8_fp8, f32 accumulation).  Same tiling as bf16.
// ---------------------------------------------------------------------

#include <hip/hip_fp8.h>

typedef __hip_fp8_e4m3 fp8_t;
typedef long i64frag;

DEVC float f8tof(fp8_t v) { return (float)v; }
DEVC fp8_t ftof8(float v) { return fp8_t(v); }

template <int BKC>
__device__ __forceinline__
void bnrelu_conv_mfma_fp8_body(
    const fp8_t* __restrict__ x, const fp8_t* __restrict__ w,
    const float* __restrict__ scale, const float* __restrict__ shift,
    float out_scale, __hip_bfloat16* __restrict__ out,
    int Bn, int H, int W, int C, int KOUT, int R, int S, int relu_on) {
  __shared__ fp8_t lA[BM][BKC];
  __shared__ fp8_t lB[96][BKC];
  __shared__ float lscale[BKC], lshift[BKC];

  const int HW = H * W;
  const long M = (long)Bn * HW;
  const long m0 = (long)blockIdx.x * BM;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wv = tid >> 6;
  const int PR = R / 2, PS = S / 2;
  const int NFRAG = (KOUT + 15) / 16;
  f32x4 acc[2][6];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 6; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int r = 0; r < R; ++r) {
    for (int s = 0; s < S; ++s) {
      for (int c0 = 0; c0 < C; c0 += BKC) {
        if (tid < BKC) {
          const int c = c0 + tid;
          lscale[tid] = c < C ? scale[c] : 0.f;
          lshift[tid] = c < C ? shift[c] : 0.f;
        }
        __syncthreads();
        constexpr int PIECES_PER_ROW = BKC / 8;   // 8 fp8 per piece
        constexpr int A_PIECES = BM * PIECES_PER_ROW;
#pragma unroll
        for (int i = 0; i < A_PIECES / 256; ++i) {
          const int flat = tid * (A_PIECES / 256) + i;
          const int arow = flat / PIECES_PER_ROW;
          const int apc = flat % PIECES_PER_ROW;
          const long p = m0 + arow;
          bool inb = false;
          long base = 0;
          if (p < M) {
            const int b = (int)(p / HW);
            const int rem = (int)(p % HW);
            const int y = rem / W + r - PR;
            const int xw = rem % W + s - PS;
            if (y >= 0 && y < H && xw >= 0 && xw < W) {
              inb = true;
              base = (((long)b * H + y) * W + xw) * C + c0 + apc * 8;
            }
          }
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            const int cc = apc * 8 + e;
            const int c = c0 + cc;
            float v = 0.f;
            if (inb && c < C) {
              v = f8tof(x[base + e]) * lscale[cc] + lshift[cc];
              if (relu_on) v = fmaxf(v, 0.f);
            }
            lA[arow][cc] = ftof8(v);
          }
        }
        for (int flat = tid; flat < NFRAG * 16 * PIECES_PER_ROW;
             flat += blockDim.x) {
          const int k = flat / PIECES_PER_ROW;
          const int bpc = flat % PIECES_PER_ROW;
          if (k < 96) {
#pragma unroll
            for (int e = 0; e < 8; ++e) {
              const int c = c0 + bpc * 8 + e;
              lB[k][bpc * 8 + e] = (k < KOUT && c < C)
                  ? w[(((long)k * R + r) * S + s) * C + c] : ftof8(0.f);
            }
          }
        }
        __syncthreads();
#pragma unroll
        for (int ks = 0; ks < BKC / 32; ++ks) {
#pragma unroll
          for (int mf = 0; mf < 2; ++mf) {
            const int arow = wv * 32 + mf * 16 + (lane & 15);
            const int kk = ks * 32 + (lane >> 4) * 8;
            const i64frag afrag = *(const i64frag*)&lA[arow][kk];
#pragma unroll
            for (int nf = 0; nf < 6; ++nf) {
              if (nf >= NFRAG) break;
              const int brow = nf * 16 + (lane & 15);
              const i64frag bfrag = *(const i64frag*)&lB[brow][kk];
              acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
                  afrag, bfrag, acc[mf][nf], 0, 0, 0);
            }
          }
        }
        __syncthreads();
      }
    }
  }
#pragma unroll
  for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 6; ++nf) {
      if (nf >= NFRAG) break;
      const int k = nf * 16 + (lane & 15);
      if (k >= KOUT) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int mrow = wv * 32 + mf * 16 + (lane >> 4) * 4 + reg;
        const long p = m0 + mrow;
        if (p < M) out[p * KOUT + k] = f2bf(acc[mf][nf][reg] * out_scale);
      }
    }
  }
}

extern "C" __global__ __launch_bounds__(256, 2)
void bnrelu_conv_mfma_fp8_fwd(
    const fp8_t* __restrict__ x, const fp8_t* __restrict__ w,
    const float* __restrict__ scale, const float* __restrict__ shift,
    float out_scale, __hip_bfloat16* __restrict__ out,
    int Bn, int H, int W, int C, int KOUT, int R, int S, int relu_on) {
  bnrelu_conv_mfma_fp8_body<32>(x, w, scale, shift, out_scale, out,
                                Bn, H, W, C, KOUT, R, S, relu_on);
}

extern "C" __global__ __launch_bounds__(256, 2)
void bnrelu_conv_mfma_fp8_fwd_bk96(
    const fp8_t* __restrict__ x, const fp8_t* __restrict__ w,
    const float* __restrict__ scale, const float* __restrict__ shift,
    float out_scale, __hip_bfloat16* __restrict__ out,
    int Bn, int H, int W, int C, int KOUT, int R, int S, int relu_on) {
  bnrelu_conv_mfma_fp8_body<96>(x, w, scale, shift, out_scale, out,
                                Bn, H, W, C, KOUT, R, S, relu_on);
}

// ---------------------------------------------------------------------
// Halo-tiled 3x3 conv (stride 1, pad 1), C == 96, KOUT <= 32.
// The bottleneck 3x3 is always 96 -> 24; the current generic kernel
// re-stages the input tile per tap (9 taps x 2 barriers).  Here one
// workgroup owns TW consecutive x-positions of ONE row y and stages the
// normalized 3-row halo [3][TW+2][96] ONCE (2 barriers total); all nine
// taps read shifted LDS views.
// Grid: (ceil(W/TW), B*H); 256 threads, 4 waves (wave w owns positions
// [w*16, w*16+16) of the tile -> 1 m-frag x (KOUT/16) n-frags).
// ---------------------------------------------------------------------

#define TW3 64
#define C3 96

extern "C" __global__ __launch_bounds__(256, 2)
void bnrelu_conv3_halo_fwd(
    const bf16_t* __restrict__ x, const bf16_t* __restrict__ w,
    const float* __restrict__ scale, const float* __restrict__ shift,
    bf16_t* __restrict__ out,
    int Bn, int H, int W, int KOUT, int relu_on) {
  __shared__ bf16_t lA[3][TW3 + 2][C3];     // 38 KB
  __shared__ bf16_t lB[9][32][C3];          // 54 KB (all taps, KOUT<=32)
  __shared__ float lscale[C3], lshift[C3];

  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wv = tid >> 6;
  const int x0 = blockIdx.x * TW3;
  const int by = blockIdx.y;            // b*H + y
  const int b = by / H, y = by % H;
  const int NFRAG = (KOUT + 15) / 16;   // <= 2

  if (tid < C3) {
    lscale[tid] = scale[tid];
    lshift[tid] = shift[tid];
  }
  __syncthreads();

  // ---- stage the 3-row halo: 3*(TW3+2) = 198 positions x 96 ch.
  // 198*12 = 2376 16B pieces; thread stages ceil(2376/256)=10 pieces.
  {
    const int PIECES = 3 * (TW3 + 2) * (C3 / 8);
    for (int flat = tid; flat < PIECES; flat += blockDim.x) {
      const int pc = flat % (C3 / 8);
      const int pos = flat / (C3 / 8);
      const int row = pos / (TW3 + 2);        // 0..2 -> y-1..y+1
      const int xi = x0 + pos % (TW3 + 2) - 1;
      const int yy = y + row - 1;
      const int c0 = pc * 8;
      float v[8];
      if (yy >= 0 && yy < H && xi >= 0 && xi < W) {
        const bf16x8 v8 = *(const bf16x8*)(x + (((long)b * H + yy) * W + xi) * C3 + c0);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const short sv = v8[e];
          float t = bf2f(*(const bf16_t*)&sv) * lscale[c0 + e] + lshift[c0 + e];
          if (relu_on) t = fmaxf(t, 0.f);
          v[e] = t;
        }
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) v[e] = 0.f;
      }
#pragma unroll
      for (int e = 0; e < 8; ++e)
        lA[row][pos % (TW3 + 2)][c0 + e] = f2bf(v[e]);
    }
    // ---- stage ALL nine weight taps: 9*KOUT*96 <= 9*32*12 pieces
    const int WP = 9 * 32 * (C3 / 8);
    for (int flat = tid; flat < WP; flat += blockDim.x) {
      const int pc = flat % (C3 / 8);
      const int kk = (flat / (C3 / 8)) % 32;
      const int tap = flat / (32 * (C3 / 8));
      const int c0 = pc * 8;
      if (kk < KOUT) {
        *(bf16x8*)&lB[tap][kk][c0] =
            *(const bf16x8*)(w + (((long)kk * 3 + tap / 3) * 3 + tap % 3) * C3 + c0);
      } else {
#pragma unroll
        for (int e = 0; e < 8; ++e) lB[tap][kk][c0 + e] = f2bf(0.f);
      }
    }
  }
  __syncthreads();

  f32x4 acc[1][2];
#pragma unroll
  for (int j = 0; j < 2; ++j) acc[0][j] = {0.f, 0.f, 0.f, 0.f};

  // wave wv owns tile positions [wv*16, wv*16+16)
#pragma unroll
  for (int tap = 0; tap < 9; ++tap) {
    const int dy = tap / 3, dx = tap % 3;
#pragma unroll
    for (int ks = 0; ks < C3 / 32; ++ks) {
      const int kk = ks * 32 + (lane >> 4) * 8;
      // A fragment: position = wv*16 + (lane&15); shifted x index dx
      const bf16x8 afrag =
          *(const bf16x8*)&lA[dy][wv * 16 + (lane & 15) + dx][kk];
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        if (nf >= NFRAG) break;
        const bf16x8 bfrag = *(const bf16x8*)&lB[tap][nf * 16 + (lane & 15)][kk];
        acc[0][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc[0][nf], 0, 0, 0);
      }
    }
  }

#pragma unroll
  for (int nf = 0; nf < 2; ++nf) {
    if (nf >= NFRAG) break;
    const int k = nf * 16 + (lane & 15);
    if (k >= KOUT) continue;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int xt = wv * 16 + (lane >> 4) * 4 + reg;   // tile position
      const int xi = x0 + xt;
      if (xi < W) out[(((long)b * H + y) * W + xi) * KOUT + k] =
          f2bf(acc[0][nf][reg]);
    }
  }
}

// ---------------------------------------------------------------------
// MX-scaled fp8 path: v_mfma_scale_f32_16x16x128_f8f6f4 (K=128/instr, the
// only instruction that reaches gfx950's ~5 PF dense fp8 rate — plain
// v_mfma_f32_16x16x32_fp8_fp8 runs at the bf16 rate, measured in r01).
// 1x1 convs only (the K-dim is C).  Operands are staged exactly like the
// non-MX fp8 kernel but in 128-channel chunks; scales are E8M0 bytes
// (here unit scale 0x7F per block — per-tensor amax scaling is already
// folded into the BN affine / output scale by the wrapper, so unit block
// scale preserves the existing numerics while using the fast instruction;
// true per-32-block scaling is a further numerics upgrade, not a speed
// requirement).
// ---------------------------------------------------------------------

typedef __attribute__((ext_vector_type(8))) int i32x8mx;

extern "C" __global__ __launch_bounds__(256, 2)
void bnrelu_conv_mfma_mx_fwd(
    const fp8_t* __restrict__ x, const fp8_t* __restrict__ w,
    const float* __restrict__ scale, const float* __restrict__ shift,
    float out_scale, __hip_bfloat16* __restrict__ out,
    int Bn, int H, int W, int C, int KOUT, int relu_on) {
  constexpr int BKC = 128;
  __shared__ fp8_t lA[BM][BKC];
  __shared__ fp8_t lB[96][BKC];
  __shared__ float lscale[BKC], lshift[BKC];

  const int HW = H * W;
  const long M = (long)Bn * HW;
  const long m0 = (long)blockIdx.x * BM;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wv = tid >> 6;
  const int NFRAG = (KOUT + 15) / 16;
  f32x4 acc[2][6];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 6; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int c0 = 0; c0 < C; c0 += BKC) {
    if (tid < BKC) {
      const int c = c0 + tid;
      lscale[tid] = c < C ? scale[c] : 0.f;
      lshift[tid] = c < C ? shift[c] : 0.f;
    }
    __syncthreads();
    constexpr int PIECES_PER_ROW = BKC / 8;
    constexpr int A_PIECES = BM * PIECES_PER_ROW;
#pragma unroll
    for (int i = 0; i < A_PIECES / 256; ++i) {
      const int flat = tid * (A_PIECES / 256) + i;
      const int arow = flat / PIECES_PER_ROW;
      const int apc = flat % PIECES_PER_ROW;
      const long p = m0 + arow;
      const bool inb = p < M;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int cc = apc * 8 + e;
        const int c = c0 + cc;
        float v = 0.f;
        if (inb && c < C) {
          v = f8tof(x[p * C + c]) * lscale[cc] + lshift[cc];
          if (relu_on) v = fmaxf(v, 0.f);
        }
        lA[arow][cc] = ftof8(v);
      }
    }
    for (int flat = tid; flat < NFRAG * 16 * PIECES_PER_ROW;
         flat += blockDim.x) {
      const int k = flat / PIECES_PER_ROW;
      const int bpc = flat % PIECES_PER_ROW;
      if (k < 96) {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const int c = c0 + bpc * 8 + e;
          lB[k][bpc * 8 + e] = (k < KOUT && c < C)
              ? w[(long)k * C + c] : ftof8(0.f);
        }
      }
    }
    __syncthreads();
    // one K=128 scaled MFMA replaces four K=32 steps; lane holds 32
    // bytes: A[i=lane&15][k=(lane>>4)*32 .. +32]
#pragma unroll
    for (int mf = 0; mf < 2; ++mf) {
      const int arow = wv * 32 + mf * 16 + (lane & 15);
      const i32x8mx afrag = *(const i32x8mx*)&lA[arow][(lane >> 4) * 32];
#pragma unroll
      for (int nf = 0; nf < 6; ++nf) {
        if (nf >= NFRAG) break;
        const i32x8mx bfrag =
            *(const i32x8mx*)&lB[nf * 16 + (lane & 15)][(lane >> 4) * 32];
        acc[mf][nf] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            afrag, bfrag, acc[mf][nf], 0, 0,
            0, 0x7F7F7F7F, 0, 0x7F7F7F7F);   // fp8 fmt, unit E8M0 scales
      }
    }
    __syncthreads();
  }
#pragma unroll
  for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
    for (int nf = 0; nf < 6; ++nf) {
      if (nf >= NFRAG) break;
      const int k = nf * 16 + (lane & 15);
      if (k >= KOUT) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int mrow = wv * 32 + mf * 16 + (lane >> 4) * 4 + reg;
        const long p = m0 + mrow;
        if (p < M) out[p * KOUT + k] = f2bf(acc[mf][nf][reg] * out_scale);
      }
    }
  }
}
