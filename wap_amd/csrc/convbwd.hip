// Backward + batch-stats kernels for the fused BN-ReLU-conv bottleneck path
// (gfx950 / CDNA4, NHWC bf16, f32 accumulation).
//
// Forward (conv_mfma.hip) computes out = conv(relu(bn(x)), w) with the BN
// affine fused into the input load.  Round-1 left the op default-OFF
// because the autograd wrapper paid for torch var_mean batch stats, an f32
// recompute of relu(bn(x)) in backward and aten.convolution_backward
// (PERF.md "End-to-end ... 254 vs 407").  These kernels remove all of
// that:
//
//   ch_stats       per-channel sum/sumsq over B*H*W (one reduction kernel)
//   bn_finalize    scale/shift/mean/invstd from the sums + running-stat
//                  update, one tiny [C] kernel (runs inside captured
//                  graphs, so replayed steps update running stats too)
//   bnconv_dgrad   dxn = conv_transpose(g_out, w) as an MFMA implicit
//                  GEMM, fused with the relu mask (recomputed from x) and
//                  the BN reductions dbeta = sum g, dgamma = sum g*xhat
//   relu_mask_reduce  the same mask+reductions standalone (for convs
//                  whose data-grad is a plain GEMM / MIOpen)
//   bnrelu_apply   materialize xn = relu(bn(x)) once for the backward-side
//                  wrw GEMM (rocBLAS 1x1 / MIOpen 3x3 — measured faster
//                  than a hand LDS-transposed wrw kernel, see PERF.md)
//   bn_bwd_dx      the final elementwise train-mode BN backward
//
// Weight-layout conventions match conv_mfma.hip: w physical [K][R][S][C]
// (channels_last), x/out physical [M=B*H*W][C].  The dgrad kernel reads a
// transposed, padded weight wT [R*S][Cp][Kp] prepared by the wrapper.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <vector>

#define WAVE 64

typedef __hip_bfloat16 cb_bf16;
typedef __attribute__((ext_vector_type(8))) short cb_bf16x8;
typedef __attribute__((ext_vector_type(4))) float cb_f32x4;

static __device__ __forceinline__ float cb_b2f(cb_bf16 v) {
  return __bfloat162float(v);
}
static __device__ __forceinline__ cb_bf16 cb_f2b(float v) {
  return __float2bfloat16(v);
}

// ------------------------------------------------------- channel stats
// x [M, C] bf16 (NHWC flat).  Grid (pixel-chunks, 8-channel groups):
// every thread of a block accumulates the SAME 8 channels over a pixel
// stripe (16-byte vector loads), then a block-wide shuffle/LDS reduction
// leaves ONE atomicAdd per channel per block (<=64 contending writers per
// address — a first version with per-thread atomics serialized ~20k adds
// per address on small-C layers and ran 10x slower than the BN it
// replaced).
__global__ void ch_stats_kernel(
    const cb_bf16* __restrict__ x, float* __restrict__ sums,
    float* __restrict__ sumsq, long M, int C) {
  __shared__ float red[4][16];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wv = tid >> 6;
  const int c0 = blockIdx.y * 8;
  // 16-byte vector loads need C %% 8 == 0 (row base p*C stays aligned)
  const bool full = (c0 + 8 <= C) && (C % 8 == 0);
  const long step = (long)gridDim.x * blockDim.x;
  float s[8] = {0.f}, q[8] = {0.f};
  for (long p = (long)blockIdx.x * blockDim.x + tid; p < M; p += step) {
    const cb_bf16* row = x + p * C + c0;
    if (full) {
      const cb_bf16x8 v8 = *(const cb_bf16x8*)row;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const short sv = v8[e];
        const float v = cb_b2f(*(const cb_bf16*)&sv);
        s[e] += v;
        q[e] += v * v;
      }
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        if (c0 + e < C) {
          const float v = cb_b2f(row[e]);
          s[e] += v;
          q[e] += v * v;
        }
      }
    }
  }
  // wave reduce each of the 16 accumulators, then across the 4 waves
#pragma unroll
  for (int e = 0; e < 8; ++e) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      s[e] += __shfl_down(s[e], off, WAVE);
      q[e] += __shfl_down(q[e], off, WAVE);
    }
  }
  if (lane == 0) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      red[wv][e] = s[e];
      red[wv][8 + e] = q[e];
    }
  }
  __syncthreads();
  if (tid < 16) {
    const float v = red[0][tid] + red[1][tid] + red[2][tid] + red[3][tid];
    const int e = tid & 7;
    if (c0 + e < C) {
      if (tid < 8) atomicAdd(sums + c0 + e, v);
      else atomicAdd(sumsq + c0 + e, v);
    }
  }
}

// ------------------------------------------------- xn materialization
// xn = relu(x*scale + shift), bf16 NHWC — backward-side input for the
// library wrw GEMM (the forward never materializes it; backward reads x
// a couple of times anyway, so one cheap write is the fast wrw route —
// the hand-written LDS-transposed wrw kernel measured 1.4 ms/call and
// was replaced by this + rocBLAS/MIOpen, see PERF.md).
__global__ void bnrelu_apply_kernel(
    const cb_bf16* __restrict__ x, const float* __restrict__ scale,
    const float* __restrict__ shift, cb_bf16* __restrict__ xn,
    long total, int C, int vec8) {
  if (vec8) {
    // C %% 8 == 0: one 16-byte load/store per thread-iteration, fully
    // coalesced; one modulo per 8 elements
    const long n8 = total >> 3;
    const int nc8 = C >> 3;
    for (long i8 = (long)blockIdx.x * blockDim.x + threadIdx.x; i8 < n8;
         i8 += (long)gridDim.x * blockDim.x) {
      const int c0 = (int)(i8 % nc8) * 8;
      cb_bf16 v8[8];
      *(cb_bf16x8*)v8 = *(const cb_bf16x8*)(x + i8 * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        v8[e] = cb_f2b(fmaxf(cb_b2f(v8[e]) * scale[c0 + e] + shift[c0 + e],
                             0.f));
      *(cb_bf16x8*)(xn + i8 * 8) = *(cb_bf16x8*)v8;
    }
  } else {
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
         i += (long)gridDim.x * blockDim.x) {
      const int c = (int)(i % C);
      xn[i] = cb_f2b(fmaxf(cb_b2f(x[i]) * scale[c] + shift[c], 0.f));
    }
  }
}

// ------------------------------------------- relu/BN mask + reductions
// g = relu'(bn(x)) * dxn, dbeta[c] += g, dgamma[c] += g*xhat — the
// epilogue of bnconv_dgrad as a standalone kernel, for convs whose
// data-grad is a plain GEMM (transition 1x1, K > 96).
__global__ void relu_mask_reduce_kernel(
    const cb_bf16* __restrict__ dxn, const cb_bf16* __restrict__ x,
    const float* __restrict__ scale, const float* __restrict__ shift,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    cb_bf16* __restrict__ g, float* __restrict__ dgamma,
    float* __restrict__ dbeta, long M, int C) {
  __shared__ float red[4][16];
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wv = tid >> 6;
  const int c0 = blockIdx.y * 8;
  const bool full = (c0 + 8 <= C) && (C % 8 == 0);
  float sc[8], sh[8], mn[8], is[8], db[8], dg[8];
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    const int c = c0 + e;
    const bool cv = c < C;
    sc[e] = cv ? scale[c] : 0.f;
    sh[e] = cv ? shift[c] : 0.f;
    mn[e] = cv ? mean[c] : 0.f;
    is[e] = cv ? invstd[c] : 0.f;
    db[e] = dg[e] = 0.f;
  }
  const long step = (long)gridDim.x * blockDim.x;
  for (long p = (long)blockIdx.x * blockDim.x + tid; p < M; p += step) {
    const cb_bf16* drow = dxn + p * C + c0;
    const cb_bf16* xrow = x + p * C + c0;
    cb_bf16* grow = g + p * C + c0;
    cb_bf16 dv8[8], xv8[8], gv8[8];
    if (full) {
      *(cb_bf16x8*)dv8 = *(const cb_bf16x8*)drow;
      *(cb_bf16x8*)xv8 = *(const cb_bf16x8*)xrow;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float xv = cb_b2f(xv8[e]);
        const float gg = (xv * sc[e] + sh[e]) > 0.f ? cb_b2f(dv8[e]) : 0.f;
        gv8[e] = cb_f2b(gg);
        db[e] += gg;
        dg[e] += gg * (xv - mn[e]) * is[e];
      }
      *(cb_bf16x8*)grow = *(cb_bf16x8*)gv8;
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        if (c0 + e < C) {
          const float xv = cb_b2f(xrow[e]);
          const float gg = (xv * sc[e] + sh[e]) > 0.f ? cb_b2f(drow[e]) : 0.f;
          grow[e] = cb_f2b(gg);
          db[e] += gg;
          dg[e] += gg * (xv - mn[e]) * is[e];
        }
      }
    }
  }
#pragma unroll
  for (int e = 0; e < 8; ++e) {
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      db[e] += __shfl_down(db[e], off, WAVE);
      dg[e] += __shfl_down(dg[e], off, WAVE);
    }
  }
  if (lane == 0) {
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      red[wv][e] = dg[e];
      red[wv][8 + e] = db[e];
    }
  }
  __syncthreads();
  if (tid < 16) {
    const float v = red[0][tid] + red[1][tid] + red[2][tid] + red[3][tid];
    const int e = tid & 7;
    if (c0 + e < C) {
      if (tid < 8) atomicAdd(dgamma + c0 + e, v);
      else atomicAdd(dbeta + c0 + e, v);
    }
  }
}

// ------------------------------------------------------- bn finalize
// One thread per channel: batch stats -> (scale, shift, mean, invstd),
// running-stat update in place (training mode).
__global__ void bn_finalize_kernel(
    const float* __restrict__ sums, const float* __restrict__ sumsq,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ running_mean, float* __restrict__ running_var,
    float* __restrict__ scale, float* __restrict__ shift,
    float* __restrict__ mean_out, float* __restrict__ invstd_out,
    float n, float eps, float momentum, int training, int C) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float mean, var;
  if (training) {
    mean = sums[c] / n;
    var = fmaxf(sumsq[c] / n - mean * mean, 0.f);
    if (momentum > 0.f) {
      const float unb = var * n / fmaxf(n - 1.f, 1.f);
      running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mean;
      running_var[c] = (1.f - momentum) * running_var[c] + momentum * unb;
    }
  } else {
    mean = running_mean[c];
    var = running_var[c];
  }
  const float inv = rsqrtf(var + eps);
  const float sc = gamma[c] * inv;
  scale[c] = sc;
  shift[c] = beta[c] - mean * sc;
  mean_out[c] = mean;
  invstd_out[c] = inv;
}

// ------------------------------------------------------------- dgrad
// g[p,c] = relu'(bn(x)[p,c]) * sum_{k,r,s} g_out[p - D(r,s), k] * w[k,r,s,c]
// plus dbeta[c] += g, dgamma[c] += g * xhat.  One (128-pixel x 64-channel)
// tile per block; K padded to Kp (mult 32), C padded to Cp (mult 64) in wT.
#define DG_BM 128
#define DG_BN 64

__global__ __launch_bounds__(256, 2)
void bnconv_dgrad_kernel(
    const cb_bf16* __restrict__ gout, const cb_bf16* __restrict__ wT,
    const cb_bf16* __restrict__ x, const float* __restrict__ scale,
    const float* __restrict__ shift, const float* __restrict__ mean,
    const float* __restrict__ invstd, cb_bf16* __restrict__ g,
    float* __restrict__ dgamma, float* __restrict__ dbeta,
    int Bn, int H, int W, int C, int K, int Kp, int R, int S, int cchunk) {
  // Whole reduction axis (all taps x all K-chunks, R*S*Kp <= 288) staged
  // ONCE, then the block loops over `cchunk` channel tiles reusing the
  // staged g_out.  cchunk is picked on the host so the grid stays >= 512
  // blocks: a pure (M x C) grid re-read g_out C/64 times (138 MB on the
  // big 1x1 layers), a pure M grid starved the chip on the small-spatial
  // blocks (32 blocks).  Each thread owns a fixed pixel row of the tile,
  // so the p -> (b,y,x) division happens once.
  __shared__ cb_bf16 lG[DG_BM][296];       // up to 9 taps x 32 K, +8 pad
  const int HW = H * W;
  const long M = (long)Bn * HW;
  const long m0 = (long)blockIdx.x * DG_BM;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wv = tid >> 6;
  const int PR = R / 2, PS = S / 2;
  const int Cp = ((C + DG_BN - 1) / DG_BN) * DG_BN;
  const int TK = R * S * Kp;               // staged row length (<= 288)

  {
    // threads t and t+128 share pixel row t%128, staging half its pieces
    const int arow = tid & (DG_BM - 1);
    const int half = tid >> 7;             // 0 or 1
    const long p = m0 + arow;
    int b = 0, y0 = 0, x0 = 0;
    const bool pv = p < M;
    if (pv) {
      b = (int)(p / HW);
      const int rem = (int)(p % HW);
      y0 = rem / W;
      x0 = rem % W;
    }
    const int npieces = TK / 8;
    for (int pc = half; pc < npieces; pc += 2) {
      const int koff = pc * 8;             // offset into the staged row
      const int tap = koff / Kp;
      const int kk = koff % Kp;            // multiple of 8 within Kp
      cb_bf16 vals[8];
      bool done = false;
      if (pv) {
        const int y = y0 - (tap / S - PR);
        const int xw = x0 - (tap % S - PS);
        if (y >= 0 && y < H && xw >= 0 && xw < W) {
          const long base = (((long)b * H + y) * W + xw) * K + kk;
          if (kk + 8 <= K) {
            *(cb_bf16x8*)vals = *(const cb_bf16x8*)(gout + base);
            done = true;
          } else {
#pragma unroll
            for (int e = 0; e < 8; ++e)
              vals[e] = kk + e < K ? gout[base + e] : cb_f2b(0.f);
            done = true;
          }
        }
      }
      if (!done) {
#pragma unroll
        for (int e = 0; e < 8; ++e) vals[e] = cb_f2b(0.f);
      }
      *(cb_bf16x8*)&lG[arow][koff] = *(cb_bf16x8*)vals;
    }
  }
  __syncthreads();

  const int cstart = blockIdx.y * cchunk * DG_BN;
  const int cend = min(C, cstart + cchunk * DG_BN);
  for (int c0 = cstart; c0 < cend; c0 += DG_BN) {
    cb_f32x4 acc[2][4];
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};
    for (int tap = 0; tap < R * S; ++tap) {
      const long wtap = ((long)tap * Cp) * Kp;
      for (int kc = 0; kc < Kp; kc += 32) {
#pragma unroll
        for (int mf = 0; mf < 2; ++mf) {
          const cb_bf16x8 afrag = *(const cb_bf16x8*)
              &lG[wv * 32 + mf * 16 + (lane & 15)][tap * Kp + kc +
                                                   (lane >> 4) * 8];
#pragma unroll
          for (int nf = 0; nf < 4; ++nf) {
            const cb_bf16x8 bfrag = *(const cb_bf16x8*)
                &wT[wtap + (long)(c0 + nf * 16 + (lane & 15)) * Kp + kc +
                    (lane >> 4) * 8];
            acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afrag, bfrag, acc[mf][nf], 0, 0, 0);
          }
        }
      }
    }
    // epilogue for this channel tile: relu mask from x, store g, reduce
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int c = c0 + nf * 16 + (lane & 15);
      const bool cv = c < C;
      const float sc = cv ? scale[c] : 0.f;
      const float sh = cv ? shift[c] : 0.f;
      const float mn = cv ? mean[c] : 0.f;
      const float is = cv ? invstd[c] : 0.f;
      float db = 0.f, dg = 0.f;
#pragma unroll
      for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) {
          const long p = m0 + wv * 32 + mf * 16 + (lane >> 4) * 4 + reg;
          if (cv && p < M) {
            const float xv = cb_b2f(x[p * C + c]);
            const float gg =
                (xv * sc + sh) > 0.f ? acc[mf][nf][reg] : 0.f;
            g[p * C + c] = cb_f2b(gg);
            db += gg;
            dg += gg * (xv - mn) * is;
          }
        }
      }
      db += __shfl_xor(db, 16, WAVE);
      db += __shfl_xor(db, 32, WAVE);
      dg += __shfl_xor(dg, 16, WAVE);
      dg += __shfl_xor(dg, 32, WAVE);
      if ((lane >> 4) == 0 && cv) {
        atomicAdd(dbeta + c, db);
        atomicAdd(dgamma + c, dg);
      }
    }
  }
}

// ---------------------------------------------------------- bn_bwd_dx
// dx = gamma*invstd * (g - dbeta/n - xhat * dgamma/n)   (training)
// dx = gamma*invstd * g                                 (eval)
__global__ void bn_bwd_dx_kernel(
    const cb_bf16* __restrict__ g, const cb_bf16* __restrict__ x,
    const float* __restrict__ gamma, const float* __restrict__ invstd,
    const float* __restrict__ mean, const float* __restrict__ dgamma,
    const float* __restrict__ dbeta, float inv_n, cb_bf16* __restrict__ dx,
    long total, int C, int training, int vec8) {
  if (vec8) {
    const long n8 = total >> 3;
    const int nc8 = C >> 3;
    for (long i8 = (long)blockIdx.x * blockDim.x + threadIdx.x; i8 < n8;
         i8 += (long)gridDim.x * blockDim.x) {
      const int c0 = (int)(i8 % nc8) * 8;
      cb_bf16 gv8[8], xv8[8], ov8[8];
      *(cb_bf16x8*)gv8 = *(const cb_bf16x8*)(g + i8 * 8);
      *(cb_bf16x8*)xv8 = *(const cb_bf16x8*)(x + i8 * 8);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int c = c0 + e;
        const float is = invstd[c];
        const float gs = gamma[c] * is;
        float v = cb_b2f(gv8[e]);
        if (training) {
          const float xhat = (cb_b2f(xv8[e]) - mean[c]) * is;
          v = v - dbeta[c] * inv_n - xhat * dgamma[c] * inv_n;
        }
        ov8[e] = cb_f2b(gs * v);
      }
      *(cb_bf16x8*)(dx + i8 * 8) = *(cb_bf16x8*)ov8;
    }
  } else {
    for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
         i += (long)gridDim.x * blockDim.x) {
      const int c = (int)(i % C);
      const float is = invstd[c];
      const float gs = gamma[c] * is;
      float v = cb_b2f(g[i]);
      if (training) {
        const float xhat = (cb_b2f(x[i]) - mean[c]) * is;
        v = v - dbeta[c] * inv_n - xhat * dgamma[c] * inv_n;
      }
      dx[i] = cb_f2b(gs * v);
    }
  }
}

// ================================================================ host
namespace {
inline hipStream_t cb_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}
__global__ void cb_zero_kernel(float* __restrict__ p, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = 0.f;
}
void cb_zero(torch::Tensor& t) {
  const long n = t.numel();
  hipLaunchKernelGGL(cb_zero_kernel, dim3((n + 255) / 256), dim3(256), 0,
                     cb_stream(), (float*)t.data_ptr(), n);
}
#define CB_OK(what) do { \
  hipError_t err = hipGetLastError(); \
  TORCH_CHECK(err == hipSuccess, what ": ", hipGetErrorString(err)); \
} while (0)
}  // namespace

std::vector<torch::Tensor> ch_stats(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16);
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast),
              "ch_stats: x must be channels_last");
  const int C = x.size(1);
  TORCH_CHECK(C <= 768, "ch_stats: C <= 768");
  const long M = x.numel() / C;
  auto buf = torch::empty({2, C}, x.options().dtype(at::kFloat));
  cb_zero(buf);  // one zero launch for both accumulators
  auto sums = buf.select(0, 0);
  auto sumsq = buf.select(0, 1);
  const int pch = (int)std::min((M + 1023) / 1024, (long)64);
  const dim3 grid(pch, (C + 7) / 8);
  hipLaunchKernelGGL(ch_stats_kernel, grid, dim3(256), 0,
                     cb_stream(), (const cb_bf16*)x.data_ptr(),
                     (float*)sums.data_ptr(), (float*)sumsq.data_ptr(), M, C);
  CB_OK("ch_stats");
  return {sums, sumsq};
}

std::vector<torch::Tensor> bn_finalize(
    torch::Tensor sums, torch::Tensor sumsq, torch::Tensor gamma,
    torch::Tensor beta, torch::Tensor running_mean, torch::Tensor running_var,
    double n, double eps, double momentum, bool training) {
  const int C = gamma.size(0);
  auto opts = gamma.options().dtype(at::kFloat);
  auto scale = torch::empty({C}, opts);
  auto shift = torch::empty({C}, opts);
  auto mean = torch::empty({C}, opts);
  auto invstd = torch::empty({C}, opts);
  TORCH_CHECK(gamma.scalar_type() == at::kFloat &&
              beta.scalar_type() == at::kFloat &&
              running_mean.scalar_type() == at::kFloat &&
              running_var.scalar_type() == at::kFloat,
              "bn_finalize: f32 BN params expected");
  hipLaunchKernelGGL(bn_finalize_kernel, dim3((C + 255) / 256), dim3(256), 0,
                     cb_stream(),
                     (const float*)sums.data_ptr(),
                     (const float*)sumsq.data_ptr(),
                     (const float*)gamma.data_ptr(),
                     (const float*)beta.data_ptr(),
                     (float*)running_mean.data_ptr(),
                     (float*)running_var.data_ptr(),
                     (float*)scale.data_ptr(), (float*)shift.data_ptr(),
                     (float*)mean.data_ptr(), (float*)invstd.data_ptr(),
                     (float)n, (float)eps, (float)momentum,
                     training ? 1 : 0, C);
  CB_OK("bn_finalize");
  return {scale, shift, mean, invstd};
}

std::vector<torch::Tensor> bnconv_dgrad(
    torch::Tensor gout, torch::Tensor wT, torch::Tensor x,
    torch::Tensor scale, torch::Tensor shift, torch::Tensor mean,
    torch::Tensor invstd, int64_t R, int64_t S) {
  TORCH_CHECK(gout.is_cuda() && gout.scalar_type() == at::kBFloat16);
  TORCH_CHECK(gout.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(wT.is_contiguous() && wT.scalar_type() == at::kBFloat16);
  const int Bn = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int K = gout.size(1);
  const int Kp = wT.size(2);
  const int Cp = wT.size(1);
  TORCH_CHECK(wT.size(0) == R * S && Kp % 32 == 0 && Kp >= K);
  TORCH_CHECK(Cp % DG_BN == 0 && Cp >= C);
  TORCH_CHECK(R * S * Kp <= 288, "bnconv_dgrad: R*S*Kp <= 288");
  auto g = at::empty_like(x);
  auto gb = torch::empty({2, C}, scale.options());
  cb_zero(gb);  // one zero launch for both reductions
  auto dgamma = gb.select(0, 0);
  auto dbeta = gb.select(0, 1);
  const long M = (long)Bn * H * W;
  const long mtiles = (M + DG_BM - 1) / DG_BM;
  const int ctiles = (C + DG_BN - 1) / DG_BN;
  int cchunk = (int)std::max((long)1, mtiles * ctiles / 512);
  cchunk = std::min(cchunk, ctiles);
  const dim3 grid(mtiles, (ctiles + cchunk - 1) / cchunk);
  hipLaunchKernelGGL(bnconv_dgrad_kernel, grid, dim3(256), 0, cb_stream(),
                     (const cb_bf16*)gout.data_ptr(),
                     (const cb_bf16*)wT.data_ptr(),
                     (const cb_bf16*)x.data_ptr(),
                     (const float*)scale.data_ptr(),
                     (const float*)shift.data_ptr(),
                     (const float*)mean.data_ptr(),
                     (const float*)invstd.data_ptr(),
                     (cb_bf16*)g.data_ptr(), (float*)dgamma.data_ptr(),
                     (float*)dbeta.data_ptr(),
                     Bn, H, W, C, K, Kp, (int)R, (int)S, cchunk);
  CB_OK("bnconv_dgrad");
  return {g, dgamma, dbeta};
}

torch::Tensor bn_bwd_dx(
    torch::Tensor g, torch::Tensor x, torch::Tensor gamma,
    torch::Tensor invstd, torch::Tensor mean, torch::Tensor dgamma,
    torch::Tensor dbeta, double n, bool training) {
  TORCH_CHECK(g.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int C = x.size(1);
  auto dx = at::empty_like(x);
  const long total = x.numel();
  const long work = (C % 8 == 0) ? total / 8 : total;
  const int blocks = (int)std::min((work + 255) / 256, (long)4096);
  hipLaunchKernelGGL(bn_bwd_dx_kernel, dim3(blocks), dim3(256), 0,
                     cb_stream(),
                     (const cb_bf16*)g.data_ptr(),
                     (const cb_bf16*)x.data_ptr(),
                     (const float*)gamma.data_ptr(),
                     (const float*)invstd.data_ptr(),
                     (const float*)mean.data_ptr(),
                     (const float*)dgamma.data_ptr(),
                     (const float*)dbeta.data_ptr(),
                     (float)(1.0 / n), (cb_bf16*)dx.data_ptr(),
                     total, C, training ? 1 : 0, C % 8 == 0 ? 1 : 0);
  CB_OK("bn_bwd_dx");
  return dx;
}

torch::Tensor bnrelu_apply(
    torch::Tensor x, torch::Tensor scale, torch::Tensor shift) {
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(x.scalar_type() == at::kBFloat16);
  const int C = x.size(1);
  auto xn = at::empty_like(x);
  const long total = x.numel();
  const long work = (C % 8 == 0) ? total / 8 : total;
  const int blocks = (int)std::min((work + 255) / 256, (long)4096);
  hipLaunchKernelGGL(bnrelu_apply_kernel, dim3(blocks), dim3(256), 0,
                     cb_stream(), (const cb_bf16*)x.data_ptr(),
                     (const float*)scale.data_ptr(),
                     (const float*)shift.data_ptr(),
                     (cb_bf16*)xn.data_ptr(), total, C,
                     C % 8 == 0 ? 1 : 0);
  CB_OK("bnrelu_apply");
  return xn;
}

std::vector<torch::Tensor> relu_mask_reduce(
    torch::Tensor dxn, torch::Tensor x, torch::Tensor scale,
    torch::Tensor shift, torch::Tensor mean, torch::Tensor invstd) {
  TORCH_CHECK(dxn.is_contiguous(at::MemoryFormat::ChannelsLast));
  TORCH_CHECK(x.is_contiguous(at::MemoryFormat::ChannelsLast));
  const int C = x.size(1);
  auto g = at::empty_like(x);
  auto gb = torch::empty({2, (long)C}, scale.options());
  cb_zero(gb);
  auto dgamma = gb.select(0, 0);
  auto dbeta = gb.select(0, 1);
  const long M = x.numel() / C;
  const int pch = (int)std::min((M + 255) / 256, (long)128);
  hipLaunchKernelGGL(relu_mask_reduce_kernel,
                     dim3(pch, (C + 7) / 8), dim3(256), 0, cb_stream(),
                     (const cb_bf16*)dxn.data_ptr(),
                     (const cb_bf16*)x.data_ptr(),
                     (const float*)scale.data_ptr(),
                     (const float*)shift.data_ptr(),
                     (const float*)mean.data_ptr(),
                     (const float*)invstd.data_ptr(),
                     (cb_bf16*)g.data_ptr(), (float*)dgamma.data_ptr(),
                     (float*)dbeta.data_ptr(), M, C);
  CB_OK("relu_mask_reduce");
  return {g, dgamma, dbeta};
}

// ===================================================================
// NCHW variants of the BN elementwise/stat kernels — used on the
// default (NCHW, MIOpen-conv) path to replace torch's BatchNorm (5
// MIOpen kernels + f32 promote/demote casts under autocast) and ReLU
// with three cast-free bf16 kernels.  NCHW makes these EASIER than the
// cl forms: one (b,c) plane is contiguous, so grid.y indexes the plane
// and the channel is one modulo per block.
// ===================================================================

__global__ void ch_stats_nchw_kernel(
    const cb_bf16* __restrict__ x, float* __restrict__ sums,
    float* __restrict__ sumsq, int B, int C, long HW) {
  __shared__ float red[4][2];
  const int c = blockIdx.y;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wv = tid >> 6;
  float s = 0.f, q = 0.f;
  const long step8 = (long)gridDim.x * blockDim.x * 8;
  for (int b = 0; b < B; ++b) {
    const cb_bf16* plane = x + ((long)b * C + c) * HW;
    for (long i = ((long)blockIdx.x * blockDim.x + tid) * 8; i < HW;
         i += step8) {
      if (i + 8 <= HW) {
        cb_bf16 v8[8];
        *(cb_bf16x8*)v8 = *(const cb_bf16x8*)(plane + i);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const float v = cb_b2f(v8[e]);
          s += v;
          q += v * v;
        }
      } else {
        for (long j = i; j < HW; ++j) {
          const float v = cb_b2f(plane[j]);
          s += v;
          q += v * v;
        }
      }
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    s += __shfl_down(s, off, WAVE);
    q += __shfl_down(q, off, WAVE);
  }
  if (lane == 0) {
    red[wv][0] = s;
    red[wv][1] = q;
  }
  __syncthreads();
  if (tid == 0) {
    atomicAdd(sums + c, red[0][0] + red[1][0] + red[2][0] + red[3][0]);
    atomicAdd(sumsq + c, red[0][1] + red[1][1] + red[2][1] + red[3][1]);
  }
}

__global__ void bnrelu_apply_nchw_kernel(
    const cb_bf16* __restrict__ x, const float* __restrict__ scale,
    const float* __restrict__ shift, cb_bf16* __restrict__ xn,
    int C, long HW) {
  const int bc = blockIdx.y;
  const int c = bc % C;
  const float sc = scale[c], sh = shift[c];
  const cb_bf16* plane = x + (long)bc * HW;
  cb_bf16* out = xn + (long)bc * HW;
  const long step8 = (long)gridDim.x * blockDim.x * 8;
  for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8; i < HW;
       i += step8) {
    if (i + 8 <= HW) {
      cb_bf16 v8[8];
      *(cb_bf16x8*)v8 = *(const cb_bf16x8*)(plane + i);
#pragma unroll
      for (int e = 0; e < 8; ++e)
        v8[e] = cb_f2b(fmaxf(cb_b2f(v8[e]) * sc + sh, 0.f));
      *(cb_bf16x8*)(out + i) = *(cb_bf16x8*)v8;
    } else {
      for (long j = i; j < HW; ++j)
        out[j] = cb_f2b(fmaxf(cb_b2f(plane[j]) * sc + sh, 0.f));
    }
  }
}

__global__ void relu_mask_reduce_nchw_kernel(
    const cb_bf16* __restrict__ dxn, const cb_bf16* __restrict__ x,
    const float* __restrict__ scale, const float* __restrict__ shift,
    const float* __restrict__ mean, const float* __restrict__ invstd,
    cb_bf16* __restrict__ g, float* __restrict__ dgamma,
    float* __restrict__ dbeta, int C, long HW) {
  __shared__ float red[4][2];
  const int bc = blockIdx.y;
  const int c = bc % C;
  const float sc = scale[c], sh = shift[c], mn = mean[c], is = invstd[c];
  const cb_bf16* dplane = dxn + (long)bc * HW;
  const cb_bf16* xplane = x + (long)bc * HW;
  cb_bf16* gplane = g + (long)bc * HW;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wv = tid >> 6;
  float db = 0.f, dg = 0.f;
  const long step8 = (long)gridDim.x * blockDim.x * 8;
  for (long i = ((long)blockIdx.x * blockDim.x + tid) * 8; i < HW;
       i += step8) {
    const long hi = min(i + 8, HW);
    for (long j = i; j < hi; ++j) {
      const float xv = cb_b2f(xplane[j]);
      const float gg = (xv * sc + sh) > 0.f ? cb_b2f(dplane[j]) : 0.f;
      gplane[j] = cb_f2b(gg);
      db += gg;
      dg += gg * (xv - mn) * is;
    }
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    db += __shfl_down(db, off, WAVE);
    dg += __shfl_down(dg, off, WAVE);
  }
  if (lane == 0) {
    red[wv][0] = db;
    red[wv][1] = dg;
  }
  __syncthreads();
  if (tid == 0) {
    atomicAdd(dbeta + c, red[0][0] + red[1][0] + red[2][0] + red[3][0]);
    atomicAdd(dgamma + c, red[0][1] + red[1][1] + red[2][1] + red[3][1]);
  }
}

__global__ void bn_bwd_dx_nchw_kernel(
    const cb_bf16* __restrict__ g, const cb_bf16* __restrict__ x,
    const float* __restrict__ gamma, const float* __restrict__ invstd,
    const float* __restrict__ mean, const float* __restrict__ dgamma,
    const float* __restrict__ dbeta, float inv_n, cb_bf16* __restrict__ dx,
    int C, long HW, int training) {
  const int bc = blockIdx.y;
  const int c = bc % C;
  const float is = invstd[c];
  const float gs = gamma[c] * is;
  const float mn = mean[c];
  const float dbn = training ? dbeta[c] * inv_n : 0.f;
  const float dgn = training ? dgamma[c] * inv_n : 0.f;
  const cb_bf16* gplane = g + (long)bc * HW;
  const cb_bf16* xplane = x + (long)bc * HW;
  cb_bf16* oplane = dx + (long)bc * HW;
  const long step8 = (long)gridDim.x * blockDim.x * 8;
  for (long i = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8; i < HW;
       i += step8) {
    if (i + 8 <= HW) {
      cb_bf16 g8[8], x8[8], o8[8];
      *(cb_bf16x8*)g8 = *(const cb_bf16x8*)(gplane + i);
      *(cb_bf16x8*)x8 = *(const cb_bf16x8*)(xplane + i);
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float xhat = (cb_b2f(x8[e]) - mn) * is;
        o8[e] = cb_f2b(gs * (cb_b2f(g8[e]) - dbn - xhat * dgn));
      }
      *(cb_bf16x8*)(oplane + i) = *(cb_bf16x8*)o8;
    } else {
      for (long j = i; j < HW; ++j) {
        const float xhat = (cb_b2f(xplane[j]) - mn) * is;
        oplane[j] = cb_f2b(gs * (cb_b2f(gplane[j]) - dbn - xhat * dgn));
      }
    }
  }
}

// ------------------------------------------------------------ hosts
std::vector<torch::Tensor> ch_stats_nchw(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == at::kBFloat16);
  const int B = x.size(0), C = x.size(1);
  const long HW = (long)x.size(2) * x.size(3);
  auto buf = torch::empty({2, C}, x.options().dtype(at::kFloat));
  cb_zero(buf);
  auto sums = buf.select(0, 0);
  auto sumsq = buf.select(0, 1);
  const int pch = (int)std::min((HW + 2047) / 2048, (long)32);
  hipLaunchKernelGGL(ch_stats_nchw_kernel, dim3(pch, C), dim3(256), 0,
                     cb_stream(), (const cb_bf16*)x.data_ptr(),
                     (float*)sums.data_ptr(), (float*)sumsq.data_ptr(),
                     B, C, HW);
  CB_OK("ch_stats_nchw");
  return {sums, sumsq};
}

torch::Tensor bnrelu_apply_nchw(
    torch::Tensor x, torch::Tensor scale, torch::Tensor shift) {
  TORCH_CHECK(x.is_contiguous() && x.scalar_type() == at::kBFloat16);
  const int B = x.size(0), C = x.size(1);
  const long HW = (long)x.size(2) * x.size(3);
  auto xn = at::empty_like(x);
  const int pch = (int)std::min((HW + 2047) / 2048, (long)16);
  hipLaunchKernelGGL(bnrelu_apply_nchw_kernel, dim3(pch, B * C), dim3(256),
                     0, cb_stream(), (const cb_bf16*)x.data_ptr(),
                     (const float*)scale.data_ptr(),
                     (const float*)shift.data_ptr(),
                     (cb_bf16*)xn.data_ptr(), C, HW);
  CB_OK("bnrelu_apply_nchw");
  return xn;
}

std::vector<torch::Tensor> relu_mask_reduce_nchw(
    torch::Tensor dxn, torch::Tensor x, torch::Tensor scale,
    torch::Tensor shift, torch::Tensor mean, torch::Tensor invstd) {
  TORCH_CHECK(dxn.is_contiguous() && x.is_contiguous());
  const int B = x.size(0), C = x.size(1);
  const long HW = (long)x.size(2) * x.size(3);
  auto g = at::empty_like(x);
  auto gb = torch::empty({2, (long)C}, scale.options());
  cb_zero(gb);
  auto dgamma = gb.select(0, 0);
  auto dbeta = gb.select(0, 1);
  const int pch = (int)std::min((HW + 2047) / 2048, (long)16);
  hipLaunchKernelGGL(relu_mask_reduce_nchw_kernel, dim3(pch, B * C),
                     dim3(256), 0, cb_stream(),
                     (const cb_bf16*)dxn.data_ptr(),
                     (const cb_bf16*)x.data_ptr(),
                     (const float*)scale.data_ptr(),
                     (const float*)shift.data_ptr(),
                     (const float*)mean.data_ptr(),
                     (const float*)invstd.data_ptr(),
                     (cb_bf16*)g.data_ptr(), (float*)dgamma.data_ptr(),
                     (float*)dbeta.data_ptr(), C, HW);
  CB_OK("relu_mask_reduce_nchw");
  return {g, dgamma, dbeta};
}

torch::Tensor bn_bwd_dx_nchw(
    torch::Tensor g, torch::Tensor x, torch::Tensor gamma,
    torch::Tensor invstd, torch::Tensor mean, torch::Tensor dgamma,
    torch::Tensor dbeta, double n, bool training) {
  TORCH_CHECK(g.is_contiguous() && x.is_contiguous());
  const int B = x.size(0), C = x.size(1);
  const long HW = (long)x.size(2) * x.size(3);
  auto dx = at::empty_like(x);
  const int pch = (int)std::min((HW + 2047) / 2048, (long)16);
  hipLaunchKernelGGL(bn_bwd_dx_nchw_kernel, dim3(pch, B * C), dim3(256), 0,
                     cb_stream(),
                     (const cb_bf16*)g.data_ptr(),
                     (const cb_bf16*)x.data_ptr(),
                     (const float*)gamma.data_ptr(),
                     (const float*)invstd.data_ptr(),
                     (const float*)mean.data_ptr(),
                     (const float*)dgamma.data_ptr(),
                     (const float*)dbeta.data_ptr(),
                     (float)(1.0 / n), (cb_bf16*)dx.data_ptr(),
                     C, HW, training ? 1 : 0);
  CB_OK("bn_bwd_dx_nchw");
  return dx;
}
