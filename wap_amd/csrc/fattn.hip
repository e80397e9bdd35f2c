// Fused coverage-attention score kernel (gfx950 / CDNA4).
//
// Per decoder step the coverage attention needs
//
//   uf[b,l,:] = bias + sum_{dy,dx} cov[b, y+dy-P, x+dx-P] * Wm[:, dy*KW+dx]
//   t[b,l,k]  = tanh(q[b,k] + ua[b,l,k] + uf[b,l,k])
//   e[b,l]    = sum_k nu[k] * t[b,l,k]
//
// where Wm is the 11x11 coverage conv with U_f already folded in
// (attention.merged_coverage_weight).  The unfused path runs this as
// im2col -> GEMM -> bias add -> attn_score = 4+ launches per timestep,
// forward AND backward, in a loop that is dispatch-bound (PERF.md).  Here
// the whole expression is ONE kernel: the conv is an MFMA implicit GEMM
// over an LDS-staged [32 x KP] im2col tile (KP = KH*KW padded to a
// multiple of 32), and the tanh/score epilogue consumes the accumulators
// in registers.
//
// The backward kernel fuses g = de*nu*(1-t^2) (= dua = duf), the per-image
// dq and global dnu reductions, and the conv data-grad
// dcov += g @ Wm^T scattered back through the conv footprint — one launch
// instead of four.  The weight grad dWm = g^T @ cols stays a host-side
// batched GEMM (it reduces over B*L, which a per-tile kernel would have to
// do with 64k atomics per block).
//
// Tiling: grid (ceil(L/32), B), 256 threads (4 waves).  BM = 32 spatial
// positions per block; wave w owns A/64 (fwd) resp. round-robin KP/16
// (bwd) 16-column MFMA fragments.  LDS rows are padded by 8 bf16 so the
// 16-byte fragment reads don't land on one bank.  f32 accumulation
// everywhere; bf16 at the MFMA operands, matching the autocast oracle.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <vector>

#define WAVE 64
#define FA_BM 32

typedef __hip_bfloat16 bf16_t;
typedef __attribute__((ext_vector_type(8))) short bf16x8f;
typedef __attribute__((ext_vector_type(4))) float f32x4f;

static __device__ __forceinline__ float fa_ldf(const float* p) { return *p; }
static __device__ __forceinline__ float fa_ldf(const bf16_t* p) {
  return __bfloat162float(*p);
}
static __device__ __forceinline__ void fa_stf(float* p, float v) { *p = v; }
static __device__ __forceinline__ void fa_stf(bf16_t* p, float v) {
  *p = __float2bfloat16(v);
}
// the score accumulates the ROUNDED t (what tcache holds), so forward,
// backward and the eager oracle all see the same tanh values
template <typename T> __device__ __forceinline__ float fa_round(float v);
template <> __device__ __forceinline__ float fa_round<float>(float v) {
  return v;
}
template <> __device__ __forceinline__ float fa_round<bf16_t>(float v) {
  return __bfloat162float(__float2bfloat16(v));
}

// --------------------------------------------------------------- forward
template <typename T>
__global__ void fattn_score_fwd_kernel(
    const float* __restrict__ cov, const bf16_t* __restrict__ wpad,
    const T* __restrict__ q, const T* __restrict__ ua,
    const float* __restrict__ nu, const float* __restrict__ bias,
    float* __restrict__ e, T* __restrict__ tcache,
    int L, int A, int Hp, int Wp, int KH, int KW, int KP) {
  extern __shared__ char smem_raw[];
  const int cstride = KP + 8;                     // bf16, bank-stagger pad
  bf16_t* cols = (bf16_t*)smem_raw;               // [FA_BM][cstride]
  float* e_s = (float*)(smem_raw + FA_BM * cstride * sizeof(bf16_t));

  const int b = blockIdx.y;
  const int m0 = blockIdx.x * FA_BM;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wv = tid >> 6;
  const int KK = KH * KW, PH = KH / 2, PW = KW / 2;

  if (tid < FA_BM) e_s[tid] = 0.f;

  // stage the im2col tile of the coverage map (f32 -> bf16 at the store,
  // like the autocast GEMM the oracle runs)
  for (int flat = tid; flat < FA_BM * (KP / 8); flat += blockDim.x) {
    const int arow = flat / (KP / 8);
    const int pc = flat % (KP / 8);
    const int l = m0 + arow;
    const int y = l / Wp, x = l % Wp;
#pragma unroll
    for (int ee = 0; ee < 8; ++ee) {
      const int kk = pc * 8 + ee;
      float v = 0.f;
      if (l < L && kk < KK) {
        const int yy = y + kk / KW - PH;
        const int xx = x + kk % KW - PW;
        if (yy >= 0 && yy < Hp && xx >= 0 && xx < Wp)
          v = cov[((long)b * Hp + yy) * Wp + xx];
      }
      cols[arow * cstride + kk] = __float2bfloat16(v);
    }
  }
  __syncthreads();

  const int nfper = A >> 6;                       // A/16 frags over 4 waves
  f32x4f acc[2][8];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  for (int kc = 0; kc < KP / 32; ++kc) {
#pragma unroll
    for (int mf = 0; mf < 2; ++mf) {
      const bf16x8f afrag = *(const bf16x8f*)
          &cols[(mf * 16 + (lane & 15)) * cstride + kc * 32 + (lane >> 4) * 8];
#pragma unroll
      for (int nf = 0; nf < 8; ++nf) {
        if (nf >= nfper) break;
        const int krow = (wv * nfper + nf) * 16 + (lane & 15);
        const bf16x8f bfrag = *(const bf16x8f*)
            &wpad[(long)krow * KP + kc * 32 + (lane >> 4) * 8];
        acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc[mf][nf], 0, 0, 0);
      }
    }
  }

  // epilogue: uf = acc + bias; t = tanh(q + ua + uf); e += nu * t
  float part[2][4];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int r = 0; r < 4; ++r) part[i][r] = 0.f;
#pragma unroll
  for (int nf = 0; nf < 8; ++nf) {
    if (nf >= nfper) break;
    const int k = (wv * nfper + nf) * 16 + (lane & 15);
    const float nuk = nu[k];
    const float qk = fa_ldf(q + (long)b * A + k);
    const float bk = bias[k];
#pragma unroll
    for (int mf = 0; mf < 2; ++mf) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int lrow = m0 + mf * 16 + (lane >> 4) * 4 + reg;
        if (lrow < L) {
          const long off = ((long)b * L + lrow) * A + k;
          const float t = tanhf(qk + fa_ldf(ua + off) + acc[mf][nf][reg] + bk);
          fa_stf(tcache + off, t);
          part[mf][reg] += nuk * fa_round<T>(t);
        }
      }
    }
  }
  // reduce over the 16 lanes of each row group, then across waves via LDS
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      float v = part[mf][reg];
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) v += __shfl_xor(v, off, WAVE);
      if ((lane & 15) == 0)
        atomicAdd(&e_s[mf * 16 + (lane >> 4) * 4 + reg], v);
    }
  __syncthreads();
  if (tid < FA_BM && m0 + tid < L) e[(long)b * L + m0 + tid] = e_s[tid];
}

// -------------------------------------------------------------- backward
template <typename T>
__global__ void fattn_score_bwd_kernel(
    const float* __restrict__ de, const T* __restrict__ tcache,
    const float* __restrict__ nu, const bf16_t* __restrict__ wTpad,
    T* __restrict__ dua, float* __restrict__ dq, float* __restrict__ dnu,
    float* __restrict__ dcov,
    int L, int A, int Hp, int Wp, int KH, int KW, int KP) {
  extern __shared__ char smem_raw[];
  const int gstride = A + 8;
  bf16_t* gtile = (bf16_t*)smem_raw;              // [FA_BM][gstride]

  const int b = blockIdx.y;
  const int m0 = blockIdx.x * FA_BM;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wv = tid >> 6;
  const int KK = KH * KW, PH = KH / 2, PW = KW / 2;

  // phase 1: g = de * nu * (1 - t^2)  -> dua (global), gtile (LDS),
  //          dq[b,k] += sum_l g, dnu[k] += sum_l t * de
  float dq_acc[2] = {0.f, 0.f}, dnu_acc[2] = {0.f, 0.f};
  for (int l = 0; l < FA_BM; ++l) {
    const int lrow = m0 + l;
    const float d = lrow < L ? de[(long)b * L + lrow] : 0.f;
    int j = 0;
    for (int k = tid; k < A; k += blockDim.x, ++j) {
      float g = 0.f;
      if (lrow < L) {
        const long off = ((long)b * L + lrow) * A + k;
        const float tv = fa_ldf(tcache + off);
        g = d * nu[k] * (1.f - tv * tv);
        fa_stf(dua + off, g);
        dq_acc[j] += g;
        dnu_acc[j] += tv * d;
      }
      gtile[l * gstride + k] = __float2bfloat16(g);
    }
  }
  {
    int j = 0;
    for (int k = tid; k < A; k += blockDim.x, ++j) {
      atomicAdd(dq + (long)b * A + k, dq_acc[j]);
      atomicAdd(dnu + k, dnu_acc[j]);
    }
  }
  __syncthreads();

  // phase 2: dcols[l,kk] = sum_a g[l,a] * Wm[a,kk]  (wTpad = Wm^T, [KP][A])
  const int nfrags = KP / 16;
  f32x4f acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};
  for (int kc = 0; kc < A / 32; ++kc) {
#pragma unroll
    for (int mf = 0; mf < 2; ++mf) {
      const bf16x8f afrag = *(const bf16x8f*)
          &gtile[(mf * 16 + (lane & 15)) * gstride + kc * 32 + (lane >> 4) * 8];
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const int nfg = wv + 4 * j;
        if (nfg >= nfrags) break;
        const bf16x8f bfrag = *(const bf16x8f*)
            &wTpad[(long)(nfg * 16 + (lane & 15)) * A + kc * 32 +
                   (lane >> 4) * 8];
        acc[mf][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afrag, bfrag, acc[mf][j], 0, 0, 0);
      }
    }
  }

  // phase 3: scatter dcols back through the conv footprint
#pragma unroll
  for (int mf = 0; mf < 2; ++mf)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int nfg = wv + 4 * j;
      if (nfg >= nfrags) break;
      const int kk = nfg * 16 + (lane & 15);
      if (kk >= KK) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int lrow = m0 + mf * 16 + (lane >> 4) * 4 + reg;
        if (lrow >= L) continue;
        const int yy = lrow / Wp + kk / KW - PH;
        const int xx = lrow % Wp + kk % KW - PW;
        if (yy >= 0 && yy < Hp && xx >= 0 && xx < Wp)
          atomicAdd(dcov + ((long)b * Hp + yy) * Wp + xx, acc[mf][j][reg]);
      }
    }
}

// ------------------------------------------------------------ host side
namespace {
inline hipStream_t fa_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}
__global__ void fa_zero_kernel(float* __restrict__ p, long n) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = 0.f;
}
void fa_zero(torch::Tensor& t) {
  const long n = t.numel();
  hipLaunchKernelGGL(fa_zero_kernel, dim3((n + 255) / 256), dim3(256), 0,
                     fa_stream(), (float*)t.data_ptr(), n);
}
}  // namespace

std::vector<torch::Tensor> fattn_score_fwd(
    torch::Tensor cov, torch::Tensor wpad, torch::Tensor q, torch::Tensor ua,
    torch::Tensor nu, torch::Tensor bias, int64_t KH, int64_t KW) {
  TORCH_CHECK(cov.is_cuda() && cov.is_contiguous() &&
              cov.scalar_type() == at::kFloat, "cov must be f32 contiguous");
  TORCH_CHECK(wpad.is_cuda() && wpad.is_contiguous() &&
              wpad.scalar_type() == at::kBFloat16, "wpad must be bf16");
  TORCH_CHECK(ua.is_contiguous() && q.is_contiguous());
  TORCH_CHECK(nu.scalar_type() == at::kFloat &&
              bias.scalar_type() == at::kFloat);
  const int B = ua.size(0), L = ua.size(1), A = ua.size(2);
  const int Hp = cov.size(1), Wp = cov.size(2);
  const int KP = wpad.size(1);
  TORCH_CHECK(cov.size(0) == B && Hp * Wp == L, "cov/ua shape mismatch");
  TORCH_CHECK(A % 64 == 0 && A <= 512, "fattn: A must be %64==0 and <=512");
  TORCH_CHECK(KP % 32 == 0 && KP <= 128 && KH * KW <= KP,
              "fattn: KP must be %32==0 and <=128");
  TORCH_CHECK(q.scalar_type() == ua.scalar_type());
  auto e = torch::empty({B, L}, ua.options().dtype(at::kFloat));
  auto tcache = torch::empty_like(ua);
  const dim3 grid((L + FA_BM - 1) / FA_BM, B);
  const int smem = FA_BM * (KP + 8) * sizeof(bf16_t) + FA_BM * sizeof(float);
  if (ua.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(fattn_score_fwd_kernel<bf16_t>, grid, dim3(256), smem,
                       fa_stream(),
                       (const float*)cov.data_ptr(),
                       (const bf16_t*)wpad.data_ptr(),
                       (const bf16_t*)q.data_ptr(),
                       (const bf16_t*)ua.data_ptr(),
                       (const float*)nu.data_ptr(),
                       (const float*)bias.data_ptr(),
                       (float*)e.data_ptr(), (bf16_t*)tcache.data_ptr(),
                       L, A, Hp, Wp, (int)KH, (int)KW, KP);
  } else {
    TORCH_CHECK(ua.scalar_type() == at::kFloat);
    hipLaunchKernelGGL(fattn_score_fwd_kernel<float>, grid, dim3(256), smem,
                       fa_stream(),
                       (const float*)cov.data_ptr(),
                       (const bf16_t*)wpad.data_ptr(),
                       (const float*)q.data_ptr(),
                       (const float*)ua.data_ptr(),
                       (const float*)nu.data_ptr(),
                       (const float*)bias.data_ptr(),
                       (float*)e.data_ptr(), (float*)tcache.data_ptr(),
                       L, A, Hp, Wp, (int)KH, (int)KW, KP);
  }
  hipError_t err = hipGetLastError();
  TORCH_CHECK(err == hipSuccess, "fattn_score_fwd: ", hipGetErrorString(err));
  return {e, tcache};
}

std::vector<torch::Tensor> fattn_score_bwd(
    torch::Tensor de, torch::Tensor tcache, torch::Tensor nu,
    torch::Tensor wTpad, int64_t Hp, int64_t Wp, int64_t KH, int64_t KW) {
  TORCH_CHECK(de.is_cuda() && de.is_contiguous() &&
              de.scalar_type() == at::kFloat);
  TORCH_CHECK(tcache.is_contiguous());
  TORCH_CHECK(wTpad.is_contiguous() && wTpad.scalar_type() == at::kBFloat16);
  const int B = tcache.size(0), L = tcache.size(1), A = tcache.size(2);
  const int KP = wTpad.size(0);
  TORCH_CHECK(wTpad.size(1) == A && A % 64 == 0 && A <= 512);
  TORCH_CHECK(KP % 32 == 0 && KP <= 128);
  auto dua = torch::empty_like(tcache);
  auto dq = torch::empty({B, A}, de.options());
  auto dnu = torch::empty({A}, de.options());
  auto dcov = torch::empty({B, Hp, Wp}, de.options());
  fa_zero(dq);
  fa_zero(dnu);
  fa_zero(dcov);
  const dim3 grid((L + FA_BM - 1) / FA_BM, B);
  const int smem = FA_BM * (A + 8) * sizeof(bf16_t);
  if (tcache.scalar_type() == at::kBFloat16) {
    hipLaunchKernelGGL(fattn_score_bwd_kernel<bf16_t>, grid, dim3(256), smem,
                       fa_stream(),
                       (const float*)de.data_ptr(),
                       (const bf16_t*)tcache.data_ptr(),
                       (const float*)nu.data_ptr(),
                       (const bf16_t*)wTpad.data_ptr(),
                       (bf16_t*)dua.data_ptr(), (float*)dq.data_ptr(),
                       (float*)dnu.data_ptr(), (float*)dcov.data_ptr(),
                       L, A, (int)Hp, (int)Wp, (int)KH, (int)KW, KP);
  } else {
    TORCH_CHECK(tcache.scalar_type() == at::kFloat);
    hipLaunchKernelGGL(fattn_score_bwd_kernel<float>, grid, dim3(256), smem,
                       fa_stream(),
                       (const float*)de.data_ptr(),
                       (const float*)tcache.data_ptr(),
                       (const float*)nu.data_ptr(),
                       (const bf16_t*)wTpad.data_ptr(),
                       (float*)dua.data_ptr(), (float*)dq.data_ptr(),
                       (float*)dnu.data_ptr(), (float*)dcov.data_ptr(),
                       L, A, (int)Hp, (int)Wp, (int)KH, (int)KW, KP);
  }
  hipError_t err = hipGetLastError();
  TORCH_CHECK(err == hipSuccess, "fattn_score_bwd: ", hipGetErrorString(err));
  return {dua, dq, dnu, dcov};
}
