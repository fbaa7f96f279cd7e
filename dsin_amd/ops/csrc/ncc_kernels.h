// Device kernels for the streaming NCC side-information search (gfx950).
// Torch-free so the file compiles standalone for .s inspection and probing:
//   hipcc --offload-arch=gfx950 -O3 -x hip -c ncc_kernels.h
// Semantics documented in ncc_search.hip (the host wrapper).
#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace dsin {

using ncbf16 = __hip_bfloat16;
using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

__device__ __forceinline__ float nb2f(ncbf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ ncbf16 nf2b(float v) { return __float2bfloat16(v); }

__device__ __forceinline__ unsigned int nfloat_flip(float f) {
  unsigned int u = __float_as_uint(f);
  return (u & 0x80000000u) ? ~u : (u | 0x80000000u);
}

__device__ __forceinline__ float nwave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return v;
}

// KITTI stats used inside the SI search (reference src/siFinder.py:62-63;
// the `variances` there are standard deviations)
__constant__ float SIF_MEAN[3] = {93.70454143384742f, 98.28243432206516f,
                                  94.84678088809876f};
__constant__ float SIF_STD[3] = {73.56493292844912f, 75.88547006820752f,
                                 76.74838442810665f};

constexpr float FOURLN2 = 2.772588722239781f;  // 4 ln 2
constexpr float NCC_EPS = 1e-10f;  // guards zero-variance windows (both paths)

constexpr int NCC_TP = 16;   // patches per workgroup (MFMA M tile)
constexpr int NCC_TJ = 64;   // cols per workgroup (4 waves x 16)
constexpr int NCC_TI = 8;    // rows looped per workgroup
constexpr int NCC_APAD = 8;  // bf16 pad per A row -> 16 distinct banks on the
                             // column-wise ds_read_b128 (stride 20 dwords)

// ---------------------------------------------------------------- stage 0
// Offset tables, computed once per call:
//   aoffs[k] = element offset of flat-k within the (3,H,W) image relative to
//              a patch origin (k = (c*ph + a)*pw + b)
//   koffs[k] = element offset of flat-k within the LDS y-window image
//              [3][YR][YCP] relative to (row i=0, col jj=0)
__global__ void ncc_offsets_kernel(unsigned int* __restrict__ aoffs,
                                   unsigned short* __restrict__ koffs,
                                   int H, int W, int ph, int pw, int YR,
                                   int YCP, int K) {
  int k = blockIdx.x * blockDim.x + threadIdx.x;
  if (k >= K) return;
  int c = k / (ph * pw), rem = k % (ph * pw);
  int a = rem / pw, b = rem % pw;
  aoffs[k] = (unsigned int)((c * H + a) * W + b);
  koffs[k] = (unsigned short)((c * YR + a) * YCP + b);
}

// ---------------------------------------------------------------- stage 1

__global__ void transform_kernel(const float* __restrict__ img,
                                 ncbf16* __restrict__ out, long long hw) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < hw; i += stride) {
    float r = (img[i] - SIF_MEAN[0]) / SIF_STD[0];
    float g = (img[hw + i] - SIF_MEAN[1]) / SIF_STD[1];
    float b = (img[2 * hw + i] - SIF_MEAN[2]) / SIF_STD[2];
    out[i] = nf2b(r + g);
    out[hw + i] = nf2b(r - g);
    out[2 * hw + i] = nf2b(0.5f * (r + b));
  }
}

// ---------------------------------------------------------------- stage 2

__global__ void patch_stats_kernel(const ncbf16* __restrict__ tx,
                                   const unsigned int* __restrict__ aoffs,
                                   float* __restrict__ psum,
                                   float* __restrict__ psum2,
                                   int H, int W, int ph, int pw, int gw,
                                   int K) {
  int p = blockIdx.x;
  long long base = (long long)((p / gw) * ph) * W + (p % gw) * pw;
  float s = 0.f, s2 = 0.f;
  for (int k = threadIdx.x; k < K; k += blockDim.x) {
    float v = nb2f(tx[base + aoffs[k]]);
    s += v;
    s2 += v * v;
  }
  s = nwave_reduce_sum(s);
  s2 = nwave_reduce_sum(s2);
  if (threadIdx.x == 0) {
    psum[p] = s;
    psum2[p] = s2;
  }
}

// ---------------------------------------------------------------- stage 3

__global__ void ysum_row_kernel(const ncbf16* __restrict__ ty,
                                float* __restrict__ s1, float* __restrict__ s2,
                                int H, int W, int pw, int Wc) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = (long long)H * Wc;
  if (i >= total) return;
  int r = i / Wc, j = i % Wc;
  float a = 0.f, b = 0.f;
  for (int c = 0; c < 3; ++c) {
    const ncbf16* row = ty + ((long long)c * H + r) * W + j;
    for (int k = 0; k < pw; ++k) {
      float v = nb2f(row[k]);
      a += v;
      b += v * v;
    }
  }
  s1[i] = a;
  s2[i] = b;
}

__global__ void ysum_col_kernel(const float* __restrict__ s1,
                                const float* __restrict__ s2,
                                float* __restrict__ sy,
                                float* __restrict__ sy2,
                                int Hc, int Wc, int ph) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = (long long)Hc * Wc;
  if (i >= total) return;
  int r = i / Wc, j = i % Wc;
  float a = 0.f, b = 0.f;
  for (int k = 0; k < ph; ++k) {
    a += s1[(long long)(r + k) * Wc + j];
    b += s2[(long long)(r + k) * Wc + j];
  }
  sy[i] = a;
  sy2[i] = b;
}

// ---------------------------------------------------------------- stage 4
// Main correlation kernel. MFMA 16x16x32 bf16; per k-chunk the B fragment is
// gathered from the LDS y-window via the koffs table (1 broadcast b128 read
// of 8 offsets + 8 u16 value reads per MFMA — no per-element index math).
// Epilogue fully predicated (no divergent branches): fast rsqrt Pearson
// normalization, inline Gaussian prior, packed (flipped-float | ~idx) u64
// running max -> one atomicMax per patch at the end.

__global__ __launch_bounds__(256)
void ncc_main_kernel(const ncbf16* __restrict__ tx, const ncbf16* __restrict__ ty,
                     const unsigned int* __restrict__ aoffs,
                     const unsigned short* __restrict__ koffs,
                     const float* __restrict__ psum,
                     const float* __restrict__ psum2,
                     const float* __restrict__ sy,
                     const float* __restrict__ sy2,
                     unsigned long long* __restrict__ best,  // (P,)
                     int H, int W, int ph, int pw, int gw, int P,
                     int Hc, int Wc, int use_mask) {
  const int K = 3 * ph * pw;
  const int KP = (K + 31) & ~31;
  const int YR = NCC_TI + ph - 1;
  const int YC = NCC_TJ + pw - 1;
  const int YCP = (YC + 8) & ~7;
  const int ASTRIDE = KP + NCC_APAD;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  ncbf16* As = reinterpret_cast<ncbf16*>(smem);                // [TP][ASTRIDE]
  ncbf16* Ys = As + NCC_TP * ASTRIDE;                          // [3][YR][YCP]
  unsigned short* Ko = reinterpret_cast<unsigned short*>(Ys + 3 * YR * YCP);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int j0 = blockIdx.x * NCC_TJ;
  const int i0 = blockIdx.y * NCC_TI;
  const int p0 = blockIdx.z * NCC_TP;

  // ---- stage LDS: A tile (16 patches x KP), y window, koffs copy ------
  for (int k = tid; k < KP; k += 256) {
    Ko[k] = (k < K) ? koffs[k] : (unsigned short)0;
  }
#pragma unroll 1
  for (int pi = 0; pi < NCC_TP; ++pi) {
    const int p = p0 + pi;
    const long long pbase =
        (p < P) ? (long long)((p / gw) * ph) * W + (long long)((p % gw) * pw)
                : 0;
    for (int k = tid; k < KP; k += 256) {
      ncbf16 v = nf2b(0.f);
      if (k < K && p < P) v = tx[pbase + aoffs[k]];
      As[pi * ASTRIDE + k] = v;
    }
  }
  {
    const int nY = 3 * YR * YC;
    for (int idx = tid; idx < nY; idx += 256) {
      int c = idx / (YR * YC), rem = idx % (YR * YC);
      int rr = rem / YC, cc = rem % YC;
      int r = i0 + rr, col = j0 + cc;
      float v = (r < H && col < W)
                    ? nb2f(ty[((long long)c * H + r) * W + col])
                    : 0.f;
      Ys[(c * YR + rr) * YCP + cc] = nf2b(v);
    }
  }
  __syncthreads();

  const int colL = lane & 15;
  const int kgrp = lane >> 4;
  const int jj = wid * 16 + colL;
  const int j = j0 + jj;

  const float fK = (float)K;
  const float invK = 1.f / fK;
  const float ish2 = 4.0f / ((float)H * (float)H);   // 1/sh^2, sh = H/2
  const float isw2 = 4.0f / ((float)W * (float)W);

  // per-reg patch constants (i-independent): stats + prior centers
  float sxv[4], xm[4], denx[4], cr[4], dj2w[4];
  bool pvalid[4];
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int prow = p0 + kgrp * 4 + reg;
    pvalid[reg] = prow < P;
    const int pc = pvalid[reg] ? prow : 0;
    sxv[reg] = psum[pc];
    xm[reg] = sxv[reg] * invK;
    denx[reg] = psum2[pc] - 2.f * xm[reg] * sxv[reg] + fK * xm[reg] * xm[reg];
    cr[reg] = ((float)(pc / gw) + 0.5f) * (float)ph;
    const float cw = ((float)(pc % gw) + 0.5f) * (float)pw;
    const float dj = (float)(j + pw / 2 - 1) - cw;
    dj2w[reg] = dj * dj * isw2;
  }

  unsigned long long bestk[4] = {0ull, 0ull, 0ull, 0ull};
  const ncbf16* __restrict__ arow = &As[colL * ASTRIDE + kgrp * 8];
  const unsigned short* __restrict__ krow = &Ko[kgrp * 8];

  for (int i = 0; i < NCC_TI; ++i) {
    const int ii = i0 + i;
    if (ii >= Hc) break;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    const ncbf16* __restrict__ ybase = &Ys[i * YCP + jj];
    for (int kt = 0; kt < KP / 32; ++kt) {
      const bf16x8 afrag =
          *reinterpret_cast<const bf16x8*>(arow + kt * 32);
      // 8 broadcast offsets (one b128 across the 16-lane group)...
      typedef __attribute__((ext_vector_type(8))) unsigned short u16x8;
      const u16x8 ko8 =
          *reinterpret_cast<const u16x8*>(krow + kt * 32);
      bf16x8 bfrag;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        ncbf16 v = ybase[ko8[e]];          // ...then 8 u16 LDS value reads
        bfrag[e] = *reinterpret_cast<__bf16*>(&v);
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc, 0, 0, 0);
    }
    // ---- predicated epilogue ----
    const bool jvalid = j < Wc;
    const long long sidx = (long long)ii * Wc + (jvalid ? j : 0);
    const float syv = sy[sidx];
    const float sy2v = sy2[sidx];
    const float ym = syv * invK;
    const float deny = sy2v - 2.f * ym * syv + fK * ym * ym;
    const float di = (float)(ii + ph / 2 - 1);
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const float num = acc[reg] - ym * sxv[reg] - xm[reg] * syv
                        + fK * xm[reg] * ym;
      float val = num * __builtin_amdgcn_rsqf(
                            fmaxf(denx[reg] * deny, NCC_EPS));
      if (use_mask) {
        const float dd = di - cr[reg];
        val *= __expf(-FOURLN2 * (dd * dd * ish2 + dj2w[reg]));
      }
      const unsigned int idx = (unsigned int)(ii * Wc + j);
      unsigned long long key = ((unsigned long long)nfloat_flip(val) << 32) |
                               (unsigned long long)(~idx);
      key = (jvalid && pvalid[reg]) ? key : 0ull;
      if (key > bestk[reg]) bestk[reg] = key;
    }
  }

  // reduce across the 16 lanes of each kgrp group (distinct cols, same rows)
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    unsigned long long k = bestk[reg];
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) {
      unsigned long long other =
          (unsigned long long)__shfl_xor((long long)k, off, 16);
      if (other > k) k = other;
    }
    const int prow = p0 + kgrp * 4 + reg;
    if (colL == 0 && prow < P && k != 0ull) atomicMax(&best[prow], k);
  }
}

// ---------------------------------------------------------------- stage 5

__global__ void scatter_kernel(const unsigned long long* __restrict__ best,
                               const float* __restrict__ y_orig,
                               float* __restrict__ y_syn,
                               long long* __restrict__ rows,
                               long long* __restrict__ cols,
                               int H, int W, int ph, int pw, int gw, int P,
                               int Wc) {
  int p = blockIdx.x;
  unsigned int idx = ~(unsigned int)(best[p] & 0xFFFFFFFFull);
  int bi = idx / Wc, bj = idx % Wc;
  if (threadIdx.x == 0) {
    rows[p] = bi;
    cols[p] = bj;
  }
  int pr = (p / gw) * ph, pc = (p % gw) * pw;
  int n = 3 * ph * pw;
  for (int k = threadIdx.x; k < n; k += blockDim.x) {
    int c = k / (ph * pw), rem = k % (ph * pw);
    int a = rem / pw, b = rem % pw;
    y_syn[((long long)c * H + pr + a) * W + pc + b] =
        y_orig[((long long)c * H + bi + a) * W + bj + b];
  }
}

}  // namespace dsin
