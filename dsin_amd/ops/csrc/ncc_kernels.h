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

constexpr int NCC_TP = 32;   // patches per workgroup (MFMA 32x32 M tile)
constexpr int NCC_TJ = 128;  // cols per workgroup (4 waves x 32)
constexpr int NCC_TI = 8;    // rows looped per workgroup
constexpr int NCC_APAD = 8;  // bf16 pad per A row for the b128 column reads

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
  // mfma_f32_32x32x16_bf16: A[row=l&31][k=(l>>5)*8+e], B[k][col=l&31],
  // C row = (reg&3) + 8*(reg>>2) + 4*(l>>5). 2x the flops per B-gather of
  // the 16x16 shape (the gather is the cost here). Two interleaved i-rows
  // keep two independent accumulator chains in flight.
  const int K = 3 * ph * pw;
  const int KP = (K + 15) & ~15;
  const int YR = NCC_TI + ph - 1;
  const int YC = NCC_TJ + pw - 1;
  const int YCP = (YC + 8) & ~7;
  const int ASTRIDE = KP + NCC_APAD;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  ncbf16* As = reinterpret_cast<ncbf16*>(smem);                // [TP][ASTRIDE]
  ncbf16* Ys = As + NCC_TP * ASTRIDE;                          // [3][YR][YCP]
  unsigned short* Ko = reinterpret_cast<unsigned short*>(Ys + 3 * YR * YCP);
  float* Pstat = reinterpret_cast<float*>(Ko + ((KP + 7) & ~7));  // [5][TP]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int j0 = blockIdx.x * NCC_TJ;
  const int i0 = blockIdx.y * NCC_TI;
  const int p0 = blockIdx.z * NCC_TP;

  const float fK = (float)K;
  const float invK = 1.f / fK;
  const float ish2 = 4.0f / ((float)H * (float)H);   // 1/sh^2, sh = H/2
  const float isw2 = 4.0f / ((float)W * (float)W);

  // ---- stage A tile (32 patches x K), y window, koffs, patch stats ----
  const int php_w = ph * pw;
  for (int k = tid; k < KP; k += 256) Ko[k] = (k < K) ? koffs[k] : 0;
  if (tid < NCC_TP) {
    const int p = min(p0 + tid, P - 1);
    const float sxv = psum[p];
    const float xm = sxv * invK;
    Pstat[tid] = sxv;
    Pstat[NCC_TP + tid] = xm;
    Pstat[2 * NCC_TP + tid] = psum2[p] - 2.f * xm * sxv + fK * xm * xm;
    Pstat[3 * NCC_TP + tid] = ((float)(p / gw) + 0.5f) * (float)ph;  // cr
    Pstat[4 * NCC_TP + tid] = ((float)(p % gw) + 0.5f) * (float)pw;  // cw
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

  const int colL = lane & 31;          // col within the wave's 32
  const int kgrp = lane >> 5;          // 0/1 (k-offset 0/8 within 16)
  const int jj = wid * 32 + colL;
  const int j = j0 + jj;
  const bool jvalid = j < Wc;

  typedef __attribute__((ext_vector_type(16))) float f32x16;
  unsigned long long bestk[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) bestk[r] = 0ull;

  const ncbf16* __restrict__ arow = &As[kgrp * 8];
  const unsigned short* __restrict__ krow = &Ko[kgrp * 8];
  typedef __attribute__((ext_vector_type(8))) unsigned short u16x8;

  for (int i = 0; i < NCC_TI; i += 2) {
    const int ii = i0 + i;
    if (ii >= Hc) break;
    const bool i1ok = (ii + 1) < Hc;
    f32x16 acc0 = {};
    f32x16 acc1 = {};
    const ncbf16* __restrict__ yb0 = &Ys[i * YCP + jj];
    const ncbf16* __restrict__ yb1 = &Ys[(i + 1) * YCP + jj];
    for (int kt = 0; kt < KP / 16; ++kt) {
      const u16x8 ko8 =
          *reinterpret_cast<const u16x8*>(krow + kt * 16);
      bf16x8 bf0, bf1;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        ncbf16 v0 = yb0[ko8[e]];
        ncbf16 v1 = yb1[ko8[e]];
        bf0[e] = *reinterpret_cast<__bf16*>(&v0);
        bf1[e] = *reinterpret_cast<__bf16*>(&v1);
      }
      // A fragment: row = colL (patch), k = kgrp*8 + e
      const bf16x8 afrag = *reinterpret_cast<const bf16x8*>(
          &arow[colL * ASTRIDE + kt * 16]);
      acc0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(afrag, bf0, acc0, 0, 0, 0);
      acc1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(afrag, bf1, acc1, 0, 0, 0);
    }
    // ---- epilogue for rows ii and ii+1 ----
#pragma unroll
    for (int half = 0; half < 2; ++half) {
      if (half == 1 && !i1ok) break;
      const int iw = ii + half;
      const long long sidx = (long long)iw * Wc + (jvalid ? j : 0);
      const float syv = sy[sidx];
      const float sy2v = sy2[sidx];
      const float ym = syv * invK;
      const float deny = sy2v - 2.f * ym * syv + fK * ym * ym;
      const float di = (float)(iw + ph / 2 - 1);
      const unsigned int idx = (unsigned int)(iw * Wc + j);
#pragma unroll
      for (int reg = 0; reg < 16; ++reg) {
        const int prow = (reg & 3) + 8 * (reg >> 2) + 4 * kgrp;
        const float av = half == 0 ? acc0[reg] : acc1[reg];
        const float sxv = Pstat[prow];
        const float xm = Pstat[NCC_TP + prow];
        const float denx = Pstat[2 * NCC_TP + prow];
        const float num = av - ym * sxv - xm * syv + fK * xm * ym;
        float val = num * __builtin_amdgcn_rsqf(fmaxf(denx * deny, NCC_EPS));
        if (use_mask) {
          const float dd = di - Pstat[3 * NCC_TP + prow];
          const float dj = (float)(j + pw / 2 - 1) - Pstat[4 * NCC_TP + prow];
          val *= __expf(-FOURLN2 * (dd * dd * ish2 + dj * dj * isw2));
        }
        unsigned long long key =
            ((unsigned long long)nfloat_flip(val) << 32) |
            (unsigned long long)(~idx);
        key = (jvalid && (p0 + prow) < P) ? key : 0ull;
        if (key > bestk[reg]) bestk[reg] = key;
      }
    }
  }

  // reduce across the 32 lanes of each half (distinct cols, same row set)
#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    unsigned long long k = bestk[reg];
#pragma unroll
    for (int off = 16; off > 0; off >>= 1) {
      unsigned long long other =
          (unsigned long long)__shfl_xor((long long)k, off, 32);
      if (other > k) k = other;
    }
    const int prow = p0 + (reg & 3) + 8 * (reg >> 2) + 4 * kgrp;
    if (colL == 0 && prow < P && k != 0ull) atomicMax(&best[prow], k);
  }
}

// ------------------------------------------------------- stage 4, variant 2
// Requires pw % 8 == 0 (the default (20,24) and every bench fallback patch).
// The v1 kernel keeps the whole (32 x K) A-tile in LDS (93 KB at K=1440 ->
// ONE workgroup per CU, 1 wave/SIMD) and gathers each B fragment with 8
// scalar LDS reads + packing (issue-bound, 105 cyc/MFMA measured). Here:
//   * A is staged in K-SLICES of NCC_SL taps (33 KB -> 3-4 workgroups/CU,
//     the load latency finally has TLP to hide under);
//   * each 8-element k-group of a row lies in ONE patch row, so a B
//     fragment is a single unaligned global u16x8 read of the L1/L2-hot
//     transformed side image (no y-window staging, no offset table, no
//     packing VALU);
//   * the k-loop is OUTER over 4-row output groups, so one LDS A fragment
//     feeds 4 MFMAs (4 independent accumulator chains).
constexpr int NCC_SL = 384;  // taps per A slice (multiple of 16; 2x25KB buffers -> 3 blocks/CU)
constexpr int NCC_TJ2 = 64;  // v2 col tile: 2 j-waves x 32 — the y window
                             // (3 x 27 x 87 bf16 = 14 KB/block) then fits
                             // L1 with two co-resident blocks; the other 2
                             // waves take the other 4-row output group

__global__ __launch_bounds__(256)
void ncc_main_v2_kernel(const ncbf16* __restrict__ tx,
                        const ncbf16* __restrict__ ty,
                        const unsigned int* __restrict__ aoffs,
                        const float* __restrict__ psum,
                        const float* __restrict__ psum2,
                        const float* __restrict__ sy,
                        const float* __restrict__ sy2,
                        unsigned long long* __restrict__ best,  // (P,)
                        int H, int W, int ph, int pw, int gw, int P,
                        int Hc, int Wc, int use_mask) {
  const int K = 3 * ph * pw;
  const int KP = (K + 15) & ~15;
  const int ASTRIDE = NCC_SL + NCC_APAD;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  ncbf16* As = reinterpret_cast<ncbf16*>(smem);      // 2 x [TP][ASTRIDE]
  float* Pstat = reinterpret_cast<float*>(As + 2 * NCC_TP * ASTRIDE);  // [5][TP]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int j0 = blockIdx.x * NCC_TJ2;
  const int i0 = blockIdx.y * NCC_TI;
  const int p0 = blockIdx.z * NCC_TP;

  const float fK = (float)K;
  const float invK = 1.f / fK;
  const float ish2 = 4.0f / ((float)H * (float)H);
  const float isw2 = 4.0f / ((float)W * (float)W);

  if (tid < NCC_TP) {
    const int p = min(p0 + tid, P - 1);
    const float sxv = psum[p];
    const float xm = sxv * invK;
    Pstat[tid] = sxv;
    Pstat[NCC_TP + tid] = xm;
    Pstat[2 * NCC_TP + tid] = psum2[p] - 2.f * xm * sxv + fK * xm * xm;
    Pstat[3 * NCC_TP + tid] = ((float)(p / gw) + 0.5f) * (float)ph;
    Pstat[4 * NCC_TP + tid] = ((float)(p % gw) + 0.5f) * (float)pw;
  }

  // slice staging: 8 threads per patch row, each stages 64 taps (8 runs of
  // 8 contiguous image elements via aoffs); zero-fills beyond K / beyond P
  const int s_pi = tid & 31;
  const int s_k0 = (tid >> 5) * 64;
  const int sp = p0 + s_pi;
  const bool s_pok = sp < P;
  const long long s_pbase =
      s_pok ? (long long)((sp / gw) * ph) * W + (long long)((sp % gw) * pw)
            : 0;
  typedef __attribute__((ext_vector_type(8))) unsigned short u16x8;
  auto stage_slice = [&](int s0, int buf) {
    ncbf16* dst = As + buf * NCC_TP * ASTRIDE;
#pragma unroll
    for (int r8 = 0; r8 < 8; ++r8) {
      const int kl = s_k0 + r8 * 8;
      const int kg = s0 + kl;
      u16x8 v = {};
      if (s_pok && kg < K)
        v = *reinterpret_cast<const u16x8*>(&tx[s_pbase + aoffs[kg]]);
      *reinterpret_cast<u16x8*>(&dst[s_pi * ASTRIDE + kl]) = v;
    }
  };

  const int colL = lane & 31;
  const int kgrp = lane >> 5;
  const int jj = (wid & 1) * 32 + colL;   // 2 j-waves cover TJ2 columns
  const int g0 = (wid >> 1) * 4;          // 2 row-group waves cover TI rows
  const int j = j0 + jj;
  const bool jvalid = j < Wc;

  typedef __attribute__((ext_vector_type(16))) float f32x16;
  unsigned long long bestk[16];
#pragma unroll
  for (int r = 0; r < 16; ++r) bestk[r] = 0ull;

  const int nslices = (KP + NCC_SL - 1) / NCC_SL;

  // per-row epilogue (identical math to the v1 kernel)
  auto epilogue_row = [&](const f32x16& acc, int iw) {
    const long long sidx = (long long)iw * Wc + (jvalid ? j : 0);
    const float syv = sy[sidx];
    const float sy2v = sy2[sidx];
    const float ym = syv * invK;
    const float deny = sy2v - 2.f * ym * syv + fK * ym * ym;
    const float di = (float)(iw + ph / 2 - 1);
    const unsigned int idx = (unsigned int)(iw * Wc + j);
#pragma unroll
    for (int reg = 0; reg < 16; ++reg) {
      const int prow = (reg & 3) + 8 * (reg >> 2) + 4 * kgrp;
      const float av = acc[reg];
      const float sxv = Pstat[prow];
      const float xm = Pstat[NCC_TP + prow];
      const float denx = Pstat[2 * NCC_TP + prow];
      const float num = av - ym * sxv - xm * syv + fK * xm * ym;
      float val = num * __builtin_amdgcn_rsqf(fmaxf(denx * deny, NCC_EPS));
      if (use_mask) {
        const float dd = di - Pstat[3 * NCC_TP + prow];
        const float dj = (float)(j + pw / 2 - 1) - Pstat[4 * NCC_TP + prow];
        val *= __expf(-FOURLN2 * (dd * dd * ish2 + dj * dj * isw2));
      }
      unsigned long long key = ((unsigned long long)nfloat_flip(val) << 32) |
                               (unsigned long long)(~idx);
      key = (jvalid && (p0 + prow) < P) ? key : 0ull;
      if (key > bestk[reg]) bestk[reg] = key;
    }
  };

  // this wave's 4-row output group (the A slices are shared by all waves)
  const int iig = i0 + g0;
  // clamped row offsets (rows beyond Hc read row Hc-1; discarded below)
  long long rofs[4];
#pragma unroll
  for (int r = 0; r < 4; ++r)
    rofs[r] = (long long)min(iig + r, Hc - 1) * W + (j0 + jj);
  f32x16 a0 = {}, a1 = {}, a2 = {}, a3 = {};
  // double-buffered slices: slice s+1 stages (scattered global reads + LDS
  // writes into the other buffer) while the MFMA loop consumes slice s; one
  // barrier per slice
  stage_slice(0, 0);
  __syncthreads();
#pragma unroll 1
  for (int s = 0; s < nslices; ++s) {
    if (s + 1 < nslices) stage_slice((s + 1) * NCC_SL, (s + 1) & 1);
    ncbf16* As_cur = As + (s & 1) * NCC_TP * ASTRIDE;
    // incremental (c, a, b) decode of this half-wave's k-group
    int dyc, dya, dyb;
    {
      int k0g = s * NCC_SL + kgrp * 8;
      dyc = k0g / (ph * pw);
      int rem = k0g % (ph * pw);
      dya = rem / pw;
      dyb = rem % pw;
    }
    const int nkt = min(NCC_SL, KP - s * NCC_SL) / 16;
    auto advance = [&]() {
      dyb += 16;
      if (dyb >= pw) {
        dyb -= pw;
        if (++dya >= ph) { dya = 0; ++dyc; }
        if (dyb >= pw) {  // pw == 8: one 16-step crosses two rows
          dyb -= pw;
          if (++dya >= ph) { dya = 0; ++dyc; }
        }
      }
      if (dyc > 2) { dyc = 2; dya = 0; }  // KP padding: A zeros cancel
    };
    // two register SETS ping-pong (no cross-iteration copies): the next
    // chunk's five loads issue before the current chunk's MFMAs
    u16x8 v0a, v1a, v2a, v3a, v0b, v1b, v2b, v3b;
    bf16x8 afa, afb;
    auto load_set = [&](int kt, u16x8& v0, u16x8& v1, u16x8& v2, u16x8& v3,
                        bf16x8& af) {
      const long long offk = ((long long)(dyc * H) + dya) * W + dyb;
      af = *reinterpret_cast<const bf16x8*>(
          &As_cur[colL * ASTRIDE + kt * 16 + kgrp * 8]);
      v0 = *reinterpret_cast<const u16x8*>(&ty[offk + rofs[0]]);
      v1 = *reinterpret_cast<const u16x8*>(&ty[offk + rofs[1]]);
      v2 = *reinterpret_cast<const u16x8*>(&ty[offk + rofs[2]]);
      v3 = *reinterpret_cast<const u16x8*>(&ty[offk + rofs[3]]);
      advance();
    };
    auto mfma_set = [&](const u16x8& v0, const u16x8& v1, const u16x8& v2,
                        const u16x8& v3, const bf16x8& af) {
      a0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          af, *reinterpret_cast<const bf16x8*>(&v0), a0, 0, 0, 0);
      a1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          af, *reinterpret_cast<const bf16x8*>(&v1), a1, 0, 0, 0);
      a2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          af, *reinterpret_cast<const bf16x8*>(&v2), a2, 0, 0, 0);
      a3 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          af, *reinterpret_cast<const bf16x8*>(&v3), a3, 0, 0, 0);
    };
    load_set(0, v0a, v1a, v2a, v3a, afa);
    int kt = 0;
    for (; kt + 2 <= nkt; kt += 2) {
      if (kt + 1 < nkt) load_set(kt + 1, v0b, v1b, v2b, v3b, afb);
      mfma_set(v0a, v1a, v2a, v3a, afa);
      if (kt + 2 < nkt) load_set(kt + 2, v0a, v1a, v2a, v3a, afa);
      mfma_set(v0b, v1b, v2b, v3b, afb);
    }
    if (kt < nkt) mfma_set(v0a, v1a, v2a, v3a, afa);  // odd tail
    __syncthreads();  // next slice's buffer fully written AND consumed
  }
  if (iig + 0 < Hc) epilogue_row(a0, iig + 0);
  if (iig + 1 < Hc) epilogue_row(a1, iig + 1);
  if (iig + 2 < Hc) epilogue_row(a2, iig + 2);
  if (iig + 3 < Hc) epilogue_row(a3, iig + 3);

#pragma unroll
  for (int reg = 0; reg < 16; ++reg) {
    unsigned long long k = bestk[reg];
#pragma unroll
    for (int off = 16; off > 0; off >>= 1) {
      unsigned long long other =
          (unsigned long long)__shfl_xor((long long)k, off, 32);
      if (other > k) k = other;
    }
    const int prow = p0 + (reg & 3) + 8 * (reg >> 2) + 4 * kgrp;
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
