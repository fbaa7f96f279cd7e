// Implicit-GEMM convolution family for CDNA4 (gfx950). Torch-free.
//
// One gather-GEMM forward kernel covers every 2D conv shape in DSIN
// (SURVEY.md section 2b, K1-K3, K15): direct conv with any stride/dilation,
// transposed conv (host zero-stuffs the input), and backward-data (host
// passes spatially-rotated ci<->co-swapped weights) — the geometry lives in
// two per-shape offset tables, not in the kernel:
//   mbase[m] = input-pixel base offset of output pixel m in the PADDED image
//   koff[k]  = offset of filter tap k = (ci, r, s) in the padded image
// so A[m][k] = xpad[mbase[m] + koff[k]] and the conv is the GEMM
//   out[M=pixels][N=couts] = A @ W^T,  W as [co][K] (the natural torch
//   weight layout flattened), bf16 MFMA 16x16x32, fp32 accumulate.
//
// Tiling: workgroup = 4 waves, tile M64 x N64 (each wave M64 x N16,
// 4 MFMA row-subtiles, acc 4x f32x4). A-chunks (64 x 32) staged in LDS with
// an 8-element row pad (conflict-free column b128 reads); W fragments read
// straight from L2 (the whole W panel is <= ~300 KB and shared by every
// M-tile workgroup). Epilogue: optional bias and activation
// (none/ReLU/leaky-0.2), bf16 store, coalesced within each 16-lane group.
//
// conv_wrw_kernel computes dW[co][k] = sum_m dy[co][m] * A[m][k] as a
// second gather-GEMM (M=K_filter, N=couts, K=pixels) with fp32 atomicAdd
// accumulation across pixel-chunk workgroups.

#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace dsin {

using cvbf16 = __hip_bfloat16;
using cv_f32x4 = __attribute__((ext_vector_type(4))) float;
using cv_bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

constexpr int CONV_TM = 64;   // output pixels per workgroup
constexpr int CONV_TN = 64;   // output channels per workgroup (4 waves x 16)
constexpr int CONV_KC = 32;   // K chunk (one MFMA K per step)
constexpr int CONV_AP = 8;    // bf16 pad per A row in LDS

__device__ __forceinline__ float cvb2f(cvbf16 v) { return __bfloat162float(v); }
__device__ __forceinline__ cvbf16 cvf2b(float v) { return __float2bfloat16(v); }

// mbase/koff setup on device (avoids per-call host loops; tables are cached
// per conv plan on the python side)
__global__ void conv_tables_kernel(int* __restrict__ mbase,
                                   int* __restrict__ koff,
                                   int M, int K, int WO, int stride,
                                   int dil, int Wp, int HpWp, int khw,
                                   int kw) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < M) {
    int oh = i / WO, ow = i % WO;
    mbase[i] = oh * stride * Wp + ow * stride;
  }
  if (i < K) {
    int ci = i / khw, rem = i % khw;
    int r = rem / kw, s = rem % kw;
    koff[i] = ci * HpWp + r * dil * Wp + s * dil;
  }
}

__global__ __launch_bounds__(256)
void conv_fwd_kernel(const cvbf16* __restrict__ xpad,   // (Ci, Hp, Wp), padded
                     const cvbf16* __restrict__ wmat,   // (Co, KP+AP) padded
                     const float* __restrict__ bias,    // (Co,) or nullptr
                     cvbf16* __restrict__ out,          // (Co, M) i.e. NCHW
                     const int* __restrict__ mbase,     // (M,)
                     const int* __restrict__ koff,      // (K,)
                     int M, int N, int K, int KP,
                     long long x_img_stride,            // Ci*Hp*Wp
                     long long o_img_stride,            // Co*M
                     int act) {                         // 0 none 1 relu 2 lrelu
  const int WSTRIDE = KP + CONV_AP;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  cvbf16* As = reinterpret_cast<cvbf16*>(smem);  // [CONV_TM][CONV_KC+CONV_AP]
  const int ASTR = CONV_KC + CONV_AP;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int m0 = blockIdx.x * CONV_TM;
  const int n0 = blockIdx.y * CONV_TN + wid * 16;
  const long long img = blockIdx.z;
  const cvbf16* x = xpad + img * x_img_stride;

  const int colL = lane & 15;
  const int kgrp = lane >> 4;

  cv_f32x4 acc[4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f},
                     {0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};

  // stage loop index precompute: element (m, k) of the current chunk
  const int sm = tid & 63;          // pixel within tile (coalesced dim)
  const int sk0 = tid >> 6;         // k within chunk, step 4
  const int gm = m0 + sm;
  const int mb = (gm < M) ? mbase[gm] : 0;

  const int ncol = n0 + colL;       // this lane's output channel
  const cvbf16* wrow = wmat + (long long)(ncol < N ? ncol : 0) * WSTRIDE;

  for (int kc = 0; kc < KP; kc += CONV_KC) {
    // ---- stage A chunk (64 x 32): thread (sm, sk0+4t) ----
    __syncthreads();
#pragma unroll
    for (int t = 0; t < CONV_KC / 4; ++t) {
      const int k = kc + sk0 + 4 * t;
      cvbf16 v = cvf2b(0.f);
      if (k < K && gm < M) v = x[mb + koff[k]];
      As[sm * ASTR + sk0 + 4 * t] = v;
    }
    __syncthreads();
    // ---- MFMA: 4 row-subtiles x K=32 ----
    const cv_bf16x8 bfrag = *reinterpret_cast<const cv_bf16x8*>(
        &wrow[kc + kgrp * 8]);
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const cv_bf16x8 afrag = *reinterpret_cast<const cv_bf16x8*>(
          &As[(mi * 16 + colL) * ASTR + kgrp * 8]);
      acc[mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[mi],
                                                        0, 0, 0);
    }
  }

  // ---- epilogue: D[row=pixel][col=cout]; row = mi*16 + kgrp*4 + reg ----
  const float bv = (bias != nullptr && ncol < N) ? bias[ncol] : 0.f;
  cvbf16* o = out + img * o_img_stride + (long long)(ncol < N ? ncol : 0) * M;
  if (ncol < N) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int m = m0 + mi * 16 + kgrp * 4 + reg;
        if (m < M) {
          float v = acc[mi][reg] + bv;
          if (act == 1) v = fmaxf(v, 0.f);
          else if (act == 2) v = fmaxf(v, 0.2f * v);
          o[m] = cvf2b(v);
        }
      }
    }
  }
}

// dW[co][k] += sum over the workgroup's pixel chunk of dy[co][m]*A[m][k].
// GEMM roles: A' (M'=filter taps K) gathered rows, B' = dy columns.
// Tile: M'64 (taps) x N'64 (couts), K' = pixels chunked by 32.
__global__ __launch_bounds__(256)
void conv_wrw_kernel(const cvbf16* __restrict__ xpad,  // (Ci, Hp, Wp)
                     const cvbf16* __restrict__ dy,    // (Co, M)
                     float* __restrict__ dw,           // (Co, K) fp32 accum
                     const int* __restrict__ mbase,
                     const int* __restrict__ koff,
                     int M, int N, int K,
                     long long x_img_stride, long long dy_img_stride,
                     int pix_chunks) {
  // blockIdx.x: tap tile; blockIdx.y: cout tile; blockIdx.z: pixel chunk*img
  extern __shared__ __attribute__((aligned(16))) char smem[];
  cvbf16* As = reinterpret_cast<cvbf16*>(smem);   // [64][32+AP] taps x pixels
  cvbf16* Bs = As + 64 * (32 + CONV_AP);          // [64][32+AP] couts x pixels
  const int ASTR = 32 + CONV_AP;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int k0 = blockIdx.x * 64;                 // tap tile base
  const int n0 = blockIdx.y * 64 + wid * 16;      // cout tile base (per wave)
  const int img = blockIdx.z / pix_chunks;
  const int pc = blockIdx.z % pix_chunks;
  const int PCHUNK = (M + pix_chunks - 1) / pix_chunks;
  const int p0 = pc * PCHUNK;
  const int p1 = min(p0 + PCHUNK, M);

  const cvbf16* x = xpad + (long long)img * x_img_stride;
  const cvbf16* g = dy + (long long)img * dy_img_stride;

  const int colL = lane & 15;
  const int kgrp = lane >> 4;

  cv_f32x4 acc[4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f},
                     {0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};

  const int sj = tid & 63;        // row of the staged tile (tap or cout)
  const int sp0 = tid >> 6;       // pixel in chunk, step 4
  const int tap = k0 + sj;
  const int ko = (tap < K) ? koff[tap] : 0;
  const int cout = blockIdx.y * 64 + sj;

  for (int pp = p0; pp < p1; pp += 32) {
    __syncthreads();
#pragma unroll
    for (int t = 0; t < 8; ++t) {
      const int p = pp + sp0 + 4 * t;
      const bool pv = p < p1;
      cvbf16 av = cvf2b(0.f), bvv = cvf2b(0.f);
      if (pv && tap < K) av = x[mbase[p] + ko];
      if (pv && cout < N) bvv = g[(long long)cout * M + p];
      As[sj * ASTR + sp0 + 4 * t] = av;
      Bs[sj * ASTR + sp0 + 4 * t] = bvv;
    }
    __syncthreads();
    const cv_bf16x8 bfrag = *reinterpret_cast<const cv_bf16x8*>(
        &Bs[(n0 - blockIdx.y * 64 + colL) * ASTR + kgrp * 8]);
    // NOTE: bfrag holds dy[cout=n0+colL][pixels kgrp*8..+7] — this is the
    // MFMA B operand B[kdim=pixel][col=cout] fragment.
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const cv_bf16x8 afrag = *reinterpret_cast<const cv_bf16x8*>(
          &As[(mi * 16 + colL) * ASTR + kgrp * 8]);
      acc[mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[mi],
                                                        0, 0, 0);
    }
  }

  // D[row=tap][col=cout]; accumulate into dw[cout][tap]
  const int nc = n0 + colL;
  if (nc < N) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int kk = k0 + mi * 16 + kgrp * 4 + reg;
        if (kk < K) atomicAdd(&dw[(long long)nc * K + kk], acc[mi][reg]);
      }
    }
  }
}

// activation-gradient helper for fused lrelu/relu epilogues:
// dyp = dy * act'(y) computed from the post-activation output y.
__global__ void act_bwd_kernel(const cvbf16* __restrict__ dy,
                               const cvbf16* __restrict__ y,
                               cvbf16* __restrict__ out,
                               long long n, int act) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float g = cvb2f(dy[i]);
    float yy = cvb2f(y[i]);
    if (act == 1) g = yy > 0.f ? g : 0.f;
    else if (act == 2) g = yy > 0.f ? g : 0.2f * g;
    out[i] = cvf2b(g);
  }
}

}  // namespace dsin
