// Streaming side-information NCC search for CDNA4 (gfx950).
//
// Reference semantics (/root/reference/src/siFinder.py:7-135,
// src/siFull_img.py:5-68, mask from src/AE.py:49,193-220): every
// non-overlapping (ph, pw) patch of the decoded image x is Pearson-correlated
// against every location of the decoded side image y (both fixed-normalized
// and H1H2H3-decorrelated), the correlation is weighted by a per-patch
// Gaussian location prior, the argmax location is found, and the winning
// patches are gathered FROM THE ORIGINAL y and scattered into y_syn.
//
// The reference materializes the (Hc, Wc, P) correlation volume and an
// equally-sized mask constant (~722 MB each at 320x960). This implementation
// streams: the correlation xy[p,i,j] is an implicit GEMM
// (M = P patches, N = Hc*Wc locations, K = 3*ph*pw) on bf16 MFMA with fp32
// accumulation; the Pearson normalization, the Gaussian prior
// (evaluated inline, never materialized) and a running packed argmax live in
// the epilogue, so nothing bigger than the transformed images ever exists.
//
// Kernel pipeline:
//   1. transform:  (3,H,W) fp32 -> bf16 t = H1H2H3((v - mean)/std)
//   2. patch_stats: per-patch sum / sum-of-squares of t_x        (P x 2 f32)
//   3. ysum_row + ysum_col: sliding-window sums of t_y           (Hc x Wc f32)
//   4. ncc_main:   MFMA correlation + Pearson + prior + packed atomic argmax
//   5. scatter:    winning (row, col) -> gather y_orig patches -> y_syn
//
// Tiling of ncc_main: workgroup = 4 waves = (16 patches) x (64 cols) x
// (8 rows, looped). A-tile (patches, 16 x Kpad bf16) and the y window
// ((TI+ph-1) x (TJ+pw-1) x 3 bf16) live in LDS; mfma_f32_16x16x32_bf16
// accumulates K in fp32. The argmax is kept per-lane as a packed
// (order-preserving-float, ~index) u64 so ties resolve to the SMALLEST index
// (TF argmax tie rule), merged per-patch with one atomicMax.

#include "common.h"

namespace dsin {

// KITTI stats used inside the SI search (reference src/siFinder.py:62-63;
// the `variances` there are standard deviations)
__constant__ float SIF_MEAN[3] = {93.70454143384742f, 98.28243432206516f,
                                  94.84678088809876f};
__constant__ float SIF_STD[3] = {73.56493292844912f, 75.88547006820752f,
                                 76.74838442810665f};

constexpr float FOURLN2 = 2.772588722239781f;  // 4 ln 2
constexpr float NCC_EPS = 1e-10f;  // guards zero-variance windows (both paths)

using f32x4 = __attribute__((ext_vector_type(4))) float;
using bf16x8 = __attribute__((ext_vector_type(8))) __bf16;

// ---------------------------------------------------------------- stage 1

__global__ void transform_kernel(const float* __restrict__ img,
                                 bf16* __restrict__ out, int64_t hw) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (; i < hw; i += stride) {
    float r = (img[i] - SIF_MEAN[0]) / SIF_STD[0];
    float g = (img[hw + i] - SIF_MEAN[1]) / SIF_STD[1];
    float b = (img[2 * hw + i] - SIF_MEAN[2]) / SIF_STD[2];
    out[i] = f2b(r + g);
    out[hw + i] = f2b(r - g);
    out[2 * hw + i] = f2b(0.5f * (r + b));
  }
}

// ---------------------------------------------------------------- stage 2

__global__ void patch_stats_kernel(const bf16* __restrict__ tx,
                                   float* __restrict__ psum,   // (P,)
                                   float* __restrict__ psum2,  // (P,)
                                   int H, int W, int ph, int pw, int gw) {
  int p = blockIdx.x;
  int pr = (p / gw) * ph, pc = (p % gw) * pw;
  int K = 3 * ph * pw;
  float s = 0.f, s2 = 0.f;
  for (int k = threadIdx.x; k < K; k += blockDim.x) {
    int c = k / (ph * pw), rem = k % (ph * pw);
    int a = rem / pw, b = rem % pw;
    float v = b2f(tx[(int64_t)c * H * W + (int64_t)(pr + a) * W + pc + b]);
    s += v;
    s2 += v * v;
  }
  s = wave_reduce_sum(s);
  s2 = wave_reduce_sum(s2);
  if (threadIdx.x == 0) {
    psum[p] = s;
    psum2[p] = s2;
  }
}

// ---------------------------------------------------------------- stage 3

__global__ void ysum_row_kernel(const bf16* __restrict__ ty,
                                float* __restrict__ s1, float* __restrict__ s2,
                                int H, int W, int pw, int Wc) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t total = (int64_t)H * Wc;
  if (i >= total) return;
  int r = i / Wc, j = i % Wc;
  float a = 0.f, b = 0.f;
  for (int c = 0; c < 3; ++c) {
    const bf16* row = ty + (int64_t)c * H * W + (int64_t)r * W + j;
    for (int k = 0; k < pw; ++k) {
      float v = b2f(row[k]);
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
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int64_t total = (int64_t)Hc * Wc;
  if (i >= total) return;
  int r = i / Wc, j = i % Wc;
  float a = 0.f, b = 0.f;
  for (int k = 0; k < ph; ++k) {
    a += s1[(int64_t)(r + k) * Wc + j];
    b += s2[(int64_t)(r + k) * Wc + j];
  }
  sy[i] = a;
  sy2[i] = b;
}

// ---------------------------------------------------------------- stage 4

constexpr int TP = 16;   // patches per workgroup (M tile)
constexpr int TJ = 64;   // cols per workgroup (N tile; 4 waves x 16)
constexpr int TI = 8;    // rows looped per workgroup
constexpr int APAD = 8;  // bf16 pad per A row: stride 20 dwords mod 64 ->
                         // 16 distinct banks for the b128 column read

__global__ __launch_bounds__(256)
void ncc_main_kernel(const bf16* __restrict__ tx, const bf16* __restrict__ ty,
                     const float* __restrict__ psum,
                     const float* __restrict__ psum2,
                     const float* __restrict__ sy,
                     const float* __restrict__ sy2,
                     unsigned long long* __restrict__ best,  // (P,)
                     int H, int W, int ph, int pw, int gw, int P,
                     int Hc, int Wc, int use_mask) {
  const int K = 3 * ph * pw;
  const int KP = (K + 31) & ~31;
  const int YR = TI + ph - 1;
  const int YC = TJ + pw - 1;
  const int YCP = (YC + 8) & ~7;  // round up, keep 16B-aligned rows
  const int ASTRIDE = KP + APAD;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* As = reinterpret_cast<bf16*>(smem);              // [TP][ASTRIDE]
  bf16* Ys = As + TP * ASTRIDE;                          // [3][YR][YCP]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int j0 = blockIdx.x * TJ;
  const int i0 = blockIdx.y * TI;
  const int p0 = blockIdx.z * TP;

  // ---- stage A tile: 16 patches x K (+ zero tail) --------------------
  const int php_w = ph * pw;
  for (int idx = tid; idx < TP * KP; idx += 256) {
    int pi = idx / KP, k = idx % KP;
    float v = 0.f;
    int p = p0 + pi;
    if (k < K && p < P) {
      int c = k / php_w, rem = k % php_w;
      int a = rem / pw, b = rem % pw;
      int pr = (p / gw) * ph + a, pc = (p % gw) * pw + b;
      v = b2f(tx[(int64_t)c * H * W + (int64_t)pr * W + pc]);
    }
    As[pi * ASTRIDE + k] = f2b(v);
  }
  // ---- stage Y window: (YR x YC) x 3 ---------------------------------
  for (int idx = tid; idx < 3 * YR * YC; idx += 256) {
    int c = idx / (YR * YC), rem = idx % (YR * YC);
    int rr = rem / YC, cc = rem % YC;
    int r = i0 + rr, col = j0 + cc;
    float v = (r < H && col < W)
                  ? b2f(ty[(int64_t)c * H * W + (int64_t)r * W + col])
                  : 0.f;
    Ys[(c * YR + rr) * YCP + cc] = f2b(v);
  }
  __syncthreads();

  const int colL = lane & 15;          // local col within the wave's 16
  const int kgrp = lane >> 4;          // 0..3
  const int jj = wid * 16 + colL;      // col within the 64-wide tile
  const int j = j0 + jj;               // absolute output col

  unsigned long long bestk[4] = {0ull, 0ull, 0ull, 0ull};

  const float fK = (float)K;
  const float invK = 1.f / fK;
  const float sh2 = 0.25f * (float)H * (float)H;  // sigma_h = H/2 (src/AE.py:208)
  const float sw2 = 0.25f * (float)W * (float)W;

  for (int i = 0; i < TI; ++i) {
    const int ii = i0 + i;
    if (ii >= Hc) break;
    f32x4 acc = {0.f, 0.f, 0.f, 0.f};
    // per-lane walking (c, a, b) position for k = kt*32 + kgrp*8
    int kbase = kgrp * 8;
    int c = 0, a = kbase / pw, b = kbase % pw;
    while (a >= ph) { a -= ph; ++c; }
    for (int kt = 0; kt < KP / 32; ++kt) {
      bf16x8 afrag =
          *reinterpret_cast<const bf16x8*>(&As[colL * ASTRIDE + kt * 32 + kbase]);
      bf16x8 bfrag;
      int cc = c, aa = a, bb = b;
      const int kcur = kt * 32 + kbase;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        bf16 v = (bf16)0.f;
        if (kcur + e < K) v = Ys[((cc * YR) + (i + aa)) * YCP + jj + bb];
        bfrag[e] = *reinterpret_cast<__bf16*>(&v);
        if (++bb == pw) {
          bb = 0;
          if (++aa == ph) { aa = 0; ++cc; }
        }
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc, 0, 0, 0);
      // advance (c, a, b) by 32
      b += 32;
      while (b >= pw) {
        b -= pw;
        if (++a == ph) { a = 0; ++c; }
      }
    }
    // ---- epilogue: Pearson + Gaussian prior + running packed argmax ----
    if (j < Wc) {
      const float syv = sy[(int64_t)ii * Wc + j];
      const float sy2v = sy2[(int64_t)ii * Wc + j];
      const float ym = syv * invK;
      const float deny = sy2v - 2.f * ym * syv + fK * ym * ym;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int prow = p0 + kgrp * 4 + reg;
        if (prow >= P) continue;
        const float sxv = psum[prow];
        const float sx2v = psum2[prow];
        const float xm = sxv * invK;
        const float denx = sx2v - 2.f * xm * sxv + fK * xm * xm;
        const float num = acc[reg] - ym * sxv - xm * syv + fK * xm * ym;
        float val = num / sqrtf(fmaxf(denx * deny, NCC_EPS));
        if (use_mask) {
          const float cr = ((float)(prow / gw) + 0.5f) * (float)ph;
          const float cw = ((float)(prow % gw) + 0.5f) * (float)pw;
          const float di = (float)(ii + ph / 2 - 1) - cr;
          const float dj = (float)(j + pw / 2 - 1) - cw;
          val *= __expf(-FOURLN2 * (di * di / sh2 + dj * dj / sw2));
        }
        const unsigned int idx = (unsigned int)(ii * Wc + j);
        const unsigned long long key =
            ((unsigned long long)float_flip(val) << 32) |
            (unsigned long long)(~idx);
        if (key > bestk[reg]) bestk[reg] = key;
      }
    }
  }

  // reduce across the 16 lanes of each kgrp group (distinct cols, same rows)
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    unsigned long long k = bestk[reg];
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) {
      unsigned long long other = __shfl_xor((long long)k, off, 16);
      if ((unsigned long long)other > k) k = other;
    }
    const int prow = p0 + kgrp * 4 + reg;
    if (colL == 0 && prow < P && k != 0ull)
      atomicMax(&best[prow], k);
  }
}

// ---------------------------------------------------------------- stage 5

__global__ void scatter_kernel(const unsigned long long* __restrict__ best,
                               const float* __restrict__ y_orig,
                               float* __restrict__ y_syn,
                               int64_t* __restrict__ rows,
                               int64_t* __restrict__ cols,
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
    y_syn[(int64_t)c * H * W + (int64_t)(pr + a) * W + pc + b] =
        y_orig[(int64_t)c * H * W + (int64_t)(bi + a) * W + bj + b];
  }
}

// ---------------------------------------------------------------- host

std::tuple<torch::Tensor, torch::Tensor, torch::Tensor> ncc_search(
    torch::Tensor x_dec, torch::Tensor y_dec, torch::Tensor y_orig,
    int64_t ph_, int64_t pw_, bool use_mask) {
  CHECK_CUDA_CONTIG(x_dec);
  CHECK_CUDA_CONTIG(y_dec);
  CHECK_CUDA_CONTIG(y_orig);
  TORCH_CHECK(x_dec.dim() == 3 && x_dec.size(0) == 3, "x_dec must be (3,H,W)");
  TORCH_CHECK(x_dec.sizes() == y_dec.sizes() && x_dec.sizes() == y_orig.sizes(),
              "x/y size mismatch (equal sizes required)");
  const int ph = (int)ph_, pw = (int)pw_;
  const int H = (int)x_dec.size(1), W = (int)x_dec.size(2);
  TORCH_CHECK(H % ph == 0 && W % pw == 0, "image must tile by the patch size");
  const int gh = H / ph, gw = W / pw, P = gh * gw;
  const int Hc = H - ph + 1, Wc = W - pw + 1;
  const int K = 3 * ph * pw;
  const int KP = (K + 31) & ~31;
  const int YR = TI + ph - 1, YC = TJ + pw - 1, YCP = (YC + 8) & ~7;
  const size_t lds = (size_t)TP * (KP + APAD) * 2 + (size_t)3 * YR * YCP * 2;
  TORCH_CHECK(lds <= 160 * 1024, "patch size too large for LDS tiling: ", lds);

  auto optsF = x_dec.options();
  auto stream = at::cuda::getCurrentCUDAStream();
  auto t_x = torch::empty({3, H, W}, optsF.dtype(torch::kBFloat16));
  auto t_y = torch::empty({3, H, W}, optsF.dtype(torch::kBFloat16));
  int64_t hw = (int64_t)H * W;
  hipLaunchKernelGGL(transform_kernel, grid1d((hw + 255) / 256 * 256, 256),
                     dim3(256), 0, stream, x_dec.data_ptr<float>(),
                     (bf16*)t_x.data_ptr(), hw);
  hipLaunchKernelGGL(transform_kernel, grid1d((hw + 255) / 256 * 256, 256),
                     dim3(256), 0, stream, y_dec.data_ptr<float>(),
                     (bf16*)t_y.data_ptr(), hw);

  auto psum = torch::empty({P}, optsF);
  auto psum2 = torch::empty({P}, optsF);
  hipLaunchKernelGGL(patch_stats_kernel, dim3(P), dim3(64), 0, stream,
                     (const bf16*)t_x.data_ptr(), psum.data_ptr<float>(),
                     psum2.data_ptr<float>(), H, W, ph, pw, gw);

  auto s1 = torch::empty({(int64_t)H * Wc}, optsF);
  auto s2 = torch::empty({(int64_t)H * Wc}, optsF);
  auto sy = torch::empty({(int64_t)Hc * Wc}, optsF);
  auto sy2 = torch::empty({(int64_t)Hc * Wc}, optsF);
  hipLaunchKernelGGL(ysum_row_kernel, grid1d((int64_t)H * Wc, 256), dim3(256),
                     0, stream, (const bf16*)t_y.data_ptr(),
                     s1.data_ptr<float>(), s2.data_ptr<float>(), H, W, pw, Wc);
  hipLaunchKernelGGL(ysum_col_kernel, grid1d((int64_t)Hc * Wc, 256), dim3(256),
                     0, stream, s1.data_ptr<float>(), s2.data_ptr<float>(),
                     sy.data_ptr<float>(), sy2.data_ptr<float>(), Hc, Wc, ph);

  auto best = torch::zeros({P}, optsF.dtype(torch::kInt64));  // u64 keys, bit-stored
  dim3 grid((Wc + TJ - 1) / TJ, (Hc + TI - 1) / TI, (P + TP - 1) / TP);
  hipLaunchKernelGGL(ncc_main_kernel, grid, dim3(256), lds, stream,
                     (const bf16*)t_x.data_ptr(), (const bf16*)t_y.data_ptr(),
                     psum.data_ptr<float>(), psum2.data_ptr<float>(),
                     sy.data_ptr<float>(), sy2.data_ptr<float>(),
                     (unsigned long long*)best.data_ptr(), H, W, ph, pw, gw, P,
                     Hc, Wc, use_mask ? 1 : 0);

  auto y_syn = torch::empty_like(y_orig);
  auto rows = torch::empty({P}, optsF.dtype(torch::kInt64));
  auto cols = torch::empty({P}, optsF.dtype(torch::kInt64));
  hipLaunchKernelGGL(scatter_kernel, dim3(P), dim3(256), 0, stream,
                     (const unsigned long long*)best.data_ptr(),
                     y_orig.data_ptr<float>(), y_syn.data_ptr<float>(),
                     rows.data_ptr<int64_t>(), cols.data_ptr<int64_t>(), H, W,
                     ph, pw, gw, P, Wc);
  return {y_syn, rows, cols};
}

// MFMA layout self-check used by the GPU tests: C = A(16x32) @ B(32x16), bf16
// inputs, one wave.
__global__ void mfma_selftest_kernel(const float* __restrict__ A,
                                     const float* __restrict__ B,
                                     float* __restrict__ C) {
  int lane = threadIdx.x & 63;
  bf16x8 a, b;
#pragma unroll
  for (int e = 0; e < 8; ++e) {
    int k = (lane >> 4) * 8 + e;
    bf16 av = f2b(A[(lane & 15) * 32 + k]);   // A[row][k]
    bf16 bv = f2b(B[k * 16 + (lane & 15)]);   // B[k][col]
    a[e] = *reinterpret_cast<__bf16*>(&av);
    b[e] = *reinterpret_cast<__bf16*>(&bv);
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    int row = (lane >> 4) * 4 + reg, col = lane & 15;
    C[row * 16 + col] = acc[reg];
  }
}

torch::Tensor mfma_selftest(torch::Tensor A, torch::Tensor B) {
  CHECK_CUDA_CONTIG(A);
  CHECK_CUDA_CONTIG(B);
  auto C = torch::empty({16, 16}, A.options());
  hipLaunchKernelGGL(mfma_selftest_kernel, dim3(1), dim3(64), 0,
                     at::cuda::getCurrentCUDAStream(), A.data_ptr<float>(),
                     B.data_ptr<float>(), C.data_ptr<float>());
  return C;
}

}  // namespace dsin
