// fp8 (OCP e4m3) variants of the gather-GEMM conv kernels (BASELINE
// config 5: the fp8 MFMA conv path). Torch-free.
//
// Same structure as the bf16 kernels in conv_kernels.h — offset-table
// geometry, branchless crossing-select staging, double-buffered LDS — with
// 1-byte elements: fragments are 8 fp8 = one i64 (ds_read_b64), the MFMA is
// v_mfma_f32_16x16x32_fp8_fp8 (fp32 accumulate; non-scaled fp8 runs at the
// bf16 rate, the win is halved staging bytes), and the LDS XOR swizzle moves
// to bits 3..5 (8-byte granules; the 8-row octave is 576 B, a multiple of
// 64, so the swizzled blocks stay inside their octave).
//
// Activations/weights are cast to e4m3 by the host (values are O(1) after
// normalization/BN, well inside e4m3 range; this is the documented
// reduced-precision path, not the default). Output stays bf16.

#pragma once
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>

namespace dsin {

using f8 = unsigned char;  // raw e4m3 byte
typedef __attribute__((ext_vector_type(8))) unsigned char u8x8;
typedef __attribute__((ext_vector_type(4))) float cv8_f32x4;
typedef long long cv8_i64;

constexpr int CONV8_TM = 64;
constexpr int CONV8_TN = 64;
constexpr int CONV8_AP = 8;   // fp8 pad per A row (bytes)

#ifdef DSIN_CONV_FP8_KERNELS
__global__ __launch_bounds__(256)
void conv_fwd_fp8_kernel(const f8* __restrict__ xpad,   // (Ci, Hp, Wp) e4m3
                         const f8* __restrict__ wmat,   // (Co, KP64+AP) 0-pad
                         const float* __restrict__ bias,
                         __hip_bfloat16* __restrict__ out,  // (Co, M)
                         const int* __restrict__ mbase,
                         const int* __restrict__ koff,
                         int M, int N, int K, int KP,
                         long long x_img_stride, long long o_img_stride,
                         int act, int WO, int stride) {
  const int KC = 64;
  const int WSTRIDE = KP + CONV8_AP;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int ASTR = KC + CONV8_AP;          // bytes per row
  const int ABUF = CONV8_TM * ASTR;        // bytes per buffer
  char* As8 = smem;                        // 2 x [TM][KC+AP] fp8 dbuf

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int m0 = blockIdx.x * CONV8_TM;
  const int n0 = blockIdx.y * CONV8_TN + wid * 16;
  const long long img = blockIdx.z;
  const f8* x = xpad + img * x_img_stride;

  const int colL = lane & 15;
  const int kgrp = lane >> 4;

  cv8_f32x4 acc[4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f},
                      {0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};

  const int sm8 = (tid & 7) * 8;
  const int sk = tid >> 3;
  const int gm0 = min(m0 + sm8, M - 1);
  const int gm7 = min(m0 + sm8 + 7, M - 1);
  const int mb0 = mbase[gm0];
  const int mb1 = mbase[gm7] - 7 * stride;
  const int cross = WO - (gm0 % WO);

  const int ncol = n0 + colL;
  const f8* wrow = wmat + (long long)(ncol < N ? ncol : 0) * WSTRIDE;

  u8x8 stage[2];
  cv8_i64 wfrag[2];

  auto load_half = [&](int k, u8x8& st) {
    const int ko = (k < K) ? koff[k] : 0;
    if (stride == 1) {
      const u8x8 a = *reinterpret_cast<const u8x8*>(&x[mb0 + ko]);
      const u8x8 b = *reinterpret_cast<const u8x8*>(&x[max(mb1 + ko, 0)]);
#pragma unroll
      for (int i = 0; i < 8; ++i) st[i] = (i < cross) ? a[i] : b[i];
    } else if (stride == 2) {
      const int b0 = mb0 + ko;
      const int b1 = max(mb1 + ko, 0);
      const u8x8 a0 = *reinterpret_cast<const u8x8*>(&x[b0]);
      const u8x8 a1 = *reinterpret_cast<const u8x8*>(&x[b0 + 8]);
      const u8x8 c0 = *reinterpret_cast<const u8x8*>(&x[b1]);
      const u8x8 c1 = *reinterpret_cast<const u8x8*>(&x[b1 + 8]);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        st[i] = (i < cross) ? a0[2 * i] : c0[2 * i];
        st[4 + i] = (4 + i < cross) ? a1[2 * i] : c1[2 * i];
      }
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int gm = min(m0 + sm8 + i, M - 1);
        st[i] = x[min(mbase[gm] + ko, (int)x_img_stride - 1)];
      }
    }
  };
  auto load_chunk = [&](int kc) {
    wfrag[0] = *reinterpret_cast<const cv8_i64*>(&wrow[kc + kgrp * 8]);
    wfrag[1] = *reinterpret_cast<const cv8_i64*>(&wrow[kc + 32 + kgrp * 8]);
    load_half(kc + sk, stage[0]);
    load_half(kc + 32 + sk, stage[1]);
  };
  // 8B-granule XOR swizzle: staging writes at 8-row stride would collide;
  // 8-row octave = 8*ASTR = 576 B (multiple of 64) keeps the map bijective
  auto aswz = [&](int m, int elem_off) -> int {
    return (m * ASTR + elem_off) ^ (((m >> 3) & 7) << 3);
  };
  int wr_off[16];
#pragma unroll
  for (int h = 0; h < 2; ++h)
#pragma unroll
    for (int i = 0; i < 8; ++i) wr_off[h * 8 + i] = aswz(sm8 + i, h * 32 + sk);
  int rd_off[8];
#pragma unroll
  for (int kk = 0; kk < 2; ++kk)
#pragma unroll
    for (int mi = 0; mi < 4; ++mi)
      rd_off[kk * 4 + mi] = aswz(mi * 16 + colL, kk * 32 + kgrp * 8);

  auto write_chunk = [&](int buf) {
    char* dst = As8 + buf * ABUF;
#pragma unroll
    for (int h = 0; h < 2; ++h)
#pragma unroll
      for (int i = 0; i < 8; ++i) dst[wr_off[h * 8 + i]] = stage[h][i];
  };
  auto mfma_chunk = [&](int kt, const cv8_i64* w2) {
    const char* cur = As8 + (kt & 1) * ABUF;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int mi = 0; mi < 4; ++mi) {
        const cv8_i64 afrag =
            *reinterpret_cast<const cv8_i64*>(&cur[rd_off[kk * 4 + mi]]);
        acc[mi] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(
            afrag, w2[kk], acc[mi], 0, 0, 0);
      }
    }
  };

  // depth-1 pipeline (fp8 staging is half the bytes of bf16; latency is
  // already covered by the MFMA phase at this depth)
  const int nchunks = KP / KC;
  load_chunk(0);
  write_chunk(0);
  cv8_i64 wcur[2] = {wfrag[0], wfrag[1]};
  __syncthreads();
  for (int kt = 0; kt < nchunks; ++kt) {
    if (kt + 1 < nchunks) load_chunk((kt + 1) * KC);
    mfma_chunk(kt, wcur);
    if (kt + 1 < nchunks) {
      wcur[0] = wfrag[0];
      wcur[1] = wfrag[1];
      write_chunk((kt + 1) & 1);
      __syncthreads();
    }
  }

  const float bv = (bias != nullptr && ncol < N) ? bias[ncol] : 0.f;
  __hip_bfloat16* o =
      out + img * o_img_stride + (long long)(ncol < N ? ncol : 0) * M;
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
          o[m] = __float2bfloat16(v);
        }
      }
    }
  }
}

// dW accumulation in fp8 inputs (x and dy both e4m3), fp32 atomics out.
__global__ __launch_bounds__(256)
void conv_wrw_fp8_kernel(const f8* __restrict__ xpad, const f8* __restrict__ dy,
                         float* __restrict__ dw,
                         const int* __restrict__ mbase,
                         const int* __restrict__ koff,
                         int M, int N, int K,
                         long long x_img_stride, long long dy_img_stride,
                         int pix_chunks, int WO, int mcontig) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int ASTR = 32 + CONV8_AP;
  const int TBUF = 64 * ASTR;
  char* As = smem;
  char* Bs = As + 2 * TBUF;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int k0 = blockIdx.x * 64;
  const int img = blockIdx.z / pix_chunks;
  const int pc = blockIdx.z % pix_chunks;
  const int PCHUNK = ((M + pix_chunks - 1) / pix_chunks + 31) & ~31;
  const int p0 = pc * PCHUNK;
  const int p1 = min(p0 + PCHUNK, M);
  // p0 may pass M when PCHUNK rounds up: the tile still stores zeros.

  const f8* x = xpad + (long long)img * x_img_stride;
  const f8* g = dy + (long long)img * dy_img_stride;

  const int colL = lane & 15;
  const int kgrp = lane >> 4;

  cv8_f32x4 acc[4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f},
                      {0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};

  const int sj = tid >> 2;
  const int sp8 = (tid & 3) * 8;
  const int tap = k0 + sj;
  const int ko = (tap < K) ? koff[tap] : 0;
  const int cout = blockIdx.y * 64 + sj;
  const long long gofs = (long long)(cout < N ? cout : 0) * M;

  u8x8 sa, sb;
  auto load_chunk = [&](int pp) {
    const int p = pp + sp8;
    const bool inb = p + 7 < p1;
    const bool avec = inb && mcontig && ((p % WO) + 8 <= WO);
    if (avec && tap < K) {
      sa = *reinterpret_cast<const u8x8*>(&x[mbase[p] + ko]);
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i)
        sa[i] = (p + i < p1 && tap < K) ? x[mbase[p + i] + ko] : (f8)0;
    }
    if (inb && cout < N) {
      sb = *reinterpret_cast<const u8x8*>(&g[gofs + p]);
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i)
        sb[i] = (p + i < p1 && cout < N) ? g[gofs + p + i] : (f8)0;
    }
  };
  auto write_chunk = [&](int buf) {
    *reinterpret_cast<u8x8*>(&As[buf * TBUF + sj * ASTR + sp8]) = sa;
    *reinterpret_cast<u8x8*>(&Bs[buf * TBUF + sj * ASTR + sp8]) = sb;
  };

  load_chunk(p0);
  write_chunk(0);
  __syncthreads();

  const int nchunks = max((p1 - p0 + 31) / 32, 0);
  for (int t = 0; t < nchunks; ++t) {
    if (t + 1 < nchunks) load_chunk(p0 + (t + 1) * 32);
    const char* ac = As + (t & 1) * TBUF;
    const char* bc = Bs + (t & 1) * TBUF;
    const cv8_i64 bfrag = *reinterpret_cast<const cv8_i64*>(
        &bc[(wid * 16 + colL) * ASTR + kgrp * 8]);
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const cv8_i64 afrag = *reinterpret_cast<const cv8_i64*>(
          &ac[(mi * 16 + colL) * ASTR + kgrp * 8]);
      acc[mi] = __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(afrag, bfrag,
                                                           acc[mi], 0, 0, 0);
    }
    if (t + 1 < nchunks) {
      write_chunk((t + 1) & 1);
      __syncthreads();
    }
  }

  const int nc = blockIdx.y * 64 + wid * 16 + colL;
  if (nc < N) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int kk = k0 + mi * 16 + kgrp * 4 + reg;
        if (kk < K)
          dw[(long long)blockIdx.z * N * K + (long long)nc * K + kk] =
              acc[mi][reg];
      }
    }
  }
}

#endif  // DSIN_CONV_FP8_KERNELS

// fused pad/stuff/cast-to-e4m3 (fp8 twin of pad_stuff_kernel)
template <typename T>
__global__ void pad_stuff_fp8_kernel(const T* __restrict__ x,
                                     f8* __restrict__ out,
                                     int C, int H, int W, int Hp, int Wp,
                                     int pt, int pl, int stride,
                                     long long n_img_out, long long n_img_in,
                                     int B) {
  long long idx = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = (long long)B * n_img_out;
  long long gstride = (long long)gridDim.x * blockDim.x;
  for (; idx < total; idx += gstride) {
    long long b = idx / n_img_out;
    long long rem = idx % n_img_out;
    int c = rem / (Hp * Wp);
    int r2 = rem % (Hp * Wp);
    int i = r2 / Wp, j = r2 % Wp;
    float v = 0.f;
    int ii = i - pt, jj = j - pl;
    if (ii >= 0 && jj >= 0) {
      if (stride == 1) {
        if (ii < H && jj < W)
          v = (float)x[b * n_img_in + ((long long)c * H + ii) * W + jj];
      } else if (ii % stride == 0 && jj % stride == 0) {
        ii /= stride;
        jj /= stride;
        if (ii < H && jj < W)
          v = (float)x[b * n_img_in + ((long long)c * H + ii) * W + jj];
      }
    }
    __hip_fp8_e4m3 q(v);
    out[idx] = q.__x;
  }
}

}  // namespace dsin
