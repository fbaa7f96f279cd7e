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
// second gather-GEMM (M=K_filter, N=couts, K=pixels); each pixel-chunk
// workgroup stores its 64x64 tile into its own partial-sum slice of
// dw[B*pix_chunks][N][K] (plain stores — no atomics) and the host reduces
// the slices with one sum(0).

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

// Packed-coordinate tables for the VIRTUAL-pad gather path (VM=1): instead
// of flat offsets into a pre-padded buffer, each entry carries 2D (or
// channel+2D) coordinates so the staging can bounds-test against the REAL
// tensor and produce zeros for the pad/stuff positions — no padded buffer,
// no pad_stuff kernel, no extra HBM round trip per conv.
//   mpack[m] = (oh*stride) << 16 | (ow*stride)        (virtual-image coords)
//   kpack[k] = ci << 20 | (r*dil) << 10 | (s*dil)
// Field limits: ow*stride < 65536, ci < 4096, tap reach < 1024 — all DSIN
// geometries (incl. 1024x2048 fp8 crops and siNet dil=128) fit.
__global__ void conv_tables_v2_kernel(int* __restrict__ mpack,
                                      int* __restrict__ kpack,
                                      int M, int K, int WO, int stride,
                                      int dil, int khw, int kw) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < M) {
    int oh = i / WO, ow = i % WO;
    mpack[i] = ((oh * stride) << 16) | (ow * stride);
  }
  if (i < K) {
    int ci = i / khw, rem = i % khw;
    int r = rem / kw, s = rem % kw;
    kpack[i] = (ci << 20) | ((r * dil) << 10) | (s * dil);
  }
}

typedef __attribute__((ext_vector_type(8))) unsigned short cv_u16x8;
typedef __attribute__((ext_vector_type(4))) unsigned short cv_u16x4;

// Stage one 8-element run from the virtual image pad(stuff(x, SV)) at
// virtual coords (hv, wv0 + i*ST), i = 0..7 (pads already subtracted by the
// caller). Elements outside the real tensor (pad region, stuff zeros,
// borders) come out 0. Vector reads only where provably in-row; the guarded
// per-element path touches only valid addresses, so no slack allocation or
// address clamping is needed.
// BRANCHLESS by construction: the border path issues 8 unconditional
// clamped scalar loads and zero-selects afterwards — an if-guarded load
// per element makes hipcc branch around each load and wait vmcnt(0) per
// element (measured: 175 vmcnt(0) in the staging, ~7x kernel slowdown).
// Every address stays inside the image: rows clamp into [0, H) and columns
// into [0, W).
template <int ST, int SV>
__device__ __forceinline__ void vstage8(const cvbf16* __restrict__ x,
                                        int H, int W, int ci, int hv,
                                        int wv0, cv_u16x8& out) {
  if (SV == 1) {
    const bool rok = (unsigned)hv < (unsigned)H;
    const long long row = ((long long)ci * H + (rok ? hv : 0)) * (long long)W;
    if (rok && wv0 >= 0 && wv0 + (ST == 1 ? 8 : 16) <= W) {
      if (ST == 1) {
        out = *reinterpret_cast<const cv_u16x8*>(&x[row + wv0]);
      } else {  // ST == 2: two contiguous reads, pick every other element
        const cv_u16x8 a0 = *reinterpret_cast<const cv_u16x8*>(&x[row + wv0]);
        const cv_u16x8 a1 =
            *reinterpret_cast<const cv_u16x8*>(&x[row + wv0 + 8]);
#pragma unroll
        for (int i = 0; i < 4; ++i) {
          out[i] = a0[2 * i];
          out[4 + i] = a1[2 * i];
        }
      }
    } else {
      const int wmax = W - 1;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int c = wv0 + i * ST;
        const cvbf16 v = x[row + min(max(c, 0), wmax)];
        out[i] = (rok && (unsigned)c < (unsigned)W)
                     ? *reinterpret_cast<const unsigned short*>(&v)
                     : (unsigned short)0;
      }
    }
  } else {  // SV == 2 (zero-stuffed input), ST == 1 always in this regime
    const bool rok = hv >= 0 && !(hv & 1) && (hv >> 1) < H;
    const long long row =
        ((long long)ci * H + (rok ? (hv >> 1) : 0)) * (long long)W;
    const int par = wv0 & 1;          // valid i have (wv0 + i) even
    const int c0 = (wv0 + par) >> 1;  // real col of the first valid element
    if (rok && c0 >= 0 && c0 + 4 <= W) {
      const cv_u16x4 v = *reinterpret_cast<const cv_u16x4*>(&x[row + c0]);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const unsigned short e =
            par ? (i >= 1 ? v[(i - 1) >> 1] : (unsigned short)0)
                : v[i >> 1];
        out[i] = ((i & 1) == par) ? e : (unsigned short)0;
      }
    } else {
      const int wmax = W - 1;
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int c = wv0 + i;
        const cvbf16 v = x[row + min(max(c >> 1, 0), wmax)];
        out[i] = (rok && !(c & 1) && (unsigned)(c >> 1) < (unsigned)W)
                     ? *reinterpret_cast<const unsigned short*>(&v)
                     : (unsigned short)0;
      }
    }
  }
}

// TM = output pixels per workgroup. 64 is the throughput tile; 32 doubles
// the workgroup count for small-M layers (e.g. the 40x120 resblock convs,
// which at TM=64 launch only ~600 workgroups on 256 CUs and run
// latency-bound at ~2.3 waves/SIMD). The host picks 32 when the TM=64 grid
// would underfill the chip.
// VM=0: flat-offset tables over a pre-padded buffer (legacy; conv3d).
// VM=1: packed-coordinate tables + VIRTUAL pad/stuff — xpad is the RAW
// (Ci, vH, vW) tensor and staging zero-masks the pad ring and stuff holes
// (vpt/vpl = virtual pads, vsv = stuff stride). Kills the pad_stuff kernel
// and the padded-buffer HBM round trip per conv. VST/VSV are the
// COMPILE-TIME conv stride / stuff stride for VM=1 (VST=0 = generic
// per-element path for narrow outputs); runtime `stride`/`vsv` args are
// used by the VM=0 path and must match VST/VSV when VM=1.
template <int TM, int VM, int VST, int VSV>
__global__ __launch_bounds__(256)
void conv_fwd_kernel(const cvbf16* __restrict__ xpad,   // see VM note above
                     const cvbf16* __restrict__ wmat,   // (Co, KP64+AP) 0-pad
                     const float* __restrict__ bias,    // (Co,) or nullptr
                     cvbf16* __restrict__ out,          // (Co, M) i.e. NCHW
                     const int* __restrict__ mbase,     // (M,)
                     const int* __restrict__ koff,      // (K,)
                     int M, int N, int K, int KP,       // KP: 64-multiple
                     long long x_img_stride,            // Ci*Hp*Wp
                     long long o_img_stride,            // Co*M
                     int act, int WO,
                     int stride, int vH, int vW, int vpt, int vpl, int vsv,
                     int oh0, int ow0, int ostep, int WOf,
                     long long o_chan) {
  // oh0/ow0/ostep/WOf/o_chan: phase-strided output placement — output pixel
  // m lands at (oh0 + ostep*(m/WO), ow0 + ostep*(m%WO)) of a WOf-wide image
  // whose channel stride is o_chan. Plain calls pass (0, 0, 1, WO, M).
  // Branchless pipeline: K chunks of 64 (two MFMA k-steps per barrier),
  // double-buffered LDS A-tile with XOR-swizzled addressing. Staging is two
  // overlapped 16B loads per 8-pixel run with a per-element crossing select
  // (handles output-row boundaries without divergence); out-of-range filter
  // taps read clamped addresses and are cancelled by the zero padding of
  // wmat. W fragments for the NEXT chunk are prefetched to registers ahead
  // of the A loads so the compiler's wait before the MFMAs is a counted
  // vmcnt, not a pipeline-draining vmcnt(0).
  const int KC = 64;
  constexpr int MI = TM / 16;       // MFMA row-subtiles per wave
  constexpr int PPT = TM / 8;       // pixel octets per k column (staging)
  constexpr int KCOV = 256 / PPT;   // k columns covered per staging pass
  constexpr int NST = 64 / KCOV;    // staging passes per 64-chunk (2 or 1)
  const int WSTRIDE = KP + CONV_AP;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int ASTR = KC + CONV_AP;
  const int ABUF = TM * ASTR;
  cvbf16* As = reinterpret_cast<cvbf16*>(smem);          // 4 x [TM][KC+AP]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int m0 = blockIdx.x * TM;
  const int n0 = blockIdx.y * CONV_TN + wid * 16;
  const long long img = blockIdx.z;
  const cvbf16* x = xpad + img * x_img_stride;

  const int colL = lane & 15;
  const int kgrp = lane >> 4;

  cv_f32x4 acc[MI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) acc[mi] = {0.f, 0.f, 0.f, 0.f};

  // staging: thread t -> 8 consecutive output pixels at k-column t / PPT
  const int sm8 = (tid % PPT) * 8;
  const int sk = tid / PPT;                // 0..KCOV-1
  const int gm0 = min(m0 + sm8, M - 1);
  const int gm7 = min(m0 + sm8 + 7, M - 1);
  const int mbv0 = mbase[gm0];
  const int mbv1 = mbase[gm7];
  // VM=0: flat base offsets; elem i sits at +i*stride from the run base.
  const int mb0 = VM ? 0 : mbv0;
  const int mb1 = VM ? 0 : mbv1 - 7 * stride;
  // VM=1: virtual coords of the two runs (A: leading row, B: trailing row)
  const int mhA = VM ? (mbv0 >> 16) : 0;
  const int mwA = VM ? (mbv0 & 0xffff) : 0;
  const int mhB = VM ? (mbv1 >> 16) : 0;
  const int mwB = VM ? ((mbv1 & 0xffff) - 7 * stride) : 0;
  // crossing point: first i whose pixel falls on the next output row
  const int cross = WO - (gm0 % WO);       // >= 8 means no crossing
  // VM=0: xbuf carries >= 16 elements of tail slack (ops/conv.py), so
  // in-range vector reads are never clamped; only negative bases are.

  const int ncol = n0 + colL;
  const cvbf16* wrow = wmat + (long long)(ncol < N ? ncol : 0) * WSTRIDE;

  typedef __attribute__((ext_vector_type(8))) unsigned short u16x8;
  u16x8 stage[NST];
  cv_bf16x8 wfrag[2];

  auto load_half = [&](int ko, u16x8& st) {
    if (VM) {
      const int kci = ko >> 20;
      const int kdh = (ko >> 10) & 1023;
      const int kdw = ko & 1023;
      if (VST != 0) {
        cv_u16x8 sa, sb;
        const int hA = mhA + kdh - vpt, wA = mwA + kdw - vpl;
        vstage8<VST ? VST : 1, VSV>(x, vH, vW, kci, hA, wA, sa);
        // the trailing-row (B) run only matters for runs that cross an
        // output row; the vote makes the skip wave-uniform (most tiles
        // sit inside one row), halving staging work on the common path
        if (__any(cross < 8)) {
          const int hB = mhB + kdh - vpt, wB = mwB + kdw - vpl;
          vstage8<VST ? VST : 1, VSV>(x, vH, vW, kci, hB, wB, sb);
#pragma unroll
          for (int i = 0; i < 8; ++i) st[i] = (i < cross) ? sa[i] : sb[i];
        } else {
          st = sa;
        }
      } else {  // generic (narrow WO / odd stride): per-element, branchless
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          const int mp = mbase[min(m0 + sm8 + i, M - 1)];
          const int hv = (mp >> 16) + kdh - vpt;
          const int wv = (mp & 0xffff) + kdw - vpl;
          bool ok;
          long long a;
          if (VSV == 2) {
            ok = hv >= 0 && !(hv & 1) && (hv >> 1) < vH && wv >= 0 &&
                 !(wv & 1) && (wv >> 1) < vW;
            a = ((long long)kci * vH + min(max(hv >> 1, 0), vH - 1)) * vW +
                min(max(wv >> 1, 0), vW - 1);
          } else {
            ok = (unsigned)hv < (unsigned)vH && (unsigned)wv < (unsigned)vW;
            a = ((long long)kci * vH + min(max(hv, 0), vH - 1)) * vW +
                min(max(wv, 0), vW - 1);
          }
          const cvbf16 v = x[a];
          st[i] = ok ? *reinterpret_cast<const unsigned short*>(&v)
                     : (unsigned short)0;
        }
      }
      return;
    }
    if (stride == 1) {
      const u16x8 a = *reinterpret_cast<const u16x8*>(&x[mb0 + ko]);
      const u16x8 b = *reinterpret_cast<const u16x8*>(
          &x[max(mb1 + ko, 0)]);
#pragma unroll
      for (int i = 0; i < 8; ++i) st[i] = (i < cross) ? a[i] : b[i];
    } else if (stride == 2) {
      const int b0 = mb0 + ko;
      const int b1 = max(mb1 + ko, 0);
      const u16x8 a0 = *reinterpret_cast<const u16x8*>(&x[b0]);
      const u16x8 a1 = *reinterpret_cast<const u16x8*>(&x[b0 + 8]);
      const u16x8 c0 = *reinterpret_cast<const u16x8*>(&x[b1]);
      const u16x8 c1 = *reinterpret_cast<const u16x8*>(&x[b1 + 8]);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        st[i] = (i < cross) ? a0[2 * i] : c0[2 * i];
        st[4 + i] = (4 + i < cross) ? a1[2 * i] : c1[2 * i];
      }
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int gm = min(m0 + sm8 + i, M - 1);
        cvbf16 v = x[min(mbase[gm] + ko, (int)x_img_stride - 1)];
        st[i] = *reinterpret_cast<unsigned short*>(&v);
      }
    }
  };
  // koff values are PREFETCHED two chunks ahead into registers: loading
  // koff[k] at staging time serializes the chain koff -> wait -> A-load and
  // was the dominant stall (each set stages every other chunk, so its next
  // k offsets are +2*KC ahead).
  auto ko_at = [&](int k) { return (k < K) ? koff[k] : 0; };
  int koA[NST], koB[NST];
#pragma unroll
  for (int h = 0; h < NST; ++h) {
    koA[h] = ko_at(h * KCOV + sk);
    koB[h] = ko_at(KC + h * KCOV + sk);
  }
  auto load_chunk = [&](int kc) {
    // W first: its consumer (the MFMA phase) waits on it with a counted
    // vmcnt that leaves the later A loads in flight
    wfrag[0] = *reinterpret_cast<const cv_bf16x8*>(&wrow[kc + kgrp * 8]);
    wfrag[1] = *reinterpret_cast<const cv_bf16x8*>(&wrow[kc + 32 + kgrp * 8]);
#pragma unroll
    for (int h = 0; h < NST; ++h) load_half(koA[h], stage[h]);
#pragma unroll
    for (int h = 0; h < NST; ++h) koA[h] = ko_at(kc + 2 * KC + h * KCOV + sk);
  };
  // A-tile byte-address XOR swizzle (see write/read pair): staging writes at
  // an 8-row stride collide on banks; rows stay 16B aligned and 128B blocks
  // stay inside one 8-row octave (8*ASTR*2 = 1152 B) so the map is bijective
  auto aswz = [&](int m, int elem_off) -> int {
    return ((m * ASTR + elem_off) * 2) ^ (((m >> 3) & 7) << 4);
  };
  char* As8 = reinterpret_cast<char*>(As);
  // chunk-invariant offsets, computed once (the staging/read address math
  // was ~12 VALU per MFMA when recomputed per chunk)
  int wr_off[NST * 8];
#pragma unroll
  for (int h = 0; h < NST; ++h)
#pragma unroll
    for (int i = 0; i < 8; ++i)
      wr_off[h * 8 + i] = aswz(sm8 + i, h * KCOV + sk);
  int rd_off[2 * MI];
#pragma unroll
  for (int kk = 0; kk < 2; ++kk)
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
      rd_off[kk * MI + mi] = aswz(mi * 16 + colL, kk * 32 + kgrp * 8);
  auto write_chunk = [&](int buf) {
    char* dst = As8 + buf * ABUF * 2;
#pragma unroll
    for (int h = 0; h < NST; ++h)
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        unsigned short u = stage[h][i];
        *reinterpret_cast<cvbf16*>(&dst[wr_off[h * 8 + i]]) =
            *reinterpret_cast<cvbf16*>(&u);
      }
  };

  // 2-deep software pipeline: 4 LDS buffers, two register stage sets.
  // Invariant at iteration kt (even): buf[kt&3] holds chunk kt, set S1
  // holds chunk kt+1; the body issues loads for chunk kt+2 into S0, MFMAs
  // chunk kt, writes S1 into buf[(kt+1)&3], barriers; odd iterations swap
  // the sets. A-prefetch therefore has ~two MFMA phases to land.
  const int nchunks = KP / KC;
  u16x8 stageB[NST];
  cv_bf16x8 wfragB[2];
  auto load_chunkB = [&](int kc) {
    wfragB[0] = *reinterpret_cast<const cv_bf16x8*>(&wrow[kc + kgrp * 8]);
    wfragB[1] = *reinterpret_cast<const cv_bf16x8*>(&wrow[kc + 32 + kgrp * 8]);
#pragma unroll
    for (int h = 0; h < NST; ++h) load_half(koB[h], stageB[h]);
#pragma unroll
    for (int h = 0; h < NST; ++h) koB[h] = ko_at(kc + 2 * KC + h * KCOV + sk);
  };
  auto write_chunkB = [&](int buf) {
    char* dst = As8 + buf * ABUF * 2;
#pragma unroll
    for (int h = 0; h < NST; ++h)
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        unsigned short u = stageB[h][i];
        *reinterpret_cast<cvbf16*>(&dst[wr_off[h * 8 + i]]) =
            *reinterpret_cast<cvbf16*>(&u);
      }
  };
  auto mfma_chunk = [&](int kt, const cv_bf16x8* w2) {
    const char* cur = As8 + (kt & 3) * ABUF * 2;
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int mi = 0; mi < MI; ++mi) {
        const cv_bf16x8 afrag = *reinterpret_cast<const cv_bf16x8*>(
            &cur[rd_off[kk * MI + mi]]);
        acc[mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, w2[kk],
                                                          acc[mi], 0, 0, 0);
      }
    }
  };

  load_chunk(0);              // -> stage/wfrag (set A)
  write_chunk(0);
  cv_bf16x8 wA[2] = {wfrag[0], wfrag[1]};
  cv_bf16x8 wB[2] = {wA[0], wA[1]};
  if (nchunks > 1) {
    load_chunkB(KC);          // chunk 1 -> set B
    wB[0] = wfragB[0];
    wB[1] = wfragB[1];
  }
  __syncthreads();

  for (int kt = 0; kt + 1 < nchunks; kt += 2) {
    // even iteration: compute chunk kt (in buf kt&3), set B holds kt+1
    if (kt + 2 < nchunks) load_chunk((kt + 2) * KC);
    mfma_chunk(kt, wA);
    write_chunkB((kt + 1) & 3);
    __syncthreads();
    // odd iteration: compute chunk kt+1, set A holds kt+2
    if (kt + 3 < nchunks) load_chunkB((kt + 3) * KC);
    mfma_chunk(kt + 1, wB);
    wA[0] = wfrag[0];   // W(kt+2): copied only after both MFMA phases so
    wA[1] = wfrag[1];   // its wait never stalls the matrix pipe
    if (kt + 2 < nchunks) {
      write_chunk((kt + 2) & 3);
      __syncthreads();
    }
    wB[0] = wfragB[0];
    wB[1] = wfragB[1];
  }
  if (nchunks & 1) {
    // odd count: last chunk sits in set A's buffer (written above)
    mfma_chunk(nchunks - 1, wA);
  }

  // ---- epilogue: D[row=pixel][col=cout]; row = mi*16 + kgrp*4 + reg ----
  const float bv = (bias != nullptr && ncol < N) ? bias[ncol] : 0.f;
  cvbf16* o = out + img * o_img_stride +
              (long long)(ncol < N ? ncol : 0) * o_chan;
  if (ncol < N) {
#pragma unroll
    for (int mi = 0; mi < MI; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int m = m0 + mi * 16 + kgrp * 4 + reg;
        if (m < M) {
          float v = acc[mi][reg] + bv;
          if (act == 1) v = fmaxf(v, 0.f);
          else if (act == 2) v = fmaxf(v, 0.2f * v);
          if (ostep == 1) {
            o[m] = cvf2b(v);
          } else {
            const int oy = oh0 + ostep * (m / WO);
            const int ox = ow0 + ostep * (m % WO);
            o[(long long)oy * WOf + ox] = cvf2b(v);
          }
        }
      }
    }
  }
}

// Direct 3x3 stride-1 convolution (the resblock / backward-data hot
// shapes: Ci a multiple of 64, dilation 1). The gather-GEMM kernel above
// re-reads every input element ~9 times through L2 (once per filter tap);
// this kernel stages each 8x8-output tile's 10x10 input halo ONCE per
// 64-channel chunk in LDS (pixel-major, channels contiguous) and runs all
// nine taps out of LDS: 9x less gather traffic and 72 MFMAs between
// barriers (one barrier per ci-chunk). W panel layout is (chunk, tap, ci)
// -- wmat_make mode 2/3.
//   out[n][oy0+py][ox0+px] = sum_{ci,r,s} xpad[ci][oy0+py+r][ox0+px+s]
//                                        * w[n][(ci,r,s)]
// vp: VIRTUAL symmetric pad — when nonzero, xpad is an UNPADDED tensor
// (Hp, Wp are its real dims) and staging clamps/zero-masks the halo
// window instead of reading a pre-padded buffer. Used where no other
// consumer needs the padded buffer (backward-data dy; no-grad forwards),
// killing the pad kernel + a full tensor round trip per conv.
__global__ __launch_bounds__(256)
void conv3x3_direct_kernel(const cvbf16* __restrict__ xpad, // (Ci, Hp, Wp)
                           const cvbf16* __restrict__ wmat, // (N, KP+AP)
                           const float* __restrict__ bias,
                           cvbf16* __restrict__ out,        // (N, HO, WO)
                           int Ci, int Hp, int Wp, int N, int HO, int WO,
                           int KP, long long x_img_stride,
                           long long o_img_stride, int act, int vp) {
  constexpr int CIC = 64;           // input channels per LDS chunk
  constexpr int XT = 11;            // LDS tile row stride in pixels (odd:
                                    // breaks 2-row bank aliasing)
  constexpr int TB = 10 * XT * CIC; // bf16 elems per LDS buffer
  extern __shared__ __attribute__((aligned(16))) char smem[];
  cvbf16* As = reinterpret_cast<cvbf16*>(smem);  // 2 x [10][XT][CIC]
  char* As8 = reinterpret_cast<char*>(As);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int tx = (WO + 7) >> 3;
  const int oy0 = (blockIdx.x / tx) * 8;
  const int ox0 = (blockIdx.x % tx) * 8;
  const long long img = blockIdx.z;
  const cvbf16* x = xpad + img * x_img_stride;

  const int colL = lane & 15;
  const int kgrp = lane >> 4;
  const int n0 = blockIdx.y * CONV_TN + wid * 16;
  const int ncol = n0 + colL;
  const int WSTRIDE = KP + CONV_AP;
  const cvbf16* wrow = wmat + (long long)(ncol < N ? ncol : 0) * WSTRIDE;

  cv_f32x4 acc[4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f},
                     {0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};

  // LDS address of (pixel p = y*XT+x, channel ci), with a 16B-granule XOR
  // swizzle on the channel group so column reads spread across banks
  auto aoff = [&](int p, int ci) -> int {
    return p * (CIC * 2) + ((((ci >> 3) ^ (p & 7)) << 4) | ((ci & 7) << 1));
  };

  typedef __attribute__((ext_vector_type(8))) unsigned short u16x8;
  // stage chunk c's 10x10xCIC halo tile into LDS buffer buf.
  // vp == 0 (pre-padded source): straight vector reads; the source buffer
  // carries >= 16 elements of tail slack (ops/conv.py) so edge overshoot
  // is safe. vp > 0: window row/cols outside [0,Hp)x[0,Wp) are zeroed; the
  // vector path is used only when its 40-element read provably stays
  // inside the tensor, else a per-element guarded path runs (first/last
  // rows of the image, left/right border tiles).
  // whole-tile interiority is uniform across the workgroup: interior tiles
  // (the vast majority) take a straight-line vector-staging path identical
  // in cost to the pre-padded vp==0 path; only border tiles pay the
  // per-row clamp logic. +16 (not +10) on the column bound keeps the
  // 16-element vector read inside the row.
  const bool tile_int =
      vp > 0 && oy0 >= vp && oy0 + 10 - vp <= Hp && ox0 >= vp &&
      ox0 + 16 - vp <= Wp;
  auto stage = [&](int c, int buf) {
    char* dst = As8 + buf * TB * 2;
    const long long nelem = (long long)Ci * Hp * Wp;
    for (int ridx = tid; ridx < CIC * 10; ridx += 256) {
      const int ci = ridx / 10, y = ridx % 10;
      if (vp == 0 || tile_int) {
        const int gy0 = oy0 - vp;  // >= 0 for tile_int; vp==0: plain base
        const int yy = min(gy0 + y, Hp - 1);  // bottom edge tiles: clamped
        const cvbf16* g =
            x + ((long long)(c * CIC + ci) * Hp + yy) * Wp + (ox0 - vp);
        const u16x8 a = *reinterpret_cast<const u16x8*>(g);
        const u16x8 b = *reinterpret_cast<const u16x8*>(g + 8);
#pragma unroll
        for (int xi = 0; xi < 10; ++xi) {
          unsigned short v = xi < 8 ? a[xi] : b[xi - 8];
          *reinterpret_cast<cvbf16*>(&dst[aoff(y * XT + xi, ci)]) =
              *reinterpret_cast<cvbf16*>(&v);
        }
        continue;
      }
      const int gy = oy0 + y - vp;
      const int cbase = ox0 - vp;
      const bool rowin = gy >= 0 && gy < Hp;
      const long long rowb =
          ((long long)(c * CIC + ci) * Hp + (rowin ? gy : 0)) * Wp;
      const long long off = rowb + cbase;
      if (rowin && cbase >= 0 && cbase + 10 <= Wp && off + 40 <= nelem) {
        const cvbf16* g = x + off;
        const u16x8 a = *reinterpret_cast<const u16x8*>(g);
        const u16x8 b = *reinterpret_cast<const u16x8*>(g + 8);
#pragma unroll
        for (int xi = 0; xi < 10; ++xi) {
          unsigned short v = xi < 8 ? a[xi] : b[xi - 8];
          *reinterpret_cast<cvbf16*>(&dst[aoff(y * XT + xi, ci)]) =
              *reinterpret_cast<cvbf16*>(&v);
        }
      } else {
        // branchless border path: clamped unconditional loads + select
        // (an if-guarded load per element serializes on vmcnt(0))
#pragma unroll
        for (int xi = 0; xi < 10; ++xi) {
          const int col = cbase + xi;
          const cvbf16 lv = x[rowb + min(max(col, 0), Wp - 1)];
          const cvbf16 v =
              (rowin && (unsigned)col < (unsigned)Wp) ? lv : cvf2b(0.f);
          *reinterpret_cast<cvbf16*>(&dst[aoff(y * XT + xi, ci)]) = v;
        }
      }
    }
  };

  // my 16 output pixels per MFMA row-subtile: row l = mi*16 + colL
  // -> pixel (py, px) = ((mi*16+colL) >> 3, (mi*16+colL) & 7)
  const int nchunks = Ci / CIC;
  // W prefetch: one 64-k tap group (2 MFMA k-steps) ahead — small register
  // sets (2 x 2 x b128) keep total pressure low enough for 3+ waves/SIMD
  cv_bf16x8 wA[2], wB[2];
  auto load_w = [&](int kbase, cv_bf16x8* wset) {
#pragma unroll
    for (int j = 0; j < 2; ++j)
      wset[j] = *reinterpret_cast<const cv_bf16x8*>(
          &wrow[kbase + j * 32 + kgrp * 8]);
  };

  stage(0, 0);
  load_w(0, wA);
  if (nchunks > 1) stage(1, 1);
  __syncthreads();

  // chunk body with COMPILE-TIME chunk parity: the W register-set rotation
  // (tap group index is 9c+tap, parity (c+tap)&1) must fold to direct
  // register references — a runtime-selected pointer would spill both sets
  auto run_chunk = [&](auto codd, int c) {
    constexpr int CO = decltype(codd)::value;
    const char* cur = As8 + CO * TB * 2;
    const int kb = c * 9 * CIC;
#pragma unroll
    for (int tap = 0; tap < 9; ++tap) {
      cv_bf16x8* wcur = ((CO + tap) & 1) ? wB : wA;
      cv_bf16x8* wnxt = ((CO + tap) & 1) ? wA : wB;
      if (tap < 8) load_w(kb + (tap + 1) * CIC, wnxt);
      else if (c + 1 < nchunks) load_w(kb + 9 * CIC, wnxt);
      const int r = tap / 3, sx = tap % 3;
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        const int cio = (kk << 5) + kgrp * 8;
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
          const int m = mi * 16 + colL;
          const int p = ((m >> 3) + r) * XT + (m & 7) + sx;
          const cv_bf16x8 afrag = *reinterpret_cast<const cv_bf16x8*>(
              &cur[aoff(p, cio)]);
          acc[mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag, wcur[kk], acc[mi], 0, 0, 0);
        }
      }
    }
  };
  constexpr std::integral_constant<int, 0> C0{};
  constexpr std::integral_constant<int, 1> C1{};
  for (int c = 0; c < nchunks; ++c) {
    if (c & 1) run_chunk(C1, c);
    else run_chunk(C0, c);
    if (c + 1 < nchunks) {
      if (c + 2 < nchunks) {
        __syncthreads();        // buffer (c&1) free only after all waves
        stage(c + 2, c & 1);    // finish the MFMA phase above
      }
      __syncthreads();
    }
  }

  // epilogue: D rows are tile pixels; row = mi*16 + kgrp*4 + reg
  const float bv = (bias != nullptr && ncol < N) ? bias[ncol] : 0.f;
  cvbf16* o = out + img * o_img_stride +
              (long long)(ncol < N ? ncol : 0) * HO * WO;
  if (ncol < N) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int m = mi * 16 + kgrp * 4 + reg;
        const int oy = oy0 + (m >> 3), ox = ox0 + (m & 7);
        if (oy < HO && ox < WO) {
          float v = acc[mi][reg] + bv;
          if (act == 1) v = fmaxf(v, 0.f);
          else if (act == 2) v = fmaxf(v, 0.2f * v);
          o[(long long)oy * WO + ox] = cvf2b(v);
        }
      }
    }
  }
}

// Direct 5x5 stride-2 convolution (encoder h2 / to_bn shapes, Ci % 64 ==
// 0). The gather path re-reads each input element 25x through L2 and
// discards half of every stride-2 staging vector; here each 8x8-output
// tile's 19x19 input halo is staged ONCE per 64-channel chunk (single
// 46 KB LDS buffer -> 3 workgroups/CU) and all 25 taps run out of LDS:
// 200 MFMAs between barriers per wave. W panel layout (chunk, tap, ci) =
// wmat_make mode 2 with khw = 25. Always VIRTUAL pad (vp = padding).
__global__ __launch_bounds__(256)
void conv5x5s2_direct_kernel(const cvbf16* __restrict__ xpad, // (Ci, Hp, Wp)
                             const cvbf16* __restrict__ wmat, // (N, KP+AP)
                             const float* __restrict__ bias,
                             cvbf16* __restrict__ out,        // (N, HO, WO)
                             int Ci, int Hp, int Wp, int N, int HO, int WO,
                             int KP, long long x_img_stride,
                             long long o_img_stride, int act, int vp) {
  constexpr int CIC = 64;
  constexpr int XT = 19;             // odd row stride breaks 2-row aliasing
  constexpr int TB = 19 * XT * CIC;  // bf16 elems, single buffer
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* As8 = smem;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int tx = (WO + 7) >> 3;
  const int oy0 = (blockIdx.x / tx) * 8;
  const int ox0 = (blockIdx.x % tx) * 8;
  const long long img = blockIdx.z;
  const cvbf16* x = xpad + img * x_img_stride;

  const int colL = lane & 15;
  const int kgrp = lane >> 4;
  const int n0 = blockIdx.y * CONV_TN + wid * 16;
  const int ncol = n0 + colL;
  const int WSTRIDE = KP + CONV_AP;
  const cvbf16* wrow = wmat + (long long)(ncol < N ? ncol : 0) * WSTRIDE;

  cv_f32x4 acc[4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f},
                     {0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};

  // 16B-granule XOR swizzle. The tap loop reads pixels at STRIDE 2 (256 B
  // = a full bank row), so the plain (p & 7) key of the 3x3 kernel only
  // takes 4 values across the 8 px lanes; folding p's low bit into bit 2
  // of the key gives 8 distinct granules for both parities.
  auto aoff = [&](int p, int ci) -> int {
    const int key = ((p >> 1) & 7) ^ ((p & 1) << 2);
    return p * (CIC * 2) + ((((ci >> 3) ^ key) << 4) | ((ci & 7) << 1));
  };

  typedef __attribute__((ext_vector_type(8))) unsigned short u16x8;
  const int gy0 = 2 * oy0 - vp;
  const int cbase = 2 * ox0 - vp;
  const bool tile_int = gy0 >= 0 && gy0 + 19 <= Hp && cbase >= 0 &&
                        cbase + 24 <= Wp;
  auto stage = [&](int c) {
    for (int ridx = tid; ridx < CIC * 19; ridx += 256) {
      const int ci = ridx / 19, y = ridx % 19;
      if (tile_int) {
        const cvbf16* g =
            x + ((long long)(c * CIC + ci) * Hp + (gy0 + y)) * Wp + cbase;
        const u16x8 a = *reinterpret_cast<const u16x8*>(g);
        const u16x8 b = *reinterpret_cast<const u16x8*>(g + 8);
        const u16x8 e = *reinterpret_cast<const u16x8*>(g + 16);
#pragma unroll
        for (int xi = 0; xi < 19; ++xi) {
          unsigned short v =
              xi < 8 ? a[xi] : (xi < 16 ? b[xi - 8] : e[xi - 16]);
          *reinterpret_cast<cvbf16*>(&As8[aoff(y * XT + xi, ci)]) =
              *reinterpret_cast<cvbf16*>(&v);
        }
      } else {  // border: clamped unconditional loads + select
        const int gy = gy0 + y;
        const bool rowin = gy >= 0 && gy < Hp;
        const long long rowb =
            ((long long)(c * CIC + ci) * Hp + (rowin ? gy : 0)) * Wp;
#pragma unroll
        for (int xi = 0; xi < 19; ++xi) {
          const int col = cbase + xi;
          const cvbf16 lv = x[rowb + min(max(col, 0), Wp - 1)];
          const cvbf16 v =
              (rowin && (unsigned)col < (unsigned)Wp) ? lv : cvf2b(0.f);
          *reinterpret_cast<cvbf16*>(&As8[aoff(y * XT + xi, ci)]) = v;
        }
      }
    }
  };

  const int nchunks = Ci / CIC;
  cv_bf16x8 wA[2], wB[2];
  auto load_w = [&](int kbase, cv_bf16x8* wset) {
#pragma unroll
    for (int j = 0; j < 2; ++j)
      wset[j] = *reinterpret_cast<const cv_bf16x8*>(
          &wrow[kbase + j * 32 + kgrp * 8]);
  };

  stage(0);
  load_w(0, wA);
  __syncthreads();

  auto run_chunk = [&](auto codd, int c) {
    constexpr int CO = decltype(codd)::value;  // (c*25)&1 == c&1 (25 odd)
    const int kb = c * 25 * CIC;
#pragma unroll
    for (int tap = 0; tap < 25; ++tap) {
      cv_bf16x8* wcur = ((CO + tap) & 1) ? wB : wA;
      cv_bf16x8* wnxt = ((CO + tap) & 1) ? wA : wB;
      if (tap < 24) load_w(kb + (tap + 1) * CIC, wnxt);
      else if (c + 1 < nchunks) load_w(kb + 25 * CIC, wnxt);
      const int r = tap / 5, sx = tap % 5;
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        const int cio = (kk << 5) + kgrp * 8;
#pragma unroll
        for (int mi = 0; mi < 4; ++mi) {
          const int m = mi * 16 + colL;
          const int p = ((m >> 3) * 2 + r) * XT + (m & 7) * 2 + sx;
          const cv_bf16x8 afrag = *reinterpret_cast<const cv_bf16x8*>(
              &As8[aoff(p, cio)]);
          acc[mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag, wcur[kk], acc[mi], 0, 0, 0);
        }
      }
    }
  };
  constexpr std::integral_constant<int, 0> D0{};
  constexpr std::integral_constant<int, 1> D1{};
  for (int c = 0; c < nchunks; ++c) {
    if (c & 1) run_chunk(D1, c);
    else run_chunk(D0, c);
    if (c + 1 < nchunks) {
      __syncthreads();   // all waves done reading the single buffer
      stage(c + 1);
      __syncthreads();
    }
  }

  const float bv = (bias != nullptr && ncol < N) ? bias[ncol] : 0.f;
  cvbf16* o = out + img * o_img_stride +
              (long long)(ncol < N ? ncol : 0) * HO * WO;
  if (ncol < N) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int m = mi * 16 + kgrp * 4 + reg;
        const int oy = oy0 + (m >> 3), ox = ox0 + (m & 7);
        if (oy < HO && ox < WO) {
          float v = acc[mi][reg] + bv;
          if (act == 1) v = fmaxf(v, 0.f);
          else if (act == 2) v = fmaxf(v, 0.2f * v);
          o[(long long)oy * WO + ox] = cvf2b(v);
        }
      }
    }
  }
}

// dW[co][k] += sum over the workgroup's pixel chunk of dy[co][m]*A[m][k].
// GEMM roles: A' (M'=filter taps K) gathered rows, B' = dy columns.
// Tile: M'64 (taps) x N'64 (couts), K' = pixels chunked by 32.
template <int VM, int VST, int VSV>
__global__ __launch_bounds__(256)
void conv_wrw_kernel(const cvbf16* __restrict__ xpad,  // (Ci, Hp, Wp)
                     const cvbf16* __restrict__ dy,    // (Co, M)
                     float* __restrict__ dw,           // (B*chunks, Co, K)
                     const int* __restrict__ mbase,
                     const int* __restrict__ koff,
                     int M, int N, int K,
                     long long x_img_stride, long long dy_img_stride,
                     int pix_chunks, int WO, int mcontig,
                     int st, int vH, int vW, int vpt, int vpl, int vsv) {
  // blockIdx.x: tap tile; blockIdx.y: cout tile; blockIdx.z: pixel chunk*img
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int ASTR = 32 + CONV_AP;
  const int TBUF = 64 * ASTR;
  cvbf16* As = reinterpret_cast<cvbf16*>(smem);   // 2 x [64 taps][32 pixels]
  cvbf16* Bs = As + 2 * TBUF;                     // 2 x [64 couts][32 pixels]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int k0 = blockIdx.x * 64;
  const int img = blockIdx.z / pix_chunks;
  const int pc = blockIdx.z % pix_chunks;
  const int PCHUNK = ((M + pix_chunks - 1) / pix_chunks + 31) & ~31;
  const int p0 = pc * PCHUNK;
  const int p1 = min(p0 + PCHUNK, M);
  // p0 may pass M when PCHUNK's 32-alignment rounds up: the tile still
  // stores (zeros) — its slice is reduced by the host's sum(0).

  const cvbf16* x = xpad + (long long)img * x_img_stride;
  const cvbf16* g = dy + (long long)img * dy_img_stride;

  const int colL = lane & 15;
  const int kgrp = lane >> 4;

  cv_f32x4 acc[4] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f},
                     {0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};

  // staging: thread t stages 8 consecutive pixels of row sj (a tap row of As
  // and a cout row of Bs in alternating quads)
  const int sj = tid >> 2;                        // 0..63 row
  const int sp8 = (tid & 3) * 8;                  // pixel offset 0/8/16/24
  const int tap = k0 + sj;
  const int ko = (tap < K) ? koff[tap] : 0;
  const int cout = blockIdx.y * 64 + sj;
  const long long gofs = (long long)(cout < N ? cout : 0) * M;

  typedef __attribute__((ext_vector_type(8))) unsigned short u16x8;
  u16x8 sa, sb;
  // VM=1: the tap's virtual-coordinate deltas, decoded once
  const int kci = VM ? (ko >> 20) : 0;
  const int kdh = VM ? ((ko >> 10) & 1023) : 0;
  const int kdw = VM ? (ko & 1023) : 0;

  auto load_chunk = [&](int pp) {
    const int p = pp + sp8;
    const bool inb = p + 7 < p1;
    if (VM) {
      if (tap < K && p < p1) {
        const int gm0 = min(p, M - 1), gm7 = min(p + 7, M - 1);
        const int mpA = mbase[gm0], mpB = mbase[gm7];
        if (VST != 0) {
          const int hA = (mpA >> 16) + kdh - vpt;
          const int wA = (mpA & 0xffff) + kdw - vpl;
          // ow is recoverable from the packed field (= ow*VST): no int div
          const int cross = WO - ((mpA & 0xffff) >> (VST == 2 ? 1 : 0));
          cv_u16x8 va, vb;
          vstage8<VST ? VST : 1, VSV>(x, vH, vW, kci, hA, wA, va);
          if (__any(cross < 8)) {
            const int hB = (mpB >> 16) + kdh - vpt;
            const int wB = (mpB & 0xffff) - 7 * VST + kdw - vpl;
            vstage8<VST ? VST : 1, VSV>(x, vH, vW, kci, hB, wB, vb);
#pragma unroll
            for (int i = 0; i < 8; ++i)
              sa[i] = (p + i < p1) ? ((i < cross) ? va[i] : vb[i])
                                   : (unsigned short)0;
          } else {
#pragma unroll
            for (int i = 0; i < 8; ++i)
              sa[i] = (p + i < p1) ? va[i] : (unsigned short)0;
          }
        } else {  // generic: per-element decode, branchless clamped loads
#pragma unroll
          for (int i = 0; i < 8; ++i) {
            const int mp = mbase[min(p + i, M - 1)];
            const int hv = (mp >> 16) + kdh - vpt;
            const int wv = (mp & 0xffff) + kdw - vpl;
            bool ok = p + i < p1;
            long long a;
            if (VSV == 2) {
              ok = ok && hv >= 0 && !(hv & 1) && (hv >> 1) < vH && wv >= 0 &&
                   !(wv & 1) && (wv >> 1) < vW;
              a = ((long long)kci * vH + min(max(hv >> 1, 0), vH - 1)) * vW +
                  min(max(wv >> 1, 0), vW - 1);
            } else {
              ok = ok && (unsigned)hv < (unsigned)vH &&
                   (unsigned)wv < (unsigned)vW;
              a = ((long long)kci * vH + min(max(hv, 0), vH - 1)) * vW +
                  min(max(wv, 0), vW - 1);
            }
            const cvbf16 v = x[a];
            sa[i] = ok ? *reinterpret_cast<const unsigned short*>(&v)
                       : (unsigned short)0;
          }
        }
      } else {
#pragma unroll
        for (int i = 0; i < 8; ++i) sa[i] = 0;
      }
    } else if (inb && mcontig && ((p % WO) + 8 <= WO) && tap < K) {
      sa = *reinterpret_cast<const u16x8*>(&x[mbase[p] + ko]);
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        cvbf16 v = cvf2b(0.f);
        if (p + i < p1 && tap < K) v = x[mbase[p + i] + ko];
        sa[i] = *reinterpret_cast<unsigned short*>(&v);
      }
    }
    if (inb && cout < N) {
      sb = *reinterpret_cast<const u16x8*>(&g[gofs + p]);
    } else {
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        cvbf16 v = cvf2b(0.f);
        if (p + i < p1 && cout < N) v = g[gofs + p + i];
        sb[i] = *reinterpret_cast<unsigned short*>(&v);
      }
    }
  };
  auto write_chunk = [&](int buf) {
    *reinterpret_cast<u16x8*>(&As[buf * TBUF + sj * ASTR + sp8]) = sa;
    *reinterpret_cast<u16x8*>(&Bs[buf * TBUF + sj * ASTR + sp8]) = sb;
  };

  load_chunk(p0);
  write_chunk(0);
  __syncthreads();

  const int nchunks = max((p1 - p0 + 31) / 32, 0);
  for (int t = 0; t < nchunks; ++t) {
    if (t + 1 < nchunks) load_chunk(p0 + (t + 1) * 32);
    const cvbf16* ac = As + (t & 1) * TBUF;
    const cvbf16* bc = Bs + (t & 1) * TBUF;
    const cv_bf16x8 bfrag = *reinterpret_cast<const cv_bf16x8*>(
        &bc[(wid * 16 + colL) * ASTR + kgrp * 8]);
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
      const cv_bf16x8 afrag = *reinterpret_cast<const cv_bf16x8*>(
          &ac[(mi * 16 + colL) * ASTR + kgrp * 8]);
      acc[mi] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(afrag, bfrag, acc[mi],
                                                        0, 0, 0);
    }
    if (t + 1 < nchunks) {
      write_chunk((t + 1) & 1);
      __syncthreads();
    }
  }

  // D[row=tap][col=cout]; store into this workgroup's slice of dw
  float* dws = dw + (long long)blockIdx.z * N * K;
  const int nc = blockIdx.y * 64 + wid * 16 + colL;
  if (nc < N) {
#pragma unroll
    for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int kk = k0 + mi * 16 + kgrp * 4 + reg;
        if (kk < K) dws[(long long)nc * K + kk] = acc[mi][reg];
      }
    }
  }
}

// activation-gradient helper for fused lrelu/relu epilogues:
// dyp = dy * act'(y) computed from the post-activation output y.
// Replaces the torch chain fill-scalar + where + cast (3 kernels) per
// conv backward with one kernel; dy may be fp32 or bf16.
template <typename T>
__global__ void act_bwd_kernel(const T* __restrict__ dy,
                               const cvbf16* __restrict__ y,
                               cvbf16* __restrict__ out,
                               long long n, int act) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float g = (float)dy[i];
    float yy = cvb2f(y[i]);
    if (act == 1) g = yy > 0.f ? g : 0.f;
    else if (act == 2) g = yy > 0.f ? g : 0.2f * g;
    out[i] = cvf2b(g);
  }
}

// builds the zero-padded bf16 W panel for conv_fwd in ONE kernel, replacing
// the torch pad+contiguous (forward) or flip+permute+reshape+pad chain
// (backward-data "rotated" weights, ~4 kernels per conv backward). mode:
//   0: out[n][k] = w1[n][k]                       (N=Co rows, K taps)
//   1: out[ci][co*khw + t] = w1[co][ci*khw + khw-1-t]
//      (full multi-radix tap reversal == flip of every spatial dim)
//   2: direct-conv layout k' = (ci/64)*64*khw + t*64 + ci%64 (channels
//      contiguous per tap per 64-chunk — see conv3x3_direct_kernel)
//   3: mode 1 + mode 2 (rotated weights in direct layout)
// k >= K -> 0.
template <typename T>
__global__ void wmat_make_kernel(const T* __restrict__ w1,
                                 cvbf16* __restrict__ out,
                                 int rows, int kout, int kin, int khw,
                                 int KPA, int mode) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long total = (long long)rows * KPA;
  long long gstride = (long long)gridDim.x * blockDim.x;
  for (; i < total; i += gstride) {
    const int n = (int)(i / KPA);
    int k = (int)(i % KPA);
    float v = 0.f;
    if (k < kout) {
      if (mode >= 2) {   // decode direct layout: k' -> (c, tap, ci_in_chunk)
        const int chunk = k / (64 * khw), rm = k % (64 * khw);
        k = (chunk * 64 + rm % 64) * khw + rm / 64;
      }
      if (mode & 1) {
        const int co = k / khw, t = k % khw;
        v = (float)w1[(long long)co * kin + n * khw + (khw - 1 - t)];
      } else {
        v = (float)w1[(long long)n * kin + k];
      }
    }
    out[i] = cvf2b(v);
  }
}

// Column-gather of a weight panel: out[n][kp] = src[n][ktab[kp]] (0 beyond
// Klocal). Builds the per-phase W panels of the phase-decomposed
// conv-transpose / stride-2 backward path from either the raw fp32 weight
// (src_stride = kin) or a cached bf16 rotated panel (src_stride = KPA).
template <typename T>
__global__ void panel_gather_kernel(const T* __restrict__ src,
                                    const int* __restrict__ ktab,
                                    cvbf16* __restrict__ out,
                                    int rows, int Klocal, int KPA,
                                    int src_stride) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  const long long total = (long long)rows * KPA;
  const long long gstride = (long long)gridDim.x * blockDim.x;
  for (; i < total; i += gstride) {
    const int n = (int)(i / KPA);
    const int k = (int)(i % KPA);
    float v = 0.f;
    if (k < Klocal) v = (float)src[(long long)n * src_stride + ktab[k]];
    out[i] = cvf2b(v);
  }
}

}  // namespace dsin
